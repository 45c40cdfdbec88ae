set -x
cd /root/repo
mkdir -p gpurun_out
# A/B matrix, cold-first: tuned vs untuned GEMM path, then GC depth sweep
timeout 600 python bench.py --steps 8 --warmup 3 > gpurun_out/b_tuned.json 2>gpurun_out/b_tuned.err
tail -1 gpurun_out/b_tuned.json
TA_DISABLE_TUNED_GEMM=1 timeout 600 python bench.py --steps 8 --warmup 3 > gpurun_out/b_untuned.json 2>gpurun_out/b_untuned.err
tail -1 gpurun_out/b_untuned.json
timeout 600 python bench.py --steps 8 --warmup 3 --no-gc > gpurun_out/b_nogc.json 2>gpurun_out/b_nogc.err
tail -1 gpurun_out/b_nogc.json
timeout 600 python bench.py --steps 8 --warmup 3 --gc-cnt 8 > gpurun_out/b_gc8.json 2>gpurun_out/b_gc8.err
tail -1 gpurun_out/b_gc8.json
timeout 600 python bench.py --steps 8 --warmup 3 --gc-cnt 16 > gpurun_out/b_gc16.json 2>gpurun_out/b_gc16.err
tail -1 gpurun_out/b_gc16.json
# repeat tuned at the end: same-box thermal drift bound
timeout 600 python bench.py --steps 8 --warmup 3 > gpurun_out/b_tuned2.json 2>gpurun_out/b_tuned2.err
tail -1 gpurun_out/b_tuned2.json
