set -x
cd /root/repo
mkdir -p gpurun_out
export PYTORCH_HIP_ALLOC_CONF=expandable_segments:True
timeout 600 python bench.py --steps 6 --warmup 2 --model llama-3-8b --gc-cnt 4 > gpurun_out/c27_l38b_gc4_exp.json 2>/dev/null
tail -1 gpurun_out/c27_l38b_gc4_exp.json
timeout 600 python bench.py --steps 6 --warmup 2 --no-gc > gpurun_out/c27_nogc_exp.json 2>/dev/null
tail -1 gpurun_out/c27_nogc_exp.json
timeout 600 python bench.py --steps 8 --warmup 3 > gpurun_out/c27_default_exp.json 2>/dev/null
tail -1 gpurun_out/c27_default_exp.json
unset PYTORCH_HIP_ALLOC_CONF
timeout 600 python bench.py --steps 6 --warmup 2 --no-gc > gpurun_out/c27_nogc_plain.json 2>/dev/null
tail -1 gpurun_out/c27_nogc_plain.json
