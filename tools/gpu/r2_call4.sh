set -x
cd /root/repo
mkdir -p gpurun_out
# 1. full GPU test suite (new: lt_gemm, tuned_linear, TP/ring changes, HF lce)
timeout 1200 python -m pytest tests/ -m gpu -q 2>&1 | tail -6
# 2. default bench after split-views fix (was 18.9-19.0k)
timeout 600 python bench.py --steps 8 --warmup 3 > gpurun_out/c4_default.json 2>gpurun_out/c4_default.err
tail -1 gpurun_out/c4_default.json
# 3. batch sweep: GEMM efficiency vs memory
timeout 600 python bench.py --steps 6 --warmup 2 --batch-size 12 --gc-cnt 14 > gpurun_out/c4_bs12.json 2>gpurun_out/c4_bs12.err
tail -1 gpurun_out/c4_bs12.json
timeout 600 python bench.py --steps 6 --warmup 2 --batch-size 16 --gc-cnt 20 > gpurun_out/c4_bs16.json 2>gpurun_out/c4_bs16.err
tail -1 gpurun_out/c4_bs16.json
# 4. long-context configs (BASELINE #3/#5 shapes, CP degenerate on 1 GPU)
timeout 900 python bench.py --mode ulysses --seq-len 32768 --batch-size 1 --steps 3 --warmup 1 > gpurun_out/c4_32k.json 2>gpurun_out/c4_32k.err
tail -1 gpurun_out/c4_32k.json
timeout 1500 python bench.py --mode 2d --seq-len 131072 --batch-size 1 --steps 2 --warmup 1 > gpurun_out/c4_128k.json 2>gpurun_out/c4_128k.err
tail -1 gpurun_out/c4_128k.json; tail -3 gpurun_out/c4_128k.err
