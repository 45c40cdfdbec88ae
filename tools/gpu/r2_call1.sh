set -x
cd /root/repo
mkdir -p gpurun_out
# 1. sanity: new GEMM path + a quick kernel-subset check
timeout 600 python -m pytest tests/ops/test_kernels_gpu.py -q -x -k "lt_gemm or tuned_linear or extension or adamw or fp16" 2>&1 | tail -5
# 2. GEMM algo search for the 7B bench shapes
timeout 1500 python benchmarks/gemm_tune.py --model llama-2-7b --bs 8 --seq 4096 --budget 25 > gpurun_out/gemm_tune_7b.log 2>&1
tail -25 gpurun_out/gemm_tune_7b.log
cp torchacc_amd/ops/gemm_algos_gfx950.json gpurun_out/ 2>/dev/null
# 3. bench A/B: tuned default vs no-gc (memory headroom check)
timeout 900 python bench.py --steps 6 --warmup 2 > gpurun_out/bench_tuned.json 2>gpurun_out/bench_tuned.err
tail -1 gpurun_out/bench_tuned.json
timeout 900 python bench.py --steps 6 --warmup 2 --no-gc > gpurun_out/bench_nogc.json 2>gpurun_out/bench_nogc.err
tail -1 gpurun_out/bench_nogc.json
