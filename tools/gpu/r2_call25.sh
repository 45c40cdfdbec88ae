set -x
cd /root/repo
mkdir -p gpurun_out
timeout 1200 python benchmarks/gemm_tune.py --model llama-2-70b --bs 2 --seq 4096 --tp 8 --budget 15 > gpurun_out/c25_tune70b.log 2>&1
tail -24 gpurun_out/c25_tune70b.log
cp torchacc_amd/ops/gemm_algos_gfx950.json gpurun_out/gemm_algos_final.json
timeout 300 python benchmarks/attn_bench.py --causal > gpurun_out/c25_attn.log 2>&1
tail -2 gpurun_out/c25_attn.log
