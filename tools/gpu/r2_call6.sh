set -x
cd /root/repo
mkdir -p gpurun_out
# numerics of the QT=64 dkv + rmsnorm grid fix + fixed HF LCE test
timeout 900 python -m pytest tests/ops/test_kernels_gpu.py -q 2>&1 | tail -3
# dkv kernel micro numbers (r01 reference: 421 TF non-causal, 331-335 causal)
timeout 600 python benchmarks/attn_bench.py > gpurun_out/c6_attn.log 2>&1
tail -20 gpurun_out/c6_attn.log
timeout 600 python benchmarks/attn_bench.py --causal > gpurun_out/c6_attn_c.log 2>&1
tail -20 gpurun_out/c6_attn_c.log
timeout 600 python bench.py --steps 8 --warmup 3 > gpurun_out/c6_default.json 2>gpurun_out/c6_default.err
tail -1 gpurun_out/c6_default.json
