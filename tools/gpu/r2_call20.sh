set -x
cd /root/repo
mkdir -p gpurun_out
timeout 600 python -m pytest tests/ops/test_kernels_gpu.py -q -k "gemv or graph_decoder or tuned_linear" 2>&1 | tail -3
timeout 600 python benchmarks/decode_bench.py --model llama-2-7b --batch 1 --prompt 128 --new 256 --graph > gpurun_out/c20_graph_b1.json 2>/dev/null
tail -1 gpurun_out/c20_graph_b1.json
timeout 600 python benchmarks/decode_bench.py --model llama-2-7b --batch 8 --prompt 128 --new 256 --graph > gpurun_out/c20_graph_b8.json 2>/dev/null
tail -1 gpurun_out/c20_graph_b8.json
timeout 600 python benchmarks/decode_bench.py --model llama-2-7b --batch 1 --prompt 128 --new 128 > gpurun_out/c20_eager_b1.json 2>/dev/null
tail -1 gpurun_out/c20_eager_b1.json
