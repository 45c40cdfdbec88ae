set -x
cd /root/repo
mkdir -p gpurun_out
timeout 600 python -m pytest tests/ops/test_kernels_gpu.py -q -k "graph_decoder or generate_gpu" 2>&1 | tail -3
timeout 600 python benchmarks/decode_bench.py --model llama-2-7b --batch 1 --prompt 128 --new 256 --graph > gpurun_out/c19_graph_b1.json 2>gpurun_out/c19_graph_b1.err
tail -1 gpurun_out/c19_graph_b1.json; tail -3 gpurun_out/c19_graph_b1.err
timeout 600 python benchmarks/decode_bench.py --model llama-2-7b --batch 8 --prompt 128 --new 256 --graph > gpurun_out/c19_graph_b8.json 2>/dev/null
tail -1 gpurun_out/c19_graph_b8.json
