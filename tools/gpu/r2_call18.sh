set -x
cd /root/repo
mkdir -p gpurun_out
timeout 600 python benchmarks/decode_bench.py --model llama-2-7b --batch 1 --prompt 128 --new 128 > gpurun_out/c18_decode_b1.json 2>/dev/null
tail -1 gpurun_out/c18_decode_b1.json
timeout 600 python benchmarks/decode_bench.py --model llama-2-7b --batch 8 --prompt 128 --new 128 > gpurun_out/c18_decode_b8.json 2>/dev/null
tail -1 gpurun_out/c18_decode_b8.json
timeout 900 python bench.py --steps 24 --warmup 4 > gpurun_out/c18_steady.json 2>/dev/null
tail -1 gpurun_out/c18_steady.json
