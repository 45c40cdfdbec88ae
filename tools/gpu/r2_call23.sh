set -x
cd /root/repo
mkdir -p gpurun_out
timeout 600 python bench.py --steps 6 --warmup 2 --model qwen2-7b > gpurun_out/c23_qwen.json 2>/dev/null
tail -1 gpurun_out/c23_qwen.json
timeout 600 python bench.py --steps 6 --warmup 2 --model llama-3-8b > gpurun_out/c23_l38b.json 2>/dev/null
tail -1 gpurun_out/c23_l38b.json
timeout 600 python bench.py --steps 6 --warmup 2 --seq-len 2048 --batch-size 16 > gpurun_out/c23_s2048.json 2>/dev/null
tail -1 gpurun_out/c23_s2048.json
timeout 600 python bench.py --steps 6 --warmup 2 --seq-len 8192 --batch-size 4 > gpurun_out/c23_s8192.json 2>/dev/null
tail -1 gpurun_out/c23_s8192.json
