set -x
cd /root/repo
mkdir -p gpurun_out
timeout 600 python bench.py --model qwen2-7b --gc-cnt 8 --steps 4 --warmup 2 > gpurun_out/c29_q8.json 2>/dev/null
tail -1 gpurun_out/c29_q8.json
timeout 600 python bench.py --model qwen2-7b --gc-cnt 4 --steps 4 --warmup 2 > gpurun_out/c29_q4.json 2>/dev/null
tail -1 gpurun_out/c29_q4.json
