set -x
cd /root/repo
mkdir -p gpurun_out
timeout 1500 python -m pytest tests/ -m gpu -q 2>&1 | tail -3
timeout 600 python -c "import __graft_entry__; __graft_entry__.smoke()" && echo SMOKE_OK
timeout 600 python bench.py --steps 10 --warmup 3 > gpurun_out/c24_default.json 2>/dev/null
tail -1 gpurun_out/c24_default.json
