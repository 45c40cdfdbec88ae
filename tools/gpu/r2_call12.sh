set -x
cd /root/repo
mkdir -p gpurun_out
timeout 1500 python -m pytest tests/ -m gpu -q 2>&1 | tail -3
# gc depth fine-tune A/B (free tokens if 6 or 4 wins at safe memory)
for gc in 6 4; do
  timeout 600 python bench.py --steps 8 --warmup 3 --gc-cnt $gc > gpurun_out/c12_gc$gc.json 2>/dev/null
  tail -1 gpurun_out/c12_gc$gc.json
done
timeout 600 python bench.py --steps 8 --warmup 3 > gpurun_out/c12_gc8.json 2>/dev/null
tail -1 gpurun_out/c12_gc8.json
