set -x
cd /root/repo
mkdir -p gpurun_out
for cfg in "--model qwen2-7b" "--model llama-3-8b" "--seq-len 2048 --batch-size 16" "--seq-len 8192 --batch-size 4"; do
  timeout 600 python bench.py --steps 6 --warmup 2 $cfg > gpurun_out/c9_$(echo $cfg | tr ' -' '__').json 2>/dev/null
  tail -1 gpurun_out/c9_$(echo $cfg | tr ' -' '__').json
done
timeout 600 python benchmarks/transformer.py --steps 20 > gpurun_out/c9_transformer.log 2>&1
tail -3 gpurun_out/c9_transformer.log
cd /tmp && export TMPDIR=/tmp && cd /root/repo
timeout 900 rocprofv3 --kernel-trace --stats -- python bench.py --steps 3 --warmup 2 > gpurun_out/c9_stats.log 2>&1
grep -A 40 "KERNEL_DISPATCH" gpurun_out/c9_stats.log | head -45 || tail -50 gpurun_out/c9_stats.log
