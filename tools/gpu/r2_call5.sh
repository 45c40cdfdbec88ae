set -x
cd /root/repo
mkdir -p gpurun_out
timeout 1200 python -m pytest tests/ -m gpu -q 2>&1 | tail -4
timeout 600 python bench.py --steps 8 --warmup 3 > gpurun_out/c5_default.json 2>gpurun_out/c5_default.err
tail -1 gpurun_out/c5_default.json
cd /tmp && export TMPDIR=/tmp && cd /root/repo
rm -rf gpurun_out/prof_r2 gpurun_out/prof_r2b
timeout 900 rocprofv3 --kernel-trace -d gpurun_out/prof_r2b -- python bench.py --steps 3 --warmup 2 > gpurun_out/c5_prof.log 2>&1
tail -2 gpurun_out/c5_prof.log | head -1
du -sh gpurun_out/prof_r2b
