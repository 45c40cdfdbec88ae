set -x
cd /root/repo
mkdir -p gpurun_out
cd /tmp && export TMPDIR=/tmp && cd /root/repo
rm -rf gpurun_out/prof_final2
timeout 900 rocprofv3 --kernel-trace -d gpurun_out/prof_final2 -- python bench.py --steps 3 --warmup 2 > gpurun_out/c22_prof.log 2>&1
grep value gpurun_out/c22_prof.log | tail -1
du -sh gpurun_out/prof_final2
timeout 900 python bench.py --mode ulysses --seq-len 32768 --batch-size 1 --steps 3 --warmup 1 > gpurun_out/c22_32k.json 2>/dev/null
tail -1 gpurun_out/c22_32k.json
timeout 1500 python bench.py --mode 2d --seq-len 131072 --batch-size 1 --steps 2 --warmup 1 > gpurun_out/c22_128k.json 2>/dev/null
tail -1 gpurun_out/c22_128k.json
