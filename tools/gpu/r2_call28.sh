set -x
cd /root/repo
mkdir -p gpurun_out
cd /tmp && export TMPDIR=/tmp && cd /root/repo
rm -rf gpurun_out/prof_l38b_gc8 gpurun_out/prof_l38b_gc4 gpurun_out/prof_final2 gpurun_out/prof_r2b
timeout 900 rocprofv3 --kernel-trace -d gpurun_out/prof_l38b_gc8 -- python bench.py --model llama-3-8b --gc-cnt 8 --steps 2 --warmup 1 > gpurun_out/c28_gc8.log 2>&1
grep value gpurun_out/c28_gc8.log | tail -1
timeout 900 rocprofv3 --kernel-trace -d gpurun_out/prof_l38b_gc4 -- python bench.py --model llama-3-8b --gc-cnt 4 --steps 2 --warmup 1 > gpurun_out/c28_gc4.log 2>&1
grep value gpurun_out/c28_gc4.log | tail -1
du -sh gpurun_out/prof_l38b_gc8 gpurun_out/prof_l38b_gc4
