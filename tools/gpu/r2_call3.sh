set -x
cd /root/repo
mkdir -p gpurun_out
# isolate lt-path overhead vs algo table (all with the new gc_cnt=8 default)
TA_GEMM_HEURISTIC=1 timeout 600 python bench.py --steps 8 --warmup 3 > gpurun_out/c3_lt_heur.json 2>/dev/null
tail -1 gpurun_out/c3_lt_heur.json
TA_DISABLE_TUNED_GEMM=1 timeout 600 python bench.py --steps 8 --warmup 3 > gpurun_out/c3_flinear.json 2>/dev/null
tail -1 gpurun_out/c3_flinear.json
timeout 600 python bench.py --steps 8 --warmup 3 > gpurun_out/c3_table_v1.json 2>/dev/null
tail -1 gpurun_out/c3_table_v1.json
# re-tune with rotating cold buffers, then bench on the new table
timeout 1200 python benchmarks/gemm_tune.py --model llama-2-7b --bs 8 --seq 4096 --budget 20 > gpurun_out/c3_tune_v2.log 2>&1
tail -22 gpurun_out/c3_tune_v2.log
cp torchacc_amd/ops/gemm_algos_gfx950.json gpurun_out/gemm_algos_v2.json
timeout 600 python bench.py --steps 8 --warmup 3 > gpurun_out/c3_table_v2.json 2>/dev/null
tail -1 gpurun_out/c3_table_v2.json
# kernel-level profile of the current default (gc8)
cd /tmp && export TMPDIR=/tmp && cd /root/repo
timeout 900 rocprofv3 --kernel-trace --stats -d gpurun_out/prof_r2 -- python bench.py --steps 3 --warmup 2 > gpurun_out/c3_prof.log 2>&1
grep -E "value|Custom|Cijk|fa_|adamw|elementwise|rmsnorm|swiglu|rope|cross_entropy|at::native" gpurun_out/c3_prof.log | head -5
ls gpurun_out/prof_r2 2>/dev/null | head
