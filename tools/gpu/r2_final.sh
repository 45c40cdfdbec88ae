set -x
cd /root/repo
mkdir -p gpurun_out
# true from-scratch provenance: wipe shipped objects AND the .so
rm -rf build torchacc_amd/_C*.so
time python -c "import __graft_entry__; __graft_entry__.build()" > gpurun_out/final_build.log 2>&1
tail -2 gpurun_out/final_build.log
ls -la torchacc_amd/_C*.so
timeout 1500 python -m pytest tests/ -m gpu -q 2>&1 | tail -3
timeout 600 python -c "import __graft_entry__; __graft_entry__.smoke()" && echo SMOKE_OK
timeout 600 python bench.py --steps 12 --warmup 3 > gpurun_out/final_bench.json 2>/dev/null
tail -1 gpurun_out/final_bench.json
timeout 900 python benchmarks/accuracy.py --steps 20 > gpurun_out/final_acc.log 2>&1
tail -2 gpurun_out/final_acc.log
timeout 900 python benchmarks/accuracy.py --model hf-llama --steps 20 > gpurun_out/final_acc_hf.log 2>&1
tail -2 gpurun_out/final_acc_hf.log
