set -x
cd /root/repo
mkdir -p gpurun_out
timeout 1500 python -m pytest tests/ -m gpu -q 2>&1 | tail -4
timeout 600 python bench.py --steps 8 --warmup 3 > gpurun_out/c7_default.json 2>gpurun_out/c7_default.err
tail -1 gpurun_out/c7_default.json
timeout 900 python benchmarks/accuracy.py --model hf-llama --steps 20 > gpurun_out/c7_acc_hf.log 2>&1
tail -3 gpurun_out/c7_acc_hf.log
timeout 600 python benchmarks/accuracy.py --steps 20 > gpurun_out/c7_acc_native.log 2>&1
tail -2 gpurun_out/c7_acc_native.log
