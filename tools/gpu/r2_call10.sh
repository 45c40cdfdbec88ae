set -x
cd /root/repo
mkdir -p gpurun_out
# fp16-native kernel validation + full suite
timeout 1500 python -m pytest tests/ -m gpu -q 2>&1 | tail -3
# driver-style torchrun rendezvous path at N=1
timeout 600 python -m torch.distributed.run --nnodes=1 --nproc-per-node 1 --master-addr 127.0.0.1 --master-port 29571 bench.py --gpus 1 --steps 3 --warmup 1 > gpurun_out/c10_torchrun.json 2>gpurun_out/c10_torchrun.err
tail -1 gpurun_out/c10_torchrun.json
# qwen2/llama-3 with the gc8 default
timeout 600 python bench.py --steps 6 --warmup 2 --model qwen2-7b > gpurun_out/c10_qwen.json 2>/dev/null
tail -1 gpurun_out/c10_qwen.json
timeout 600 python bench.py --steps 6 --warmup 2 --model llama-3-8b > gpurun_out/c10_l38b.json 2>/dev/null
tail -1 gpurun_out/c10_l38b.json
# default sanity after fp16 templating (bf16 instantiation must be unchanged)
timeout 600 python bench.py --steps 8 --warmup 3 > gpurun_out/c10_default.json 2>/dev/null
tail -1 gpurun_out/c10_default.json
