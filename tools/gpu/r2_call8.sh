set -x
cd /root/repo
mkdir -p gpurun_out
# build provenance: wipe the shipped .so and rebuild from source on the box
rm -f torchacc_amd/_C*.so
python -c "import __graft_entry__; __graft_entry__.build()" > gpurun_out/c8_build.log 2>&1
tail -3 gpurun_out/c8_build.log
ls -la torchacc_amd/_C*.so
python -c "import torchacc_amd._C as C; print('freshly built ext loads:', hasattr(C, 'fa_forward'))"
# full validation on the fresh build
timeout 1200 python -m pytest tests/ -m gpu -q 2>&1 | tail -3
timeout 600 python -c "import __graft_entry__; __graft_entry__.smoke()" && echo SMOKE_OK
timeout 600 python bench.py --steps 12 --warmup 3 > gpurun_out/c8_bench.json 2>gpurun_out/c8_bench.err
tail -1 gpurun_out/c8_bench.json
timeout 900 python benchmarks/accuracy.py --model hf-llama --steps 20 > gpurun_out/c8_acc_hf.log 2>&1
tail -2 gpurun_out/c8_acc_hf.log
