set -x
cd /root/repo
mkdir -p gpurun_out
timeout 1500 python -m pytest tests/ -m gpu -q 2>&1 | tail -3
timeout 600 python bench.py --steps 8 --warmup 3 > gpurun_out/c26_default.json 2>/dev/null
tail -1 gpurun_out/c26_default.json
timeout 600 python - > gpurun_out/c26_fp16_decode.log 2>&1 <<'PY'
import torch, time, json
from torchacc_amd.models import LlamaForCausalLM, llama_2_7b
torch.manual_seed(0)
with torch.device("cuda"):
    model = LlamaForCausalLM(llama_2_7b()).to(torch.float16).eval()
ids = torch.randint(0, 32000, (1, 128), device="cuda")
from torchacc_amd.models.generation import GraphDecoder
dec = GraphDecoder(model, 1, 520)
dec.decode(ids, 8)
torch.cuda.synchronize()
t0 = time.perf_counter(); out = dec.decode(ids, 256); torch.cuda.synchronize()
dt = time.perf_counter() - t0
print(json.dumps({"fp16_graph_decode_tok_s": 256/dt, "ms_per_token": dt/256*1000}))
PY
tail -1 gpurun_out/c26_fp16_decode.log
