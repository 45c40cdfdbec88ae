"""KV-cache generation vs naive full-recompute decode (native models)."""
import pytest
import torch


def _naive_greedy(model, ids, n):
    out = ids
    for _ in range(n):
        logits = model(out)           # [b, s, vocab]
        nxt = logits[:, -1].argmax(-1, keepdim=True)
        out = torch.cat([out, nxt], dim=1)
    return out


@pytest.mark.parametrize("family", ["llama", "qwen2"])
def test_generate_matches_naive_decode(family):
    torch.manual_seed(0)
    if family == "llama":
        from torchacc_amd.models import LlamaForCausalLM, llama_tiny
        model = LlamaForCausalLM(llama_tiny())
    else:
        from torchacc_amd.models import Qwen2ForCausalLM, qwen2_tiny
        model = Qwen2ForCausalLM(qwen2_tiny())
    model.eval()
    ids = torch.randint(0, 1024, (2, 12))
    got = model.generate(ids, max_new_tokens=8)
    want = _naive_greedy(model, ids, 8)
    assert got.shape == (2, 20)
    assert torch.equal(got, want), (got[:, 12:], want[:, 12:])


def test_generate_eos_stops():
    from torchacc_amd.models import LlamaForCausalLM, llama_tiny
    torch.manual_seed(0)
    model = LlamaForCausalLM(llama_tiny()).eval()
    ids = torch.randint(0, 1024, (1, 8))
    full = model.generate(ids, max_new_tokens=16)
    eos = int(full[0, 10])  # force the 3rd generated token to be "eos"
    out = model.generate(ids, max_new_tokens=16, eos_token_id=eos)
    assert out.shape[1] <= full.shape[1]
    assert eos in out[0, 8:].tolist()


def test_generate_sampling_runs():
    from torchacc_amd.models import LlamaForCausalLM, llama_tiny
    torch.manual_seed(0)
    model = LlamaForCausalLM(llama_tiny()).eval()
    ids = torch.randint(0, 1024, (2, 8))
    torch.manual_seed(1)
    out = model.generate(ids, max_new_tokens=6, temperature=0.8, top_k=20)
    assert out.shape == (2, 14)
