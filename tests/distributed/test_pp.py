"""Pipeline parallelism: 2-stage split on gloo vs single-process training,
plus a skip-connection model exercising cross-stage value propagation
(reference tests/standalone/pipeline.py SkipNet)."""
import pytest
import torch

from tests.utils.distributed import run_multiprocess


class SkipNet(torch.nn.Module):
    """First block's output feeds both block2 and the head (skip across the
    stage boundary)."""

    def __init__(self):
        super().__init__()
        self.block1 = torch.nn.Linear(16, 16)
        self.block2 = torch.nn.Linear(16, 16)
        self.block3 = torch.nn.Linear(16, 16)
        self.head = torch.nn.Linear(32, 1)

    def forward(self, x, labels=None):
        h1 = torch.relu(self.block1(x))
        h2 = torch.relu(self.block2(h1))
        h3 = torch.relu(self.block3(h2))
        out = self.head(torch.cat([h3, h1], dim=-1))
        if labels is not None:
            return torch.nn.functional.mse_loss(out, labels)
        return out


def _pp_llama_worker(rank, world, q):
    import torchacc_amd as ta
    from torchacc_amd.models import LlamaForCausalLM, llama_tiny
    cfg = ta.Config()
    cfg.dist.pp.size = world
    cfg.dist.pp.num_micro_batches = 2
    cfg.dist.pp.input_names = ["input_ids", "labels"]
    torch.manual_seed(0)
    model = LlamaForCausalLM(llama_tiny())
    model = ta.accelerate(model, config=cfg)
    opt = torch.optim.AdamW(model.parameters(), lr=1e-3)
    torch.manual_seed(42)
    losses = []
    for _ in range(4):
        ids = torch.randint(0, 1024, (4, 32))
        loss = model.forward_backward(ids, labels=ids)
        opt.step()
        opt.zero_grad()
        losses.append(float(loss))
    q.put((rank, losses))


def test_pp2_llama_matches_single():
    import multiprocessing as mp
    ctx = mp.get_context("spawn")
    q = ctx.Queue()
    run_multiprocess(_pp_llama_worker, world_size=2, args=(q,))
    results = {}
    while not q.empty():
        r, losses = q.get()
        results[r] = losses
    assert len(results) == 2
    # broadcast_loss: both stages report identical losses
    assert results[0] == pytest.approx(results[1], abs=1e-5)

    # single-process baseline with the same data order
    from torchacc_amd.models import LlamaForCausalLM, llama_tiny
    torch.manual_seed(0)
    model = LlamaForCausalLM(llama_tiny())
    opt = torch.optim.AdamW(model.parameters(), lr=1e-3)
    torch.manual_seed(42)
    base = []
    for _ in range(4):
        ids = torch.randint(0, 1024, (4, 32))
        # PP averages the per-micro-batch losses; micro-batches are the
        # dim-0 halves
        l1 = model(ids[:2], labels=ids[:2])
        l2 = model(ids[2:], labels=ids[2:])
        ((l1 + l2) / 2).backward()
        opt.step()
        opt.zero_grad()
        base.append(float((l1 + l2) / 2))
    assert results[0] == pytest.approx(base, abs=2e-4)


def _pp_skip_worker(rank, world, q):
    import torchacc_amd as ta
    cfg = ta.Config()
    cfg.dist.pp.size = world
    cfg.dist.pp.num_micro_batches = 2
    cfg.dist.pp.input_names = ["x", "labels"]
    cfg.dist.pp.split_points = ["block3"]
    torch.manual_seed(0)
    model = SkipNet()
    model = ta.accelerate(model, config=cfg)
    opt = torch.optim.SGD(model.parameters(), lr=0.1)
    torch.manual_seed(1)
    losses = []
    for _ in range(5):
        x = torch.randn(4, 16)
        y = torch.randn(4, 1)
        loss = model.forward_backward(x, labels=y)
        opt.step()
        opt.zero_grad()
        losses.append(float(loss))
    q.put((rank, losses))


def test_pp2_skip_connection():
    import multiprocessing as mp
    ctx = mp.get_context("spawn")
    q = ctx.Queue()
    run_multiprocess(_pp_skip_worker, world_size=2, args=(q,))
    results = {}
    while not q.empty():
        r, losses = q.get()
        results[r] = losses
    assert len(results) == 2
    assert results[0] == pytest.approx(results[1], abs=1e-6)

    torch.manual_seed(0)
    model = SkipNet()
    opt = torch.optim.SGD(model.parameters(), lr=0.1)
    torch.manual_seed(1)
    base = []
    for _ in range(5):
        x = torch.randn(4, 16)
        y = torch.randn(4, 1)
        l1 = model(x[:2], labels=y[:2])
        l2 = model(x[2:], labels=y[2:])
        ((l1 + l2) / 2).backward()
        opt.step()
        opt.zero_grad()
        base.append(float((l1 + l2) / 2))
    assert results[0] == pytest.approx(base, abs=1e-5)


class TiedLM(torch.nn.Module):
    """Input embedding reused as the LM-head weight: the tied parameter is
    consumed by stage 0 (embedding) and the last stage (logits), so PP must
    sum its grad across both stages."""

    def __init__(self, vocab=64, hidden=32):
        super().__init__()
        self.embed = torch.nn.Embedding(vocab, hidden)
        self.block1 = torch.nn.Linear(hidden, hidden)
        self.block2 = torch.nn.Linear(hidden, hidden)

    def forward(self, input_ids, labels=None):
        x = torch.relu(self.block1(self.embed(input_ids)))
        x = torch.relu(self.block2(x))
        logits = torch.nn.functional.linear(x, self.embed.weight)
        if labels is None:
            return logits
        return torch.nn.functional.cross_entropy(
            logits.view(-1, logits.size(-1)), labels.reshape(-1))


def _pp_tied_worker(rank, world, q):
    import torchacc_amd as ta
    cfg = ta.Config()
    cfg.dist.pp.size = world
    cfg.dist.pp.num_micro_batches = 2
    cfg.dist.pp.input_names = ["input_ids", "labels"]
    cfg.dist.pp.split_points = ["block2"]
    torch.manual_seed(0)
    model = TiedLM()
    model = ta.accelerate(model, config=cfg)
    opt = torch.optim.SGD(model.parameters(), lr=0.1)
    torch.manual_seed(7)
    ids = torch.randint(0, 64, (4, 8))
    losses = []
    for _ in range(4):
        loss = model.forward_backward(ids, labels=ids)
        opt.step()
        opt.zero_grad()
        losses.append(float(loss))
    q.put((rank, losses))


def test_pp2_tied_embedding_matches_single_process():
    import multiprocessing as mp
    ctx = mp.get_context("spawn")
    q = ctx.Queue()
    run_multiprocess(_pp_tied_worker, world_size=2, args=(q,))
    results = {}
    for _ in range(2):
        r, losses = q.get()
        results[r] = losses
    assert results[0] == pytest.approx(results[1], abs=1e-6)

    torch.manual_seed(0)
    model = TiedLM()
    opt = torch.optim.SGD(model.parameters(), lr=0.1)
    torch.manual_seed(7)
    ids = torch.randint(0, 64, (4, 8))
    base = []
    for _ in range(4):
        loss = model(ids, labels=ids)
        loss.backward()
        opt.step()
        opt.zero_grad()
        base.append(float(loss))
    assert results[0] == pytest.approx(base, abs=1e-5)


def _pp_infer_worker(rank, world, outdir):
    import numpy as np
    import os
    import torchacc_amd as ta
    from torchacc_amd.models import LlamaForCausalLM, llama_tiny
    cfg = ta.Config()
    cfg.dist.pp.size = world
    cfg.dist.pp.num_micro_batches = 2
    cfg.dist.pp.input_names = ["input_ids"]
    torch.manual_seed(0)
    model = LlamaForCausalLM(llama_tiny())
    model = ta.accelerate(model, config=cfg)
    torch.manual_seed(9)
    ids = torch.randint(0, 1024, (4, 16))
    out = model(ids)  # inference schedule (no labels -> logits)
    if rank == world - 1:
        np.save(os.path.join(outdir, "pp_logits.npy"),
                out.detach().float().numpy())


def test_pp2_inference_matches_single_process(tmp_path):
    """executor.forward (inference schedule): last stage's concatenated
    logits must match the single-process model."""
    run_multiprocess(_pp_infer_worker, world_size=2, args=(str(tmp_path),))
    import numpy as np
    got = np.load(tmp_path / "pp_logits.npy")

    torch.manual_seed(0)
    from torchacc_amd.models import LlamaForCausalLM, llama_tiny
    model = LlamaForCausalLM(llama_tiny())
    torch.manual_seed(9)
    ids = torch.randint(0, 1024, (4, 16))
    with torch.no_grad():
        ref = model(ids).float().numpy()
    assert got.shape == ref.shape
    assert abs(got - ref).max() < 1e-4, abs(got - ref).max()


def _pp_hf_worker(rank, world, q):
    """HF transformers Llama split on the decoder ModuleList (no fx;
    reference capability dist/pp/pipeline.py:38-44)."""
    import torchacc_amd as ta
    from transformers.models.llama.configuration_llama import LlamaConfig
    from transformers.models.llama.modeling_llama import LlamaForCausalLM
    hf_cfg = LlamaConfig(
        vocab_size=256, hidden_size=64, intermediate_size=128,
        num_hidden_layers=4, num_attention_heads=4, num_key_value_heads=4,
        max_position_embeddings=64, attn_implementation="eager",
        tie_word_embeddings=False)
    cfg = ta.Config()
    cfg.dist.pp.size = world
    cfg.dist.pp.num_micro_batches = 2
    cfg.dist.pp.input_names = ["input_ids", "labels"]
    cfg.compute.disable_kernel_patches = True
    torch.manual_seed(0)
    model = LlamaForCausalLM(hf_cfg)
    model = ta.accelerate(model, config=cfg)
    opt = torch.optim.AdamW(model.parameters(), lr=1e-3)
    torch.manual_seed(42)
    losses = []
    for _ in range(4):
        ids = torch.randint(0, 256, (4, 32))
        loss = model.forward_backward(ids, labels=ids)
        opt.step()
        opt.zero_grad()
        losses.append(float(loss))
    q.put((rank, losses))


def test_pp2_hf_llama_matches_single():
    import multiprocessing as mp
    ctx = mp.get_context("spawn")
    q = ctx.Queue()
    run_multiprocess(_pp_hf_worker, world_size=2, args=(q,))
    results = {}
    while not q.empty():
        r, losses = q.get()
        results[r] = losses
    assert len(results) == 2
    assert results[0] == pytest.approx(results[1], abs=1e-5)

    # single-process baseline: same micro-batch split, same loss formula
    # as the PP last stage (shifted fused linear-CE)
    from transformers.models.llama.configuration_llama import LlamaConfig
    from transformers.models.llama.modeling_llama import LlamaForCausalLM
    from torchacc_amd.ops.cross_entropy import linear_cross_entropy
    hf_cfg = LlamaConfig(
        vocab_size=256, hidden_size=64, intermediate_size=128,
        num_hidden_layers=4, num_attention_heads=4, num_key_value_heads=4,
        max_position_embeddings=64, attn_implementation="eager",
        tie_word_embeddings=False)
    torch.manual_seed(0)
    model = LlamaForCausalLM(hf_cfg)
    opt = torch.optim.AdamW(model.parameters(), lr=1e-3)
    torch.manual_seed(42)

    def loss_fn(ids):
        h = model.model(input_ids=ids).last_hidden_state
        hs = h[:, :-1, :].reshape(-1, h.shape[-1])
        return linear_cross_entropy(hs, model.lm_head.weight,
                                    ids[:, 1:].reshape(-1))

    base = []
    for _ in range(4):
        ids = torch.randint(0, 256, (4, 32))
        l1 = loss_fn(ids[:2])
        l2 = loss_fn(ids[2:])
        ((l1 + l2) / 2).backward()
        opt.step()
        opt.zero_grad()
        base.append(float((l1 + l2) / 2))
    assert results[0] == pytest.approx(base, abs=2e-4)
