"""Tensor parallelism: Megatron-style column/row-parallel linears.

The reference expressed TP as GSPMD ``mark_sharding`` annotations and let the
XLA compiler insert collectives (dist/tp.py:4-5) — no eager equivalent
exists, so this module implements the idiomatic eager design: explicit
column/row sharding with all-reduce on the TP group over xGMI
(SURVEY.md §7 step 8). Exposed:

- :class:`ColumnParallelLinear` / :class:`RowParallelLinear` /
  :class:`VocabParallelEmbedding`
- :func:`parallelize_module(model, config)`: module surgery for the native
  Llama family (q/k/v/gate/up -> column, o/down -> row, attention heads
  divided by tp); generic models can use the classes directly.
- :func:`mark_sharding` / :class:`Mesh` shims for reference API
  compatibility (no-ops that explain the eager design).
"""
import math
import torch
import torch.distributed as dist
import torch.nn as nn

from ..utils.logger import logger


class _CopyToTP(torch.autograd.Function):
    """Identity forward; all-reduce gradient over the TP group (input of a
    column-parallel layer)."""

    @staticmethod
    def forward(ctx, x, group):
        ctx.group = group
        return x

    @staticmethod
    def backward(ctx, grad):
        if dist.get_world_size(ctx.group) > 1:
            grad = grad.contiguous()
            dist.all_reduce(grad, group=ctx.group)
        return grad, None


class _ReduceFromTP(torch.autograd.Function):
    """All-reduce forward; identity gradient (output of a row-parallel
    layer)."""

    @staticmethod
    def forward(ctx, x, group):
        if dist.get_world_size(group) > 1:
            x = x.contiguous()
            dist.all_reduce(x, group=group)
        return x

    @staticmethod
    def backward(ctx, grad):
        return grad, None


def copy_to_tp_region(x, group):
    return _CopyToTP.apply(x, group)


def reduce_from_tp_region(x, group):
    return _ReduceFromTP.apply(x, group)


class ColumnParallelLinear(nn.Module):
    """y_local = x @ W_local^T with W row-sharded (output features split).

    ``gather_output=False`` keeps the sharded output (fed to a row-parallel
    layer or head-sharded attention)."""

    def __init__(self, in_features, out_features, group, bias=False,
                 gather_output=False, dtype=None, device=None):
        super().__init__()
        self.group = group
        self.tp = dist.get_world_size(group)
        assert out_features % self.tp == 0
        self.in_features = in_features
        self.out_features = out_features
        self.out_local = out_features // self.tp
        self.gather_output = gather_output
        self.weight = nn.Parameter(
            torch.empty(self.out_local, in_features, dtype=dtype,
                        device=device))
        self.bias = nn.Parameter(
            torch.zeros(self.out_local, dtype=dtype, device=device)) \
            if bias else None
        nn.init.kaiming_uniform_(self.weight, a=math.sqrt(5))

    @classmethod
    def from_linear(cls, lin: nn.Linear, group,
                    sections=None) -> "ColumnParallelLinear":
        """``sections``: for merged projections (qkv / gate_up), shard each
        section independently and concatenate the per-rank slices so the
        local layout stays [q_r; k_r; v_r]."""
        m = cls(lin.in_features, lin.out_features, group,
                bias=lin.bias is not None, dtype=lin.weight.dtype,
                device=lin.weight.device)
        rank = dist.get_rank(group)
        with torch.no_grad():
            if sections is None:
                m.weight.copy_(lin.weight.chunk(m.tp, dim=0)[rank])
                if lin.bias is not None:
                    m.bias.copy_(lin.bias.chunk(m.tp, dim=0)[rank])
            else:
                wparts = lin.weight.split(list(sections), dim=0)
                m.weight.copy_(torch.cat(
                    [p.chunk(m.tp, dim=0)[rank] for p in wparts], dim=0))
                if lin.bias is not None:
                    bparts = lin.bias.split(list(sections), dim=0)
                    m.bias.copy_(torch.cat(
                        [p.chunk(m.tp, dim=0)[rank] for p in bparts],
                        dim=0))
        return m

    def forward(self, x):
        x = copy_to_tp_region(x, self.group)
        y = torch.nn.functional.linear(x, self.weight, self.bias)
        if self.gather_output and self.tp > 1:
            from ..ops.context_parallel.utils import \
                gather_forward_split_backward
            y = gather_forward_split_backward(y, -1, self.group)
        return y


class RowParallelLinear(nn.Module):
    """y = all_reduce(x_local @ W_local^T) with W column-sharded (input
    features split); expects the input already sharded on the last dim
    (``input_is_parallel=True``, the usual pairing with a column layer)."""

    def __init__(self, in_features, out_features, group, bias=False,
                 input_is_parallel=True, dtype=None, device=None):
        super().__init__()
        self.group = group
        self.tp = dist.get_world_size(group)
        assert in_features % self.tp == 0
        self.in_features = in_features
        self.in_local = in_features // self.tp
        self.out_features = out_features
        self.input_is_parallel = input_is_parallel
        self.weight = nn.Parameter(
            torch.empty(out_features, self.in_local, dtype=dtype,
                        device=device))
        self.bias = nn.Parameter(
            torch.zeros(out_features, dtype=dtype, device=device)) \
            if bias else None
        nn.init.kaiming_uniform_(self.weight, a=math.sqrt(5))

    @classmethod
    def from_linear(cls, lin: nn.Linear, group) -> "RowParallelLinear":
        m = cls(lin.in_features, lin.out_features, group,
                bias=lin.bias is not None, dtype=lin.weight.dtype,
                device=lin.weight.device)
        rank = dist.get_rank(group)
        with torch.no_grad():
            m.weight.copy_(lin.weight.chunk(m.tp, dim=1)[rank])
            if lin.bias is not None:
                m.bias.copy_(lin.bias)
        return m

    def forward(self, x):
        if not self.input_is_parallel:
            from ..ops.context_parallel.utils import \
                split_forward_gather_backward
            x = split_forward_gather_backward(x, -1, self.group)
        y = torch.nn.functional.linear(x, self.weight)
        y = reduce_from_tp_region(y, self.group)
        if self.bias is not None:
            y = y + self.bias
        return y


class VocabParallelEmbedding(nn.Module):
    """Embedding with the vocab dimension sharded: out-of-partition tokens
    contribute zero and the partial embeddings are all-reduced."""

    def __init__(self, num_embeddings, embedding_dim, group, dtype=None,
                 device=None):
        super().__init__()
        self.group = group
        self.tp = dist.get_world_size(group)
        assert num_embeddings % self.tp == 0
        self.num_embeddings = num_embeddings
        self.local_vocab = num_embeddings // self.tp
        self.vocab_start = dist.get_rank(group) * self.local_vocab
        self.weight = nn.Parameter(
            torch.empty(self.local_vocab, embedding_dim, dtype=dtype,
                        device=device))
        nn.init.normal_(self.weight, std=0.02)

    @classmethod
    def from_embedding(cls, emb: nn.Embedding, group):
        m = cls(emb.num_embeddings, emb.embedding_dim, group,
                dtype=emb.weight.dtype, device=emb.weight.device)
        rank = dist.get_rank(group)
        with torch.no_grad():
            m.weight.copy_(emb.weight.chunk(m.tp, dim=0)[rank])
        return m

    def forward(self, ids):
        mask = (ids >= self.vocab_start) & \
               (ids < self.vocab_start + self.local_vocab)
        local = (ids - self.vocab_start).masked_fill(~mask, 0)
        out = torch.nn.functional.embedding(local, self.weight)
        out = out * mask.unsqueeze(-1)
        return reduce_from_tp_region(out, self.group)


def parallelize_module(model: nn.Module, config) -> nn.Module:
    """Shard the native Llama family over the TP group (q/k/v/gate/up
    column, o/down row, heads divided). Non-Llama models: apply the classes
    manually."""
    mesh = config.get_mesh()
    group = mesh.get_tp_proc_group()
    tp = mesh.get_tp_num()
    if tp == 1 or group is None:
        return model
    from ..models.llama import LlamaAttention, LlamaMLP
    from ..models.qwen2 import Qwen2Attention
    n_attn = n_mlp = 0
    for mod in model.modules():
        if isinstance(mod, (LlamaAttention, Qwen2Attention)):
            assert mod.num_heads % tp == 0, \
                f"attention heads {mod.num_heads} not divisible by tp {tp}"
            assert mod.num_kv_heads % tp == 0, \
                (f"kv heads {mod.num_kv_heads} not divisible by tp {tp}; "
                 "use a smaller tp degree for this GQA config")
            mod.q_proj = ColumnParallelLinear.from_linear(mod.q_proj, group)
            mod.k_proj = ColumnParallelLinear.from_linear(mod.k_proj, group)
            mod.v_proj = ColumnParallelLinear.from_linear(mod.v_proj, group)
            mod.o_proj = RowParallelLinear.from_linear(mod.o_proj, group)
            mod.num_heads //= tp
            mod.num_kv_heads //= tp
            n_attn += 1
        elif isinstance(mod, LlamaMLP):
            mod.gate_proj = ColumnParallelLinear.from_linear(
                mod.gate_proj, group)
            mod.up_proj = ColumnParallelLinear.from_linear(mod.up_proj,
                                                           group)
            mod.down_proj = RowParallelLinear.from_linear(mod.down_proj,
                                                          group)
            n_mlp += 1
    if n_attn == 0 and n_mlp == 0:
        logger.warning(
            "parallelize_module: no Llama attention/MLP modules found; "
            "model left unsharded (use Column/RowParallelLinear directly)")
    else:
        logger.info("TP=%d: sharded %d attention + %d MLP blocks", tp,
                    n_attn, n_mlp)
    return model


# ---- reference-API shims ---------------------------------------------------

class Mesh:  # noqa: N801 - name kept from reference (xs.Mesh alias)
    """Placeholder for the reference's GSPMD mesh alias; the eager backend
    expresses TP through the parallel modules above."""

    def __init__(self, *a, **k):
        raise NotImplementedError(
            "GSPMD meshes do not exist on the eager MI355X backend; use "
            "dist.tp.parallelize_module or the *ParallelLinear classes")


def mark_sharding(*args, **kwargs):
    raise NotImplementedError(
        "GSPMD mark_sharding does not exist on the eager MI355X backend; "
        "use dist.tp.parallelize_module or the *ParallelLinear classes")
