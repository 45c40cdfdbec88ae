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
        from ..ops.linear import tuned_linear
        y = tuned_linear(x, self.weight, self.bias)
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
        from ..ops.linear import tuned_linear
        y = tuned_linear(x, self.weight)
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


class _VocabParallelLinearCE(torch.autograd.Function):
    """Fused linear + cross-entropy over a vocab-sharded lm_head: each rank
    holds logits [N, V/tp]; the global log-sum-exp and the target logit are
    assembled with three small [N]-vector all-reduces per chunk (Megatron
    vocab-parallel loss semantics). Full [N, V] logits are never
    materialized on any rank — the 70B config's biggest single activation.
    """

    CHUNK = 8192

    @staticmethod
    def forward(ctx, x, weight, target, group, ignore_index):
        tp = dist.get_world_size(group)
        v_local = weight.shape[0]
        vocab_start = dist.get_rank(group) * v_local
        N = x.shape[0]
        total = x.new_zeros((), dtype=torch.float32)
        nvalid = torch.zeros((), dtype=torch.long, device=x.device)
        lse_all = torch.empty(N, dtype=torch.float32, device=x.device)
        C = _VocabParallelLinearCE.CHUNK
        for s in range(0, N, C):
            e = min(N, s + C)
            lf = (x[s:e] @ weight.t()).float()
            gmax = lf.max(-1).values
            if tp > 1:
                dist.all_reduce(gmax, op=dist.ReduceOp.MAX, group=group)
            sumexp = (lf - gmax.unsqueeze(-1)).exp().sum(-1)
            if tp > 1:
                dist.all_reduce(sumexp, group=group)
            lse = gmax + sumexp.log()
            tgt = target[s:e]
            valid = tgt != ignore_index
            in_part = valid & (tgt >= vocab_start) & \
                (tgt < vocab_start + v_local)
            tlocal = (tgt - vocab_start).masked_fill(~in_part, 0)
            tlogit = lf.gather(1, tlocal.unsqueeze(-1)).squeeze(-1) * in_part
            if tp > 1:
                dist.all_reduce(tlogit, group=group)
            total += ((lse - tlogit) * valid).sum()
            nvalid += valid.sum()
            lse_all[s:e] = lse
        ctx.save_for_backward(x, weight, target, lse_all, nvalid)
        ctx.group = group
        ctx.ignore_index = ignore_index
        ctx.vocab_start = vocab_start
        return total / nvalid.clamp_min(1).float()

    @staticmethod
    def backward(ctx, dloss):
        x, weight, target, lse_all, nvalid = ctx.saved_tensors
        group = ctx.group
        tp = dist.get_world_size(group)
        v_local = weight.shape[0]
        vocab_start = ctx.vocab_start
        N = x.shape[0]
        scale = dloss.float() / nvalid.clamp_min(1).float()
        dx = torch.empty_like(x)
        dw = torch.zeros_like(weight, dtype=torch.float32)
        C = _VocabParallelLinearCE.CHUNK
        for s in range(0, N, C):
            e = min(N, s + C)
            lf = (x[s:e] @ weight.t()).float()
            soft = (lf - lse_all[s:e].unsqueeze(-1)).exp()
            tgt = target[s:e]
            valid = tgt != ctx.ignore_index
            in_part = valid & (tgt >= vocab_start) & \
                (tgt < vocab_start + v_local)
            tlocal = (tgt - vocab_start).masked_fill(~in_part, 0)
            soft.scatter_add_(
                1, tlocal.unsqueeze(-1),
                -in_part.unsqueeze(-1).to(soft.dtype))
            dl = (soft * (scale * valid).unsqueeze(-1)).to(x.dtype)
            dx[s:e] = dl @ weight
            dw += (dl.t() @ x[s:e]).float()
        if tp > 1:
            # each rank's dl covers only its vocab shard: the true dx sums
            # the partial products over the group
            dist.all_reduce(dx, group=group)
        return dx, dw.to(weight.dtype), None, None, None


def vocab_parallel_linear_cross_entropy(x, weight, target, group,
                                        ignore_index: int = -100):
    return _VocabParallelLinearCE.apply(x, weight, target, group,
                                        ignore_index)


def parallelize_module(model: nn.Module, config) -> nn.Module:
    """Shard a Llama-family model over the TP group: q/k/v/gate/up column,
    o/down row, heads divided, embed_tokens + lm_head vocab-parallel with
    the CE loss computed on sharded logits. Matching is duck-typed on the
    projection attribute names, so native models AND HF transformers
    modules are both covered (the reference expressed TP as GSPMD
    annotations over any model, dist/tp.py:4-5; this is the eager
    equivalent). Other models: apply the classes manually."""
    mesh = config.get_mesh()
    group = mesh.get_tp_proc_group()
    tp = mesh.get_tp_num()
    if tp == 1 or group is None:
        return model
    n_attn = n_mlp = 0
    for mod in model.modules():
        if all(hasattr(mod, a) for a in ("q_proj", "k_proj", "v_proj",
                                         "o_proj")) and \
                isinstance(mod.q_proj, nn.Linear):
            assert mod.q_proj.out_features % tp == 0, \
                f"q_proj {mod.q_proj.out_features} not divisible by tp {tp}"
            assert mod.k_proj.out_features % tp == 0, \
                (f"kv projection {mod.k_proj.out_features} not divisible by "
                 f"tp {tp}; use a smaller tp degree for this GQA config")
            mod.q_proj = ColumnParallelLinear.from_linear(mod.q_proj, group)
            mod.k_proj = ColumnParallelLinear.from_linear(mod.k_proj, group)
            mod.v_proj = ColumnParallelLinear.from_linear(mod.v_proj, group)
            mod.o_proj = RowParallelLinear.from_linear(mod.o_proj, group)
            if hasattr(mod, "num_heads"):
                mod.num_heads //= tp
            if hasattr(mod, "num_kv_heads"):
                mod.num_kv_heads //= tp
            n_attn += 1
        elif all(hasattr(mod, a) for a in ("gate_proj", "up_proj",
                                           "down_proj")) and \
                isinstance(mod.gate_proj, nn.Linear):
            mod.gate_proj = ColumnParallelLinear.from_linear(
                mod.gate_proj, group)
            mod.up_proj = ColumnParallelLinear.from_linear(mod.up_proj,
                                                           group)
            mod.down_proj = RowParallelLinear.from_linear(mod.down_proj,
                                                          group)
            n_mlp += 1
    _parallelize_embedding_and_head(model, group, tp)
    if n_attn == 0 and n_mlp == 0:
        logger.warning(
            "parallelize_module: no Llama attention/MLP modules found; "
            "model left unsharded (use Column/RowParallelLinear directly)")
    else:
        logger.info("TP=%d: sharded %d attention + %d MLP blocks", tp,
                    n_attn, n_mlp)
    return model


def _parallelize_embedding_and_head(model: nn.Module, group, tp: int):
    """Vocab-shard embed_tokens and lm_head (the 70B config's fattest
    replicated blocks) wherever the model exposes them under the standard
    names; the loss is computed vocab-parallel by the model forward (native
    models) or by gathering logits (``gather_output=True`` inference)."""
    holder = None
    for mod in model.modules():
        if hasattr(mod, "embed_tokens") and \
                isinstance(getattr(mod, "embed_tokens"), nn.Embedding):
            holder = mod
            break
    head_holder = None
    for mod in model.modules():
        if hasattr(mod, "lm_head") and \
                isinstance(getattr(mod, "lm_head"), nn.Linear):
            head_holder = mod
            break
    if holder is None and head_holder is None:
        return
    tied = (holder is not None and head_holder is not None and
            holder.embed_tokens.weight is head_holder.lm_head.weight)
    if holder is not None and \
            holder.embed_tokens.num_embeddings % tp == 0:
        holder.embed_tokens = VocabParallelEmbedding.from_embedding(
            holder.embed_tokens, group)
    if head_holder is not None and \
            head_holder.lm_head.out_features % tp == 0:
        head = ColumnParallelLinear.from_linear(head_holder.lm_head, group)
        head.gather_output = True  # inference path returns full logits
        if tied and isinstance(holder.embed_tokens, VocabParallelEmbedding):
            head.weight = holder.embed_tokens.weight
        head_holder.lm_head = head
        head_holder._tp_group = group


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
