"""Pipeline splitting for HF-transformers causal LMs without fx.

transformers >= 5 removed its fx tracer (HFTracer), so the reference's
trace-and-split path (reference dist/pp/pipeline.py:38-44,
utils/trace.py:41-80) cannot cover HF models anymore. Every HF causal LM
exposes the same skeleton — ``model.embed_tokens``, ``model.layers``
(ModuleList of decoder blocks), ``model.norm``, ``model.rotary_emb``,
``lm_head`` — so the stages can be built directly on the ModuleList: the
decoder layers are partitioned uniformly, stage 0 prepends the embedding,
the last stage appends final norm + lm_head + loss (via the fused
linear-cross-entropy, never materializing [b, s, vocab] logits during
training). The hand-built SplitResult feeds the same PipeExecutor as the
fx path; only ``hidden_states`` crosses stage boundaries and each stage
recomputes the (parameter-free) rotary tables locally instead of shipping
them through the pipe.
"""
from typing import Optional

import torch
import torch.nn as nn

from ...utils.logger import logger
from ...utils.utils import partition_uniform
from .utils import SplitResult, StageSpec


def _hf_base(model):
    """(base_model, lm_head) for an HF causal LM, else (None, None)."""
    base = getattr(model, "model", None)
    head = getattr(model, "lm_head", None)
    if base is None or head is None:
        return None, None
    for attr in ("embed_tokens", "layers", "norm", "rotary_emb"):
        if not hasattr(base, attr):
            return None, None
    if not isinstance(base.layers, nn.ModuleList):
        return None, None
    return base, head


def is_hf_splittable(model) -> bool:
    return _hf_base(model)[0] is not None


class HFPipeStage(nn.Module):
    """One pipeline stage of an HF causal LM (see module docstring)."""

    def __init__(self, model, start: int, end: int, first: bool,
                 last: bool):
        super().__init__()
        base, head = _hf_base(model)
        self.first = first
        self.last = last
        if first:
            self.embed_tokens = base.embed_tokens
        self.layers = nn.ModuleList(list(base.layers[start:end]))
        self.rotary_emb = base.rotary_emb
        if last:
            self.norm = base.norm
            self.lm_head = head
        # flash/sdpa paths are causal with mask=None; eager attention gets
        # its causal mask from LlamaModel.forward, so the stage must build
        # it (the base model's forward never runs under PP)
        self.attn_impl = getattr(model.config, "_attn_implementation",
                                 "eager")

    def forward(self, x, labels: Optional[torch.Tensor] = None):
        h = self.embed_tokens(x) if self.first else x
        s = h.shape[1]
        pos = torch.arange(s, device=h.device).unsqueeze(0)
        cos_sin = self.rotary_emb(h, position_ids=pos)
        mask = None
        if self.attn_impl == "eager":
            mask = torch.full((s, s), torch.finfo(h.dtype).min,
                              device=h.device, dtype=h.dtype).triu(1)
            mask = mask[None, None]
        for layer in self.layers:
            out = layer(h, attention_mask=mask, position_ids=pos,
                        position_embeddings=cos_sin)
            h = out[0] if isinstance(out, tuple) else out
        if not self.last:
            return h
        h = self.norm(h)
        if labels is not None:
            from ...ops.cross_entropy import linear_cross_entropy
            hs = h[:, :-1, :].reshape(-1, h.shape[-1])
            tg = labels[:, 1:].reshape(-1)
            return linear_cross_entropy(hs, self.lm_head.weight, tg)
        return self.lm_head(h)


def split_hf_model(model, num_stages: int, input_names=None) -> SplitResult:
    base, head = _hf_base(model)
    assert base is not None
    n_layers = len(base.layers)
    assert n_layers >= num_stages, \
        f"{n_layers} decoder layers cannot fill {num_stages} stages"
    bounds = partition_uniform(n_layers, num_stages)
    submods = []
    specs = []
    tied = (getattr(model.config, "tie_word_embeddings", False) and
            base.embed_tokens.weight is head.weight)
    for s in range(num_stages):
        first = s == 0
        last = s == num_stages - 1
        submods.append(HFPipeStage(model, bounds[s], bounds[s + 1], first,
                                   last))
        spec = StageSpec()
        if first:
            spec.inputs.append(("batch", "input_ids"))
        else:
            spec.inputs.append(("value", s - 1))
            spec.recv_vids = [s - 1]
        if last:
            spec.inputs.append(("batch", "labels"))
        else:
            spec.outputs = [(0, s)]
            spec.send_vids = [s]
        if last:
            spec.outputs = [(0, num_stages - 1)]
        specs.append(spec)
    final_vid = num_stages - 1
    tied_groups = []
    if tied:
        tied_groups.append({0: "embed_tokens.weight",
                            num_stages - 1: "lm_head.weight"})
    logger.info("HF pipeline split: %d layers -> %s per stage", n_layers,
                [bounds[i + 1] - bounds[i] for i in range(num_stages)])
    return SplitResult(
        submodules=submods, specs=specs, final_vids=[final_vid],
        final_structure=("value", final_vid),
        batch_keys=["input_ids", "labels"], attrs={},
        tied_groups=tied_groups)
