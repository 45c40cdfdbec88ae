"""Fused AdamW for MI355X (+ syncfree found_inf gating).

Replaces torch_xla.amp.syncfree.AdamW (reference utils/patch.py:55-57): the
HIP kernel performs the whole AdamW update for each (large, flat) param in
one grid-stride launch, keeps fp32 exp_avg/exp_avg_sq (and optional fp32
master weights for bf16/fp16 params), and accepts a device-side
``found_inf`` flag: when nonzero the update is a device-side no-op, so fp16
loss scaling never host-syncs. The step counter lives on the host (bias
correction after a skipped step drifts by that one step — warmup-only,
matching syncfree semantics closely enough for training parity).
"""
from typing import Optional

import torch

from ._backend import dispatch


class AdamW(torch.optim.Optimizer):

    # torch.amp.GradScaler.step() sees this and hands us device-side
    # ``grad_scale``/``found_inf`` tensors instead of host-syncing on
    # found_inf.item() — the fully syncfree fp16 path (reference
    # torch_xla.amp.syncfree semantics, utils/patch.py:55-57)
    _step_supports_amp_scaling = True

    def __init__(self, params, lr=1e-3, betas=(0.9, 0.999), eps=1e-8,
                 weight_decay=1e-2, use_master_weights: bool = True):
        defaults = dict(lr=lr, betas=betas, eps=eps,
                        weight_decay=weight_decay)
        super().__init__(params, defaults)
        self.use_master_weights = use_master_weights

    @torch.no_grad()
    def step(self, closure=None, found_inf: Optional[torch.Tensor] = None):
        loss = None
        if closure is not None:
            with torch.enable_grad():
                loss = closure()
        # attributes installed by torch.amp.GradScaler.step (see
        # _step_supports_amp_scaling): grad_scale is the CURRENT scale when
        # grads are still scaled (None once unscale_ ran); found_inf is the
        # device-side overflow flag
        if found_inf is None:
            found_inf = getattr(self, "found_inf", None)
        grad_scale = getattr(self, "grad_scale", None)
        inv_scale = None
        if grad_scale is not None:
            inv_scale = grad_scale.double().reciprocal().float()
        for group in self.param_groups:
            beta1, beta2 = group["betas"]
            lr = group["lr"]
            eps = group["eps"]
            wd = group["weight_decay"]
            params, grads, exp_avgs, exp_avg_sqs, masters, steps = \
                [], [], [], [], [], []
            for p in group["params"]:
                if p.grad is None:
                    continue
                state = self.state[p]
                if len(state) == 0:
                    state["step"] = 0.0
                    state["exp_avg"] = torch.zeros(
                        p.shape, dtype=torch.float32, device=p.device)
                    state["exp_avg_sq"] = torch.zeros(
                        p.shape, dtype=torch.float32, device=p.device)
                    if self.use_master_weights and p.dtype in (
                            torch.bfloat16, torch.float16):
                        state["master"] = p.detach().float().clone()
                state["step"] += 1.0
                params.append(p)
                grads.append(p.grad)
                exp_avgs.append(state["exp_avg"])
                exp_avg_sqs.append(state["exp_avg_sq"])
                masters.append(state.get("master"))
                steps.append(state["step"])
            if not params:
                continue
            ext = dispatch(params[0])
            if ext is not None:
                fi = found_inf if found_inf is not None else \
                    torch.zeros((), dtype=torch.float32,
                                device=params[0].device)
                master_list = [
                    m if m is not None else torch.empty(
                        0, device=params[0].device) for m in masters
                ]
                grads = [
                    g if g.dtype == p.dtype else g.to(p.dtype)
                    for g, p in zip(grads, params)
                ]
                if inv_scale is not None:
                    # grads still carry the loss scale: unscale device-side
                    grads = torch._foreach_mul(grads, inv_scale)
                ext.fused_adamw(params, grads, exp_avgs, exp_avg_sqs,
                                master_list, steps, fi, lr, beta1, beta2,
                                eps, wd)
            else:
                if found_inf is not None and bool(found_inf != 0):
                    continue
                for p, g, m, v, mw, t in zip(params, grads, exp_avgs,
                                             exp_avg_sqs, masters, steps):
                    gf = g.float()
                    if inv_scale is not None:
                        gf = gf * inv_scale
                    m.mul_(beta1).add_(gf, alpha=1 - beta1)
                    v.mul_(beta2).addcmul_(gf, gf, value=1 - beta2)
                    bc1 = 1 - beta1 ** t
                    bc2 = 1 - beta2 ** t
                    denom = (v / bc2).sqrt_().add_(eps)
                    upd = (m / bc1) / denom
                    tgtf = mw if mw is not None else (
                        p if p.dtype == torch.float32 else p.float())
                    tgtf.mul_(1 - lr * wd).add_(upd, alpha=-lr)
                    if tgtf is not p:
                        p.copy_(tgtf.to(p.dtype))
        return loss


# torch_xla.amp.syncfree-compatible alias
SyncFreeAdamW = AdamW
