"""Pipeline executor: interprets the 1F1B instruction schedule
(reference dist/pp/executor.py:25-725).

Eager-MI355X simplifications vs the reference: RCCL send/recv are native
(no preserved-send hack), no graph cuts (`maybe_sync` gone), activations
and grads move as tensor lists with an int64 meta header (p2p.py).

Backward correctness with pass-through values (skip connections): a value
produced at stage p and consumed at stage c > p rides through every
intermediate stage's send list; during backward each stage accumulates the
grad received from downstream into the grad it forwards upstream, plus any
local autograd contribution if the value was also consumed here.
"""
from typing import Any, Dict, List

import torch
import torch.distributed as dist

from ...utils.logger import logger
from . import p2p
from .microbatch import bind_args_to_kwargs, split_microbatches
from .schedule import (BackwardPass, ForwardPass, LoadMicroBatch,
                       OptimizerStep, RecvActivation, RecvGrad, ReduceGrads,
                       SendActivation, SendGrad, create_scheduler)
from .utils import SplitResult


class PipeExecutor:

    def __init__(self, split_result: SplitResult, stage_id: int, mesh,
                 config, device, forward_signature):
        self.sr = split_result
        self.stage_id = stage_id
        self.mesh = mesh
        self.config = config
        self.device = device
        self.stages = mesh.get_pp_num()
        self.spec = split_result.specs[stage_id]
        self.module = split_result.submodules[stage_id]
        self.num_micro = config.dist.pp.num_micro_batches
        self.broadcast_loss = config.dist.pp.broadcast_loss
        self.forward_signature = forward_signature
        self.prev_rank = mesh.stage_to_global(stage_id - 1) \
            if stage_id > 0 else None
        self.next_rank = mesh.stage_to_global(stage_id + 1) \
            if stage_id < self.stages - 1 else None
        self.producer_stage = {}
        for s, spec in enumerate(split_result.specs):
            for _idx, vid in spec.outputs:
                self.producer_stage[vid] = s
        # tied parameters (same Parameter object in >1 stage): sum grads
        # across the owning stages after every backward so each stage's
        # optimizer applies an identical update
        self.tied_groups = []
        for grp in split_result.tied_groups:
            if stage_id in grp:
                self.tied_groups.append({
                    "name": grp[stage_id],
                    "peers": sorted(s for s in grp if s != stage_id),
                })
        if self.tied_groups and config.dist.fsdp.size > 1:
            raise NotImplementedError(
                "tied parameters across pipeline stages are not supported "
                "in combination with FSDP (the tied grad lives inside a "
                "flattened shard); untie the weights or disable FSDP")
        self._reset()

    def _reset(self):
        M = self.num_micro
        self.env: List[Dict[int, torch.Tensor]] = [dict() for _ in range(M)]
        self.recv_leaves: List[Dict[int, torch.Tensor]] = \
            [dict() for _ in range(M)]
        self.micro_kwargs: List[Dict[str, Any]] = [dict() for _ in range(M)]
        self.losses: List[torch.Tensor] = []
        self.final_outputs: List[Any] = [None] * M

    # ---- instruction handlers ------------------------------------------

    def _exec_load(self, mb: int):
        pass  # micro_kwargs are pre-split in forward/forward_backward

    def _resolve_inputs(self, mb: int):
        args = []
        for kind, val in self.spec.inputs:
            if kind == "batch":
                v = self.micro_kwargs[mb].get(val)
                if isinstance(v, torch.Tensor):
                    v = v.to(self.device)
                args.append(v)
            elif kind == "value":
                args.append(self.env[mb][val])
            elif kind == "attr":
                t = self.sr.attrs[val]
                if isinstance(t, torch.Tensor) and t.device != self.device:
                    t = t.to(self.device)
                    self.sr.attrs[val] = t
                args.append(t)
            else:
                args.append(val)
        return args

    def _exec_forward(self, mb: int, output_fn=None):
        args = self._resolve_inputs(mb)
        out = self.module(*args)
        outs = out if isinstance(out, (tuple, list)) else (out,)
        for idx, vid in self.spec.outputs:
            self.env[mb][vid] = outs[idx]
        if self.stage_id == self.stages - 1:
            final = self._build_final(mb)
            if output_fn is not None:
                final = output_fn(final)
            self.final_outputs[mb] = final

    def _build_final(self, mb: int):
        struct = self.sr.final_structure

        def resolve(entry):
            kind, val = entry
            if kind == "value":
                return self.env[mb][val]
            if kind == "batch":
                return self.micro_kwargs[mb].get(val)
            return val

        if isinstance(struct, list):
            return tuple(resolve(e) for e in struct)
        return resolve(struct)

    def _exec_send_activation(self, mb: int):
        tensors = [self.env[mb][vid] for vid in self.spec.send_vids]
        p2p.send_tensors([t.detach() for t in tensors], self.next_rank,
                         self.device)

    def _exec_recv_activation(self, mb: int):
        tensors = p2p.recv_tensors(self.prev_rank, self.device)
        assert len(tensors) == len(self.spec.recv_vids)
        for vid, t in zip(self.spec.recv_vids, tensors):
            if t.is_floating_point():
                t.requires_grad_(True)
                self.recv_leaves[mb][vid] = t
            self.env[mb][vid] = t

    def _float_send_vids(self, mb: int) -> List[int]:
        return [
            vid for vid in self.spec.send_vids
            if self.env[mb].get(vid) is not None and
            self.env[mb][vid].is_floating_point()
        ]

    def _exec_send_grad(self, mb: int):
        grads = []
        pt = getattr(self, "_passthrough_grads", {})
        for vid in self.spec.recv_vids:
            leaf = self.recv_leaves[mb].get(vid)
            if leaf is None:
                continue  # non-float: no grad channel
            g = leaf.grad
            if vid in pt:
                g = pt[vid] if g is None else g + pt[vid]
            if g is None:
                g = torch.zeros_like(leaf)
            grads.append(g)
            leaf.grad = None
        p2p.send_tensors(grads, self.prev_rank, self.device,
                         with_meta=False)
        self.recv_leaves[mb].clear()
        self._passthrough_grads = {}

    def _grad_bufs(self, mb: int, float_send_vids) -> List[torch.Tensor]:
        return [
            torch.empty_like(self.env[mb][vid]) for vid in float_send_vids
        ]

    # ---- public API -----------------------------------------------------

    def forward_backward(self, *args, output_fn=None, **kwargs):
        """One full training step over num_micro_batches micro-batches.
        Returns the aggregated loss on every rank."""
        self._reset()
        full_kwargs = bind_args_to_kwargs(self.forward_signature, args,
                                          kwargs)
        self.micro_kwargs = split_microbatches(full_kwargs, self.num_micro)
        sched = create_scheduler("train", self.num_micro, self.stages,
                                 self.stage_id)
        self._exec_schedule_train(sched, output_fn)
        self._sync_tied_grads()
        return self._aggregate_total_loss()

    def _sync_tied_grads(self):
        """Sum the gradient of every tied parameter over its owning stages
        (p2p exchange on the pp group; contributions added in stage order so
        all owners compute bit-identical totals)."""
        if not self.tied_groups:
            return
        params = dict(self.module.named_parameters())
        for grp in self.tied_groups:
            p = params[grp["name"]]
            g = p.grad if p.grad is not None else torch.zeros_like(p)
            g = g.contiguous()
            contrib = {self.stage_id: g}
            ops = []
            for s in grp["peers"]:
                peer = self.mesh.stage_to_global(s)
                buf = torch.empty_like(g)
                contrib[s] = buf
                if self.stage_id < s:
                    ops.append(dist.P2POp(dist.isend, g, peer))
                    ops.append(dist.P2POp(dist.irecv, buf, peer))
                else:
                    ops.append(dist.P2POp(dist.irecv, buf, peer))
                    ops.append(dist.P2POp(dist.isend, g, peer))
            for w in dist.batch_isend_irecv(ops):
                w.wait()
            total = None
            for s in sorted(contrib):
                total = contrib[s].clone() if total is None \
                    else total + contrib[s]
            p.grad = total

    def _exec_schedule_train(self, sched, output_fn):
        pending_grads: Dict[int, List[torch.Tensor]] = {}
        float_sends: Dict[int, List[int]] = {}
        for cmds in sched.steps():
            # steady-state fusion: SendActivation(f) + RecvGrad(b) in one
            # batched p2p (both toward next_rank) — deadlock-free by
            # construction
            fuse_pair = None
            send_i = next((i for i, c in enumerate(cmds)
                           if isinstance(c, SendActivation)), None)
            recvg_i = next((i for i, c in enumerate(cmds)
                            if isinstance(c, RecvGrad)), None)
            if send_i is not None and recvg_i is not None:
                fuse_pair = (cmds[send_i].micro_batch,
                             cmds[recvg_i].micro_batch)
            for cmd in cmds:
                mb = cmd.micro_batch
                if isinstance(cmd, LoadMicroBatch):
                    pass
                elif isinstance(cmd, RecvActivation):
                    self._exec_recv_activation(mb)
                elif isinstance(cmd, ForwardPass):
                    self._exec_forward(mb, output_fn)
                    float_sends[mb] = self._float_send_vids(mb)
                elif isinstance(cmd, SendActivation):
                    if fuse_pair is not None:
                        fmb, bmb = fuse_pair
                        acts = [self.env[fmb][vid].detach()
                                for vid in self.spec.send_vids]
                        bufs = self._grad_bufs(bmb, float_sends[bmb])
                        p2p.send_acts_recv_grads(acts, bufs, self.next_rank,
                                                 self.device)
                        pending_grads[bmb] = bufs
                    else:
                        self._exec_send_activation(mb)
                elif isinstance(cmd, RecvGrad):
                    if fuse_pair is not None and mb == fuse_pair[1]:
                        pass  # already received in the fused op
                    else:
                        bufs = self._grad_bufs(mb, float_sends[mb])
                        pending_grads[mb] = p2p.recv_into(bufs,
                                                          self.next_rank)
                elif isinstance(cmd, BackwardPass):
                    self._backward_mb(mb, pending_grads.pop(mb, None),
                                      float_sends.pop(mb, []))
                elif isinstance(cmd, SendGrad):
                    self._exec_send_grad(mb)
                elif isinstance(cmd, (ReduceGrads, OptimizerStep)):
                    pass

    def _backward_mb(self, mb, grads_from_next, float_send_vids):
        if self.stage_id == self.stages - 1:
            loss = self.final_outputs[mb]
            assert isinstance(loss, torch.Tensor) and loss.dim() == 0, \
                ("PP training requires the last stage to produce a scalar "
                 "loss (model computes loss, or pass output_fn)")
            self.losses.append(loss.detach())
            (loss / self.num_micro).backward()
        else:
            assert grads_from_next is not None
            own_t, own_g = [], []
            passthrough: Dict[int, torch.Tensor] = {}
            for vid, g in zip(float_send_vids, grads_from_next):
                if self.producer_stage[vid] == self.stage_id:
                    own_t.append(self.env[mb][vid])
                    own_g.append(g)
                else:
                    passthrough[vid] = g
            if own_t:
                torch.autograd.backward(own_t, own_g)
            self._passthrough_grads = passthrough
        self.env[mb].clear()
        self.final_outputs[mb] = None

    def forward(self, *args, output_fn=None, **kwargs):
        """Inference over micro-batches; returns concatenated outputs on the
        last stage (None elsewhere)."""
        self._reset()
        full_kwargs = bind_args_to_kwargs(self.forward_signature, args,
                                          kwargs)
        self.micro_kwargs = split_microbatches(full_kwargs, self.num_micro)
        sched = create_scheduler("infer", self.num_micro, self.stages,
                                 self.stage_id)
        with torch.no_grad():
            for cmds in sched.steps():
                for cmd in cmds:
                    mb = cmd.micro_batch
                    if isinstance(cmd, RecvActivation):
                        self._exec_recv_activation(mb)
                    elif isinstance(cmd, ForwardPass):
                        self._exec_forward(mb, output_fn)
                    elif isinstance(cmd, SendActivation):
                        self._exec_send_activation(mb)
        if self.stage_id != self.stages - 1:
            return None
        outs = self.final_outputs
        if isinstance(outs[0], torch.Tensor):
            return torch.cat(outs, dim=0)
        if isinstance(outs[0], tuple):
            return tuple(
                torch.cat([o[i] for o in outs], dim=0)
                if isinstance(outs[0][i], torch.Tensor) else outs[0][i]
                for i in range(len(outs[0])))
        return outs

    def _aggregate_total_loss(self):
        """Scale/average the per-micro-batch losses, all-reduce over dp+fsdp
        groups, broadcast from the last stage over the pp group
        (reference executor.py:283-321)."""
        if self.stage_id == self.stages - 1:
            loss = torch.stack(self.losses).mean() if self.losses else \
                torch.zeros((), device=self.device)
            loss = loss.to(self.device)
            for group, num in ((self.mesh.get_fsdp_proc_group(),
                                self.mesh.get_fsdp_num()),
                               (self.mesh.get_dp_proc_group(),
                                self.mesh.get_dp_num())):
                if group is not None and num > 1:
                    dist.all_reduce(loss, group=group)
                    loss /= num
        else:
            loss = torch.zeros((), device=self.device)
        if self.broadcast_loss and self.stages > 1:
            src = self.mesh.stage_to_global(self.stages - 1)
            dist.broadcast(loss, src=src,
                           group=self.mesh.get_pp_proc_group())
        return loss
