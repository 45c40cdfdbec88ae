"""Pipeline graph splitting (reference dist/pp/utils.py:12-350).

Approach: fx-trace the model (utils/trace.py), assign every top-level node a
stage with a split callback (module-name split points, explicit
``pipe_split()`` markers, or automatic balanced splitting over repeated
transformer blocks), partition with torch.fx.passes.split_module, then build
a VALUE-FLOW SPEC instead of rewriting the graph for pass-through values:

- every cross-stage value gets an id; stage s sends to s+1 exactly the
  values produced at stages <= s that are still needed by stages > s (so
  skip connections ride through intermediate stages — the reference's
  _propagate_output :85-239 — but threaded by the executor, not the graph);
- stage inputs are classified as ('batch', key) for placeholders (loaded
  from the micro-batch at that stage — Note [PP input_tensor_attr],
  reference :329-350) or ('value', vid).
"""
import re
from dataclasses import dataclass, field
from typing import Any, Dict, List, Optional, Tuple

import torch
import torch.fx as fx
from torch.fx.passes.split_module import split_module

from ...utils.logger import logger


def pipe_split():
    """User marker: insert between two statements of a custom forward to cut
    a stage boundary there (reference :12-77). Traced as a leaf."""
    return None


class PipeSplitWrapper(torch.nn.Module):
    """Wrap a module so a pipe_split() marker is emitted before its forward
    (annotate_split_points support)."""

    def __init__(self, mod: torch.nn.Module):
        super().__init__()
        self.mod = mod

    def forward(self, *args, **kwargs):
        pipe_split()
        return self.mod(*args, **kwargs)


def annotate_split_points(model: torch.nn.Module, split_points: List[str]):
    for qualname in split_points:
        parent_name, _, attr = qualname.rpartition(".")
        parent = model.get_submodule(parent_name) if parent_name else model
        setattr(parent, attr, PipeSplitWrapper(getattr(parent, attr)))
    return model


@dataclass
class StageSpec:
    """Per-stage IO contract used by the executor."""
    inputs: List[Tuple[str, Any]] = field(default_factory=list)
    # ('batch', key) | ('value', vid)
    outputs: List[Tuple[int, int]] = field(default_factory=list)
    # (submodule output tuple index, vid)
    recv_vids: List[int] = field(default_factory=list)  # from stage-1
    send_vids: List[int] = field(default_factory=list)  # to stage+1


@dataclass
class SplitResult:
    submodules: List[fx.GraphModule]
    specs: List[StageSpec]
    final_vids: List[int]       # vids of the model's final output(s)
    final_structure: Any        # how to rebuild the output from final_vids
    batch_keys: List[str]
    attrs: Dict[str, Any] = field(default_factory=dict)
    tied_groups: List[Dict[int, str]] = field(default_factory=list)
    # parameters shared by >1 stage (e.g. embedding tied to the LM head):
    # each entry maps stage_id -> qualified param name inside that stage's
    # submodule; the executor all-reduces their grads across those stages
    # after every backward (reference keeps tied weights consistent via the
    # compiler; eager PP must sum the per-stage grads explicitly)
    # top-level get_attr values (buffers like rope tables) consumed by
    # stage inputs — the reference moved these into consumer submodules
    # (move_single_param_to_callee, trace.py:95-176); we hand them to the
    # executor directly


def _module_stack_top(node: fx.Node) -> Optional[str]:
    stack = node.meta.get("nn_module_stack")
    if not stack:
        return None
    # first entry is the outermost submodule qualname
    first_key = next(iter(stack))
    val = stack[first_key]
    if isinstance(val, tuple):
        return val[0] if isinstance(val[0], str) else first_key
    return first_key


def _auto_split_points(gm: fx.GraphModule, num_stages: int) -> List[str]:
    """Balanced split over repeated blocks ('<prefix>.<i>' submodules),
    weighting each block by parameter count."""
    block_re = re.compile(r"^(.*\.)?(\d+)$")
    blocks: List[str] = []
    for node in gm.graph.nodes:
        stack = node.meta.get("nn_module_stack")
        if not stack:
            continue
        for qualname in stack:
            name = qualname if isinstance(qualname, str) else str(qualname)
            m = block_re.match(name)
            if m and (not blocks or blocks[-1] != name):
                if name not in blocks:
                    blocks.append(name)
                break
    if len(blocks) < num_stages:
        raise ValueError(
            f"auto-split found only {len(blocks)} repeated blocks for "
            f"{num_stages} stages; pass pp.split_points explicitly")
    from ...utils.utils import partition_uniform
    bounds = partition_uniform(len(blocks), num_stages)
    return [blocks[b] for b in bounds[1:-1]]


def split(gm: fx.GraphModule, num_stages: int,
          split_points: Optional[List[str]] = None) -> SplitResult:
    if num_stages == 1:
        raise ValueError("split() requires num_stages > 1")
    if not split_points:
        split_points = _auto_split_points(gm, num_stages)
    assert len(split_points) == num_stages - 1, \
        f"need {num_stages - 1} split points, got {split_points}"
    logger.info("pp split points: %s", split_points)

    remaining = list(split_points)
    stage_of_node: Dict[fx.Node, int] = {}
    cur = 0
    for node in gm.graph.nodes:
        if node.op in ("placeholder", "output"):
            continue
        if node.op == "call_function" and getattr(
                node.target, "__name__", "") == "pipe_split":
            cur += 1
            stage_of_node[node] = cur
            continue
        top = None
        stack = node.meta.get("nn_module_stack")
        if stack:
            names = [k if isinstance(k, str) else str(k) for k in stack]
        else:
            names = []
        if remaining:
            sp = remaining[0]
            if any(n == sp or n.startswith(sp + ".") for n in names):
                cur += 1
                remaining.pop(0)
        stage_of_node[node] = cur
    n_found = cur + 1
    assert n_found == num_stages, \
        (f"split produced {n_found} stages, wanted {num_stages}; "
         f"unmatched split points: {remaining}")

    part = split_module(gm, None,
                        lambda n: stage_of_node.get(n, 0),
                        keep_original_order=True)

    # drop pipe_split leftovers inside partitions
    for name, sub in part.named_children():
        if not isinstance(sub, fx.GraphModule):
            continue
        changed = False
        for node in list(sub.graph.nodes):
            if node.op == "call_function" and getattr(
                    node.target, "__name__", "") == "pipe_split":
                sub.graph.erase_node(node)
                changed = True
        if changed:
            sub.recompile()

    # ---- build the value-flow spec from the main (partitioned) graph ----
    submod_names = [f"submod_{i}" for i in range(num_stages)]
    submods = [getattr(part, n) for n in submod_names]

    vid_of: Dict[Tuple[str, int], int] = {}   # (submod_name, out_idx) -> vid
    producer_stage: Dict[int, int] = {}
    next_vid = [0]

    def get_vid(sub_name: str, idx: int) -> int:
        key = (sub_name, idx)
        if key not in vid_of:
            vid_of[key] = next_vid[0]
            producer_stage[vid_of[key]] = submod_names.index(sub_name)
            next_vid[0] += 1
        return vid_of[key]

    attrs: Dict[str, Any] = {}

    def classify_arg(arg) -> Tuple[str, Any]:
        if isinstance(arg, fx.Node):
            if arg.op == "placeholder":
                return ("batch", arg.target)
            if arg.op == "get_attr":
                if arg.target not in attrs:
                    obj = part
                    for piece in arg.target.split("."):
                        obj = getattr(obj, piece)
                    attrs[arg.target] = obj
                return ("attr", arg.target)
            if arg.op == "call_module" and arg.target in submod_names:
                # single-output submodule used directly
                return ("value", get_vid(arg.target, 0))
            if arg.op == "call_function" and arg.target.__name__ == \
                    "getitem":
                src, idx = arg.args
                assert isinstance(src, fx.Node) and \
                    src.target in submod_names, \
                    f"unsupported main-graph value {arg.format_node()}"
                return ("value", get_vid(src.target, idx))
            raise ValueError(
                f"unsupported cross-stage arg: {arg.format_node()}")
        return ("const", arg)

    specs = [StageSpec() for _ in range(num_stages)]
    consumers: Dict[int, List[int]] = {}
    batch_keys: List[str] = []
    for node in part.graph.nodes:
        if node.op == "placeholder":
            batch_keys.append(node.target)
    for node in part.graph.nodes:
        if node.op == "call_module" and node.target in submod_names:
            stage = submod_names.index(node.target)
            for a in node.args:
                kind, val = classify_arg(a)
                specs[stage].inputs.append((kind, val))
                if kind == "value":
                    consumers.setdefault(val, []).append(stage)
    # final output structure
    out_node = next(n for n in part.graph.nodes if n.op == "output")
    final_vids: List[int] = []

    def map_out(a):
        kind, val = classify_arg(a)
        if kind == "value":
            if val not in final_vids:
                final_vids.append(val)
            consumers.setdefault(val, []).append(num_stages - 1 + 10**6)
            return ("value", val)
        return (kind, val)

    raw_out = out_node.args[0]
    if isinstance(raw_out, (tuple, list)):
        final_structure = [map_out(a) for a in raw_out]
    else:
        final_structure = map_out(raw_out)

    # send set per stage: values produced at <= s, consumed at > s (final
    # outputs count as consumed by the virtual last consumer)
    for s in range(num_stages - 1):
        send = []
        for vid, prod in producer_stage.items():
            if prod <= s:
                last_use = max(consumers.get(vid, [prod]))
                if last_use > s:
                    send.append(vid)
        send.sort()
        specs[s].send_vids = send
        specs[s + 1].recv_vids = send
    for (sub_name, idx), vid in vid_of.items():
        st = submod_names.index(sub_name)
        specs[st].outputs.append((idx, vid))
    for s in range(num_stages):
        specs[s].outputs.sort()

    # ---- move top-level attrs (params/buffers) into consuming stages ----
    # split_module hoists get_attr to the main graph; a parameter consumed
    # only via that path (e.g. lm_head.weight feeding the fused
    # linear-cross-entropy) would otherwise never belong to any stage and
    # never be trained (the reference's move_single_param_to_callee,
    # utils/trace.py:95-176).
    from collections import defaultdict
    attr_consumers = defaultdict(set)
    for st, spec in enumerate(specs):
        for kind, val in spec.inputs:
            if kind == "attr":
                attr_consumers[val].add(st)
    for st, spec in enumerate(specs):
        sub = submods[st]
        phs = [n for n in sub.graph.nodes if n.op == "placeholder"]
        new_inputs = []
        changed = False
        for idx, (kind, val) in enumerate(spec.inputs):
            if kind != "attr":
                new_inputs.append((kind, val))
                continue
            orig = attrs[val]
            san = "_attr_" + val.replace(".", "_")
            if isinstance(orig, torch.nn.Parameter):
                # shared across stages is fine: the SAME Parameter object
                # is registered in every consuming stage and the executor
                # sums its grad across those stages after backward
                sub.register_parameter(san, orig)
            elif isinstance(orig, torch.Tensor):
                sub.register_buffer(
                    san,
                    orig if len(attr_consumers[val]) == 1 else orig.clone())
            else:
                setattr(sub, san, orig)
            ph = phs[idx]
            with sub.graph.inserting_before(ph):
                ga = sub.graph.get_attr(san)
            ph.replace_all_uses_with(ga)
            sub.graph.erase_node(ph)
            changed = True
        if changed:
            spec.inputs = new_inputs
            sub.recompile()
        else:
            spec.inputs = new_inputs

    # ---- tied-parameter detection (same Parameter object in >1 stage,
    # e.g. input embedding reused as the LM head weight) ----
    by_id: Dict[int, Dict[int, str]] = {}
    for st, sub in enumerate(submods):
        for name, prm in sub.named_parameters():
            by_id.setdefault(id(prm), {})[st] = name
    tied_groups = [g for g in by_id.values() if len(g) > 1]
    if tied_groups:
        logger.info("pp: %d tied parameter group(s) across stages: %s",
                    len(tied_groups),
                    [sorted(g.keys()) for g in tied_groups])

    return SplitResult(submodules=submods, specs=specs,
                       final_vids=final_vids,
                       final_structure=final_structure,
                       batch_keys=batch_keys, attrs=attrs,
                       tied_groups=tied_groups)


def output_index_map(sub: fx.GraphModule) -> int:
    """Number of outputs a partition produces."""
    out = next(n for n in sub.graph.nodes if n.op == "output")
    val = out.args[0]
    if isinstance(val, (tuple, list)):
        return len(val)
    return 1
