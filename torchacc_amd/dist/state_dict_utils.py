"""FSDP checkpoint machinery: sharded save/load, optimizer-state APIs,
offline consolidation and resharding.

Reimplements the semantics of the reference's dist/state_dict_utils.py
(get_layer_full_info :51, all_gather_state :157, broadcast_processed_state
:179, load/save_checkpoints :245-318, consolidate_sharded_model_checkpoints
:321, reshard :422-551, consolidate_and_reshard_* :552-738) against this
framework's flat-param units. Checkpoint layout kept reference-compatible:

- per rank:  rank-R-of-W-model.pth     = {"model": ..., "shard_metadata": ...}
             rank-R-of-W-optimizer.pth = {"optimizer": ..., "shard_metadata": ...}
- shards are 1/W slices of each unit's flat tensor padded to a multiple of
  W * 128 (PAD_MULTIPLE), concatenation order = rank order;
- consolidation emits a standard full state_dict (+ layer_info.pickle with
  the per-unit name/shape/numel map).
"""
import os
import pickle
import re
import threading
from typing import Dict, List, Optional, Tuple

import torch
import torch.distributed as dist

from ..utils.logger import logger
from .fsdp import PAD_MULTIPLE, FullyShardedDataParallel

MODEL_NAME_PATTERN = "rank-*-of-*-model.pth"
OPTIM_NAME_PATTERN = "rank-*-of-*-optimizer.pth"


# ---------------------------------------------------------------------------
# online save/load
# ---------------------------------------------------------------------------

def _unwrap(fsdp):
    """Accept a DistributedParallel composite or the FSDP wrapper itself."""
    if not isinstance(fsdp, FullyShardedDataParallel) and \
            getattr(fsdp, "fsdp_wrapper", None) is not None:
        return fsdp.fsdp_wrapper
    return fsdp


def save_sharded_checkpoint(fsdp: FullyShardedDataParallel, optimizer,
                            ckpt_dir: str):
    """Every rank writes its model + optimizer shard files."""
    fsdp = _unwrap(fsdp)
    os.makedirs(ckpt_dir, exist_ok=True)
    r, w = fsdp.shard_rank, fsdp.ws
    torch.save(fsdp.sharded_state_dict(),
               os.path.join(ckpt_dir, f"rank-{r}-of-{w}-model.pth"))
    if optimizer is not None:
        torch.save(
            sharded_optim_state_dict(fsdp, optimizer),
            os.path.join(ckpt_dir, f"rank-{r}-of-{w}-optimizer.pth"))


def load_sharded_checkpoint(fsdp: FullyShardedDataParallel, optimizer,
                            ckpt_dir: str):
    fsdp = _unwrap(fsdp)
    r, w = fsdp.shard_rank, fsdp.ws
    payload = torch.load(
        os.path.join(ckpt_dir, f"rank-{r}-of-{w}-model.pth"),
        map_location="cpu", weights_only=False)
    fsdp.load_sharded_state_dict(payload)
    if optimizer is not None:
        opt_payload = torch.load(
            os.path.join(ckpt_dir, f"rank-{r}-of-{w}-optimizer.pth"),
            map_location="cpu", weights_only=False)
        load_sharded_optim_state_dict(fsdp, optimizer, opt_payload)


# ---------------------------------------------------------------------------
# optimizer-state APIs (reference fsdp.py:243-578)
# ---------------------------------------------------------------------------

def _unit_of_param(fsdp, p) -> Optional[int]:
    for i, u in enumerate(fsdp.units):
        if u.shard is p:
            return i
    return None


def sharded_optim_state_dict(fsdp: FullyShardedDataParallel,
                             optimizer) -> dict:
    """Per-rank optimizer state keyed by unit name."""
    fsdp = _unwrap(fsdp)
    state = {}
    for group in optimizer.param_groups:
        for p in group["params"]:
            ui = _unit_of_param(fsdp, p)
            if ui is None:
                continue
            st = optimizer.state.get(p, {})
            state[fsdp.units[ui].name] = {
                k: (v.detach().cpu() if isinstance(v, torch.Tensor) else v)
                for k, v in st.items()
            }
    groups = [{k: v for k, v in g.items() if k != "params"}
              for g in optimizer.param_groups]
    return {
        "optimizer": {"state": state, "param_groups": groups},
        "shard_metadata": fsdp.shard_metadata(),
    }


def load_sharded_optim_state_dict(fsdp: FullyShardedDataParallel, optimizer,
                                  payload: dict):
    fsdp = _unwrap(fsdp)
    state = payload["optimizer"]["state"]
    by_name = {u.name: u for u in fsdp.units}
    for group in optimizer.param_groups:
        for p in group["params"]:
            ui = _unit_of_param(fsdp, p)
            if ui is None:
                continue
            name = fsdp.units[ui].name
            if name not in state:
                continue
            st = {}
            for k, v in state[name].items():
                if isinstance(v, torch.Tensor):
                    st[k] = v.to(p.device)
                else:
                    st[k] = v
            optimizer.state[p] = st
    del by_name


def full_optim_state_dict(fsdp: FullyShardedDataParallel, optimizer) -> dict:
    """All-gather every unit's shard states and unflatten into per-param
    entries (reference fsdp.py:291-424). Returned on every rank."""
    fsdp = _unwrap(fsdp)
    full_state: Dict[str, dict] = {}
    for p_group in optimizer.param_groups:
        for p in p_group["params"]:
            ui = _unit_of_param(fsdp, p)
            if ui is None:
                continue
            u = fsdp.units[ui]
            st = optimizer.state.get(p, {})
            gathered: Dict[str, torch.Tensor] = {}
            for k, v in st.items():
                if isinstance(v, torch.Tensor) and v.numel() == \
                        u.shard_numel:
                    if u.ws > 1:
                        buf = torch.empty(u.padded_numel, dtype=v.dtype,
                                          device=v.device)
                        dist.all_gather_into_tensor(buf, v.contiguous(),
                                                    group=u.group)
                    else:
                        buf = v
                    gathered[k] = buf
                else:
                    gathered[k] = v
            # unflatten per param
            seen = set()
            for mod, attr, off, shape, n in u.entries:
                if off in seen:
                    continue
                seen.add(off)
                name = fsdp._param_full_name(mod, attr)
                entry = {}
                for k, v in gathered.items():
                    if isinstance(v, torch.Tensor) and v.numel() == \
                            u.padded_numel:
                        entry[k] = v[off:off + n].view(shape).cpu().clone()
                    else:
                        entry[k] = v
                full_state[name] = entry
    groups = [{k: v for k, v in g.items() if k != "params"}
              for g in optimizer.param_groups]
    return {"state": full_state, "param_groups": groups}


def to_torch_optim_state_dict(full_state: dict,
                              model: torch.nn.Module) -> dict:
    """Convert the name-keyed full optimizer state into torch's index-keyed
    ``Optimizer.state_dict()`` format, loadable into a PLAIN (non-FSDP)
    optimizer built over ``model.parameters()`` — the reference's
    unflatten-per-layer interop (reference dist/fsdp.py:291-424). The fused
    AdamW's fp32 ``master`` entries are dropped (a plain optimizer rebuilds
    them from the loaded weights)."""
    name_to_idx = {n: i for i, (n, _p) in
                   enumerate(model.named_parameters())}
    state = {}
    for name, entry in full_state["state"].items():
        if name not in name_to_idx:
            continue
        state[name_to_idx[name]] = {
            k: v for k, v in entry.items() if k != "master"
        }
    groups = []
    for g in full_state["param_groups"]:
        gg = dict(g)
        gg["params"] = sorted(name_to_idx.values())
        groups.append(gg)
    return {"state": state, "param_groups": groups}


def from_torch_optim_state_dict(torch_sd: dict,
                                model: torch.nn.Module) -> dict:
    """Inverse of :func:`to_torch_optim_state_dict`: index-keyed plain
    optimizer state -> the name-keyed format consumed by
    :func:`optim_state_dict_to_load`."""
    idx_to_name = {i: n for i, (n, _p) in
                   enumerate(model.named_parameters())}
    state = {}
    for idx, entry in torch_sd["state"].items():
        name = idx_to_name.get(int(idx))
        if name is not None:
            state[name] = dict(entry)
    groups = [{k: v for k, v in g.items() if k != "params"}
              for g in torch_sd["param_groups"]]
    return {"state": state, "param_groups": groups}


def optim_state_dict_to_load(fsdp: FullyShardedDataParallel, full_state: dict,
                             optimizer=None) -> dict:
    """Reshard a full optimizer state dict for this rank's shards; if
    ``optimizer`` is given, also install it (reference fsdp.py:426-578)."""
    fsdp = _unwrap(fsdp)
    state = full_state["state"]
    out_state = {}
    for u in fsdp.units:
        shard_state: Dict[str, torch.Tensor] = {}
        scalar_state = {}
        # flatten per-key across the unit's params in offset order
        keys = None
        seen = set()
        ordered = []
        for mod, attr, off, shape, n in u.entries:
            if off in seen:
                continue
            seen.add(off)
            name = fsdp._param_full_name(mod, attr)
            ordered.append((off, name, shape, n))
        ordered.sort()
        for off, name, shape, n in ordered:
            entry = state[name]
            if keys is None:
                keys = list(entry.keys())
            for k in keys:
                v = entry[k]
                if isinstance(v, torch.Tensor) and v.numel() == n:
                    shard_state.setdefault(k, []).append(v.reshape(-1))
                else:
                    scalar_state[k] = v
        final = {}
        for k, parts in shard_state.items():
            flat = torch.cat(parts)
            pad = u.padded_numel - flat.numel()
            if pad > 0:
                flat = torch.cat([flat, flat.new_zeros(pad)])
            final[k] = flat[u.rank * u.shard_numel:(u.rank + 1) *
                            u.shard_numel].clone()
        final.update(scalar_state)
        out_state[u.name] = final
    if optimizer is not None:
        for group in optimizer.param_groups:
            for p in group["params"]:
                ui = _unit_of_param(fsdp, p)
                if ui is None:
                    continue
                st = {
                    k: (v.to(p.device) if isinstance(v, torch.Tensor) else v)
                    for k, v in out_state[fsdp.units[ui].name].items()
                }
                optimizer.state[p] = st
    return {"state": out_state, "param_groups": full_state["param_groups"]}


# ---------------------------------------------------------------------------
# offline consolidation / resharding (CLI backend)
# ---------------------------------------------------------------------------

def _expand_pattern(ckpt_dir: str, pattern: str) -> List[str]:
    """rank-*-of-*-model.pth -> sorted per-rank paths, validated count."""
    rx = re.escape(pattern).replace("\\*", r"(\d+)")
    out = []
    for f in os.listdir(ckpt_dir):
        m = re.fullmatch(rx, f)
        if m:
            out.append((int(m.group(1)), int(m.group(2)), f))
    if not out:
        raise FileNotFoundError(
            f"no files matching {pattern} under {ckpt_dir}")
    world = out[0][1]
    assert all(w == world for _, w, _ in out), "mixed world sizes"
    assert len(out) == world, \
        f"found {len(out)} shards, expected {world}"
    out.sort()
    return [os.path.join(ckpt_dir, f) for _, _, f in out]


def load_checkpoints(ckpt_dir: str, pattern: str) -> List[dict]:
    """Threaded multi-file load (reference :245-283)."""
    paths = _expand_pattern(ckpt_dir, pattern)
    results: List[Optional[dict]] = [None] * len(paths)

    def worker(i, p):
        results[i] = torch.load(p, map_location="cpu", weights_only=False)

    threads = [
        threading.Thread(target=worker, args=(i, p))
        for i, p in enumerate(paths)
    ]
    for t in threads:
        t.start()
    for t in threads:
        t.join()
    assert all(r is not None for r in results)
    return results  # type: ignore


def consolidate_sharded_model_checkpoints(ckpt_dir: str,
                                          pattern: str = MODEL_NAME_PATTERN,
                                          save_path: Optional[str] = None
                                          ) -> Tuple[dict, dict]:
    """Concat shards -> unpad -> unflatten into a full state_dict; writes
    layer_info.pickle beside save_path (reference :321-366)."""
    ckpts = load_checkpoints(ckpt_dir, pattern)
    meta = ckpts[0]["shard_metadata"]
    world = meta["world_size"]
    full_sd = {}
    layer_info = []
    for ui, umeta in enumerate(meta["units"]):
        uname = umeta["unit_name"]
        flat = torch.cat([c["model"][uname] for c in ckpts])
        assert flat.numel() == umeta["padded_numel"], \
            (flat.numel(), umeta["padded_numel"])
        for pmeta in umeta["params"]:
            t = flat[pmeta["offset"]:pmeta["offset"] + pmeta["numel"]]
            full_sd[pmeta["name"]] = t.view(pmeta["shape"]).clone()
            layer_info.append({
                "unit": uname,
                "name": pmeta["name"],
                "shape": pmeta["shape"],
                "numel": pmeta["numel"],
                "offset": pmeta["offset"],
            })
    # buffers saved under __buffer__.
    for key, val in ckpts[0]["model"].items():
        if key.startswith("__buffer__."):
            full_sd[key[len("__buffer__."):]] = val
    if save_path:
        os.makedirs(os.path.dirname(save_path) or ".", exist_ok=True)
        torch.save(full_sd, save_path)
        with open(os.path.join(os.path.dirname(save_path) or ".",
                               "layer_info.pickle"), "wb") as f:
            pickle.dump(layer_info, f)
    return full_sd, meta


def consolidate_sharded_optimizer_checkpoints(
        ckpt_dir: str, pattern: str = OPTIM_NAME_PATTERN,
        save_path: Optional[str] = None) -> Tuple[dict, dict]:
    """reference :368-420"""
    ckpts = load_checkpoints(ckpt_dir, pattern)
    meta = ckpts[0]["shard_metadata"]
    full_state = {}
    for umeta in meta["units"]:
        uname = umeta["unit_name"]
        states = [c["optimizer"]["state"][uname] for c in ckpts]
        merged_keys = states[0].keys()
        flat_by_key = {}
        scalars = {}
        for k in merged_keys:
            v0 = states[0][k]
            if isinstance(v0, torch.Tensor) and v0.numel() == \
                    umeta["shard_numel"]:
                flat_by_key[k] = torch.cat([s[k] for s in states])
            else:
                scalars[k] = v0
        for pmeta in umeta["params"]:
            entry = dict(scalars)
            for k, flat in flat_by_key.items():
                entry[k] = flat[pmeta["offset"]:pmeta["offset"] +
                                pmeta["numel"]].view(
                                    pmeta["shape"]).clone()
            full_state[pmeta["name"]] = entry
    out = {
        "state": full_state,
        "param_groups": ckpts[0]["optimizer"]["param_groups"],
    }
    if save_path:
        torch.save(out, save_path)
    return out, meta


def _reshard_flat(full_sd: dict, meta: dict, reshard_num: int,
                  value_of) -> List[dict]:
    """Rebuild per-rank flat shards for a new world size."""
    out = [dict() for _ in range(reshard_num)]
    for umeta in meta["units"]:
        uname = umeta["unit_name"]
        parts = []
        for pmeta in sorted(umeta["params"], key=lambda p: p["offset"]):
            parts.append(value_of(pmeta).reshape(-1))
        flat = torch.cat(parts)
        pad_to = reshard_num * PAD_MULTIPLE
        padded = ((flat.numel() + pad_to - 1) // pad_to) * pad_to
        if padded > flat.numel():
            flat = torch.cat([flat, flat.new_zeros(padded - flat.numel())])
        shard = padded // reshard_num
        for r in range(reshard_num):
            out[r][uname] = flat[r * shard:(r + 1) * shard].clone()
    return out


def _resharded_metadata(meta: dict, reshard_num: int) -> List[dict]:
    metas = []
    for r in range(reshard_num):
        m = {
            "world_size": reshard_num,
            "rank": r,
            "pad_multiple": PAD_MULTIPLE,
            "flat_dtype": meta.get("flat_dtype", "torch.bfloat16"),
            "units": [],
            "buffers": meta.get("buffers", []),
        }
        for umeta in meta["units"]:
            pad_to = reshard_num * PAD_MULTIPLE
            padded = ((umeta["total_numel"] + pad_to - 1) // pad_to) * pad_to
            m["units"].append({
                "unit_name": umeta["unit_name"],
                "total_numel": umeta["total_numel"],
                "padded_numel": padded,
                "shard_numel": padded // reshard_num,
                "params": umeta["params"],
            })
        metas.append(m)
    return metas


def reshard_model_dict(full_sd: dict, meta: dict, reshard_num: int,
                       out_dir: str):
    shards = _reshard_flat(full_sd, meta, reshard_num,
                           lambda p: full_sd[p["name"]])
    metas = _resharded_metadata(meta, reshard_num)
    for r in range(reshard_num):
        model_sd = shards[r]
        for bname in meta.get("buffers", []):
            if bname in full_sd:
                model_sd[f"__buffer__.{bname}"] = full_sd[bname]
        torch.save({"model": model_sd, "shard_metadata": metas[r]},
                   os.path.join(out_dir,
                                f"rank-{r}-of-{reshard_num}-model.pth"))
    logger.info("wrote %d resharded model files to %s", reshard_num, out_dir)


def reshard_optim_dict(full_opt: dict, meta: dict, reshard_num: int,
                       out_dir: str):
    state = full_opt["state"]
    # determine tensor keys from any entry
    any_entry = next(iter(state.values()))
    tensor_keys = [
        k for k, v in any_entry.items()
        if isinstance(v, torch.Tensor) and v.numel() > 1 or
        (isinstance(v, torch.Tensor) and v.dim() > 0)
    ]
    scalar_keys = [k for k in any_entry.keys() if k not in tensor_keys]
    metas = _resharded_metadata(meta, reshard_num)
    per_rank_state = [dict() for _ in range(reshard_num)]
    for umeta in meta["units"]:
        uname = umeta["unit_name"]
        for r in range(reshard_num):
            per_rank_state[r][uname] = {}
        for k in tensor_keys:
            shards = _reshard_flat(
                {}, {"units": [umeta]}, reshard_num,
                lambda p, _k=k: state[p["name"]][_k].float())
            for r in range(reshard_num):
                per_rank_state[r][uname][k] = shards[r][uname]
        for k in scalar_keys:
            v = state[next(p["name"] for p in umeta["params"])][k]
            for r in range(reshard_num):
                per_rank_state[r][uname][k] = v
    for r in range(reshard_num):
        torch.save(
            {
                "optimizer": {
                    "state": per_rank_state[r],
                    "param_groups": full_opt["param_groups"],
                },
                "shard_metadata": metas[r],
            },
            os.path.join(out_dir,
                         f"rank-{r}-of-{reshard_num}-optimizer.pth"))
    logger.info("wrote %d resharded optimizer files to %s", reshard_num,
                out_dir)


def consolidate_and_reshard_fsdp_model_dict(ckpt_dir, out_dir,
                                            reshard_num: int,
                                            save_consolidated=True):
    os.makedirs(out_dir, exist_ok=True)
    full_sd, meta = consolidate_sharded_model_checkpoints(
        ckpt_dir, MODEL_NAME_PATTERN,
        os.path.join(out_dir, "consolidated_model.pth")
        if save_consolidated else None)
    if reshard_num and reshard_num > 0:
        reshard_model_dict(full_sd, meta, reshard_num, out_dir)
    return full_sd


def consolidate_and_reshard_fsdp_optim_dict(ckpt_dir, out_dir,
                                            reshard_num: int,
                                            save_consolidated=True):
    os.makedirs(out_dir, exist_ok=True)
    full_opt, meta = consolidate_sharded_optimizer_checkpoints(
        ckpt_dir, OPTIM_NAME_PATTERN,
        os.path.join(out_dir, "consolidated_optimizer.pth")
        if save_consolidated else None)
    if reshard_num and reshard_num > 0:
        reshard_optim_dict(full_opt, meta, reshard_num, out_dir)
    return full_opt


def consolidate_and_reshard_fsdp_checkpoint(ckpt_dir, out_dir,
                                            reshard_num: int,
                                            ckpt_type: str = "all"):
    """reference :719-738"""
    if ckpt_type in ("all", "model"):
        consolidate_and_reshard_fsdp_model_dict(ckpt_dir, out_dir,
                                                reshard_num)
    if ckpt_type in ("all", "optimizer"):
        consolidate_and_reshard_fsdp_optim_dict(ckpt_dir, out_dir,
                                                reshard_num)
