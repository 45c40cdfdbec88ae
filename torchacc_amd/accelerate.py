"""The one-call API (reference accelerate.py:49-149).

accelerate(model, dataloader=None, config=None) ->
    (wrapped_model, wrapped_loader) or wrapped_model

Eager-ROCm flow: validate config -> init process group + comm warm-up ->
device select -> AsyncLoader -> optional SDPA swap / HF kernel patches ->
TP module surgery -> DistributedParallel (PP -> FSDP -> DP) -> gradient
checkpointing -> move to device.
"""
from typing import Optional

import torch
import torch.nn.functional as F

from . import dist as ta_dist
from .async_loader import AsyncLoader
from .config import Config
from .utils.checkpoint import gradient_checkpoint
from .utils.logger import logger


def accelerate(model: torch.nn.Module, dataloader=None,
               config: Optional[Config] = None):
    config = config or Config()
    ta_dist.init_process_group(config)
    config.validate()
    mesh = config.get_mesh()
    ta_dist.init_comm_context(config)

    if torch.cuda.is_available():
        device = torch.device("cuda", ta_dist.local_rank())
        torch.cuda.set_device(device)
    else:
        device = torch.device("cpu")

    loader = None
    if dataloader is not None:
        dl_cfg = config.dataloader
        loader = AsyncLoader(
            dataloader, device, buckets=dl_cfg.buckets,
            max_length=dl_cfg.max_length, num_buckets=dl_cfg.num_buckets,
            pad_value_dict=dl_cfg.pad_value_dict)

    if config.compute.acc_scaled_dot_attn:
        from .ops.scaled_dot_product_attention import \
            scaled_dot_product_attention
        F.scaled_dot_product_attention = scaled_dot_product_attention
        logger.info("replaced F.scaled_dot_product_attention")

    if not config.compute.disable_kernel_patches:
        from .utils import patch
        patch.apply_fused_kernel_patches(model)

    # pure-precision cast before flattening/sharding: the FSDP engine shards
    # in the compute dtype (fp32 masters live in the fused-AdamW state)
    if config.compute.bf16:
        model = model.to(torch.bfloat16)
    elif config.compute.fp16:
        model = model.to(torch.float16)

    if config.dist.tp.size > 1:
        from .dist import tp
        model = tp.parallelize_module(model, config)

    model = model.to(device)

    if config.is_distributed_parallel() or config.dist.fsdp.wrap_layer_cls:
        model = ta_dist.DistributedParallel(model, config)

    if config.memory.gc:
        target = model
        from .dist.parallel_module import ParallelModule
        inner = model._get_underlay_model() if isinstance(
            model, ParallelModule) else model
        if config.memory.gc_cls:
            gradient_checkpoint(inner, config.memory.gc_cls,
                                config.memory.gc_cnt,
                                config.memory.gc_selective_attn)
        else:
            logger.warning("memory.gc set without gc_cls: wrapping the root "
                           "module in one checkpoint region")
            wrapped = gradient_checkpoint(
                inner, selective_attn=config.memory.gc_selective_attn)
            if isinstance(model, ParallelModule):
                model._update_underlay_model(wrapped)
            else:
                model = wrapped

    if loader is not None:
        return model, loader
    return model


def broadcast_master_param(model: torch.nn.Module, config: Config) -> None:
    """Broadcast rank-0 params to the world (reference accelerate.py:142)."""
    import torch.distributed as dist
    if not dist.is_initialized() or dist.get_world_size() == 1:
        return
    with torch.no_grad():
        for p in model.parameters():
            dist.broadcast(p.data, src=0)
