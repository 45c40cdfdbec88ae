"""HF Trainer bridge (reference core/accelerate_hf_trainer.py:14-78).

The reference's HF path needed only ``import torchacc`` +
``patch_llama/patch_qwen`` before building the Trainer
(benchmarks/accuracy/run_clm.py:59-62): the import-time flash-attention
patch routed HF models to the accelerated kernels and HF's own ``--fsdp``
config drove sharding. Here:

- :func:`accelerate_hf_trainer(enable=True)` applies the flash-attention
  and fused-kernel patches so any HF Trainer run uses the CDNA4 kernels;
- sharding: either HF's own FSDP flags (torch-native FSDP on ROCm works
  unmodified), or wrap the model with torchacc_amd.accelerate() before
  handing it to the Trainer for this framework's flat-param engine.
"""
from .utils import patch
from .utils.logger import logger


def accelerate_hf_trainer(enable: bool = True) -> bool:
    if not enable:
        logger.info("accelerate_hf_trainer(enable=False): no patches")
        return False
    ok = patch.patch_fa()
    ok = patch.apply_fused_kernel_patches() or ok
    patch.patch_llama()
    patch.patch_qwen()
    if ok:
        logger.info("HF Trainer bridge active: flash-attention + fused "
                    "kernels patched")
    return ok


# reference-compatible names
def patch_accelerate():
    """The reference patched the `accelerate` package's TPU detection so HF
    drove torch_xla; the eager ROCm backend is plain CUDA-API to HF, so
    nothing needs patching."""
    return False


def patch_transformers():
    return accelerate_hf_trainer(True)
