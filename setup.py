"""Build the in-tree CDNA4 HIP extension: python setup.py build_ext --inplace

Targets gfx950 (MI355X) only — set via PYTORCH_ROCM_ARCH (no multi-arch
fatbins, no CUDA path).
"""
import os

os.environ.setdefault("PYTORCH_ROCM_ARCH", "gfx950")

from setuptools import setup  # noqa: E402
from torch.utils.cpp_extension import BuildExtension, CUDAExtension  # noqa: E402

ROOT = os.path.dirname(os.path.abspath(__file__))
CSRC = os.path.join(ROOT, "torchacc_amd", "csrc")

sources = [
    os.path.join(CSRC, f) for f in (
        "bindings.cpp",
        "elementwise.hip",
        "cross_entropy.hip",
        "adamw.hip",
        "flash_attn_fwd.hip",
        "flash_attn_bwd.hip",
        "flash_attn_varlen.hip",
        "flash_attn_extra_fwd.hip",
        "flash_attn_extra_bwd.hip",
        "gemm.hip",
    )
]

setup(
    name="torchacc_amd",
    version="0.1.0",
    packages=["torchacc_amd"],
    ext_modules=[
        CUDAExtension(
            name="torchacc_amd._C",
            sources=sources,
            libraries=["hipblaslt"],
            extra_compile_args={
                "cxx": ["-O3", "-std=c++17"],
                "nvcc": ["-O3", "-std=c++17"],
            },
        )
    ],
    cmdclass={"build_ext": BuildExtension.with_options(use_ninja=True)},
    entry_points={
        "console_scripts": [
            "consolidate_and_reshard_fsdp_ckpts = "
            "torchacc_amd.utils.consolidate_and_reshard_ckpts:main",
        ]
    },
)
