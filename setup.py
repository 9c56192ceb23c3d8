"""In-tree build of the smartcal_amd HIP extension for MI355X (gfx950).

    PYTORCH_ROCM_ARCH=gfx950 python setup.py build_ext --inplace

Sources are pure HIP (.hip) — no CUDA, no hipify. torch's BuildExtension
drives hipcc for .hip sources on a ROCm build.
"""

import os
from pathlib import Path

from setuptools import setup

os.environ.setdefault("PYTORCH_ROCM_ARCH", "gfx950")

from torch.utils.cpp_extension import BuildExtension, CUDAExtension  # noqa: E402

ROOT = Path(__file__).resolve().parent
CSRC = ROOT / "smartcal_amd" / "ops" / "csrc"

sources = [str(CSRC / f) for f in [
    "bindings.hip",
    "fused_linear.hip",
    "gemm_f32.hip",
    "elementwise.hip",
    "enet_solver.hip",
    "als_sweep.hip",
    "conv2d.hip",
    "per.hip",
    "fused_linear_bf16.hip",
    "attention.hip",
    "cgemm.hip",
    "coherency.hip",
    "hessianres.hip",
    "two_loop.hip",
    "gather_sum.hip",
]]

setup(
    name="smartcal_amd",
    version="0.1.0",
    packages=["smartcal_amd"],
    ext_modules=[
        CUDAExtension(
            name="smartcal_amd.ops._hip",
            sources=sources,
            extra_compile_args={
                "cxx": ["-O3"],
                "nvcc": ["-O3", "--offload-arch=gfx950", "-std=c++17"],
            },
        )
    ],
    cmdclass={"build_ext": BuildExtension.with_options(use_ninja=True)},
)
