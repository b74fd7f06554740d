"""Build the photon_amd HIP extension IN-TREE for gfx950 (MI355X).

    PYTORCH_ROCM_ARCH=gfx950 python setup.py build_ext --inplace

The resulting photon_amd/ops/_photon_hip.so is git-ignored but travels with
repo snapshots to GPU boxes (it is NOT gpurun-ignored).
"""

import os
from pathlib import Path

from setuptools import setup

os.environ.setdefault("PYTORCH_ROCM_ARCH", "gfx950")

from torch.utils.cpp_extension import BuildExtension, CUDAExtension  # noqa: E402

ROOT = Path(__file__).parent
HIP_DIR = ROOT / "photon_amd" / "ops" / "hip"

# torch's hipify caches *_hip.hip / *_hip.h copies and does NOT re-hipify
# when only an included header changed — which silently builds stale device
# code (cost us a debugging session). Always clear the generated copies.
for stale in HIP_DIR.glob("*_hip.hip"):
    stale.unlink()
for stale in HIP_DIR.glob("*_hip.h"):
    stale.unlink()

sources = [
    str(HIP_DIR / "bindings.cpp"),
    str(HIP_DIR / "layernorm.hip"),
    str(HIP_DIR / "cross_entropy.hip"),
    str(HIP_DIR / "optim.hip"),
    str(HIP_DIR / "attention.hip"),
    str(HIP_DIR / "linear_lt.hip"),
    str(HIP_DIR / "bias_grad.hip"),
    str(HIP_DIR / "fused_linear_fn.cpp"),
    str(HIP_DIR / "debug.hip"),
]

setup(
    name="photon_amd_hip",
    ext_modules=[
        CUDAExtension(
            name="photon_amd.ops._photon_hip",
            sources=sources,
            extra_compile_args={
                "cxx": ["-O3", "-std=c++17"],
                "nvcc": ["-O3", "-std=c++17"],
            },
            libraries=["hipblaslt"],
        )
    ],
    cmdclass={"build_ext": BuildExtension.with_options(use_ninja=True)},
)
