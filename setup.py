"""Build the gfx950 HIP kernel extension in-tree.

Usage:  PYTORCH_ROCM_ARCH=gfx950 python setup.py build_ext --inplace

The resulting _kernels .so lands inside inference_gateway_amd/ops/ so the
gpurun snapshot carries it to the GPU box (JIT caches under ~/.cache do not
travel).
"""
import os
from pathlib import Path

from setuptools import setup

os.environ.setdefault("PYTORCH_ROCM_ARCH", "gfx950")

from torch.utils import cpp_extension  # noqa: E402

ROOT = Path(__file__).parent
CSRC = ROOT / "inference_gateway_amd" / "ops" / "csrc"

sources = [str(CSRC / "bindings.cpp")] + sorted(
    str(p) for p in CSRC.glob("*.hip")
)

setup(
    name="inference_gateway_amd_kernels",
    ext_modules=[
        cpp_extension.CUDAExtension(
            name="inference_gateway_amd.ops._kernels",
            sources=sources,
            extra_compile_args={
                "cxx": ["-O3", "-std=c++17"],
                "nvcc": [
                    "-O3",
                    "-std=c++17",
                    "--offload-arch=gfx950",
                ],
            },
        )
    ],
    cmdclass={"build_ext": cpp_extension.BuildExtension},
)
