"""Build the in-tree HIP extension for gfx950 (MI355X).

    PYTORCH_ROCM_ARCH=gfx950 python setup.py build_ext --inplace

The resulting _mcdp_C*.so lands inside the package directory so it travels
with repo snapshots (it is git-ignored; history stays source-only).
"""
import os
from pathlib import Path

from setuptools import setup

os.environ.setdefault("PYTORCH_ROCM_ARCH", "gfx950")

from torch.utils import cpp_extension  # noqa: E402

ROOT = Path(__file__).resolve().parent
CSRC = ROOT / "csrc"

# exclude the *_hip.hip copies torch's hipify pass generates in-place
sources = sorted(
    str(p) for p in CSRC.glob("*.hip") if not p.name.endswith("_hip.hip")
) + [str(CSRC / "bindings.cpp")]

setup(
    name="mlx_cuda_distributed_pretraining_amd",
    version="0.1.0",
    packages=["mlx_cuda_distributed_pretraining_amd"],
    ext_modules=[
        cpp_extension.CUDAExtension(
            name="mlx_cuda_distributed_pretraining_amd._mcdp_C",
            sources=sources,
            extra_compile_args={
                "cxx": ["-O3", "-std=c++17"],
                "nvcc": ["-O3", "-std=c++17"],
            },
        )
    ],
    cmdclass={"build_ext": cpp_extension.BuildExtension},
)
