"""Build the in-tree HIP extension (gfx950 only):
    PYTORCH_ROCM_ARCH=gfx950 python setup.py build_ext --inplace
produces midgpt_amd/ops/_C*.so (travels with the repo snapshot)."""
import os

os.environ.setdefault("PYTORCH_ROCM_ARCH", "gfx950")

from setuptools import setup  # noqa: E402
from torch.utils.cpp_extension import BuildExtension, CUDAExtension  # noqa: E402

setup(
    name="midgpt_amd_ext",
    ext_modules=[
        CUDAExtension(
            name="midgpt_amd.ops._C",
            sources=["midgpt_amd/ops/csrc/ext.hip"],
            libraries=["hipblaslt"],
            extra_compile_args={
                "cxx": ["-O3", "-std=c++17"],
                "nvcc": ["-O3", "-std=c++17"],
            },
        )
    ],
    cmdclass={"build_ext": BuildExtension},
)
