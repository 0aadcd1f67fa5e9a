"""Build the matrel_amd HIP extension in-tree (matrel_amd/_C.so).

    PYTORCH_ROCM_ARCH=gfx950 python setup.py build_ext --inplace

gfx950 (MI355X) only - no multi-arch fatbins, no CUDA path.
"""

import os

from setuptools import setup

os.environ.setdefault("PYTORCH_ROCM_ARCH", "gfx950")

from torch.utils.cpp_extension import BuildExtension, CUDAExtension  # noqa

ROOT = os.path.dirname(os.path.abspath(__file__))
CSRC = os.path.join(ROOT, "matrel_amd", "csrc")

sources = [
    os.path.join(CSRC, "ext.cpp"),
    os.path.join(CSRC, "gemm.hip"),
    os.path.join(CSRC, "elementwise.hip"),
    os.path.join(CSRC, "reduce.hip"),
    os.path.join(CSRC, "transpose.hip"),
    os.path.join(CSRC, "spmm.hip"),
    os.path.join(CSRC, "spgemm.hip"),
    os.path.join(CSRC, "gemm_mx8.hip"),
    os.path.join(CSRC, "gemm_mx4.hip"),
    os.path.join(CSRC, "gemm_mx6.hip"),
    os.path.join(CSRC, "mxconvert.hip"),
]

setup(
    name="matrel_amd",
    version="0.1.0",
    packages=["matrel_amd"],
    ext_modules=[
        CUDAExtension(
            name="matrel_amd._C",
            sources=sources,
            extra_compile_args={
                "cxx": ["-O3", "-std=c++17"],
                "nvcc": ["-O3", "-std=c++17"],
            },
        )
    ],
    cmdclass={"build_ext": BuildExtension.with_options(no_python_abi_suffix=True)},
)
