"""In-tree build of the gfx950 HIP extension.

Usage:
    PYTORCH_ROCM_ARCH=gfx950 python setup.py build_ext --inplace

Produces msrflute_amd/_C.*.so next to the package sources so the built
artifact travels with the tree (no JIT cache dependence).
"""

import os

os.environ.setdefault("PYTORCH_ROCM_ARCH", "gfx950")

from setuptools import setup
from torch.utils.cpp_extension import BuildExtension, CUDAExtension

ROOT = os.path.dirname(os.path.abspath(__file__))

ext = CUDAExtension(
    name="msrflute_amd._C",
    sources=[
        "msrflute_amd/csrc/bindings.cpp",
        "msrflute_amd/csrc/flat_ops.hip",
        "msrflute_amd/csrc/fused_cnn.hip",
        "msrflute_amd/csrc/fused_cnn_mega.hip",
        "msrflute_amd/csrc/lstm_seq.hip",
        "msrflute_amd/csrc/gru_seq.hip",
    ],
    extra_compile_args={
        "cxx": ["-O3", "-std=c++17"],
        "nvcc": ["-O3", "-std=c++17", "--offload-arch=gfx950"],
    },
)

setup(
    name="msrflute_amd",
    version="0.1.0",
    packages=["msrflute_amd"],
    ext_modules=[ext],
    cmdclass={"build_ext": BuildExtension.with_options(no_python_abi_suffix=False)},
)
