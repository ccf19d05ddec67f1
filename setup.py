"""Build the in-tree gfx950 HIP extension: python setup.py build_ext --inplace

The resulting zaremba_amd/_hip*.so travels with the repo snapshot to GPU
boxes (no JIT cache involved). hipcc cross-compiles without a GPU.
"""

import os

from setuptools import setup

os.environ.setdefault("PYTORCH_ROCM_ARCH", "gfx950")

from torch.utils.cpp_extension import BuildExtension, CUDAExtension  # noqa: E402

here = os.path.dirname(os.path.abspath(__file__))

ext = CUDAExtension(
    name="zaremba_amd._hip",
    sources=[
        "zaremba_amd/csrc/ext_bind.hip",
        "zaremba_amd/csrc/gemm.hip",
        "zaremba_amd/csrc/lstm.hip",
        "zaremba_amd/csrc/lstm_persistent.hip",
        "zaremba_amd/csrc/elementwise.hip",
    ],
    extra_compile_args={
        "cxx": ["-O3", "-std=c++17"],
        "nvcc": ["-O3", "-std=c++17", "--offload-arch=gfx950"],
    },
)

setup(
    name="zaremba_amd",
    version="0.1.0",
    packages=["zaremba_amd"],
    ext_modules=[ext],
    cmdclass={"build_ext": BuildExtension},
)
