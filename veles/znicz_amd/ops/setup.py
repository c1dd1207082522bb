"""Build the _mzk HIP extension in-tree for gfx950.

    python veles/znicz_amd/ops/setup.py build_ext --inplace

hipcc cross-compiles without a GPU; the resulting _mzk*.so lives next to
this file so it travels with the repo snapshot to GPU boxes.
"""

import os
import sys

os.environ.setdefault("PYTORCH_ROCM_ARCH", "gfx950")

from setuptools import setup  # noqa: E402
from torch.utils.cpp_extension import BuildExtension, CUDAExtension  # noqa: E402

HERE = os.path.dirname(os.path.abspath(__file__))
os.chdir(HERE)

ext = CUDAExtension(
    name="_mzk",
    sources=[
        "bindings.cpp",
        "hip/gemm.hip",
        "hip/gemm8.hip",
        "hip/elementwise.hip",
        "hip/softmax.hip",
        "hip/pool_lrn.hip",
    ],
    extra_compile_args={
        "cxx": ["-O3", "-std=c++17"],
        "nvcc": ["-O3", "-std=c++17", "--offload-arch=gfx950"],
    },
)

if __name__ == "__main__":
    if len(sys.argv) == 1:
        sys.argv += ["build_ext", "--inplace"]
    setup(name="mzk", version="0.1", ext_modules=[ext],
          cmdclass={"build_ext": BuildExtension.with_options(
              no_python_abi_suffix=False)})
