"""In-tree build of the gfx950 HIP extension eventgrad_amd._core.

    python setup.py build_ext --inplace

The .so lands inside eventgrad_amd/ so it travels with the repo snapshot
(no JIT cache dependency). hipcc cross-compiles without a GPU.
"""

import os

os.environ.setdefault("PYTORCH_ROCM_ARCH", "gfx950")

from setuptools import setup  # noqa: E402
from torch.utils.cpp_extension import BuildExtension, CUDAExtension  # noqa: E402

ROOT = os.path.dirname(os.path.abspath(__file__))
SRC = [
    "csrc/bindings.cpp",
    "csrc/engine.hip",
    "csrc/elementwise.hip",
    "csrc/gemm.hip",
    "csrc/conv.hip",
    "csrc/bn.hip",
    "csrc/pool.hip",
    "csrc/loss.hip",
    "csrc/topk.hip",
]

setup(
    name="eventgrad_amd_core",
    ext_modules=[
        CUDAExtension(
            name="eventgrad_amd._core",
            sources=SRC,
            extra_compile_args={
                "cxx": ["-O3", "-std=c++17"],
                "nvcc": ["-O3", "-std=c++17", "--offload-arch=gfx950"],
            },
        )
    ],
    cmdclass={"build_ext": BuildExtension},
)
