"""Compute ops: hand-written CDNA4 HIP kernels on GPU, torch reference on CPU.

Dispatch policy (fail-loud):
  * tensors on a HIP device ("cuda" in torch-ROCm) ALWAYS run the native
    kernels from ``eventgrad_amd._core``; if the extension is missing the op
    raises instead of silently falling back to eager PyTorch;
  * CPU tensors run plain fp32 PyTorch — this is both the test oracle for
    every HIP kernel and the backend for CPU/gloo multi-process tests.
"""

from .backend import native, native_available, require_native  # noqa: F401
from . import functional  # noqa: F401
