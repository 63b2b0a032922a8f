"""Op library.

Replaces the reference's native dependency surface (libnd4j CUDA kernels +
cuDNN bindings, reference pom.xml:104-128) with hand-written HIP/CDNA4
kernels for gfx950, dispatched per-device:

- CUDA (= ROCm/HIP) tensors: custom autograd Functions backed by the in-tree
  `_C` HIP extension.  If the extension is missing on a GPU machine, ops FAIL
  LOUDLY — there is deliberately no silent eager fallback on GPU.
- CPU tensors: plain PyTorch fp32 ops (the reference's nd4j-native CPU
  backend analog), used for the world_size=1 plumbing config and as the
  numerics reference in tests.
"""

from . import functional  # noqa: F401
from .backend import has_hip_ext, hip_ext  # noqa: F401
