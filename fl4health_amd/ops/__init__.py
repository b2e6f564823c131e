"""Device-op layer: hand-written CDNA4 HIP kernels with CPU reference fallbacks.

On a GPU (ROCm) device every op REQUIRES the in-tree `fl4health_amd._C`
extension — if it is missing the op raises instead of silently falling back
to eager PyTorch. On CPU tensors the ops run a pure-PyTorch reference
implementation (used by the CPU test suite and as the numerics oracle).
"""

from fl4health_amd.ops import functional

__all__ = ["functional"]
