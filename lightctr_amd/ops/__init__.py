"""Compute ops: hand-written HIP/CDNA4 kernels on GPU, torch reference
implementations on CPU (which double as the numerics test oracles).

Policy (enforced): on a CUDA/HIP device the native extension MUST be present —
ops raise ImportError rather than silently falling back to eager PyTorch, so a
GPU run always exercises the gfx950 kernels.
"""

from ._extension import hip_ops, has_hip_ops, require_hip_ops

__all__ = ["hip_ops", "has_hip_ops", "require_hip_ops"]
