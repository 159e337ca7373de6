"""Loader for the in-tree HIP extension (lightctr_amd.ops._hip_ops).

The extension is built in-tree by ``python setup.py build_ext --inplace``
(driven by __graft_entry__.build()) so the .so ships with the repo snapshot
to GPU boxes. On a machine with a GPU, ops FAIL LOUDLY if the extension is
missing — no silent eager fallback on the compute path.
"""

from __future__ import annotations

import torch  # noqa: F401  -- loads libc10/libtorch before the extension

_hip_ops = None
_import_error: Exception | None = None

try:
    from . import _hip_ops as _ext  # type: ignore

    _hip_ops = _ext
except ImportError as e:  # extension not built (CPU-only dev box is fine)
    _import_error = e


def has_hip_ops() -> bool:
    return _hip_ops is not None


def require_hip_ops():
    if _hip_ops is None:
        raise ImportError(
            "lightctr_amd HIP extension (_hip_ops) is not built but a GPU "
            "op was requested. Build it in-tree with "
            "`PYTORCH_ROCM_ARCH=gfx950 python setup.py build_ext --inplace`. "
            f"Original import error: {_import_error!r}"
        )
    return _hip_ops


class _LazyOps:
    """Attribute access proxy: hip_ops.fm_forward(...) resolves at call time."""

    def __getattr__(self, name):
        return getattr(require_hip_ops(), name)


hip_ops = _LazyOps()


def sort_ids(fids, upper: int):
    """(sorted, perm int64) for nonnegative int32 ids known to be < upper —
    drop-in for torch.sort(fids) on the GPU hot path, but radix-sorts only
    the live bit range (2-3 passes instead of 4; see sort_kernels.hip)."""
    end_bit = max(1, int(upper - 1).bit_length())
    return require_hip_ops().radix_sort_index(fids, end_bit)
