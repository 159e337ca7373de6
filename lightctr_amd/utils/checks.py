"""Runtime invariant checks (the reference's pervasive
asserts — NaN/Inf wire validity distributed_algo_abst.h:73-75, buffer
bounds buffer.h:107, leak checkpoint memory_pool.h:34-36 — mapped to the
GPU setting).

Always-on checks live at the wire/API boundaries (PS NaN drops, binding
shape/dtype CHKs, FFM field-range guard). The heavier batch validators
here are enabled with LIGHTCTR_CHECK=1 (the GPU analog of running under
a sanitizer: one pass over every batch before it reaches the kernels,
catching malformed CSR / out-of-range ids / non-finite values that
would otherwise corrupt device state silently).
"""

from __future__ import annotations

import os

import torch

CHECKS_ON = os.environ.get("LIGHTCTR_CHECK", "0") == "1"


def validate_csr_batch(row_ptr: torch.Tensor, fids: torch.Tensor,
                       vals: torch.Tensor, num_features: int,
                       fields: torch.Tensor | None = None,
                       num_fields: int | None = None) -> None:
    """Full batch validation (LIGHTCTR_CHECK=1 only — costs syncs)."""
    if not CHECKS_ON:
        return
    assert row_ptr.dim() == 1 and int(row_ptr[0]) == 0, "row_ptr[0] != 0"
    assert bool((row_ptr[1:] >= row_ptr[:-1]).all()), \
        "row_ptr must be non-decreasing"
    nnz = int(row_ptr[-1])
    assert fids.numel() == nnz and vals.numel() == nnz, \
        f"nnz mismatch: row_ptr says {nnz}, fids {fids.numel()}"
    if nnz:
        fmin, fmax = int(fids.min()), int(fids.max())
        assert 0 <= fmin and fmax < num_features, \
            f"fid out of range: [{fmin}, {fmax}] vs F={num_features}"
        assert bool(torch.isfinite(vals).all()), "non-finite feature value"
    if fields is not None and num_fields is not None and nnz:
        gmax = int(fields.max())
        assert 0 <= int(fields.min()) and gmax < num_fields, \
            f"field out of range: max {gmax} vs nfields={num_fields}"


def validate_finite(name: str, *tensors: torch.Tensor) -> None:
    """Parameter-state validity sweep (LIGHTCTR_CHECK=1 only)."""
    if not CHECKS_ON:
        return
    for i, t in enumerate(tensors):
        assert bool(torch.isfinite(t).all()), \
            f"{name}[{i}]: non-finite parameter state"
