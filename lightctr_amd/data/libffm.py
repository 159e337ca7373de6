"""libffm-format dataset loader.

Parses the reference's text format ``label field:fid:val`` per line
(semantics of /root/reference/LightCTR/fm_algo_abst.h:70-107 — re-implemented,
not translated) into a CSR batch representation:

    row_ptr : int32 [n_rows + 1]
    fields  : int32 [nnz]
    fids    : int32 [nnz]
    vals    : float32 [nnz]
    labels  : float32 [n_rows]  (0/1)

All GPU kernels in lightctr_amd.ops consume this CSR layout directly.
"""

from __future__ import annotations

from dataclasses import dataclass

import numpy as np
import torch


@dataclass
class LibffmDataset:
    row_ptr: torch.Tensor  # int32 [N+1]
    fields: torch.Tensor  # int32 [nnz]
    fids: torch.Tensor  # int32 [nnz]
    vals: torch.Tensor  # float32 [nnz]
    labels: torch.Tensor  # float32 [N]

    @property
    def num_rows(self) -> int:
        return self.labels.numel()

    @property
    def nnz(self) -> int:
        return self.fids.numel()

    @property
    def num_features(self) -> int:
        return int(self.fids.max().item()) + 1 if self.nnz else 0

    @property
    def num_fields(self) -> int:
        return int(self.fields.max().item()) + 1 if self.nnz else 0

    def to(self, device) -> "LibffmDataset":
        return LibffmDataset(
            self.row_ptr.to(device),
            self.fields.to(device),
            self.fids.to(device),
            self.vals.to(device),
            self.labels.to(device),
        )

    def slice_rows(self, start: int, end: int) -> "LibffmDataset":
        """Contiguous row slice (for minibatching / rank sharding)."""
        rp = self.row_ptr[start : end + 1]
        lo = int(rp[0].item())
        hi = int(rp[-1].item())
        return LibffmDataset(
            (rp - lo).contiguous(),
            self.fields[lo:hi],
            self.fids[lo:hi],
            self.vals[lo:hi],
            self.labels[start:end],
        )


def load_libffm(path: str, max_rows: int | None = None) -> LibffmDataset:
    try:  # native C++ parser (ops extension); Python fallback below
        from ..ops._extension import has_hip_ops, require_hip_ops

        if has_hip_ops():
            rp, fl, fi, v, lb = require_hip_ops().parse_libffm(
                path, -1 if max_rows is None else max_rows)
            return LibffmDataset(rp, fl, fi, v, lb)
    except ImportError:
        pass
    row_ptr = [0]
    fields: list[int] = []
    fids: list[int] = []
    vals: list[float] = []
    labels: list[float] = []
    with open(path, "r") as f:
        for line in f:
            parts = line.split()
            if not parts:
                continue
            labels.append(float(parts[0]))
            for tok in parts[1:]:
                a, b, c = tok.split(":")
                fields.append(int(a))
                fids.append(int(b))
                vals.append(float(c))
            row_ptr.append(len(fids))
            if max_rows is not None and len(labels) >= max_rows:
                break
    return LibffmDataset(
        torch.tensor(row_ptr, dtype=torch.int32),
        torch.tensor(fields, dtype=torch.int32),
        torch.tensor(fids, dtype=torch.int32),
        torch.tensor(np.asarray(vals, dtype=np.float32)),
        torch.tensor(np.asarray(labels, dtype=np.float32)),
    )
