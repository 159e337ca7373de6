"""Dense CSV loader (MNIST-style: label,pix,pix,...) — parity with the
reference DL loader (/root/reference/LightCTR/dl_algo_abst.h:179-228),
which feeds the CNN/RNN/VAE trainers. Values are scaled to [0,1] when
`scale` is set (the reference divides by 255)."""

from __future__ import annotations

import numpy as np
import torch


def load_dense_csv(path: str, scale: float = 1.0 / 255.0,
                   max_rows: int | None = None):
    """Returns (X [N, D] float32, y [N] int64)."""
    labels = []
    rows = []
    with open(path) as f:
        for line in f:
            parts = line.strip().rstrip(",").split(",")
            if len(parts) < 2:
                continue
            labels.append(int(float(parts[0])))
            rows.append(np.asarray(parts[1:], dtype=np.float32))
            if max_rows is not None and len(rows) >= max_rows:
                break
    X = torch.from_numpy(np.stack(rows)) * scale
    y = torch.tensor(labels, dtype=torch.int64)
    return X, y
