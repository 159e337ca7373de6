"""Per-shape wgrad kernel debug: maxdiff + mismatch location pattern."""
import sys
from pathlib import Path

sys.path.insert(0, str(Path(__file__).resolve().parent.parent))

import torch

from lightctr_amd.ops import hip_ops


def check(M, N, K):
    g = torch.Generator().manual_seed(M + K)
    At = (torch.randn(K, M, generator=g) * 0.5).to(torch.bfloat16).cuda()
    Bt = (torch.randn(K, N, generator=g) * 0.5).to(torch.bfloat16).cuda()
    C = hip_ops.gemm_wgrad_bf16(At, Bt, M, N, K)
    ref = At.float().t() @ Bt.float()
    d = (C - ref).abs()
    tol = 2e-2 * K ** 0.5
    bad = (d > tol)
    print(f"{M}x{N}x{K}: maxdiff={float(d.max()):.4f} tol={tol:.3f} "
          f"bad={int(bad.sum())}/{M * N}")
    if bad.any():
        idx = bad.nonzero()
        rows = torch.unique(idx[:, 0])
        cols = torch.unique(idx[:, 1])
        print(f"   bad rows: n={rows.numel()} head={rows[:8].tolist()} "
              f"min={int(rows.min())} max={int(rows.max())}")
        print(f"   bad cols: n={cols.numel()} head={cols[:8].tolist()} "
              f"min={int(cols.min())} max={int(cols.max())}")
        r, c = int(idx[0, 0]), int(idx[0, 1])
        print(f"   C[{r},{c}]={float(C[r, c]):.4f} ref={float(ref[r, c]):.4f}")


def main():
    for shape in [(16, 16, 1024), (128, 128, 1024), (128, 128, 2048),
                  (256, 624, 4096), (128, 256, 2048), (256, 256, 1056),
                  (144, 112, 1024)]:
        check(*shape)


if __name__ == "__main__":
    main()
