"""GBM training throughput on MI355X (LDS-histogram boosted trees).

One "step" = one boosting round on a fixed synthetic matrix (the
non-neural bench convention). Prints rounds/sec and wall per round.
"""
import argparse
import sys
import time
from pathlib import Path

sys.path.insert(0, str(Path(__file__).resolve().parent.parent))

import torch

from lightctr_amd.models.gbm import GBMHyper, GBMModel


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--rows", type=int, default=200_000)
    ap.add_argument("--cols", type=int, default=128)
    ap.add_argument("--rounds", type=int, default=40)
    ap.add_argument("--depth", type=int, default=6)
    args = ap.parse_args()

    dev = "cuda" if torch.cuda.is_available() else "cpu"
    g = torch.Generator().manual_seed(3)
    X = torch.randn(args.rows, args.cols, generator=g).to(dev)
    w = torch.randn(args.cols, generator=g)
    y = ((X.cpu() @ w + 0.5 * (X.cpu()[:, 0] * X.cpu()[:, 1]))
         > 0).float().to(dev)

    m = GBMModel(GBMHyper(n_rounds=args.rounds, max_depth=args.depth,
                          n_classes=2, seed=5), device=dev)
    if dev == "cuda":
        torch.cuda.synchronize()
    t0 = time.perf_counter()
    m.fit(X, y, log=None)
    if dev == "cuda":
        torch.cuda.synchronize()
    dt = time.perf_counter() - t0
    p = m.predict_proba(X)
    from lightctr_amd.utils.metrics import auc_score

    auc = auc_score(p.cpu(), y.cpu())
    print(f"rows={args.rows} cols={args.cols} depth={args.depth} "
          f"rounds={args.rounds}: {dt:.2f}s total, "
          f"{dt / args.rounds * 1000:.1f} ms/round, "
          f"{args.rounds / dt:.1f} rounds/s, train AUC={auc:.4f}")


if __name__ == "__main__":
    main()
