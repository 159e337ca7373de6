"""Inference (serving) throughput: predict_proba on synthetic
Criteo-shaped batches, per sparse model. Forward-only, hipGraph-free,
batch-at-a-time — the BatchPredictor serving path.
"""
import argparse
import sys
import time
from pathlib import Path

sys.path.insert(0, str(Path(__file__).resolve().parent.parent))

import torch

from lightctr_amd.data.synthetic import SyntheticCriteo


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--model", default="fm",
                    choices=["fm", "ffm", "nfm", "widedeep"])
    ap.add_argument("--batch", type=int, default=65536)
    ap.add_argument("--features", type=int, default=1 << 24)
    ap.add_argument("--reps", type=int, default=50)
    args = ap.parse_args()

    dev = "cuda" if torch.cuda.is_available() else "cpu"
    gen = SyntheticCriteo(num_features=args.features, seed=7, device=dev)
    row_ptr, fields, fids, vals, labels = gen.batch(args.batch)

    if args.model == "fm":
        from lightctr_amd.models.fm import FMHyper, FMModel

        m = FMModel(FMHyper(num_features=args.features, k=16), device=dev)
        fn = lambda: m.predict_proba(row_ptr, fids, vals)
    elif args.model == "ffm":
        from lightctr_amd.models.ffm import FFMHyper, FFMModel

        # bf16 compute mirror (config #3 serving precision)
        m = FFMModel(FFMHyper(num_features=args.features, num_fields=39,
                              k=8, dtype="bf16"), device=dev)
        fn = lambda: m.predict_proba(row_ptr, fields, fids, vals)
    elif args.model == "nfm":
        from lightctr_amd.models.nfm import NFMHyper, NFMModel

        m = NFMModel(NFMHyper(num_features=args.features, k=16,
                              hidden=(64,)), device=dev)
        fn = lambda: m.predict_proba(row_ptr, fids, vals)
    else:
        from lightctr_amd.models.wide_deep import (WideDeepHyper,
                                                   WideDeepModel)

        m = WideDeepModel(WideDeepHyper(num_features=args.features,
                                        num_fields=39, k=16,
                                        hidden=(256, 128)), device=dev)
        fn = lambda: m.predict_proba(row_ptr, fids, vals)

    for _ in range(5):
        fn()
    if dev == "cuda":
        torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(args.reps):
        fn()
    if dev == "cuda":
        torch.cuda.synchronize()
    dt = (time.perf_counter() - t0) / args.reps
    print(f"{args.model}: {args.batch / dt / 1e6:.1f}M predictions/sec "
          f"({dt * 1e3:.3f} ms / {args.batch}-row batch)")


if __name__ == "__main__":
    main()
