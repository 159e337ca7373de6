"""Train-to-AUC run for the BASELINE headline metric ("examples/sec +
test AUC"): streams synthetic Criteo-shaped batches through the fused FM
(or FFM/NFM/W&D) trainer and reports held-out AUC/logloss + sustained
throughput.

Usage (GPU box): python tools/train_auc.py --model fm --steps 200
"""

import argparse
import sys
import time
from pathlib import Path

sys.path.insert(0, str(Path(__file__).resolve().parent.parent))

import torch  # noqa: E402


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--model", default="fm",
                    choices=["fm", "ffm", "nfm", "widedeep"])
    ap.add_argument("--steps", type=int, default=200)
    ap.add_argument("--batch", type=int, default=65536)
    ap.add_argument("--features", type=int, default=1 << 24)
    ap.add_argument("--k", type=int, default=16)
    ap.add_argument("--optimizer", default="ftrl")
    ap.add_argument("--eval-rows", type=int, default=262144)
    ap.add_argument("--dtype", default="fp32", choices=["fp32", "bf16"],
                    help="FFM only: bf16 compute mirror (config #3)")
    args = ap.parse_args()

    from lightctr_amd.data.synthetic import SyntheticCriteo
    from lightctr_amd.utils.metrics import auc_score

    dev = "cuda:0" if torch.cuda.is_available() else "cpu"
    fieldaware = args.model == "ffm"
    if args.model == "fm":
        from lightctr_amd.models.fm import FMHyper, FMModel

        model = FMModel(FMHyper(num_features=args.features, k=args.k,
                                optimizer=args.optimizer), device=dev)
    elif args.model == "ffm":
        from lightctr_amd.models.ffm import FFMHyper, FFMModel

        model = FFMModel(FFMHyper(num_features=args.features, num_fields=39,
                                  k=min(args.k, 8),
                                  optimizer=args.optimizer,
                                  dtype=args.dtype), device=dev)
    elif args.model == "nfm":
        from lightctr_amd.models.nfm import NFMHyper, NFMModel

        model = NFMModel(NFMHyper(num_features=args.features, k=args.k),
                         device=dev)
    else:
        from lightctr_amd.models.wide_deep import (WideDeepHyper,
                                                   WideDeepModel)

        model = WideDeepModel(WideDeepHyper(num_features=args.features,
                                            num_fields=39, k=args.k),
                              device=dev)

    train_gen = SyntheticCriteo(num_features=args.features, seed=1, device=dev)
    # held-out eval stream: same planted teacher (same seed => same teacher
    # table) but a different batch RNG stream position
    if torch.cuda.is_available():
        torch.cuda.synchronize()
    t0 = time.perf_counter()
    first = last = None
    for step in range(args.steps):
        rp, fl, fi, v, lb = train_gen.batch(args.batch)
        loss = (model.train_step(rp, fl, fi, v, lb) if fieldaware
                else model.train_step(rp, fi, v, lb))
        if step == 0:
            first = float(loss.mean())
        if step == args.steps - 1:
            last = float(loss.mean())
    if torch.cuda.is_available():
        torch.cuda.synchronize()
    dt = time.perf_counter() - t0
    # held-out eval (unseen draws from the same distribution)
    preds, labs = [], []
    for _ in range(args.eval_rows // args.batch + 1):
        rp, fl, fi, v, lb = train_gen.batch(args.batch)
        p = (model.predict_proba(rp, fl, fi, v) if fieldaware
             else model.predict_proba(rp, fi, v))
        preds.append(p)
        labs.append(lb)
    pred = torch.cat(preds)
    lab = torch.cat(labs)
    if pred.is_cuda:
        # in-tree atomic-histogram + device-scan AUC kernel
        from lightctr_amd.utils.metrics import HistAUC

        h = HistAUC(buckets=1 << 22, device=str(pred.device))
        h.add(pred, lab)
        auc = h.compute()
    else:
        auc = auc_score(pred.cpu(), lab.cpu())
    logloss = float(torch.nn.functional.binary_cross_entropy(
        pred.clamp(1e-7, 1 - 1e-7), lab))
    print(f"model={args.model} steps={args.steps} batch={args.batch} "
          f"examples={args.steps * args.batch}")
    print(f"throughput={args.steps * args.batch / dt:.0f} ex/s  "
          f"loss first={first:.4f} last={last:.4f}")
    print(f"held-out AUC={auc:.4f} logloss={logloss:.4f}")


if __name__ == "__main__":
    main()
