"""CLI entrypoint — runtime model/role selection.

Replaces the reference's compile-time role/test selection
(/root/reference/main.cpp `-DTEST_*` / `-DMASTER/-DPS/-DWORKER` ladder +
Makefile per-role binaries + LightCTR_* env vars) with one
command:

    python -m lightctr_amd.cli train --model fm --data path.csv --epochs 5
    python -m lightctr_amd.cli train --model ffm|nfm|widedeep|gbm ...
    python -m lightctr_amd.cli predict --model fm --load m.pt --data t.csv

Distributed modes are launched via torch.distributed.run (one rank per
GPU), selecting --mode ring|sharded|ps at runtime.
"""

from __future__ import annotations

import argparse
import json
import sys

import torch


def _device(args) -> str:
    if args.device:
        return args.device
    return "cuda:0" if torch.cuda.is_available() else "cpu"


def _load_data(args):
    from .data import load_libffm
    from .data.synthetic import SyntheticCriteo
    from .data import LibffmDataset

    if args.data == "synthetic":
        gen = SyntheticCriteo(num_features=args.features, seed=args.seed,
                              device="cpu")
        row_ptr, fields, fids, vals, labels = gen.batch(args.rows)
        return LibffmDataset(row_ptr, fields, fids, vals, labels)
    return load_libffm(args.data)


def cmd_train(args) -> int:
    dev = _device(args)
    ds = _load_data(args)
    F = max(args.features, ds.num_features)
    if args.model == "fm":
        from .models.fm import FMHyper, FMTrainer

        tr = FMTrainer(ds, FMHyper(num_features=F, k=args.k,
                                   optimizer=args.optimizer, lr=args.lr),
                       device=dev, batch_size=args.batch,
                       epochs=args.epochs)
    elif args.model == "ffm":
        from .models.ffm import FFMHyper, FFMTrainer

        tr = FFMTrainer(ds, FFMHyper(num_features=F,
                                     num_fields=ds.num_fields, k=args.k,
                                     optimizer=args.optimizer, lr=args.lr),
                        device=dev, batch_size=args.batch,
                        epochs=args.epochs)
    elif args.model == "nfm":
        from .models.nfm import NFMHyper, NFMTrainer

        tr = NFMTrainer(ds, NFMHyper(num_features=F, k=args.k,
                                     optimizer=args.optimizer, lr=args.lr),
                        device=dev, batch_size=args.batch,
                        epochs=args.epochs)
    elif args.model == "widedeep":
        from .models.wide_deep import WideDeepHyper, WideDeepTrainer

        tr = WideDeepTrainer(ds, WideDeepHyper(num_features=F,
                                               num_fields=ds.num_fields,
                                               k=args.k,
                                               optimizer=args.optimizer,
                                               lr=args.lr),
                             device=dev, batch_size=args.batch,
                             epochs=args.epochs)
    else:
        print(f"unknown model {args.model}", file=sys.stderr)
        return 2
    tr.train(log=lambda msg: print(msg, flush=True))
    metrics = tr.evaluate()
    print(json.dumps({"final": metrics}))
    if args.save:
        tr.model.save(args.save)
        print(f"saved -> {args.save}")
    return 0


def cmd_predict(args) -> int:
    dev = _device(args)
    ds = _load_data(args).to(torch.device(dev))
    F = max(args.features, ds.num_features)
    if args.model == "fm":
        from .models.fm import FMHyper, FMModel

        model = FMModel(FMHyper(num_features=F, k=args.k,
                                optimizer=args.optimizer), device=dev)
    elif args.model == "ffm":
        from .models.ffm import FFMHyper, FFMModel

        model = FFMModel(FFMHyper(num_features=F,
                                  num_fields=ds.num_fields, k=args.k),
                         device=dev)
    else:
        print(f"unknown model {args.model}", file=sys.stderr)
        return 2
    if args.load:
        model.load(args.load)
    from .predict.predictor import BatchPredictor

    bp = BatchPredictor(model, batch_size=args.batch)
    pred = bp.predict_csr(ds)
    rep = bp.report(pred, ds.labels, dump_path=args.dump)
    print(json.dumps(rep))
    return 0


def build_parser():
    ap = argparse.ArgumentParser(prog="lightctr_amd")
    sub = ap.add_subparsers(dest="cmd", required=True)
    for name, fn in (("train", cmd_train), ("predict", cmd_predict)):
        p = sub.add_parser(name)
        p.set_defaults(fn=fn)
        p.add_argument("--model", default="fm")
        p.add_argument("--data", default="synthetic",
                       help="libffm file path or 'synthetic'")
        p.add_argument("--rows", type=int, default=4096,
                       help="rows for synthetic data")
        p.add_argument("--features", type=int, default=1 << 16)
        p.add_argument("--k", type=int, default=16)
        p.add_argument("--epochs", type=int, default=5)
        p.add_argument("--batch", type=int, default=256)
        p.add_argument("--lr", type=float, default=0.1)
        p.add_argument("--optimizer", default="adagrad")
        p.add_argument("--device", default=None)
        p.add_argument("--seed", type=int, default=1234)
        p.add_argument("--save", default=None)
        p.add_argument("--load", default=None)
        p.add_argument("--dump", default=None, help="score dump path")
    return ap


def main(argv=None) -> int:
    args = build_parser().parse_args(argv)
    return args.fn(args)


if __name__ == "__main__":
    sys.exit(main())
