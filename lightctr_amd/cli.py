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
    elif args.model == "lr":
        from .models.lr import LRHyper, LRTrainer

        tr = LRTrainer(ds, LRHyper(num_features=F, optimizer=args.optimizer,
                                   lr=args.lr), device=dev,
                       batch_size=args.batch, epochs=args.epochs)
    else:
        return _train_other(args, dev)
    guard = None
    if args.watchdog:
        from .utils.watchdog import TrainGuard

        guard = TrainGuard(tr.model, ckpt_path=args.ckpt or None,
                           soft_s=args.watchdog_soft_s,
                           dead_s=args.watchdog_dead_s,
                           ckpt_every=args.ckpt_every,
                           log=lambda m: print(m, flush=True)).start()
        if args.resume:
            guard.maybe_resume()
        # one heartbeat per train_step, whichever trainer loop runs it
        inner = tr.model.train_step

        def guarded_step(*a, **kw):
            return guard.run_step(inner, *a, **kw)

        tr.model.train_step = guarded_step
    try:
        tr.train(log=lambda msg: print(msg, flush=True))
    finally:
        if guard is not None:
            tr.model.train_step = inner
            guard.stop()
    metrics = tr.evaluate()
    print(json.dumps({"final": metrics}))
    if args.save:
        tr.model.save(args.save)
        print(f"saved -> {args.save}")
    return 0


def _dense_data(args):
    """(X, y) from a dense CSV or a synthetic quadrant-image set."""
    import torch as t

    if args.data == "synthetic":
        g = t.Generator().manual_seed(args.seed)
        size = 16
        X = t.rand(args.rows, size * size, generator=g) * 0.2
        y = t.randint(0, 4, (args.rows,), generator=g)
        hs = size // 2
        img = X.view(args.rows, size, size)
        for i in range(args.rows):
            q = int(y[i])
            img[i, (q // 2) * hs:(q // 2 + 1) * hs,
                (q % 2) * hs:(q % 2 + 1) * hs] += 0.8
        return X, y
    from .data import load_dense_csv

    return load_dense_csv(args.data, max_rows=args.rows or None)


def _train_other(args, dev) -> int:
    """The rest of the zoo (reference main.cpp TEST_* coverage): gbm, cnn,
    rnn, vae, gmm, embed, plsa."""
    import math

    import torch as t

    if args.model == "gbm":
        from .models.gbm import GBMHyper, GBMModel
        from .utils.metrics import auc_score

        X, y = _dense_data(args)
        ncls = int(y.max()) + 1
        m = GBMModel(GBMHyper(n_rounds=args.epochs * 5, max_depth=5,
                              n_classes=max(2, ncls), seed=args.seed),
                     device=dev)
        m.fit(X.to(dev), y.float().to(dev),
              log=lambda msg: print(msg, flush=True))
        p = m.predict_proba(X.to(dev))
        if p.dim() == 1:
            print(json.dumps({"final": {"auc": auc_score(p.cpu(),
                                                         y.float())}}))
        else:
            acc = float((p.argmax(dim=1).cpu() == y).float().mean())
            print(json.dumps({"final": {"accuracy": acc}}))
        if args.save:
            m.save(args.save)
        return 0
    if args.model == "cnn":
        from .models.cnn import CNNHyper, CNNModel

        X, y = _dense_data(args)
        side = int(math.isqrt(X.shape[1]))
        ncls = int(y.max()) + 1
        m = CNNModel(CNNHyper(in_shape=(1, side, side), n_classes=ncls,
                              seed=args.seed), device=dev)
        Xi = X.view(-1, 1, side, side).to(dev)
        yi = y.to(dev)
        for ep in range(args.epochs):
            perm = t.randperm(Xi.shape[0])
            tot, nb = 0.0, 0
            for s0 in range(0, Xi.shape[0], args.batch):
                idx = perm[s0:s0 + args.batch]
                tot += m.train_step(Xi[idx], yi[idx])
                nb += 1
            print(f"epoch {ep}: loss={tot / max(nb, 1):.5f}", flush=True)
        acc = float((m.predict_proba(Xi).argmax(dim=1) == yi)
                    .float().mean())
        print(json.dumps({"final": {"accuracy": acc}}))
        if args.save:
            m.save(args.save)
        return 0
    if args.model == "rnn":
        from .models.rnn import RNNHyper, RNNModel

        X, y = _dense_data(args)
        side = int(math.isqrt(X.shape[1]))
        ncls = int(y.max()) + 1
        m = RNNModel(RNNHyper(in_dim=side, seq_len=side, n_classes=ncls,
                              seed=args.seed), device=dev)
        Xi = X.view(-1, side, side).to(dev)
        yi = y.to(dev)
        for ep in range(args.epochs):
            perm = t.randperm(Xi.shape[0])
            tot, nb = 0.0, 0
            for s0 in range(0, Xi.shape[0], args.batch):
                idx = perm[s0:s0 + args.batch]
                tot += m.train_step(Xi[idx], yi[idx])
                nb += 1
            print(f"epoch {ep}: loss={tot / max(nb, 1):.5f}", flush=True)
        acc = float((m.predict_proba(Xi).argmax(dim=1) == yi)
                    .float().mean())
        print(json.dumps({"final": {"accuracy": acc}}))
        if args.save:
            m.save(args.save)
        return 0
    if args.model == "vae":
        from .models.vae import VAEHyper, VAEModel

        X, _ = _dense_data(args)
        Xi = X.to(dev)
        m = VAEModel(VAEHyper(in_dim=X.shape[1], seed=args.seed),
                     device=dev)
        for ep in range(args.epochs):
            stats = {}
            for s0 in range(0, Xi.shape[0], args.batch):
                stats = m.train_step(Xi[s0:s0 + args.batch])
            print(f"epoch {ep}: {stats}", flush=True)
        print(json.dumps({"final": stats}))
        if args.save:
            m.save(args.save)
        return 0
    if args.model == "gmm":
        from .models.gmm import GMMHyper, GMMModel

        X, _ = _dense_data(args)
        m = GMMModel(GMMHyper(n_components=max(2, args.k), seed=args.seed),
                     device=dev)
        m.fit(X.to(dev), log=lambda msg: print(msg, flush=True))
        print(json.dumps({"final": {"weights": m.weights.cpu().tolist()}}))
        if args.save:
            m.save(args.save)
        return 0
    if args.model == "embed":
        from .models.embedding import (EmbedHyper, EmbedModel,
                                       vocab_from_tokens)

        if args.data == "synthetic":
            import torch as t2

            g = t2.Generator().manual_seed(args.seed)
            toks = []
            for _ in range(max(200, args.rows)):
                grp = "a" if t2.rand(1, generator=g) < 0.5 else "b"
                toks.extend(f"{grp}{int(i)}"
                            for i in t2.randperm(5, generator=g))
        else:
            toks = open(args.data).read().split()
        vocab, counts = vocab_from_tokens(toks)
        m = EmbedModel(vocab, counts, EmbedHyper(dim=min(args.k * 4, 64),
                                                 seed=args.seed),
                       device=dev)
        ids = t.tensor([m.word2id[tok] for tok in toks])
        m.train_stream(ids, epochs=args.epochs,
                       log=lambda msg: print(msg, flush=True))
        m.normalize()
        print(json.dumps({"final": {"vocab": len(vocab),
                                    "top": m.most_similar(vocab[0], 3)}}))
        if args.save:
            m.save(args.save)
        return 0
    if args.model == "plsa":
        from .models.plsa import PLSAHyper, PLSAModel

        if args.data == "synthetic":
            g = t.Generator().manual_seed(args.seed)
            docs, words, cnts = [], [], []
            for d in range(200):
                for _ in range(30):
                    w = int(t.randint(0, 10, (1,), generator=g)) \
                        + (d % 2) * 10
                    docs.append(d)
                    words.append(w)
                    cnts.append(1.0)
            n_docs, n_words = 200, 20
        else:
            vocab = {}
            docs, words, cnts = [], [], []
            lines = open(args.data).read().splitlines()
            for d, line in enumerate(lines):
                for tok in line.split():
                    wid = vocab.setdefault(tok, len(vocab))
                    docs.append(d)
                    words.append(wid)
                    cnts.append(1.0)
            n_docs, n_words = len(lines), len(vocab)
        m = PLSAModel(PLSAHyper(n_topics=max(2, args.k), seed=args.seed),
                      device=dev)
        m.fit(t.tensor(docs), t.tensor(words), t.tensor(cnts), n_docs,
              n_words, log=lambda msg: print(msg, flush=True))
        print(json.dumps({"final": {"topics": [m.top_words(z, 5)
                                               for z in
                                               range(m.h.n_topics)]}}))
        if args.save:
            m.save(args.save)
        return 0
    print(f"unknown model {args.model}", file=sys.stderr)
    return 2


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
    elif args.model == "nfm":
        from .models.nfm import NFMHyper, NFMModel

        model = NFMModel(NFMHyper(num_features=F, k=args.k,
                                  optimizer=args.optimizer), device=dev)
    elif args.model == "widedeep":
        from .models.wide_deep import WideDeepHyper, WideDeepModel

        model = WideDeepModel(WideDeepHyper(num_features=F,
                                            num_fields=ds.num_fields,
                                            k=args.k,
                                            optimizer=args.optimizer),
                              device=dev)
    elif args.model == "lr":
        from .models.lr import LRHyper, LRModel

        model = LRModel(LRHyper(num_features=F, optimizer=args.optimizer),
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
        # §5.3 failure detection: heartbeat watchdog + checkpoint-restart
        p.add_argument("--watchdog", action="store_true", default=True,
                       help="stall watchdog + checkpoint-abort (default on)")
        p.add_argument("--no-watchdog", dest="watchdog",
                       action="store_false")
        p.add_argument("--watchdog-soft-s", type=float, default=30.0)
        p.add_argument("--watchdog-dead-s", type=float, default=120.0)
        p.add_argument("--ckpt", default=None,
                       help="checkpoint path for the watchdog + periodic "
                            "saves")
        p.add_argument("--ckpt-every", type=int, default=0,
                       help="checkpoint every N steps (0 = only on death)")
        p.add_argument("--resume", action="store_true",
                       help="resume from --ckpt if it exists")
    return ap


def main(argv=None) -> int:
    args = build_parser().parse_args(argv)
    return args.fn(args)


if __name__ == "__main__":
    sys.exit(main())
