"""Isolated loop over ONE stage of the FM sorted pipeline for single-
counter rocprofv3 --pmc runs. --stage apply|optimizer|emit."""
import argparse
import sys
import time
from pathlib import Path

sys.path.insert(0, str(Path(__file__).resolve().parent.parent))

import torch

from lightctr_amd.data.synthetic import SyntheticCriteo
from lightctr_amd.models.fm import FMHyper, FMModel
from lightctr_amd.ops._extension import require_hip_ops, sort_ids


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--reps", type=int, default=30)
    ap.add_argument("--stage", default="apply",
                    choices=["apply", "optimizer", "emit"])
    ap.add_argument("--chunk", type=int, default=0)
    args = ap.parse_args()

    ops = require_hip_ops()
    h = FMHyper(num_features=1 << 24, k=16, optimizer="ftrl")
    m = FMModel(h, device="cuda")
    gen = SyntheticCriteo(num_features=h.num_features, seed=5, device="cuda")
    row_ptr, _f, fids, vals, labels = gen.batch(65536)
    B = 65536
    pred, sumVX = ops.fm_forward(row_ptr, fids, vals, m.W, m.V)
    _, dpred = ops.logloss_grad(pred, labels, 1.0 / B)
    sorted_fids, perm = sort_ids(fids, h.num_features)
    gw, gv = ops.fm_backward_emit(row_ptr, fids, vals, m.V, sumVX, dpred)

    if args.stage == "apply":
        fn = lambda: ops.fm_sorted_apply(sorted_fids, perm, gw, gv, m.gradW,
                                         m.gradV, m.touched, args.chunk)
    elif args.stage == "emit":
        fn = lambda: ops.fm_backward_emit(row_ptr, fids, vals, m.V, sumVX,
                                          dpred)
    else:
        ops.fm_sorted_apply(sorted_fids, perm, gw, gv, m.gradW, m.gradV,
                            m.touched, 0)
        m.count.zero_()
        ops.bitmap_compact(m.touched, m.uniq, m.count)
        live = m.uniq[: min(m.uniq.numel(), fids.numel())]

        def fn():
            ops.fm_ftrl_apply(live, m.count, m.W, m.V, m.zW, m.nW, m.zV,
                              m.nV, m.gradW, m.gradV, h.ftrl_alpha,
                              h.ftrl_beta, h.ftrl_l1, h.ftrl_l2, 1, h.lr,
                              h.eps, h.l2)
    for _ in range(3):
        fn()
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(args.reps):
        fn()
    torch.cuda.synchronize()
    dt = (time.perf_counter() - t0) / args.reps
    print(f"{args.stage}: {dt * 1e6:.1f} us/call")


if __name__ == "__main__":
    main()
