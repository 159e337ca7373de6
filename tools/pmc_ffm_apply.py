"""Isolated FFM fused-apply loop for single-counter PMC runs."""
import argparse
import sys
import time
from pathlib import Path

sys.path.insert(0, str(Path(__file__).resolve().parent.parent))

import torch

from lightctr_amd.data.synthetic import SyntheticCriteo
from lightctr_amd.models.ffm import FFMHyper, FFMModel
from lightctr_amd.ops._extension import require_hip_ops, sort_ids


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--reps", type=int, default=25)
    ap.add_argument("--fused", type=int, default=1)
    args = ap.parse_args()
    ops = require_hip_ops()
    F, nf, K = 1 << 24, 39, 8
    m = FFMModel(FFMHyper(num_features=F, num_fields=nf, k=K,
                          optimizer="ftrl", dtype="bf16"), device="cuda")
    gen = SyntheticCriteo(num_features=F, seed=5, device="cuda")
    row_ptr, fields, fids, vals, labels = gen.batch(65536)
    pred = ops.ffm_forward(row_ptr, fields, fids, vals, m.W, m._Vc)
    _, dpred = ops.logloss_grad(pred, labels, 1.0 / 65536)
    gw, gblocks = ops.ffm_row_emit(row_ptr, fields, fids, vals, m._Vc,
                                   dpred, scale=65536.0)
    sorted_fids, perm = sort_ids(fids, F)
    gv = m.gradV.view(F, -1)
    Vv = m.V.view(F, -1)
    nVv = m.nV.view(F, -1)
    Vhv = m.Vh.view(F, -1)

    def fn():
        if args.fused:
            ops.ffm_blocks_apply_f16(sorted_fids, perm, gblocks, gw,
                                     m.gradW, gv, m.touched,
                                     inv_scale=1.0 / 65536.0, opt_mode=3,
                                     V=Vv, W=m.W, nW=m.nW, zW=m.zW,
                                     nV=nVv, Vh=Vhv, p0=0.15, p1=1.0,
                                     p2=1e-4, p3=1e-4, q0=0.1, q1=1e-8,
                                     q2=1e-5)
        else:
            ops.ffm_blocks_apply_f16(sorted_fids, perm, gblocks, gw,
                                     m.gradW, gv, m.touched,
                                     inv_scale=1.0 / 65536.0)
    for _ in range(3):
        fn()
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(args.reps):
        fn()
    torch.cuda.synchronize()
    print(f"apply fused={args.fused}: "
          f"{(time.perf_counter() - t0) / args.reps * 1e6:.1f} us")


if __name__ == "__main__":
    main()
