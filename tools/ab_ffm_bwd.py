"""A/B the FFM backward variants (full train_step and isolated backward
pipeline per mode) on one GPU. Round-2 check of the vectorized row_emit
(docs/STATUS.md lever 1): float4+half2 staging, field-map direct quad
emit, dwordx2 block stores.
"""
import argparse
import sys
import time
from pathlib import Path

sys.path.insert(0, str(Path(__file__).resolve().parent.parent))

import torch

from lightctr_amd.data.synthetic import SyntheticCriteo
from lightctr_amd.models.ffm import FFMHyper, FFMModel
from lightctr_amd.ops._extension import require_hip_ops, sort_ids


def time_fn(fn, reps, warmup=3):
    for _ in range(warmup):
        fn()
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(reps):
        fn()
    torch.cuda.synchronize()
    return (time.perf_counter() - t0) / reps


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--reps", type=int, default=20)
    ap.add_argument("--batch", type=int, default=65536)
    ap.add_argument("--modes", default="sorted,rowemit,blocks")
    ap.add_argument("--dtype", default="fp32", choices=["fp32", "bf16"])
    args = ap.parse_args()

    ops = require_hip_ops()
    F, nf, K = 1 << 24, 39, 8
    m = FFMModel(FFMHyper(num_features=F, num_fields=nf, k=K,
                          dtype=args.dtype), device="cuda")
    gen = SyntheticCriteo(num_features=F, seed=5, device="cuda")
    row_ptr, fields, fids, vals, labels = gen.batch(args.batch)
    nnz = fids.numel()
    B = args.batch

    pred = ops.ffm_forward(row_ptr, fields, fids, vals, m.W, m._Vc)
    _, dpred = ops.logloss_grad(pred, labels, 1.0 / B)
    sorted_fids, perm = sort_ids(fids, F)
    row_of_entry = ops.row_index(row_ptr, nnz)

    # isolated phase timings
    t = time_fn(lambda: ops.ffm_forward(row_ptr, fields, fids, vals, m.W,
                                        m._Vc), args.reps)
    print(f"forward                : {t * 1e3:7.3f} ms")
    t = time_fn(lambda: sort_ids(fids, F), args.reps)
    print(f"sort_ids               : {t * 1e3:7.3f} ms")

    for mode in args.modes.split(","):
        if mode == "sorted":
            def bwd():
                ops.ffm_sorted_backward(sorted_fids, perm, row_of_entry,
                                        row_ptr, fields, fids, vals, m.V,
                                        dpred, m.gradW, m.gradV, m.touched)
            t = time_fn(bwd, args.reps)
            print(f"bwd[sorted]            : {t * 1e3:7.3f} ms")
        elif mode == "rowemit":
            def emit():
                return ops.ffm_row_emit(row_ptr, fields, fids, vals, m._Vc,
                                        dpred, scale=65536.0)
            te = time_fn(emit, args.reps)
            gw, gblocks = emit()
            gv = m.gradV.view(F, -1)

            def apply():
                ops.ffm_blocks_apply_f16(sorted_fids, perm, gblocks, gw,
                                         m.gradW, gv, m.touched,
                                         inv_scale=1.0 / 65536.0)
            ta = time_fn(apply, args.reps)
            print(f"bwd[rowemit] emit      : {te * 1e3:7.3f} ms")
            print(f"bwd[rowemit] apply     : {ta * 1e3:7.3f} ms")
            print(f"bwd[rowemit] total     : {(te + ta) * 1e3:7.3f} ms")
        elif mode == "blocks":
            def emit_b():
                return ops.ffm_block_emit(row_of_entry, row_ptr, fields,
                                          fids, vals, m.V, dpred)
            te = time_fn(emit_b, args.reps)
            gw, gblocks = emit_b()
            gv = m.gradV.view(F, -1)

            def apply_b():
                ops.ffm_blocks_apply(sorted_fids, perm, gblocks, gw, m.gradW,
                                     gv, m.touched)
            ta = time_fn(apply_b, args.reps)
            print(f"bwd[blocks] emit       : {te * 1e3:7.3f} ms")
            print(f"bwd[blocks] apply      : {ta * 1e3:7.3f} ms")
            print(f"bwd[blocks] total      : {(te + ta) * 1e3:7.3f} ms")

    # full train_step per mode
    for mode in args.modes.split(","):
        for fused in ((False, True) if mode == "rowemit" else (False,)):
            m2 = FFMModel(FFMHyper(num_features=F, num_fields=nf, k=K,
                                  optimizer="ftrl",
                                  dtype=args.dtype), device="cuda")
            m2.backward_mode = mode
            m2.fused_apply = fused
            t = time_fn(lambda: m2.train_step(row_ptr, fields, fids, vals,
                                              labels), args.reps)
            exs = B / t
            tag = mode + ("+fused" if fused else "")
            print(f"step[{tag:14s}]  : {t * 1e3:7.3f} ms  "
                  f"({exs / 1e6:.1f}M ex/s)")


if __name__ == "__main__":
    main()
