"""Micro-bench of the FM sorted-backward pipeline stages (GPU).

Times each stage and the {gather-apply, scatter-emit} x chunk-size matrix
to locate the real bound of fm_sorted_apply (hypothesis: hot-feature
atomic flushes, one per (subgroup, run), not the gv gathers — larger
chunks divide the flush count per hot fid).
"""
import sys
import time
from pathlib import Path

sys.path.insert(0, str(Path(__file__).resolve().parent.parent))

import torch

from lightctr_amd.data.synthetic import SyntheticCriteo
from lightctr_amd.models.fm import FMHyper, FMModel
from lightctr_amd.ops._extension import require_hip_ops, sort_ids


def t(fn, reps=20):
    for _ in range(3):
        fn()
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(reps):
        fn()
    torch.cuda.synchronize()
    return (time.perf_counter() - t0) / reps * 1e6  # us


def main():
    ops = require_hip_ops()
    h = FMHyper(num_features=1 << 24, k=16, optimizer="ftrl")
    m = FMModel(h, device="cuda")
    gen = SyntheticCriteo(num_features=h.num_features, seed=5,
                          device="cuda")
    row_ptr, _fields, fids, vals, labels = gen.batch(65536)
    nnz = fids.numel()
    B = row_ptr.numel() - 1

    pred, sumVX = ops.fm_forward(row_ptr, fids, vals, m.W, m.V)
    loss, dpred = ops.logloss_grad(pred, labels, 1.0 / B)
    print(f"nnz={nnz} uniq={torch.unique(fids).numel()}")

    print(f"fwd        {t(lambda: ops.fm_forward(row_ptr, fids, vals, m.W, m.V)):8.1f} us")
    print(f"sort       {t(lambda: sort_ids(fids, h.num_features)):8.1f} us")
    sorted_fids, perm = sort_ids(fids, h.num_features)
    print(f"inv_perm   {t(lambda: ops.inv_perm_i32(perm)):8.1f} us")
    pos = ops.inv_perm_i32(perm)
    print(f"emit seq   {t(lambda: ops.fm_backward_emit(row_ptr, fids, vals, m.V, sumVX, dpred)):8.1f} us")
    print(f"emit scat  {t(lambda: ops.fm_backward_emit(row_ptr, fids, vals, m.V, sumVX, dpred, pos)):8.1f} us")

    gw_s, gv_s = ops.fm_backward_emit(row_ptr, fids, vals, m.V, sumVX, dpred)
    gw_p, gv_p = ops.fm_backward_emit(row_ptr, fids, vals, m.V, sumVX, dpred,
                                      pos)
    for chunk in (-4, -3, -2, -1, 256, 384, 512):
        a = t(lambda: ops.fm_sorted_apply(sorted_fids, perm, gw_s, gv_s,
                                          m.gradW, m.gradV, m.touched,
                                          chunk))
        b = t(lambda: ops.fm_sorted_apply(sorted_fids, None, gw_p, gv_p,
                                          m.gradW, m.gradV, m.touched,
                                          chunk))
        tag = {-1: "segscan", -2: "segscan4", -3: "walk-1buf",
               -4: "walk-pp2"}.get(chunk, f"walk{chunk}")
        print(f"apply {tag:8s}  gather {a:8.1f} us   seq {b:8.1f} us")

    # correctness cross-check: gather vs scatter paths agree
    m.gradW.zero_(); m.gradV.zero_(); m.touched.zero_()
    ops.fm_sorted_apply(sorted_fids, perm, gw_s, gv_s, m.gradW, m.gradV,
                        m.touched, 256)
    gW1, gV1 = m.gradW.clone(), m.gradV.clone()
    m.gradW.zero_(); m.gradV.zero_(); m.touched.zero_()
    ops.fm_sorted_apply(sorted_fids, None, gw_p, gv_p, m.gradW, m.gradV,
                        m.touched, 512)
    dmax = (gV1 - m.gradV).abs().max().item()
    print(f"walk-gather vs walk-seq gradV maxdiff {dmax:.3e}")
    t1 = m.touched.clone()
    m.gradW.zero_(); m.gradV.zero_(); m.touched.zero_()
    ops.fm_sorted_apply(sorted_fids, perm, gw_s, gv_s, m.gradW, m.gradV,
                        m.touched, -1)
    dmax = (gV1 - m.gradV).abs().max().item()
    dw = (gW1 - m.gradW).abs().max().item()
    tok = bool(torch.equal(t1, m.touched))
    print(f"segscan vs walk gradV maxdiff {dmax:.3e} gradW {dw:.3e} touched_eq {tok}")
    m.gradW.zero_(); m.gradV.zero_(); m.touched.zero_()
    ops.fm_sorted_apply(sorted_fids, perm, gw_s, gv_s, m.gradW, m.gradV,
                        m.touched, -2)
    dmax = (gV1 - m.gradV).abs().max().item()
    dw = (gW1 - m.gradW).abs().max().item()
    tok = bool(torch.equal(t1, m.touched))
    print(f"segscan4 vs walk gradV maxdiff {dmax:.3e} gradW {dw:.3e} touched_eq {tok}")

    # full-step A/B: fused mode-3 apply vs two-phase (default ftrl pairing)
    for fused in (False, True):
        mm = FMModel(h, device="cuda")
        mm.fused_apply = fused
        tt = t(lambda: mm.train_step(row_ptr, fids, vals, labels))
        print(f"step fused={int(fused)}  {tt:8.1f} us  "
              f"({65536 / tt:.1f}M ex/s)")


if __name__ == "__main__":
    main()
