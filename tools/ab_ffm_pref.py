"""Within-process alternating A/B of the FFM fused-apply optimizer-state
prefetch (LCTR_FFM_PREF 0/1, re-read per launch) on the isolated apply
loop + an equivalence check of the two paths on a fresh model pair.
"""
import os
import sys
import time
from pathlib import Path

sys.path.insert(0, str(Path(__file__).resolve().parent.parent))

import torch  # noqa: E402

from lightctr_amd.data.synthetic import SyntheticCriteo  # noqa: E402
from lightctr_amd.models.ffm import FFMHyper, FFMModel  # noqa: E402
from lightctr_amd.ops._extension import require_hip_ops, sort_ids  # noqa


def main():
    ops = require_hip_ops()
    F, nf, K, B = 1 << 24, 39, 8, 65536
    gen = SyntheticCriteo(num_features=F, seed=5, device="cuda")
    batch = gen.batch(B)
    row_ptr, fields, fids, vals, labels = batch

    # equivalence: two fresh models (small F for fast init), 3 steps
    Fs = 1 << 20
    gen_s = SyntheticCriteo(num_features=Fs, seed=7, device="cuda")
    batch_s = gen_s.batch(16384)
    states = {}
    for v in ("1", "0"):
        os.environ["LCTR_FFM_PREF"] = v
        m = FFMModel(FFMHyper(num_features=Fs, num_fields=nf, k=K,
                              optimizer="ftrl", dtype="bf16"),
                     device="cuda")
        for _ in range(3):
            m.train_step(*batch_s)
        torch.cuda.synchronize()
        states[v] = (m.W.clone(), m.V.clone())
    dW = (states["1"][0] - states["0"][0]).abs().max().item()
    dV = (states["1"][1] - states["0"][1]).abs().max().item()
    print(f"equivalence after 3 steps: dW={dW:.2e} dV={dV:.2e} "
          f"{'OK' if dW < 1e-5 and dV < 1e-5 else 'FAIL'}")

    # isolated apply A/B
    m = FFMModel(FFMHyper(num_features=F, num_fields=nf, k=K,
                          optimizer="ftrl", dtype="bf16"), device="cuda")
    pred = ops.ffm_forward(row_ptr, fields, fids, vals, m.W, m._Vc)
    _, dpred = ops.logloss_grad(pred, labels, 1.0 / B)
    gw, gblocks = ops.ffm_row_emit(row_ptr, fields, fids, vals, m._Vc,
                                   dpred, scale=float(B))
    sorted_fids, perm = sort_ids(fids, F)
    gv = m.gradV.view(F, -1)
    Vv = m.V.view(F, -1)
    nVv = m.nV.view(F, -1)
    Vhv = m.Vh.view(F, -1)

    def fn():
        ops.ffm_blocks_apply_f16(sorted_fids, perm, gblocks, gw, m.gradW,
                                 gv, m.touched, inv_scale=1.0 / B,
                                 opt_mode=3, V=Vv, W=m.W, nW=m.nW,
                                 zW=m.zW, nV=nVv, Vh=Vhv, p0=0.15, p1=1.0,
                                 p2=1e-4, p3=1e-4, q0=0.1, q1=1e-8,
                                 q2=1e-5)

    def bench(reps=25, warmup=3):
        for _ in range(warmup):
            fn()
        torch.cuda.synchronize()
        t0 = time.perf_counter()
        for _ in range(reps):
            fn()
        torch.cuda.synchronize()
        return (time.perf_counter() - t0) / reps

    bench(40, 10)  # ramp
    for env, vals in (("LCTR_FFM_WPE", ("0", "8")),):
        os.environ["LCTR_FFM_PREF"] = "0"
        res = {}
        for p in range(8):
            for v in vals:
                os.environ[env] = v
                t = bench()
                res.setdefault(v, []).append(t)
                print(f"p{p} {env}={v}: {t*1e6:8.1f} us")
        os.environ.pop(env, None)
        for v, ts in sorted(res.items()):
            ts = sorted(ts)[1:-1]
            print(f"TRIMMED {env}={v}: {sum(ts)/len(ts)*1e6:.1f} us")


if __name__ == "__main__":
    main()
