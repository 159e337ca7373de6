"""Within-process alternating A/B of the FM walk's waves-per-EU cap
(LCTR_FM_APPLY_WPE 0/6, launcher reads env per call): isolated
fm_sorted_apply loop at the flagship config + a 3-step model
equivalence check."""
import os
import sys
import time
from pathlib import Path

sys.path.insert(0, str(Path(__file__).resolve().parent.parent))

import torch  # noqa: E402

from lightctr_amd.data.synthetic import SyntheticCriteo  # noqa: E402
from lightctr_amd.models.fm import FMHyper, FMModel  # noqa: E402
from lightctr_amd.ops._extension import require_hip_ops, sort_ids  # noqa


def main():
    ops = require_hip_ops()

    # equivalence at small F
    Fs = 1 << 20
    gen_s = SyntheticCriteo(num_features=Fs, seed=7, device="cuda")
    rp_s, _fs, fid_s, val_s, lab_s = gen_s.batch(16384)
    states = {}
    for v in ("6", "0"):
        os.environ["LCTR_FM_APPLY_WPE"] = v
        m = FMModel(FMHyper(num_features=Fs, k=16, optimizer="ftrl"),
                    device="cuda")
        for _ in range(3):
            m.train_step(rp_s, fid_s, val_s, lab_s)
        torch.cuda.synchronize()
        states[v] = (m.W.clone(), m.V.clone())
    dW = (states["6"][0] - states["0"][0]).abs().max().item()
    dV = (states["6"][1] - states["0"][1]).abs().max().item()
    print(f"equivalence: dW={dW:.2e} dV={dV:.2e} "
          f"{'OK' if dW < 1e-5 and dV < 1e-5 else 'FAIL'}")

    h = FMHyper(num_features=1 << 24, k=16, optimizer="ftrl")
    m = FMModel(h, device="cuda")
    gen = SyntheticCriteo(num_features=h.num_features, seed=5,
                          device="cuda")
    row_ptr, _f, fids, vals, labels = gen.batch(65536)
    pred, sumVX = ops.fm_forward(row_ptr, fids, vals, m.W, m.V)
    _, dpred = ops.logloss_grad(pred, labels, 1.0 / 65536)
    sorted_fids, perm = sort_ids(fids, h.num_features)
    gw, gv = ops.fm_backward_emit(row_ptr, fids, vals, m.V, sumVX, dpred)

    def fn():
        ops.fm_sorted_apply(sorted_fids, perm, gw, gv, m.gradW, m.gradV,
                            m.touched, 0)

    def bench(reps=40, warmup=5):
        for _ in range(warmup):
            fn()
        torch.cuda.synchronize()
        t0 = time.perf_counter()
        for _ in range(reps):
            fn()
        torch.cuda.synchronize()
        return (time.perf_counter() - t0) / reps

    bench(60, 10)  # ramp
    res = {}
    for p in range(8):
        for v in ("0", "6"):
            os.environ["LCTR_FM_APPLY_WPE"] = v
            t = bench()
            res.setdefault(v, []).append(t)
            print(f"p{p} WPE={v}: {t*1e6:8.1f} us")
    for v, ts in sorted(res.items()):
        ts = sorted(ts)[1:-1]
        print(f"TRIMMED WPE={v}: {sum(ts)/len(ts)*1e6:.1f} us")


if __name__ == "__main__":
    main()
