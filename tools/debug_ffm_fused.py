"""One-step fused-vs-two-phase FFM apply diff, classified by whether the
fid's sorted run crosses a chunk boundary (chunk = 64 after the sweep)."""
import os
import sys
from pathlib import Path

sys.path.insert(0, str(Path(__file__).resolve().parent.parent))

import torch

from lightctr_amd.data.synthetic import SyntheticCriteo
from lightctr_amd.models.ffm import FFMHyper, FFMModel

CHUNK = int(os.environ.get("LCTR_FFM_APPLY_CHUNK", "64"))


def main():
    h = FFMHyper(num_features=1 << 14, num_fields=39, k=8,
                 optimizer="adagrad", seed=21)
    gen = SyntheticCriteo(num_features=1 << 14, seed=9, device="cuda:0")
    a = FFMModel(h, device="cuda:0")
    a.fused_apply = True
    b = FFMModel(h, device="cuda:0")
    b.fused_apply = False
    for step in range(3):
        row_ptr, fields, fids, vals, labels = gen.batch(2048)
        a.train_step(row_ptr, fields, fids, vals, labels)
        b.train_step(row_ptr, fields, fids, vals, labels)
        print(f"step {step}: dV={float((a.V - b.V).abs().max()):.3e} "
              f"dW={float((a.W - b.W).abs().max()):.3e} "
              f"dnV={float((a.nV - b.nV).abs().max()):.3e} "
              f"dnW={float((a.nW - b.nW).abs().max()):.3e} "
              f"dgV={float((a.gradV - b.gradV).abs().max()):.3e} "
              f"a.gV={float(a.gradV.abs().max()):.3e} "
              f"b.gV={float(b.gradV.abs().max()):.3e} "
              f"a.tc={int(a.touched.ne(0).sum())} "
              f"b.tc={int(b.touched.ne(0).sum())}")

    dW = (a.W - b.W).abs()
    dV = (a.V - b.V).abs().amax(dim=(1, 2))
    badW = (dW > 1e-6).nonzero().flatten()
    badV = (dV > 1e-6).nonzero().flatten()
    print(f"badW fids: {badW.numel()}  badV fids: {badV.numel()}")

    sorted_fids, _ = torch.sort(fids)
    sf = sorted_fids.cpu()
    # run spans
    spans = {}
    start = 0
    for i in range(1, sf.numel() + 1):
        if i == sf.numel() or int(sf[i]) != int(sf[i - 1]):
            spans[int(sf[i - 1])] = (start, i - 1)
            start = i
    n_interior = n_boundary = 0
    for f in badV[:2000].tolist():
        s, t = spans.get(f, (-1, -2))
        interior = (s // CHUNK) == (t // CHUNK) and s >= 0
        if interior:
            n_interior += 1
        else:
            n_boundary += 1
    print(f"badV classified: interior={n_interior} boundary={n_boundary}")
    # sample details
    for f in badV[:5].tolist():
        s, t = spans.get(f, (-1, -2))
        print(f"  fid {f}: run [{s},{t}] chunk {s // CHUNK}..{t // CHUNK} "
              f"len {t - s + 1} dV={float(dV[f]):.2e} dW={float(dW[f]):.2e}"
              f" aV0={float(a.V[f].flatten()[0]):.6f}"
              f" bV0={float(b.V[f].flatten()[0]):.6f}")
    # untouched fids must be identical
    touched = set(spans.keys())
    untouched_bad = [f for f in badV.tolist() if f not in touched]
    print(f"untouched-but-different: {len(untouched_bad)}")


if __name__ == "__main__":
    main()
