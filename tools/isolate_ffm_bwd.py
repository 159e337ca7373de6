"""Run ONLY the FFM sorted-backward kernel in a loop (plus its one-time
setup) so PMC counter runs profile a single kernel with minimal tool
surface — the full-bench rocprofv3 --pmc run crashed the profiler in
round 1 (gpurun_out/pmc.log).
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


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--reps", type=int, default=20)
    ap.add_argument("--batch", type=int, default=65536)
    args = ap.parse_args()

    ops = require_hip_ops()
    F, nf, K = 1 << 24, 39, 8
    m = FFMModel(FFMHyper(num_features=F, num_fields=nf, k=K),
                 device="cuda")
    gen = SyntheticCriteo(num_features=F, seed=5, device="cuda")
    row_ptr, fields, fids, vals, labels = gen.batch(args.batch)
    pred = ops.ffm_forward(row_ptr, fields, fids, vals, m.W, m.V)
    _, dpred = ops.logloss_grad(pred, labels, 1.0 / args.batch)
    sorted_fids, perm = sort_ids(fids, F)
    row_of_entry = ops.row_index(row_ptr, fids.numel())

    def bwd():
        ops.ffm_sorted_backward(sorted_fids, perm, row_of_entry, row_ptr,
                                fields, fids, vals, m.V, dpred, m.gradW,
                                m.gradV, m.touched)

    for _ in range(3):
        bwd()
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(args.reps):
        bwd()
    torch.cuda.synchronize()
    dt = (time.perf_counter() - t0) / args.reps
    print(f"ffm_sorted_backward: {dt * 1e3:.2f} ms/call")


if __name__ == "__main__":
    main()
