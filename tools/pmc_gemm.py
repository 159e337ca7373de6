"""Minimal single-kernel GEMM loop for rocprofv3 --pmc counter runs
(single counters only — large sets crash the tool on this pool).

Usage: [LCTR_GEMM_P8=0|1 LCTR_GEMM_P8_LITE=0|1] \
    rocprofv3 --pmc SQ_LDS_BANK_CONFLICT -d out -- \
    python tools/pmc_gemm.py --m 4096 --n 4096 --k 4096 --reps 20
"""
import argparse
import sys
import time
from pathlib import Path

sys.path.insert(0, str(Path(__file__).resolve().parent.parent))

import torch


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--m", type=int, default=4096)
    ap.add_argument("--n", type=int, default=4096)
    ap.add_argument("--k", type=int, default=4096)
    ap.add_argument("--reps", type=int, default=20)
    ap.add_argument("--transA", type=int, default=0)
    ap.add_argument("--transB", type=int, default=0)
    ap.add_argument("--wgrad-direct", action="store_true",
                    help="call gemm_wgrad_bf16 directly (bypass dispatch)")
    args = ap.parse_args()

    from lightctr_amd.ops import hip_ops

    M, N, K = args.m, args.n, args.k
    tA, tB = args.transA, args.transB
    g = torch.Generator().manual_seed(0)
    A = (torch.randn(K, M) if tA else torch.randn(M, K)) \
        .to(torch.bfloat16).cuda()
    Bst = (torch.randn(K, N) if tB else torch.randn(N, K)) \
        .to(torch.bfloat16).cuda()
    if args.wgrad_direct:
        fn = lambda: hip_ops.gemm_wgrad_bf16(A, Bst, M, N, K)
    else:
        fn = lambda: hip_ops.gemm_bf16(A, Bst, None, M, N, K, tA, tB, 0,
                                       False)
    for _ in range(3):
        fn()
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(args.reps):
        fn()
    torch.cuda.synchronize()
    dt = (time.perf_counter() - t0) / args.reps
    print(f"{M}x{N}x{K} tA={tA} tB={tB}: {dt * 1e3:.3f} ms  "
          f"{2.0 * M * N * K / dt / 1e12:.1f} TF")


if __name__ == "__main__":
    main()
