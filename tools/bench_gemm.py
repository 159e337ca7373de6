"""Micro-benchmark of the hand-written MFMA bf16 GEMM vs torch (hipBLASLt).

Usage (GPU box): python tools/bench_gemm.py
"""

import sys
import time
from pathlib import Path

sys.path.insert(0, str(Path(__file__).resolve().parent.parent))

import torch  # noqa: E402


def bench(fn, iters=20, warmup=5):
    for _ in range(warmup):
        fn()
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(iters):
        fn()
    torch.cuda.synchronize()
    return (time.perf_counter() - t0) / iters


def main():
    from lightctr_amd.ops import hip_ops

    shapes = [(4096, 4096, 4096), (8192, 8192, 8192),
              (65536, 256, 624), (65536, 1, 256)]  # MLP shapes incl. W&D l1
    print(f"{'M':>7} {'N':>6} {'K':>6} {'ours_ms':>9} {'ours_TF':>8} "
          f"{'torch_ms':>9} {'torch_TF':>9}")
    for M, N, K in shapes:
        g = torch.Generator().manual_seed(0)
        A = (torch.randn(M, K, generator=g)).to(torch.bfloat16).cuda()
        Bst = (torch.randn(N, K, generator=g)).to(torch.bfloat16).cuda()
        flops = 2.0 * M * N * K

        t_ours = bench(lambda: hip_ops.gemm_bf16(A, Bst, None, M, N, K, 0,
                                                 0, 0, False))
        t_torch = bench(lambda: A @ Bst.t())
        print(f"{M:>7} {N:>6} {K:>6} {t_ours*1e3:9.3f} "
              f"{flops/t_ours/1e12:8.1f} {t_torch*1e3:9.3f} "
              f"{flops/t_torch/1e12:9.1f}")


if __name__ == "__main__":
    main()
