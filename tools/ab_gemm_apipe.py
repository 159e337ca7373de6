"""Within-process A/B of the gemm256 phase-pipelined A-read schedule
(LCTR_GEMM_APIPE=0/1/2 — the launcher re-reads the env per launch, so
one process can interleave variants: two passes each to bound drift).

Also checks numerics of each variant against torch bf16 matmul.
"""

import os
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

    shapes = [(4096, 4096, 4096), (8192, 8192, 8192)]
    data = {}
    for M, N, K in shapes:
        g = torch.Generator().manual_seed(0)
        A = torch.randn(M, K, generator=g).to(torch.bfloat16).cuda()
        Bst = torch.randn(N, K, generator=g).to(torch.bfloat16).cuda()
        data[(M, N, K)] = (A, Bst)

    # numerics first: each variant vs torch
    for v in ("0", "1", "2"):
        os.environ["LCTR_GEMM_APIPE"] = v
        for (M, N, K), (A, Bst) in data.items():
            C = hip_ops.gemm_bf16(A, Bst, None, M, N, K, 0, 0, 0, False)
            ref = (A @ Bst.t()).float()
            md = (C - ref).abs().max().item()
            rel = md / ref.abs().max().item()
            flag = "OK" if rel < 2e-2 else "FAIL"
            print(f"numerics APIPE={v} {M}x{N}x{K}: maxdiff={md:.4f} "
                  f"rel={rel:.2e} {flag}")

    # two interleaved timing passes
    for rep in range(2):
        for v in ("0", "1", "2"):
            os.environ["LCTR_GEMM_APIPE"] = v
            for (M, N, K), (A, Bst) in data.items():
                t = bench(lambda: hip_ops.gemm_bf16(A, Bst, None, M, N, K,
                                                    0, 0, 0, False))
                tf = 2.0 * M * N * K / t / 1e12
                print(f"pass{rep} APIPE={v} {M}x{N}x{K}: "
                      f"{t*1e3:7.3f} ms {tf:7.1f} TF")


if __name__ == "__main__":
    main()
