"""Within-process alternating A/B of the phase-pipelined A-read GEMM
schedules (launchers re-read env per launch):
  part 1: gemm256  LCTR_GEMM_APIPE   0 vs 2  (4096^3, 8192^3)
  part 2: gemm256p8 LCTR_GEMM_P8_APIPE 0 vs 1 (W&D tail shape + 4096^3,
          forced via LCTR_GEMM_P8=1)
Numerics of every variant are checked against torch bf16 matmul first.
"""
import os
import sys
import time
from pathlib import Path

sys.path.insert(0, str(Path(__file__).resolve().parent.parent))

import torch  # noqa: E402


def bench(fn, iters=30, warmup=3):
    for _ in range(warmup):
        fn()
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(iters):
        fn()
    torch.cuda.synchronize()
    return (time.perf_counter() - t0) / iters


def sweep(tag, env, vals, shapes, hip_ops):
    data = {}
    for M, N, K in shapes:
        g = torch.Generator().manual_seed(0)
        data[(M, N, K)] = (
            torch.randn(M, K, generator=g).to(torch.bfloat16).cuda(),
            torch.randn(N, K, generator=g).to(torch.bfloat16).cuda())
    for v in vals:
        os.environ[env] = v
        for (M, N, K), (A, B) in data.items():
            C = hip_ops.gemm_bf16(A, B, None, M, N, K, 0, 0, 0, False)
            ref = (A @ B.t()).float()
            rel = (C - ref).abs().max().item() / ref.abs().max().item()
            print(f"{tag} numerics {env}={v} {M}x{N}x{K}: rel={rel:.2e} "
                  f"{'OK' if rel < 2e-2 else 'FAIL'}")
    res = {}
    for (M, N, K), (A, B) in data.items():
        fn = lambda: hip_ops.gemm_bf16(A, B, None, M, N, K, 0, 0, 0, False)
        os.environ[env] = vals[0]
        bench(fn, 60, 10)  # clock ramp burn
        for p in range(8):
            for v in vals:
                os.environ[env] = v
                t = bench(fn)
                res.setdefault((M, N, K, v), []).append(t)
                print(f"{tag} {M}x{N}x{K} p{p} {env}={v}: {t*1e3:7.3f} ms "
                      f"{2.0*M*N*K/t/1e12:7.1f} TF")
    for (M, N, K, v), ts in sorted(res.items()):
        ts = sorted(ts)[1:-1]
        m = sum(ts) / len(ts)
        print(f"{tag} TRIMMED {M}x{N}x{K} {env}={v}: {m*1e3:.3f} ms "
              f"{2.0*M*N*K/m/1e12:.1f} TF")


def main():
    from lightctr_amd.ops import hip_ops

    which = sys.argv[1] if len(sys.argv) > 1 else "both"
    if which in ("g256", "both"):
        os.environ.pop("LCTR_GEMM_P8", None)
        sweep("g256", "LCTR_GEMM_APIPE", ["0", "2"],
              [(4096, 4096, 4096), (8192, 8192, 8192)], hip_ops)
    if which in ("p8", "both"):
        os.environ["LCTR_GEMM_P8"] = "1"
        sweep("p8", "LCTR_GEMM_P8_APIPE", ["0", "1"],
              [(65536, 256, 624), (4096, 4096, 4096)], hip_ops)
        os.environ.pop("LCTR_GEMM_P8", None)


if __name__ == "__main__":
    main()
