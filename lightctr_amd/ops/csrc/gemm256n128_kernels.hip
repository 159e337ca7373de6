// 256(M) x 128(N)-tile MFMA bf16 GEMM for MI355X/gfx950 — the gemm256
// phase-split schedule (double-buffered glds staging, per-phase prefetch,
// ONE barrier + drain per K-tile, s_setprio MFMA clusters) adapted to the
// MLP's mid shapes where N % 256 != 0 (e.g. Wide&Deep fwd2 [65536,128,256]
// and dX1 [65536,624,256]): those otherwise fall to the single-buffered
// 128x128 generalist, which serializes staging latency against math every
// K-tile (measured ~74 us / ~58 TF on fwd2). The last block column guards
// a partial-N B tile with zero-filled scalar staging.
//
// Wave geometry: 8 waves as 4M x 2N, 64x64 per wave = acc[4][4]; B
// fragments for the whole tile are register-held across phases; phase p
// computes fragment row fm = p while prefetching tile t+1's half-tiles
// (p0: A-top, p1: A-bot, p2: B).
#include "common.h"

namespace lightctr {

typedef __attribute__((ext_vector_type(8))) __bf16 bf16x8;
typedef __attribute__((ext_vector_type(4))) float f32x4;

#define GN128_BK 64

__device__ __forceinline__ int swz_n128(int row, int ke) {
  return ke ^ ((row & 7) << 3);
}

// stage a full 128-row x 64-col bf16 half-tile with 512 threads
__device__ __forceinline__ void stage_half_n128(
    const __bf16* __restrict__ gbase, long stride, __bf16* dst) {
  const int tid = threadIdx.x;
  const int lane = tid & 63;
  const int wave = tid >> 6;
#pragma unroll
  for (int it = 0; it < 2; ++it) {
    const int seg = wave * 2 + it;  // 16 segments of 8 rows
    const int row = seg * 8 + (lane >> 3);
    const int ke = swz_n128(row, (lane & 7) * 8);
    const __bf16* src = gbase + (long)row * stride + ke;
    __builtin_amdgcn_global_load_lds(
        (const __attribute__((address_space(1))) void*)src,
        (__attribute__((address_space(3))) void*)(dst + seg * 512), 16, 0,
        0);
  }
}

// scalar zero-filled staging for a partial-N B tile (rows >= nrem zeroed)
__device__ __forceinline__ void stage_b_tail_n128(
    const __bf16* __restrict__ g, long stride, int k0, int nrem,
    __bf16* dst) {
  const int tid = threadIdx.x;
  const int r = tid >> 1;  // 0..255 -> rows 0..127 handled by tid/2 < 128
  if (r >= 128) return;
  const int kk = (tid & 1) * 32;
  for (int u = 0; u < 32; ++u) {
    const int ke = kk + u;
    __bf16 v = (__bf16)0.f;
    if (r < nrem) v = g[(long)r * stride + k0 + ke];
    dst[r * GN128_BK + swz_n128(r, ke)] = v;
  }
}

__device__ __forceinline__ float act_n128(float v, int act) {
  if (act == 1) return fmaxf(v, 0.f);
  if (act == 2) return sigmoidf_clamped(v);
  return v;
}

__global__ __launch_bounds__(512, 1) void gemm256n128_bf16_kernel(
    const __bf16* __restrict__ A, const __bf16* __restrict__ Bst,
    const float* __restrict__ bias, float* __restrict__ C,
    __bf16* __restrict__ Cbf, int M, int N, int K, int act) {
  // buffer b: A(256x64) then B(128x64); two buffers
  __shared__ __bf16 smem[2 * (256 + 128) * GN128_BK];
  const int lane = threadIdx.x & 63;
  const int wave = threadIdx.x >> 6;
  const int wm = wave >> 1;  // 0..3
  const int wn = wave & 1;   // 0..1
  const int M0 = blockIdx.y * 256;
  const int N0 = blockIdx.x * 128;
  const int nrem = min(128, N - N0);  // partial last column tile
  const bool b_full = nrem == 128;

  f32x4 acc[4][4] = {};
  const int NT = K / GN128_BK;

  __bf16* bufA0 = smem;
  // prologue: stage tile 0
  stage_half_n128(A + (long)M0 * K, K, bufA0);
  stage_half_n128(A + (long)(M0 + 128) * K, K, bufA0 + 128 * GN128_BK);
  if (b_full) {
    stage_half_n128(Bst + (long)N0 * K, K, bufA0 + 256 * GN128_BK);
  } else {
    stage_b_tail_n128(Bst + (long)N0 * K, K, 0, nrem,
                      bufA0 + 256 * GN128_BK);
  }
  asm volatile("s_waitcnt vmcnt(0)" ::: "memory");
  __syncthreads();

  int cur = 0;
  for (int t = 0; t < NT; ++t) {
    __bf16* curA = smem + (size_t)cur * 384 * GN128_BK;
    __bf16* curB = curA + 256 * GN128_BK;
    __bf16* nxtA = smem + (size_t)(cur ^ 1) * 384 * GN128_BK;
    __bf16* nxtB = nxtA + 256 * GN128_BK;
    const int k0 = (t + 1) * GN128_BK;
    const bool more = (t + 1) < NT;

    // B fragments for this wave's 64 columns, whole K-tile, in registers
    bf16x8 b[4][2];
#pragma unroll
    for (int fn = 0; fn < 4; ++fn)
#pragma unroll
      for (int kc = 0; kc < 2; ++kc) {
        const int rb = wn * 64 + fn * 16 + (lane & 15);
        const int ks = kc * 32 + (lane >> 4) * 8;
        b[fn][kc] =
            *(const bf16x8*)&curB[rb * GN128_BK + swz_n128(rb, ks)];
      }
#pragma unroll
    for (int p = 0; p < 4; ++p) {
      if (more) {  // prefetch tile t+1, one piece per phase
        if (p == 0) {
          stage_half_n128(A + (long)M0 * K + k0, K, nxtA);
        } else if (p == 1) {
          stage_half_n128(A + (long)(M0 + 128) * K + k0, K,
                          nxtA + 128 * GN128_BK);
        } else if (p == 2) {
          if (b_full)
            stage_half_n128(Bst + (long)N0 * K + k0, K, nxtB);
          else
            stage_b_tail_n128(Bst + (long)N0 * K, K, k0, nrem, nxtB);
        }
      }
      bf16x8 a[2];
#pragma unroll
      for (int kc = 0; kc < 2; ++kc) {
        const int ra = wm * 64 + p * 16 + (lane & 15);
        const int ks = kc * 32 + (lane >> 4) * 8;
        a[kc] = *(const bf16x8*)&curA[ra * GN128_BK + swz_n128(ra, ks)];
      }
      __builtin_amdgcn_s_setprio(1);
#pragma unroll
      for (int fn = 0; fn < 4; ++fn)
#pragma unroll
        for (int kc = 0; kc < 2; ++kc)
          acc[p][fn] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
              a[kc], b[fn][kc], acc[p][fn], 0, 0, 0);
      __builtin_amdgcn_s_setprio(0);
    }
    asm volatile("s_waitcnt vmcnt(0)" ::: "memory");
    __syncthreads();
    cur ^= 1;
  }

  // epilogue (C/D map: col=lane&15, row=(lane>>4)*4+r)
#pragma unroll
  for (int fm = 0; fm < 4; ++fm) {
#pragma unroll
    for (int fn = 0; fn < 4; ++fn) {
      const int col = N0 + wn * 64 + fn * 16 + (lane & 15);
      if (col >= N) continue;
      const float bv = bias ? bias[col] : 0.f;
#pragma unroll
      for (int r = 0; r < 4; ++r) {
        const int row = M0 + wm * 64 + fm * 16 + (lane >> 4) * 4 + r;
        const float v = act_n128(acc[fm][fn][r] + bv, act);
        C[(size_t)row * N + col] = v;
        if (Cbf) Cbf[(size_t)row * N + col] = (__bf16)v;
      }
    }
  }
}

bool gemm256n128_eligible(int M, int N, int K, int transA, int transB) {
  // N%256 shapes go to the 256x256 kernel; this one takes the N>=128
  // leftovers with any N (partial last tile guarded). K%8 keeps glds
  // row stride 16B-aligned.
  return transA == 0 && transB == 0 && M % 256 == 0 && N >= 128 &&
         N % 256 != 0 && K % GN128_BK == 0 && K >= 128;
}

void gemm256n128_bf16_launch(const void* A, const void* Bst,
                             const float* bias, float* C, void* Cbf, int M,
                             int N, int K, int act, hipStream_t stream) {
  dim3 block(512);
  dim3 grid((N + 127) / 128, M / 256);
  hipLaunchKernelGGL(gemm256n128_bf16_kernel, grid, block, 0, stream,
                     (const __bf16*)A, (const __bf16*)Bst, bias, C,
                     (__bf16*)Cbf, M, N, K, act);
}

}  // namespace lightctr
