// Hand-written MFMA bf16 GEMM for MI355X / gfx950 (CDNA4).
//
// Replaces the reference's scalar Matrix::Multiply
// (/root/reference/LightCTR/util/matrix.h:216-235) as the dense engine's
// GEMM — redesigned for the CDNA4 matrix cores:
//   * v_mfma_f32_16x16x32_bf16 per-wave tiles, fp32 accumulation
//   * 128x128 block tile, 4 waves (2x2), 64x64 per wave = 4x4 fragments
//   * K-tile 64 staged through LDS, +8-element row pad (16 B) so the
//     ds_read_b128 fragment reads are bank-conflict-free (lanes read 16
//     different rows at the same k-range; pad makes row stride 144 B,
//     gcd(36,64)=4 -> 16 distinct banks per 16-lane group)
//   * B is ALWAYS consumed as Bst[N,K] ("transB" torch-Linear layout), so
//     both operand fragments are contiguous-k 16 B LDS reads; the model
//     keeps both W[out,in] and W^T[in,out] bf16 copies (HBM3E is abundant)
//     so forward AND dgrad hit this fast path; wgrad uses TRANSA=1
//   * fused epilogue: optional bias add + activation (relu/sigmoid) +
//     optional bf16 mirror of C for the next layer's input.
//
// Correctness-first structure (the "step-3" ladder shape of the CDNA4
// guide); deliberate headroom: global_load_lds staging, 8-phase schedule.
#include "common.h"

namespace lightctr {

typedef __attribute__((ext_vector_type(8))) __bf16 bf16x8;
typedef __attribute__((ext_vector_type(4))) float f32x4;

#define GEMM_BM 128
#define GEMM_BN 128
#define GEMM_BK 64
#define GEMM_PAD 8  // bf16 elements = 16 B

__device__ __forceinline__ float act_apply(float v, int act) {
  if (act == 1) return fmaxf(v, 0.f);
  if (act == 2) return sigmoidf_clamped(v);
  return v;
}

// A logical [M,K]: TRANSA=0 -> stored row-major [M,K]; TRANSA=1 -> stored
// [K,M] (i.e. logical A[m,k] = Aptr[k*M + m]; the wgrad path where A = dY^T).
// B logical [K,N]: TRANSB=0 -> stored [N,K] (torch-Linear weight layout, the
// fast contiguous-k path used by forward and dgrad); TRANSB=1 -> stored
// [K,N] row-major (wgrad's activation operand).
template <int TRANSA, int TRANSB>
__global__ __launch_bounds__(256) void gemm_bf16_kernel(
    const __bf16* __restrict__ A, const __bf16* __restrict__ Bst,
    const float* __restrict__ bias, float* __restrict__ C,
    __bf16* __restrict__ Cbf, int M, int N, int K, int act) {
  __shared__ __bf16 As[GEMM_BM][GEMM_BK + GEMM_PAD];
  __shared__ __bf16 Bs[GEMM_BN][GEMM_BK + GEMM_PAD];

  const int tid = threadIdx.x;
  const int lane = tid & 63;
  const int wave = tid >> 6;
  const int wm = wave >> 1, wn = wave & 1;
  const int M0 = blockIdx.y * GEMM_BM;
  const int N0 = blockIdx.x * GEMM_BN;

  f32x4 acc[4][4] = {};

  for (int k0 = 0; k0 < K; k0 += GEMM_BK) {
    // ---- stage A tile ----
    if (TRANSA == 0) {
      // thread t: row = t/2, k-half = (t%2)*32, 32 bf16 = 64 B contiguous
      const int m = tid >> 1;
      const int kk = (tid & 1) * 32;
      const int gm = M0 + m;
#pragma unroll
      for (int u = 0; u < 32; u += 8) {
        bf16x8 v = {};
        if (gm < M && k0 + kk + u + 7 < K) {
          v = *(const bf16x8*)&A[(size_t)gm * K + k0 + kk + u];
        } else if (gm < M) {
          for (int e = 0; e < 8; ++e)
            if (k0 + kk + u + e < K)
              ((__bf16*)&v)[e] = A[(size_t)gm * K + k0 + kk + u + e];
        }
        *(bf16x8*)&As[m][kk + u] = v;
      }
    } else {
      // stored [K,M]: thread t: k = t/8, m-part = (t%8)*16; contiguous in m,
      // scatter-transposed into As
      const int kk = tid >> 3;
      const int mp = (tid & 7) * 16;
      const int gk = k0 + kk;
      // two k rows per thread (BK=64, 32 k-rows covered per 256 threads pass)
#pragma unroll
      for (int kr = 0; kr < 2; ++kr) {
        const int kcur = kk + kr * 32;
        const int gkc = k0 + kcur;
        for (int e = 0; e < 16; ++e) {
          const int gm = M0 + mp + e;
          __bf16 v = (__bf16)0.f;
          if (gkc < K && gm < M) v = A[(size_t)gkc * M + gm];
          As[mp + e][kcur] = v;
        }
      }
      (void)gk;
    }
    // ---- stage B tile ----
    if (TRANSB == 0) {
      // stored [N,K]: same contiguous-k pattern as the TRANSA=0 A load
      const int n = tid >> 1;
      const int kk = (tid & 1) * 32;
      const int gn = N0 + n;
#pragma unroll
      for (int u = 0; u < 32; u += 8) {
        bf16x8 v = {};
        if (gn < N && k0 + kk + u + 7 < K) {
          v = *(const bf16x8*)&Bst[(size_t)gn * K + k0 + kk + u];
        } else if (gn < N) {
          for (int e = 0; e < 8; ++e)
            if (k0 + kk + u + e < K)
              ((__bf16*)&v)[e] = Bst[(size_t)gn * K + k0 + kk + u + e];
        }
        *(bf16x8*)&Bs[n][kk + u] = v;
      }
    } else {
      // stored [K,N] row-major: coalesced along n, scatter-transpose to Bs
      const int kk = tid >> 3;
      const int np = (tid & 7) * 16;
#pragma unroll
      for (int kr = 0; kr < 2; ++kr) {
        const int kcur = kk + kr * 32;
        const int gkc = k0 + kcur;
        for (int e = 0; e < 16; ++e) {
          const int gn = N0 + np + e;
          __bf16 v = (__bf16)0.f;
          if (gkc < K && gn < N) v = Bst[(size_t)gkc * N + gn];
          Bs[np + e][kcur] = v;
        }
      }
    }
    __syncthreads();

    // ---- MFMA over the K tile (2 chunks of 32) ----
#pragma unroll
    for (int kc = 0; kc < GEMM_BK; kc += 32) {
      const int ks = kc + (lane >> 4) * 8;  // this lane's 8-deep k slice
      bf16x8 a[4], b[4];
#pragma unroll
      for (int f = 0; f < 4; ++f) {
        a[f] = *(const bf16x8*)&As[wm * 64 + f * 16 + (lane & 15)][ks];
        b[f] = *(const bf16x8*)&Bs[wn * 64 + f * 16 + (lane & 15)][ks];
      }
#pragma unroll
      for (int i = 0; i < 4; ++i)
#pragma unroll
        for (int j = 0; j < 4; ++j)
          acc[i][j] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
              a[i], b[j], acc[i][j], 0, 0, 0);
    }
    __syncthreads();
  }

  // ---- epilogue: bias + act + store (C/D map: col=lane&15,
  // row=(lane>>4)*4+r) ----
#pragma unroll
  for (int i = 0; i < 4; ++i) {
#pragma unroll
    for (int j = 0; j < 4; ++j) {
      const int col = N0 + wn * 64 + j * 16 + (lane & 15);
      if (col >= N) continue;
      const float bv = bias ? bias[col] : 0.f;
#pragma unroll
      for (int r = 0; r < 4; ++r) {
        const int row = M0 + wm * 64 + i * 16 + (lane >> 4) * 4 + r;
        if (row >= M) continue;
        const float v = act_apply(acc[i][j][r] + bv, act);
        C[(size_t)row * N + col] = v;
        if (Cbf) Cbf[(size_t)row * N + col] = (__bf16)v;
      }
    }
  }
}

void gemm_bf16_launch(const void* A, const void* Bst, const float* bias,
                      float* C, void* Cbf, int M, int N, int K, int transA,
                      int transB, int act, hipStream_t stream) {
  dim3 block(256);
  dim3 grid((N + GEMM_BN - 1) / GEMM_BN, (M + GEMM_BM - 1) / GEMM_BM);
#define LAUNCH_GEMM(TA, TB)                                                 \
  hipLaunchKernelGGL((gemm_bf16_kernel<TA, TB>), grid, block, 0, stream,    \
                     (const __bf16*)A, (const __bf16*)Bst, bias, C,         \
                     (__bf16*)Cbf, M, N, K, act)
  if (transA == 0 && transB == 0) LAUNCH_GEMM(0, 0);
  else if (transA == 1 && transB == 0) LAUNCH_GEMM(1, 0);
  else if (transA == 0 && transB == 1) LAUNCH_GEMM(0, 1);
  else LAUNCH_GEMM(1, 1);
#undef LAUNCH_GEMM
}

}  // namespace lightctr
