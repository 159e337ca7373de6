// Hand-written MFMA bf16 GEMM for MI355X / gfx950 (CDNA4).
//
// Replaces the reference's scalar Matrix::Multiply
// (/root/reference/LightCTR/util/matrix.h:216-235) as the dense engine's
// GEMM — redesigned for the CDNA4 matrix cores:
//   * v_mfma_f32_16x16x32_bf16 per-wave tiles, fp32 accumulation
//   * 128x128 block tile, 4 waves (2x2), 64x64 per wave = 4x4 fragments
//   * K-tile 64 staged with __builtin_amdgcn_global_load_lds width-16
//     (direct HBM->LDS DMA; measured +69% over plain-load staging on this
//     structure per the CDNA4 guide ladder) when the tile is full and the
//     row stride is 16B-aligned; scalar staging fallback for tails and
//     transposed-layout operands
//   * LDS image is LINEAR (glds writes lane-linear) with an XOR swizzle
//     folded into the SOURCE address and the fragment READ address
//     (elem_k ^= (row&7)<<3) so the 16-lane ds_read_b128 groups spread
//     over 8 bank slots instead of hitting 2 (the T2 bank-conflict fix,
//     both-sides-or-neither rule)
//   * B is ALWAYS consumed as Bst[N,K] ("transB" torch-Linear layout); the
//     model keeps both W[out,in] and W^T[in,out] bf16 copies so forward
//     AND dgrad hit the fast path; wgrad uses TRANSA=1 (+TRANSB=1)
//   * fused epilogue: optional bias add + activation (relu/sigmoid) +
//     optional bf16 mirror of C for the next layer's input.
#include "common.h"
#include "launchers.h"

namespace lightctr {

bool gemm256_eligible(int M, int N, int K, int transA, int transB);
void gemm256_bf16_launch(const void* A, const void* Bst, const float* bias,
                         float* C, void* Cbf, int M, int N, int K, int act,
                         hipStream_t stream);
bool gemm256v2_eligible(int M, int N, int K, int transA, int transB);
void gemm256v2_bf16_launch(const void* A, const void* Bst, const float* bias,
                           float* C, void* Cbf, int M, int N, int K, int act,
                           hipStream_t stream);

typedef __attribute__((ext_vector_type(8))) __bf16 bf16x8;
typedef __attribute__((ext_vector_type(4))) float f32x4;

#define GEMM_BM 128
#define GEMM_BN 128
#define GEMM_BK 64

__device__ __forceinline__ float act_apply(float v, int act) {
  if (act == 1) return fmaxf(v, 0.f);
  if (act == 2) return sigmoidf_clamped(v);
  return v;
}

// element-index swizzle: flip k-elem bits 3..5 by row bits 0..2 (16B units)
__device__ __forceinline__ int swz(int row, int ke) {
  return ke ^ ((row & 7) << 3);
}

// Stage a [ROWS x 64] bf16 tile (global row-major, row stride `stride`
// elems) into LDS `dst` (linear [ROWS][64] + swizzle) with glds. Full tiles
// only; whole 256-thread block participates; ROWS = 128.
__device__ __forceinline__ void stage_glds(const __bf16* __restrict__ gbase,
                                           long stride, __bf16* dst) {
  const int tid = threadIdx.x;
  const int lane = tid & 63;
  const int wave = tid >> 6;
#pragma unroll
  for (int it = 0; it < 4; ++it) {
    const int seg = wave * 4 + it;  // 8 rows per segment
    const int row = seg * 8 + (lane >> 3);
    const int ke = swz(row, (lane & 7) * 8);
    const __bf16* src = gbase + (long)row * stride + ke;
    // LDS dest is wave-uniform base; HW writes lane*16 bytes per lane,
    // which is exactly this mapping's (row, ke) order (lane-linear image)
    __builtin_amdgcn_global_load_lds(
        (const __attribute__((address_space(1))) void*)src,
        (__attribute__((address_space(3))) void*)(dst + seg * 512), 16, 0,
        0);
  }
}

// Scalar staging for tails / transposed layouts; writes the same swizzled
// image. rows/cols guarded against (max_row, K).
template <int TRANS>
__device__ __forceinline__ void stage_scalar(const __bf16* __restrict__ g,
                                             int base_row, int k0, int K,
                                             int max_row, long stride,
                                             __bf16* dst) {
  const int tid = threadIdx.x;
  if (TRANS == 0) {
    // row-major [row, k]: thread t covers row=t/2, k-half=(t%2)*32
    const int r = tid >> 1;
    const int kk = (tid & 1) * 32;
    const int grow = base_row + r;
    for (int u = 0; u < 32; ++u) {
      const int ke = kk + u;
      __bf16 v = (__bf16)0.f;
      if (grow < max_row && k0 + ke < K)
        v = g[(long)grow * stride + k0 + ke];
      dst[r * 64 + swz(r, ke)] = v;
    }
  } else {
    // stored [K, rows]: thread t covers k=t/8 (+32), rowpart=(t%8)*16;
    // fast path loads the 16 contiguous elements as two bf16x8 vectors
    // (the wgrad operands are large and almost always fully in-bounds)
    typedef __attribute__((ext_vector_type(8))) __bf16 v8;
    const int kk = tid >> 3;
    const int rp = (tid & 7) * 16;
    for (int kr = 0; kr < 2; ++kr) {
      const int ke = kk + kr * 32;
      const int gk = k0 + ke;
      if (gk < K && base_row + rp + 15 < max_row &&
          (((uintptr_t)&g[(long)gk * stride + base_row + rp]) & 15) == 0) {
        const v8 lo = *(const v8*)&g[(long)gk * stride + base_row + rp];
        const v8 hi = *(const v8*)&g[(long)gk * stride + base_row + rp + 8];
#pragma unroll
        for (int e = 0; e < 8; ++e) {
          dst[(rp + e) * 64 + swz(rp + e, ke)] = lo[e];
          dst[(rp + 8 + e) * 64 + swz(rp + 8 + e, ke)] = hi[e];
        }
      } else {
        for (int e = 0; e < 16; ++e) {
          const int r = rp + e;
          const int grow = base_row + r;
          __bf16 v = (__bf16)0.f;
          if (gk < K && grow < max_row) v = g[(long)gk * stride + grow];
          dst[r * 64 + swz(r, ke)] = v;
        }
      }
    }
  }
}

// SPLITK: gridDim.z > 1 partitions the K loop across z-blocks, each
// accumulating its partial tile into C with fp32 atomics (C pre-zeroed by
// the launcher). Used for the wgrad shapes (tiny M,N and K = batch) where
// a single tile otherwise serializes hundreds of K-steps on a handful of
// CUs. bias/act/bf16-emit are not supported with split-K (wgrad needs
// neither).
template <int TRANSA, int TRANSB>
__global__ __launch_bounds__(256) void gemm_bf16_kernel(
    const __bf16* __restrict__ A, const __bf16* __restrict__ Bst,
    const float* __restrict__ bias, float* __restrict__ C,
    __bf16* __restrict__ Cbf, int M, int N, int K, int act, int kchunk) {
  // single __shared__ object (A tile then B tile) — guide §5 trap 4(a)
  __shared__ __bf16 smem[2 * GEMM_BM * GEMM_BK];
  __bf16* As = smem;
  __bf16* Bs = smem + GEMM_BM * GEMM_BK;

  const int lane = threadIdx.x & 63;
  const int wave = threadIdx.x >> 6;
  const int wm = wave >> 1, wn = wave & 1;
  const int M0 = blockIdx.y * GEMM_BM;
  const int N0 = blockIdx.x * GEMM_BN;

  // glds eligibility (full tile + 16B-aligned row stride + 16B base)
  const bool a_glds = (TRANSA == 0) && (M0 + GEMM_BM <= M) && (K % 8 == 0);
  const bool b_glds = (TRANSB == 0) && (N0 + GEMM_BN <= N) && (K % 8 == 0);

  f32x4 acc[4][4] = {};

  const int kbeg = (kchunk > 0) ? blockIdx.z * kchunk : 0;
  const int kend = (kchunk > 0) ? min(K, kbeg + kchunk) : K;
  for (int k0 = kbeg; k0 < kend; k0 += GEMM_BK) {
    const bool k_full = (k0 + GEMM_BK <= kend) && (k0 + GEMM_BK <= K);
    if (a_glds && k_full) {
      stage_glds(A + (long)M0 * K + k0, K, As);
    } else {
      stage_scalar<TRANSA>(A, M0, k0, K, M, TRANSA ? M : K, As);
    }
    if (b_glds && k_full) {
      stage_glds(Bst + (long)N0 * K + k0, K, Bs);
    } else {
      stage_scalar<TRANSB>(Bst, N0, k0, K, N, TRANSB ? N : K, Bs);
    }
    __syncthreads();

#pragma unroll
    for (int kc = 0; kc < GEMM_BK; kc += 32) {
      const int ks = kc + (lane >> 4) * 8;  // this lane's 8-deep k slice
      bf16x8 a[4], b[4];
#pragma unroll
      for (int f = 0; f < 4; ++f) {
        const int ra = wm * 64 + f * 16 + (lane & 15);
        const int rb = wn * 64 + f * 16 + (lane & 15);
        a[f] = *(const bf16x8*)&As[ra * 64 + swz(ra, ks)];
        b[f] = *(const bf16x8*)&Bs[rb * 64 + swz(rb, ks)];
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
        if (kchunk > 0) {
          atomicAdd(&C[(size_t)row * N + col], acc[i][j][r]);
        } else {
          const float v = act_apply(acc[i][j][r] + bv, act);
          C[(size_t)row * N + col] = v;
          if (Cbf) Cbf[(size_t)row * N + col] = (__bf16)v;
        }
      }
    }
  }
}

void gemm_bf16_launch(const void* A, const void* Bst, const float* bias,
                      float* C, void* Cbf, int M, int N, int K, int transA,
                      int transB, int act, hipStream_t stream) {
  // v2 (counted-vmcnt k-chunk pipeline) measured 858/1026 TF vs v1's
  // 992/1133 at 4k/8k^3: one barrier per 32-K chunk costs more than the
  // barrier-spanning loads buy at this geometry. v1 stays preferred; v2
  // kept (correct, race-screened) as the documented experiment.
  if (gemm256p8_eligible(M, N, K, transA, transB)) {
    // 8-phase counted-vmcnt schedule (round 2). Measured (gemm_p8/old.txt):
    // 2x the round-1 kernel on K-tail shapes (65536x256x624: 555 vs 299
    // TF — the masked-glds tail staging + counted waits), but ~8% behind
    // it on big square K%64==0 shapes (904 vs 996 @4k^3) where the extra
    // per-phase barriers outweigh the counted drain. Auto: p8 for tail
    // shapes, round-1 kernel otherwise. LCTR_GEMM_P8=1/0 forces/disables.
    static const int p8mode = [] {
      const char* e = getenv("LCTR_GEMM_P8");
      if (!e) return 2;  // auto
      return e[0] == '0' ? 0 : 1;
    }();
    const bool use_p8 =
        p8mode == 1 || (p8mode == 2 && (K % 64 != 0));
    if (use_p8) {
      gemm256p8_bf16_launch(A, Bst, bias, C, Cbf, M, N, K, act, stream);
      return;
    }
  }
  if (gemm256_eligible(M, N, K, transA, transB)) {
    gemm256_bf16_launch(A, Bst, bias, C, Cbf, M, N, K, act, stream);
    return;
  }
  if (gemm256n128_eligible(M, N, K, transA, transB)) {
    gemm256n128_bf16_launch(A, Bst, bias, C, Cbf, M, N, K, act, stream);
    return;
  }
  // degenerate shapes (the MLP's 1-wide output layer): direct kernels
  // instead of 128x128 tiles at ~99% waste (see nn_kernels.hip)
  if (N == 1 && transA == 0) {
    gemv_n1_launch(A, Bst, bias, C, Cbf, M, K, act, stream);
    return;
  }
  if (M == 1 && transA == 1 && transB == 1 && bias == nullptr &&
      act == 0 && Cbf == nullptr) {
    wrowsum_m1_launch(A, Bst, C, K, N, stream);
    return;
  }
  // wgrad128 first: it beats small_wgrad even on tiny outputs
  // (64x16x65536: 65 vs 93 us); small_wgrad keeps the shapes wgrad128
  // cannot take (M or N not 16-aligned)
  if (gemm_wgrad_eligible(M, N, K, transA, transB) && bias == nullptr &&
      act == 0 && Cbf == nullptr) {
    // K-major x K-major wgrad: linear glds staging + ds_read_b64_tr_b16
    // cooperative-transpose fragment reads. Measured vs the generic
    // scalar-transposed-staging path at K=65536: 256x512 67 vs 92 us,
    // 128x256 40 vs 59, 256x624 (partial strip) 119 vs 126.
    // LCTR_WGRAD128=0 reverts.
    static const bool wg = [] {
      const char* e = getenv("LCTR_WGRAD128");
      return !(e && e[0] == '0');
    }();
    if (wg) {
      gemm_wgrad_bf16_launch(A, Bst, C, M, N, K, stream);
      return;
    }
  }
  if (transA == 1 && transB == 1 && (long)M * N <= 4096 && K >= 8192 &&
      bias == nullptr && act == 0 && Cbf == nullptr) {
    small_wgrad_launch(A, Bst, C, M, N, K, stream);
    return;
  }
  if (K == 1 && transA == 0) {
    outer_k1_launch(A, Bst, bias, C, Cbf, M, N, act, stream);
    return;
  }
  dim3 block(256);
  dim3 grid((N + GEMM_BN - 1) / GEMM_BN, (M + GEMM_BM - 1) / GEMM_BM);
  // split-K when the output grid is far too small to fill 256 CUs and K is
  // deep (the wgrad regime): target >= 512 z*xy blocks
  int kchunk = 0;
  const int xy = (int)(grid.x * grid.y);
  if (xy < 64 && K >= 4096 && Cbf == nullptr && bias == nullptr &&
      act == 0) {
    // split-K fan-out target (blocks in flight); LCTR_SPLITK_TARGET
    // overrides for sweeps
    static const int sk_target = [] {
      const char* e = getenv("LCTR_SPLITK_TARGET");
      return e ? atoi(e) : 512;
    }();
    int zw = min(64, max(2, sk_target / xy));
    kchunk = ((K + zw - 1) / zw + GEMM_BK - 1) / GEMM_BK * GEMM_BK;
    grid.z = (K + kchunk - 1) / kchunk;
    LCTR_CHECK_HIP(hipMemsetAsync(C, 0, (size_t)M * N * sizeof(float),
                                  stream));
  }
#define LAUNCH_GEMM(TA, TB)                                                 \
  hipLaunchKernelGGL((gemm_bf16_kernel<TA, TB>), grid, block, 0, stream,    \
                     (const __bf16*)A, (const __bf16*)Bst, bias, C,         \
                     (__bf16*)Cbf, M, N, K, act, kchunk)
  if (transA == 0 && transB == 0) LAUNCH_GEMM(0, 0);
  else if (transA == 1 && transB == 0) LAUNCH_GEMM(1, 0);
  else if (transA == 0 && transB == 1) LAUNCH_GEMM(0, 1);
  else LAUNCH_GEMM(1, 1);
#undef LAUNCH_GEMM
}

}  // namespace lightctr
