// Weight-gradient GEMM for MI355X/gfx950: C[M,N] = sum_k At[k,m]*Bt[k,n]
// with BOTH operands stored K-major (At = [K,M] row-major, Bt = [K,N] —
// the dZ / activation layouts of the MLP backward, reference
// fullyconnLayer.h:166-179 weight-delta accumulation).
//
// The generic kernel's TRANSA/TRANSB path stages these with per-element
// scalar LDS writes (measured 125 us for 256x624x65536 — 9x off the
// ~14 us traffic floor: VALU/LDS-write bound, guide common-mistake #1).
// This kernel instead:
//   * stages [64k x 128row] panels with global_load_lds into the PLAIN
//     linear [k][128] image (fully coalesced 16 B pieces)
//   * reads MFMA fragments with the hardware transpose read
//     ds_read_b64_tr_b16. Probe-verified semantics
//     (tools/debug_wgrad_probe.py): within each 16-lane group, lane i
//     loads 4 contiguous elements at its address and the hardware
//     redistributes them as out(lane 4*(i&3)+t, elem i>>2) =
//     load[i][t]. Feeding addr_i = (k0g + (i>>2))*128 + col0 + 4*(i&3)
//     therefore hands every lane its fragment column — k-major operands
//     consumed straight from a row-major image, no scalar transposes
//     anywhere (the "attention-V recipe" from the CDNA4 guide)
//   * split-K over gridDim.z with fp32 atomic accumulation into C
//     (pre-zeroed by the launcher), as in the generic kernel.
// Eligibility (launcher): transA && transB, M%16==0, N%16==0, M%8==0
// row strides (always true given %16), K >= 1024. Partial K-tails are
// staged by a scalar blocked path.
#include <cstdlib>

#include "common.h"

namespace lightctr {

typedef __attribute__((ext_vector_type(8))) __bf16 wg_bf16x8;
typedef __attribute__((ext_vector_type(4))) __bf16 wg_bf16x4;
typedef __attribute__((ext_vector_type(4))) float wg_f32x4;

#define WG_BM 128
#define WG_BN 128
#define WG_BK 64

// Stage a [BK x 128] K-major panel (g = base of [K, ld] array, columns
// col0..col0+127, k rows k0..k0+BK) into the plain linear [k][128]
// image with glds: 16 instructions x 64 lanes x 16 B, fully coalesced.
// Caller guarantees range + 16 B alignment.
__device__ __forceinline__ void wg_stage_glds(const __bf16* __restrict__ g,
                                              long ld, int k0, int col0,
                                              __bf16* dst) {
  const int tid = threadIdx.x;
  const int lane = tid & 63;
  const int wave = tid >> 6;
#pragma unroll
  for (int it = 0; it < 4; ++it) {
    const int instr = wave * 4 + it;         // 16 instructions
    const int k = instr * 4 + (lane >> 4);   // 4 k-rows per instruction
    const int m0 = (lane & 15) * 8;
    const __bf16* src = g + (long)(k0 + k) * ld + col0 + m0;
    __builtin_amdgcn_global_load_lds(
        (const __attribute__((address_space(1))) void*)src,
        (__attribute__((address_space(3))) void*)(dst + instr * 512), 16, 0,
        0);
  }
}

// Edge staging (K-tails / partial column strips): vector 16 B copies for
// fully in-range chunks, per-element zero-padded fill only at the
// boundary chunk — the linear image makes the write a single
// ds_write_b128, so a partial strip stages at near-glds speed instead
// of the 8192-scalar-store pass that made the first cut 2x slower on
// the 624-column W&D shape.
__device__ __forceinline__ void wg_stage_scalar(const __bf16* __restrict__ g,
                                                long ld, int k0, int col0,
                                                int K, int ncols,
                                                __bf16* dst) {
  const int tid = threadIdx.x;
#pragma unroll
  for (int it = 0; it < 4; ++it) {
    const int t = tid + it * 256;  // 1024 16 B slots
    const int k = t >> 4;
    const int m0 = (t & 15) * 8;
    const int gk = k0 + k;
    if (gk < K && col0 + m0 + 7 < ncols) {
      *(wg_bf16x8*)(dst + k * 128 + m0) =
          *(const wg_bf16x8*)(g + (long)gk * ld + col0 + m0);
    } else {
      for (int e = 0; e < 8; ++e) {
        const int c = col0 + m0 + e;
        __bf16 v = (__bf16)0.f;
        if (gk < K && c < ncols) v = g[(long)gk * ld + c];
        dst[k * 128 + m0 + e] = v;
      }
    }
  }
}

__device__ __forceinline__ wg_bf16x8 wg_frag(const __bf16* base, int fb,
                                             int kc, int lane) {
  // fragment (16-row block fb, k-chunk kc) from the LINEAR [k][128]
  // image via the cooperative transpose read. Within a 16-lane group
  // (i = lane & 15), hardware redistributes lane i's 4 loaded elements
  // as out(lane 4*(i&3)+t, elem i>>2) = load[i][t]; with
  //   addr_i = (k0g + (i>>2)) * 128 + fb*16 + 4*(i&3)
  // every lane ends up with its column m = fb*16 + (l&15) at
  // k = k0g + (0..3); the second read at +4 rows gives k +4..7.
  const int i = lane & 15;
  const int k0g = kc * 32 + (lane >> 4) * 8;
  const __bf16* p = base + (k0g + (i >> 2)) * 128 + fb * 16 + 4 * (i & 3);
  const wg_bf16x4 lo = __builtin_amdgcn_ds_read_tr16_b64_v4bf16(
      (__attribute__((address_space(3))) wg_bf16x4*)p);
  const wg_bf16x4 hi = __builtin_amdgcn_ds_read_tr16_b64_v4bf16(
      (__attribute__((address_space(3))) wg_bf16x4*)(p + 4 * 128));
  wg_bf16x8 out;
  out[0] = lo.x; out[1] = lo.y; out[2] = lo.z; out[3] = lo.w;
  out[4] = hi.x; out[5] = hi.y; out[6] = hi.z; out[7] = hi.w;
  return out;
}

__global__ __launch_bounds__(256) void gemm_wgrad_bf16_kernel(
    const __bf16* __restrict__ At, const __bf16* __restrict__ Bt,
    float* __restrict__ C, int M, int N, int K, int kchunk) {
  __shared__ __bf16 smem[2 * WG_BK * WG_BM];  // As | Bs (16 KB each)
  __bf16* As = smem;
  __bf16* Bs = smem + WG_BK * WG_BM;
  const int lane = threadIdx.x & 63;
  const int wave = threadIdx.x >> 6;
  const int wm = wave >> 1, wn = wave & 1;  // 2x2 waves, 64x64 each
  const int M0 = blockIdx.y * WG_BM;
  const int N0 = blockIdx.x * WG_BN;
  const bool a_full = M0 + WG_BM <= M;
  const bool b_full = N0 + WG_BN <= N;

  wg_f32x4 acc[4][4] = {};

  const int kbeg = (kchunk > 0) ? blockIdx.z * kchunk : 0;
  const int kend = (kchunk > 0) ? min(K, kbeg + kchunk) : K;
  for (int k0 = kbeg; k0 < kend; k0 += WG_BK) {
    const bool k_full = k0 + WG_BK <= kend;
    if (a_full && k_full)
      wg_stage_glds(At, M, k0, M0, As);
    else
      wg_stage_scalar(At, M, k0, M0, min(kend, K), M, As);
    if (b_full && k_full)
      wg_stage_glds(Bt, N, k0, N0, Bs);
    else
      wg_stage_scalar(Bt, N, k0, N0, min(kend, K), N, Bs);
    __syncthreads();

#pragma unroll
    for (int kc = 0; kc < 2; ++kc) {
      wg_bf16x8 a[4], b[4];
#pragma unroll
      for (int f = 0; f < 4; ++f) {
        a[f] = wg_frag(As, wm * 4 + f, kc, lane);
        b[f] = wg_frag(Bs, wn * 4 + f, kc, lane);
      }
#pragma unroll
      for (int fm = 0; fm < 4; ++fm)
#pragma unroll
        for (int fn = 0; fn < 4; ++fn)
          acc[fm][fn] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
              a[fm], b[fn], acc[fm][fn], 0, 0, 0);
    }
    __syncthreads();
  }

  // C write (atomic when split-K): D map col = lane&15, row = (l>>4)*4+r
#pragma unroll
  for (int fm = 0; fm < 4; ++fm) {
#pragma unroll
    for (int fn = 0; fn < 4; ++fn) {
      const int col = N0 + wn * 64 + fn * 16 + (lane & 15);
      if (col >= N) continue;
#pragma unroll
      for (int r = 0; r < 4; ++r) {
        const int row = M0 + wm * 64 + fm * 16 + (lane >> 4) * 4 + r;
        if (row >= M) continue;
        if (kchunk > 0)
          atomicAdd(&C[(size_t)row * N + col], acc[fm][fn][r]);
        else
          C[(size_t)row * N + col] = acc[fm][fn][r];
      }
    }
  }
}

// Debug probe: mode 0 = stage a [64][128] panel from g via wg_stage_glds
// and dump the LDS image linearly; mode 1 = copy g directly into LDS
// (identity image) — both then dump wave-0 fragments read via wg_frag.
__global__ void wg_probe_kernel(const __bf16* __restrict__ g, int ld,
                                __bf16* __restrict__ out_lds,
                                float* __restrict__ out_frag, int mode) {
  __shared__ __bf16 s[WG_BK * WG_BM];
  if (mode == 0) {
    wg_stage_glds(g, ld, 0, 0, s);
    asm volatile("s_waitcnt vmcnt(0)" ::: "memory");
    __syncthreads();
  } else {
    for (int e = threadIdx.x; e < WG_BK * WG_BM; e += 256) s[e] = g[e];
    __syncthreads();
  }
  for (int e = threadIdx.x; e < WG_BK * WG_BM; e += 256) out_lds[e] = s[e];
  const int lane = threadIdx.x & 63;
  if (threadIdx.x < 64) {
    for (int fb = 0; fb < 2; ++fb)
      for (int kc = 0; kc < 2; ++kc) {
        const wg_bf16x8 f = wg_frag(s, fb, kc, lane);
        for (int e = 0; e < 8; ++e)
          out_frag[(((size_t)fb * 2 + kc) * 64 + lane) * 8 + e] =
              (float)f[e];
      }
  }
}

void wg_probe_launch(const void* g, int ld, void* out_lds, float* out_frag,
                     int mode, hipStream_t stream) {
  hipLaunchKernelGGL(wg_probe_kernel, dim3(1), dim3(256), 0, stream,
                     (const __bf16*)g, ld, (__bf16*)out_lds, out_frag, mode);
}

bool gemm_wgrad_eligible(int M, int N, int K, int transA, int transB) {
  // both operands K-major; blocked glds staging wants 16-col row blocks
  // and 16 B-aligned 8-element half-rows (M%8, N%8; %16 keeps the edge
  // logic out of the fragment path)
  return transA == 1 && transB == 1 && M % 16 == 0 && N % 16 == 0 &&
         K >= 1024;
}

void gemm_wgrad_bf16_launch(const void* At, const void* Bt, float* C, int M,
                            int N, int K, hipStream_t stream) {
  dim3 block(256);
  dim3 grid((N + WG_BN - 1) / WG_BN, (M + WG_BM - 1) / WG_BM);
  int kchunk = 0;
  const int xy = (int)(grid.x * grid.y);
  static const int sk_target = [] {
    const char* e = getenv("LCTR_SPLITK_TARGET");
    return e ? atoi(e) : 512;
  }();
  if (xy < 256 && K >= 2048) {
    int zw = min(64, max(1, sk_target / xy));
    if (zw > 1) {
      kchunk = ((K + zw - 1) / zw + WG_BK - 1) / WG_BK * WG_BK;
      grid.z = (K + kchunk - 1) / kchunk;
      LCTR_CHECK_HIP(
          hipMemsetAsync(C, 0, (size_t)M * N * sizeof(float), stream));
    }
  }
  hipLaunchKernelGGL(gemm_wgrad_bf16_kernel, grid, block, 0, stream,
                     (const __bf16*)At, (const __bf16*)Bt, C, M, N, K,
                     kchunk);
}

}  // namespace lightctr
