// 256x256-tile 8-phase counted-vmcnt MFMA bf16 GEMM for MI355X/gfx950.
//
// Round-2 schedule upgrade over gemm256_kernels.hip (which drains
// vmcnt(0) + __syncthreads once per K-tile): the CDNA4 guide's 8-phase
// template structure — per 64-deep K-tile, FOUR phases each of
//   { ds_read register subtile | stage one half-tile (global_load_lds) |
//     raw s_barrier | counted s_waitcnt lgkmcnt | s_setprio-wrapped
//     16x MFMA | raw s_barrier }
// with the VM counter drained by a COUNTED s_waitcnt once per K-tile
// (never vmcnt(0) in steady state), so prefetch loads stay in flight
// across barriers. Raw s_barrier only — __syncthreads would fold the
// pending LDS-DMA into its fence and drain vmcnt(0) (guide §6 rule on
// glds + barriers).
//
// Staging pipeline (2 LDS buffers = 4 independent half-tile-pair slots;
// A parity slots are freed at the end of the owning tile's compute, B
// parity slots right after the tile-start b-fragment register reads, so
// B can be staged TWO tiles ahead):
//   prologue: stage tile0 (A+B) and B(tile1);            vmcnt(4)
//   tile t:   p0 stage A-top(t+1), p1 stage A-bot(t+1),
//             p2 stage B-top(t+2), p3 stage B-bot(t+2);  vmcnt(4)
// The boundary vmcnt(4) leaves B(t+2)'s four loads in flight and
// guarantees (in-order VM retirement) that A(t+1) and B(t+1) landed.
//
// K-tail tiles (K % 64 != 0, K % 8 == 0) are staged by the same glds
// path with per-lane masking on the in-range 16 B chunks plus an LDS
// zero-fill of the out-of-range chunk slots — same 2-instruction vmcnt
// footprint as a full stage, so the counted waits stay exact.
//
// B fragments for the whole tile are read once into registers at phase
// 0 (8x ds_read_b128); each phase reads its A quadrant (4x ds_read_b128)
// and issues 16 MFMA (2 fm x 4 fn x 2 kc).
#include <cstdlib>

#include "common.h"

namespace lightctr {

typedef __attribute__((ext_vector_type(8))) __bf16 p8bf16x8;
typedef __attribute__((ext_vector_type(4))) float p8f32x4;

#define P8_BK 64

__device__ __forceinline__ int p8swz(int row, int ke) {
  return ke ^ ((row & 7) << 3);
}

// stage one 128-row x 64-col bf16 half-tile with 512 threads (2 glds each)
__device__ __forceinline__ void p8_stage_half(const __bf16* __restrict__ gbase,
                                              long stride, __bf16* dst) {
  const int tid = threadIdx.x;
  const int lane = tid & 63;
  const int wave = tid >> 6;
#pragma unroll
  for (int it = 0; it < 2; ++it) {
    const int seg = wave * 2 + it;  // 16 segments of 8 rows
    const int row = seg * 8 + (lane >> 3);
    const int ke = p8swz(row, (lane & 7) * 8);
    const __bf16* src = gbase + (long)row * stride + ke;
    __builtin_amdgcn_global_load_lds(
        (const __attribute__((address_space(1))) void*)src,
        (__attribute__((address_space(3))) void*)(dst + seg * 512), 16, 0, 0);
  }
}

// K-tail half-tile stage: the swizzled source chunk of each lane is either
// fully inside [0, krem) (8-element chunks, krem % 8 == 0) -> masked glds,
// or fully outside -> 16 B LDS zero store. Every row keeps krem/8 >= 1
// active glds lanes, so both glds instructions always issue and the
// per-wave vmcnt bookkeeping matches p8_stage_half exactly.
__device__ __forceinline__ void p8_stage_half_tail(
    const __bf16* __restrict__ gbase, long stride, int krem, __bf16* dst) {
  const int tid = threadIdx.x;
  const int lane = tid & 63;
  const int wave = tid >> 6;
#pragma unroll
  for (int it = 0; it < 2; ++it) {
    const int seg = wave * 2 + it;
    const int row = seg * 8 + (lane >> 3);
    const int ke = p8swz(row, (lane & 7) * 8);
    if (ke >= krem) {
      // this lane's linear LDS slot holds an out-of-range source chunk
      *(p8f32x4*)(dst + seg * 512 + lane * 8) = p8f32x4{0.f, 0.f, 0.f, 0.f};
    }
    const __bf16* src = gbase + (long)row * stride + ke;
    if (ke < krem) {
      __builtin_amdgcn_global_load_lds(
          (const __attribute__((address_space(1))) void*)src,
          (__attribute__((address_space(3))) void*)(dst + seg * 512), 16, 0,
          0);
    }
  }
}

__device__ __forceinline__ float p8_act(float v, int act) {
  if (act == 1) return fmaxf(v, 0.f);
  if (act == 2) return sigmoidf_clamped(v);
  return v;
}

// LITE: same staging pipeline and counted boundary vmcnt, but only TWO
// barriers per K-tile (after phase 1 — publishing every wave's b-frag
// consumption before B(t+2) overwrites the slot — and at the tile
// boundary) instead of the template's eight; the compiler pipelines the
// phases freely as in the round-1 kernel. A/B lever for the big square
// shapes where the 8-barrier variant measured 9% behind round 1.
// DEEPA: A triple-buffered (3 parity slots, 160 KB LDS total) so A can
// stage TWO tiles ahead like B; the boundary wait loosens from
// vmcnt(4) to vmcnt(8) (one full tile of loads stays in flight).
// APIPE (LCTR_GEMM_P8_APIPE=1, LITE only): phase-pipeline the per-phase
// A-fragment ds_reads one MFMA cluster ahead through two named register
// buffers, same transform as gemm256's LCTR_GEMM_APIPE (which measured
// +2.3% @8192^3 — profiles/r2_09_gemm_apipe.txt). The a-reads target
// curA, which no staging touches during the tile, so issuing them
// before the previous phase's MFMA cluster is hazard-free; the LITE
// mid-barrier constraint (B parity-slot overwrite) only orders b-frag
// reads vs phase-2 staging and is preserved.
template <bool TAIL, bool LITE, bool XCD, bool DEEPA = false,
          bool APIPE = false>
__global__ __launch_bounds__(512, 1) void gemm256p8_bf16_kernel(
    const __bf16* __restrict__ A, const __bf16* __restrict__ Bst,
    const float* __restrict__ bias, float* __restrict__ C,
    __bf16* __restrict__ Cbf, int M, int N, int K, int act) {
  // LDS: A parity slots (2 or 3) then B parity 0/1 (16384 els each)
  __shared__ __bf16 smem[(DEEPA ? 5 : 4) * 256 * P8_BK];
  const int lane = threadIdx.x & 63;
  const int wave = threadIdx.x >> 6;
  const int wm = wave >> 2;  // 0..1
  const int wn = wave & 3;   // 0..3
  int bx = blockIdx.x, by = blockIdx.y;
  if constexpr (XCD) {
    // T1 XCD-aware remap (bijective form, guide §5.5): default blockIdx
    // round-robins the 8 XCDs, so neighbor tiles that share A rows / B
    // cols land on different private L2s; give each XCD a contiguous
    // grid chunk instead
    const int nwg = gridDim.x * gridDim.y;
    const int wg = by * gridDim.x + bx;
    const int q = nwg / 8, r = nwg % 8;
    const int xcd = wg % 8, idx = wg / 8;
    const int nw = (xcd < r ? xcd * (q + 1) : r * (q + 1) + (xcd - r) * q)
                   + idx;
    by = nw / gridDim.x;
    bx = nw - by * gridDim.x;
  }
  const int M0 = by * 256;
  const int N0 = bx * 256;

  p8f32x4 acc[8][4] = {};

  const int NT = (K + P8_BK - 1) / P8_BK;
  const int krem = K - (NT - 1) * P8_BK;  // in (0, 64], % 8 == 0

  __bf16* Abuf0 = smem;
  __bf16* Abuf1 = smem + 16384;
  __bf16* Abuf2 = DEEPA ? smem + 2 * 16384 : nullptr;
  __bf16* Bbuf0 = smem + (DEEPA ? 3 : 2) * 16384;
  __bf16* Bbuf1 = Bbuf0 + 16384;

#define P8_ABUF(tt)                                                        \
  (DEEPA ? ((tt) % 3 == 0 ? Abuf0 : ((tt) % 3 == 1 ? Abuf1 : Abuf2))      \
         : (((tt) & 1) ? Abuf1 : Abuf0))
#define P8_STAGE_A(tt, h)                                                   \
  do {                                                                      \
    __bf16* dst_ = P8_ABUF(tt) + (h) * 8192;                                \
    const __bf16* g_ = A + (long)(M0 + (h) * 128) * K + (tt) * P8_BK;       \
    if (!TAIL || (tt) < NT - 1)                                             \
      p8_stage_half(g_, K, dst_);                                           \
    else                                                                    \
      p8_stage_half_tail(g_, K, krem, dst_);                                \
  } while (0)
#define P8_STAGE_B(tt, h)                                                   \
  do {                                                                      \
    __bf16* dst_ = (((tt) & 1) ? Bbuf1 : Bbuf0) + (h) * 8192;               \
    const __bf16* g_ = Bst + (long)(N0 + (h) * 128) * K + (tt) * P8_BK;     \
    if (!TAIL || (tt) < NT - 1)                                             \
      p8_stage_half(g_, K, dst_);                                           \
    else                                                                    \
      p8_stage_half_tail(g_, K, krem, dst_);                                \
  } while (0)

  // prologue: tile 0 (+ tile 1's B; DEEPA also tile 1's A) ahead
  P8_STAGE_A(0, 0);
  P8_STAGE_A(0, 1);
  P8_STAGE_B(0, 0);
  P8_STAGE_B(0, 1);
  if (NT > 1) {
    if (DEEPA) {
      P8_STAGE_A(1, 0);
      P8_STAGE_A(1, 1);
    }
    P8_STAGE_B(1, 0);
    P8_STAGE_B(1, 1);
    if (DEEPA)
      asm volatile("s_waitcnt vmcnt(8)" ::: "memory");
    else
      asm volatile("s_waitcnt vmcnt(4)" ::: "memory");
  } else {
    asm volatile("s_waitcnt vmcnt(0)" ::: "memory");
  }
  __builtin_amdgcn_s_barrier();

  for (int t = 0; t < NT; ++t) {
    __bf16* curA = P8_ABUF(t);
    __bf16* curB = (t & 1) ? Bbuf1 : Bbuf0;
    const bool sA = DEEPA ? (t + 2 < NT) : (t + 1 < NT);
    const bool sB = t + 2 < NT;

    // ---- phase 0: b-frags (whole tile) + A quadrant 0 ----
    p8bf16x8 b[4][2];
#pragma unroll
    for (int fn = 0; fn < 4; ++fn)
#pragma unroll
      for (int kc = 0; kc < 2; ++kc) {
        const int rb = wn * 64 + fn * 16 + (lane & 15);
        const int ks = kc * 32 + (lane >> 4) * 8;
        b[fn][kc] = *(const p8bf16x8*)&curB[rb * P8_BK + p8swz(rb, ks)];
      }
    p8bf16x8 a[2][2];
#pragma unroll
    for (int fm = 0; fm < 2; ++fm)
#pragma unroll
      for (int kc = 0; kc < 2; ++kc) {
        const int ra = wm * 128 + fm * 16 + (lane & 15);
        const int ks = kc * 32 + (lane >> 4) * 8;
        a[fm][kc] = *(const p8bf16x8*)&curA[ra * P8_BK + p8swz(ra, ks)];
      }
    if (sA) P8_STAGE_A(DEEPA ? t + 2 : t + 1, 0);

#define P8_READ_A_PH(P, ARR)                                             \
  _Pragma("unroll") for (int fm = 0; fm < 2; ++fm)                       \
  _Pragma("unroll") for (int kc = 0; kc < 2; ++kc) {                     \
    const int ra = wm * 128 + ((P)*2 + fm) * 16 + (lane & 15);           \
    const int ks = kc * 32 + (lane >> 4) * 8;                            \
    ARR[fm][kc] = *(const p8bf16x8*)&curA[ra * P8_BK + p8swz(ra, ks)];   \
  }
#define P8_MFMA_PH(P, ARR)                                               \
  __builtin_amdgcn_s_setprio(1);                                         \
  _Pragma("unroll") for (int fm = 0; fm < 2; ++fm)                       \
  _Pragma("unroll") for (int fn = 0; fn < 4; ++fn)                       \
  _Pragma("unroll") for (int kc = 0; kc < 2; ++kc) acc[(P)*2 + fm][fn] = \
      __builtin_amdgcn_mfma_f32_16x16x32_bf16(ARR[fm][kc], b[fn][kc],    \
                                              acc[(P)*2 + fm][fn], 0, 0, \
                                              0);                        \
  __builtin_amdgcn_s_setprio(0);

    if constexpr (APIPE && LITE) {
      p8bf16x8 a2[2][2];
      P8_READ_A_PH(1, a2);  // in flight under phase-0 MFMAs
      P8_MFMA_PH(0, a);
      if (sA) P8_STAGE_A(DEEPA ? t + 2 : t + 1, 1);
      P8_READ_A_PH(2, a);
      P8_MFMA_PH(1, a2);
      // LITE mid-barrier: every wave is past its phase-0 MFMA issue
      // (b-frag registers populated), so phase 2 may overwrite the
      // B(t) parity slot with B(t+2)
      asm volatile("" ::: "memory");
      __builtin_amdgcn_s_barrier();
      asm volatile("" ::: "memory");
      if (sB) P8_STAGE_B(t + 2, 0);
      P8_READ_A_PH(3, a2);
      P8_MFMA_PH(2, a);
      if (sB) P8_STAGE_B(t + 2, 1);
      P8_MFMA_PH(3, a2);
      // tile boundary: counted drain + publishing barrier (as below)
      if (DEEPA && sA && sB)
        asm volatile("s_waitcnt vmcnt(8)" ::: "memory");
      else if (!DEEPA && sB)
        asm volatile("s_waitcnt vmcnt(4)" ::: "memory");
      else if (DEEPA && sB)
        asm volatile("s_waitcnt vmcnt(4)" ::: "memory");
      else
        asm volatile("s_waitcnt vmcnt(0)" ::: "memory");
      __builtin_amdgcn_s_barrier();
      continue;
    }

    if constexpr (!LITE) {
      asm volatile("s_waitcnt lgkmcnt(8)" ::: "memory");
      __builtin_amdgcn_s_barrier();
      asm volatile("s_waitcnt lgkmcnt(0)" ::: "memory");
      __builtin_amdgcn_sched_barrier(0);
    }
    __builtin_amdgcn_s_setprio(1);
#pragma unroll
    for (int fm = 0; fm < 2; ++fm)
#pragma unroll
      for (int fn = 0; fn < 4; ++fn)
#pragma unroll
        for (int kc = 0; kc < 2; ++kc)
          acc[fm][fn] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
              a[fm][kc], b[fn][kc], acc[fm][fn], 0, 0, 0);
    __builtin_amdgcn_s_setprio(0);
    if constexpr (!LITE) __builtin_amdgcn_s_barrier();

    // ---- phases 1..3: A quadrant p ----
#pragma unroll
    for (int p = 1; p < 4; ++p) {
#pragma unroll
      for (int fm = 0; fm < 2; ++fm)
#pragma unroll
        for (int kc = 0; kc < 2; ++kc) {
          const int ra = wm * 128 + (p * 2 + fm) * 16 + (lane & 15);
          const int ks = kc * 32 + (lane >> 4) * 8;
          a[fm][kc] = *(const p8bf16x8*)&curA[ra * P8_BK + p8swz(ra, ks)];
        }
      if (p == 1) {
        if (sA) P8_STAGE_A(DEEPA ? t + 2 : t + 1, 1);
      } else if (p == 2) {
        if (sB) P8_STAGE_B(t + 2, 0);
      } else {
        if (sB) P8_STAGE_B(t + 2, 1);
      }
      if constexpr (!LITE) {
        __builtin_amdgcn_s_barrier();
        asm volatile("s_waitcnt lgkmcnt(0)" ::: "memory");
        __builtin_amdgcn_sched_barrier(0);
      }
      __builtin_amdgcn_s_setprio(1);
#pragma unroll
      for (int fm = 0; fm < 2; ++fm)
#pragma unroll
        for (int fn = 0; fn < 4; ++fn)
#pragma unroll
          for (int kc = 0; kc < 2; ++kc)
            acc[p * 2 + fm][fn] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
                a[fm][kc], b[fn][kc], acc[p * 2 + fm][fn], 0, 0, 0);
      __builtin_amdgcn_s_setprio(0);
      if (p == 1 && LITE) {
        // LITE mid-barrier: every wave is past its phase-0 MFMA issue
        // (b-frag registers populated), so phase 2 may overwrite the
        // B(t) parity slot with B(t+2)
        asm volatile("" ::: "memory");
        __builtin_amdgcn_s_barrier();
        asm volatile("" ::: "memory");
      }
      if (p < 3) {
        if constexpr (!LITE) __builtin_amdgcn_s_barrier();
      } else {
        // tile boundary: counted drain, then the barrier that publishes
        // every wave's landed stages. DEEPA: A(t+2)+B(t+2) may stay in
        // flight (vmcnt(8)); else only B(t+2) (vmcnt(4)).
        if (DEEPA && sA && sB)
          asm volatile("s_waitcnt vmcnt(8)" ::: "memory");
        else if (!DEEPA && sB)
          asm volatile("s_waitcnt vmcnt(4)" ::: "memory");
        else if (DEEPA && sB)
          asm volatile("s_waitcnt vmcnt(4)" ::: "memory");
        else
          asm volatile("s_waitcnt vmcnt(0)" ::: "memory");
        __builtin_amdgcn_s_barrier();
      }
    }
  }
#undef P8_STAGE_A
#undef P8_STAGE_B
#undef P8_READ_A_PH
#undef P8_MFMA_PH

  // epilogue (C/D map: col = lane & 15, row = (lane>>4)*4 + r); bias
  // hoisted to one load per fn column (the round-1 epilogue reloaded it
  // per fragment: 20+ vmcnt(0) stalls)
  float bv[4];
#pragma unroll
  for (int fn = 0; fn < 4; ++fn) {
    const int col = N0 + wn * 64 + fn * 16 + (lane & 15);
    bv[fn] = bias ? bias[col] : 0.f;
  }
#pragma unroll
  for (int fm = 0; fm < 8; ++fm) {
#pragma unroll
    for (int fn = 0; fn < 4; ++fn) {
      const int col = N0 + wn * 64 + fn * 16 + (lane & 15);
#pragma unroll
      for (int r = 0; r < 4; ++r) {
        const int row = M0 + wm * 128 + fm * 16 + (lane >> 4) * 4 + r;
        const float v = p8_act(acc[fm][fn][r] + bv[fn], act);
        C[(size_t)row * N + col] = v;
        if (Cbf) Cbf[(size_t)row * N + col] = (__bf16)v;
      }
    }
  }
}

bool gemm256p8_eligible(int M, int N, int K, int transA, int transB) {
  return transA == 0 && transB == 0 && M % 256 == 0 && N % 256 == 0 &&
         K % 8 == 0 && K >= 32;
}

void gemm256p8_bf16_launch(const void* A, const void* Bst, const float* bias,
                           float* C, void* Cbf, int M, int N, int K, int act,
                           hipStream_t stream) {
  dim3 block(512);
  dim3 grid(N / 256, M / 256);
  // LITE measured faster than the 8-barrier variant at every shape
  // (596 vs 555 TF on the W&D 65536x256x624 tail, 949 vs 904 @4k^3 —
  // profiles/r2_04_gemm_p8lite.txt): default on, =0 reverts.
  static const bool lite = [] {
    const char* e = getenv("LCTR_GEMM_P8_LITE");
    return !(e && e[0] == '0');
  }();
  static const bool xcd = [] {
    const char* e = getenv("LCTR_GEMM_P8_XCD");
    return e && e[0] == '1';
  }();
  const bool tail = (K % P8_BK) != 0;
  // phase-pipelined A reads (LITE only; A/B via LCTR_GEMM_P8_APIPE=0/1,
  // re-read per launch for in-process sweeps)
  const char* ea = getenv("LCTR_GEMM_P8_APIPE");
  const bool apipe = ea && ea[0] == '1';
#define P8_LAUNCH(T, L, X)                                                    hipLaunchKernelGGL((gemm256p8_bf16_kernel<T, L, X>), grid, block, 0,                           stream, (const __bf16*)A, (const __bf16*)Bst, bias, C,                      (__bf16*)Cbf, M, N, K, act)
#define P8_LAUNCH_AP(T)                                                  \
  hipLaunchKernelGGL((gemm256p8_bf16_kernel<T, true, false, false,       \
                                            true>),                     \
                     grid, block, 0, stream, (const __bf16*)A,           \
                     (const __bf16*)Bst, bias, C, (__bf16*)Cbf, M, N, K, \
                     act)
  if (!tail && lite && !xcd && apipe) P8_LAUNCH_AP(false);
  else if (tail && lite && !xcd && apipe) P8_LAUNCH_AP(true);
  else if (!tail && lite && !xcd) P8_LAUNCH(false, true, false);
  else if (!tail && lite && xcd) P8_LAUNCH(false, true, true);
  else if (!tail && !lite && !xcd) P8_LAUNCH(false, false, false);
  else if (!tail && !lite && xcd) P8_LAUNCH(false, false, true);
  else if (tail && lite && !xcd) P8_LAUNCH(true, true, false);
  else if (tail && lite && xcd) P8_LAUNCH(true, true, true);
  else if (tail && !lite && !xcd) P8_LAUNCH(true, false, false);
  else P8_LAUNCH(true, false, true);
#undef P8_LAUNCH
#undef P8_LAUNCH_AP
}

}  // namespace lightctr
