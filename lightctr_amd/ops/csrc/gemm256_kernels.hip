// 256x256-tile phase-split MFMA bf16 GEMM for MI355X/gfx950.
//
// An experiment toward the CDNA4 guide's 8-phase 256^2 template: bigger
// tile (2x the compute per byte of the 128^2 kernel), 8 waves (2Mx4N),
// per-wave 128x64 output (8x4 fragments), double-buffered 128 KiB LDS
// staged by global_load_lds, ONE barrier + one vmcnt drain per K-tile
// (vs 2 barriers per tile in the 128^2 kernel), compute split into 4
// phases (one 2x4-fragment quadrant x full K-tile each) wrapped in
// s_setprio so the CU scheduler favors MFMA-entering waves (T5; needs the
// phase role-split to matter). B fragments for the whole tile are kept in
// registers across phases (read once per tile).
//
// Eligibility: TRANSA=TRANSB=0, M%256==0, N%256==0, K%64==0, K%8==0 —
// dispatched from gemm_bf16_launch; everything else takes the 128^2 path.
#include <cstdlib>

#include "common.h"

namespace lightctr {

typedef __attribute__((ext_vector_type(8))) __bf16 bf16x8;
typedef __attribute__((ext_vector_type(4))) float f32x4;

#define G256_BK 64

__device__ __forceinline__ int swz256(int row, int ke) {
  return ke ^ ((row & 7) << 3);
}

// stage one 128-row x 64-col bf16 half-tile with 512 threads (2 glds each)
__device__ __forceinline__ void stage_half_512(
    const __bf16* __restrict__ gbase, long stride, __bf16* dst) {
  const int tid = threadIdx.x;
  const int lane = tid & 63;
  const int wave = tid >> 6;  // 0..7
#pragma unroll
  for (int it = 0; it < 2; ++it) {
    const int seg = wave * 2 + it;  // 16 segments of 8 rows
    const int row = seg * 8 + (lane >> 3);
    const int ke = swz256(row, (lane & 7) * 8);
    const __bf16* src = gbase + (long)row * stride + ke;
    __builtin_amdgcn_global_load_lds(
        (const __attribute__((address_space(1))) void*)src,
        (__attribute__((address_space(3))) void*)(dst + seg * 512), 16, 0,
        0);
  }
}

// scalar staging for the K-tail tile (row-major source, 512 threads):
// thread t covers row=t/2, k-half=(t%2)*32; zero-fills out-of-range K
__device__ __forceinline__ void stage_tail_512(
    const __bf16* __restrict__ g, long stride, int k0, int K, __bf16* dst) {
  const int tid = threadIdx.x;
  const int r = tid >> 1;  // 0..255
  const int kk = (tid & 1) * 32;
  for (int u = 0; u < 32; ++u) {
    const int ke = kk + u;
    __bf16 v = (__bf16)0.f;
    if (k0 + ke < K) v = g[(long)r * stride + k0 + ke];
    dst[r * G256_BK + swz256(r, ke)] = v;
  }
}

__device__ __forceinline__ float act_apply256(float v, int act) {
  if (act == 1) return fmaxf(v, 0.f);
  if (act == 2) return sigmoidf_clamped(v);
  return v;
}

// BURST: issue all four of tile t+1's half-tile stages at phase 0 so the
// tile-end vmcnt(0) drain waits on loads ~3 phases old instead of on the
// half-tile just issued in phase 3 (A/B via LCTR_GEMM_BURST=1).
//
// PIPE (LCTR_GEMM_APIPE=1): software-pipeline the per-phase A-fragment
// ds_reads one phase ahead. The baseline schedule (ISA-audited) emits
// `ds_read_b128 x4; s_waitcnt lgkmcnt(0); v_mfma x16` per phase — the
// LDS latency of the 4 reads is exposed on every phase because the
// s_setprio intrinsics and the staging branches pin the reads to their
// own phase. Issuing phase p+1's reads into named ping-pong buffers
// BEFORE phase p's MFMA cluster lets the compiler's waitcnt pass emit a
// counted `lgkmcnt(4)` instead (4 newer reads left in flight), hiding
// the read latency under the previous 16-MFMA cluster.

// phase-p A-fragment read into a named buffer (compile-time P keeps the
// acc[] indices constant — runtime indexing would spill to scratch)
#define G256_READ_A(P, AARR)                                             \
  _Pragma("unroll") for (int fm = 0; fm < 2; ++fm)                       \
  _Pragma("unroll") for (int kc = 0; kc < 2; ++kc) {                     \
    const int ra = wm * 128 + ((P)*2 + fm) * 16 + (lane & 15);           \
    const int ks = kc * 32 + (lane >> 4) * 8;                            \
    AARR[fm][kc] = *(const bf16x8*)&curA[ra * G256_BK + swz256(ra, ks)]; \
  }

#define G256_MFMA(P, AARR)                                               \
  __builtin_amdgcn_s_setprio(1);                                         \
  _Pragma("unroll") for (int fm = 0; fm < 2; ++fm)                       \
  _Pragma("unroll") for (int fn = 0; fn < 4; ++fn)                       \
  _Pragma("unroll") for (int kc = 0; kc < 2; ++kc) acc[(P)*2 + fm][fn] = \
      __builtin_amdgcn_mfma_f32_16x16x32_bf16(AARR[fm][kc], b[fn][kc],   \
                                              acc[(P)*2 + fm][fn], 0, 0, \
                                              0);                        \
  __builtin_amdgcn_s_setprio(0);

// PIPE=2 additionally makes the per-phase staging branch-free on the
// full-tile path: the last tile re-stages itself (k0 clamped) instead
// of branching on `more`, so the whole tile body is one basic block and
// the waitcnt pass can count across all four phases.
template <bool TAIL, bool BURST, int PIPE>
__global__ __launch_bounds__(512, 1) void gemm256_bf16_kernel(
    const __bf16* __restrict__ A, const __bf16* __restrict__ Bst,
    const float* __restrict__ bias, float* __restrict__ C,
    __bf16* __restrict__ Cbf, int M, int N, int K, int act) {
  // one __shared__ object: [buf][A(256x64) then B(256x64)]
  __shared__ __bf16 smem[2 * 2 * 256 * G256_BK];
  const int lane = threadIdx.x & 63;
  const int wave = threadIdx.x >> 6;
  const int wm = wave >> 2;      // 0..1
  const int wn = wave & 3;       // 0..3
  // (XCD-aware tile remap was tried here and measured -5% at 4k^3 and 8k^3
  // on this schedule — 1 block/CU with 512-thread blocks leaves little
  // cross-block L2 reuse to recover; plain mapping kept.)
  const int M0 = blockIdx.y * 256;
  const int N0 = blockIdx.x * 256;

  f32x4 acc[8][4] = {};

  const int NT = (K + G256_BK - 1) / G256_BK;
  // buffer b: A at smem + b*32768, B at A + 16384 elems
  // prologue: stage tile 0 (4 half-tiles: A-top, A-bot, B-top, B-bot)
  if (!TAIL || G256_BK <= K) {
    stage_half_512(A + (long)M0 * K, K, smem);
    stage_half_512(A + (long)(M0 + 128) * K, K, smem + 128 * G256_BK);
    stage_half_512(Bst + (long)N0 * K, K, smem + 256 * G256_BK);
    stage_half_512(Bst + (long)(N0 + 128) * K, K, smem + 384 * G256_BK);
  } else {
    stage_tail_512(A + (long)M0 * K, K, 0, K, smem);
    stage_tail_512(Bst + (long)N0 * K, K, 0, K, smem + 256 * G256_BK);
  }
  asm volatile("s_waitcnt vmcnt(0)" ::: "memory");
  __syncthreads();

  int cur = 0;
  for (int t = 0; t < NT; ++t) {
    __bf16* curA = smem + cur * 512 * G256_BK;
    __bf16* curB = curA + 256 * G256_BK;
    __bf16* nxtA = smem + (cur ^ 1) * 512 * G256_BK;
    __bf16* nxtB = nxtA + 256 * G256_BK;
    const int k0 = (t + 1) * G256_BK;
    const bool more = (t + 1) < NT;
    // B fragments for this wave's 64 columns, whole K-tile, kept in regs
    bf16x8 b[4][2];
#pragma unroll
    for (int fn = 0; fn < 4; ++fn)
#pragma unroll
      for (int kc = 0; kc < 2; ++kc) {
        const int rb = wn * 64 + fn * 16 + (lane & 15);
        const int ks = kc * 32 + (lane >> 4) * 8;
        b[fn][kc] = *(const bf16x8*)&curB[rb * G256_BK + swz256(rb, ks)];
      }
    // prefetch half-tile(s) of tile t+1 for phase p (glds issue keeps
    // the one-stage-per-phase cadence; BURST front-loads all four)
    auto stage_phase = [&](int p) {
      if (!more) return;
      const bool next_full = !TAIL || (k0 + G256_BK <= K);
      if (next_full) {
        if (BURST) {
          if (p == 0) {
            stage_half_512(A + (long)M0 * K + k0, K, nxtA);
            stage_half_512(A + (long)(M0 + 128) * K + k0, K,
                           nxtA + 128 * G256_BK);
            stage_half_512(Bst + (long)N0 * K + k0, K, nxtB);
            stage_half_512(Bst + (long)(N0 + 128) * K + k0, K,
                           nxtB + 128 * G256_BK);
          }
        } else if (p == 0) {
          stage_half_512(A + (long)M0 * K + k0, K, nxtA);
        } else if (p == 1) {
          stage_half_512(A + (long)(M0 + 128) * K + k0, K,
                         nxtA + 128 * G256_BK);
        } else if (p == 2) {
          stage_half_512(Bst + (long)N0 * K + k0, K, nxtB);
        } else {
          stage_half_512(Bst + (long)(N0 + 128) * K + k0, K,
                         nxtB + 128 * G256_BK);
        }
      } else {  // K tail: zero-filled scalar staging covers all 256
                // rows per call — phases 0 (A) and 2 (B) only
        if (p == 0)
          stage_tail_512(A + (long)M0 * K, K, k0, K, nxtA);
        else if (p == 2)
          stage_tail_512(Bst + (long)N0 * K, K, k0, K, nxtB);
      }
    };
    // branch-free staging (PIPE=2, full tiles only): clamp so the last
    // tile harmlessly re-stages itself into the unused nxt buffer
    const long k0c = (!TAIL && PIPE == 2)
                         ? (more ? (long)k0 : (long)k0 - G256_BK)
                         : (long)k0;
    auto stage_flat = [&](int p) {
      if (p == 0) stage_half_512(A + (long)M0 * K + k0c, K, nxtA);
      else if (p == 1)
        stage_half_512(A + (long)(M0 + 128) * K + k0c, K,
                       nxtA + 128 * G256_BK);
      else if (p == 2)
        stage_half_512(Bst + (long)N0 * K + k0c, K, nxtB);
      else
        stage_half_512(Bst + (long)(N0 + 128) * K + k0c, K,
                       nxtB + 128 * G256_BK);
    };
    if (PIPE == 2 && !TAIL) {
      bf16x8 aA[2][2], aB[2][2];
      G256_READ_A(0, aA);
      stage_flat(0);
      G256_READ_A(1, aB);
      G256_MFMA(0, aA);
      stage_flat(1);
      G256_READ_A(2, aA);
      G256_MFMA(1, aB);
      stage_flat(2);
      G256_READ_A(3, aB);
      G256_MFMA(2, aA);
      stage_flat(3);
      G256_MFMA(3, aB);
      asm volatile("s_waitcnt vmcnt(0)" ::: "memory");
      __syncthreads();
      cur ^= 1;
      continue;
    }
    if (PIPE) {
      bf16x8 aA[2][2], aB[2][2];
      G256_READ_A(0, aA);
      stage_phase(0);
      G256_READ_A(1, aB);  // in flight under phase-0 MFMAs
      G256_MFMA(0, aA);
      stage_phase(1);
      G256_READ_A(2, aA);
      G256_MFMA(1, aB);
      stage_phase(2);
      G256_READ_A(3, aB);
      G256_MFMA(2, aA);
      stage_phase(3);
      G256_MFMA(3, aB);
      asm volatile("s_waitcnt vmcnt(0)" ::: "memory");
      __syncthreads();
      cur ^= 1;
      continue;
    }
    // 4 phases: quadrant p = fm in [2p, 2p+2), all fn, full K-tile
#pragma unroll
    for (int p = 0; p < 4; ++p) {
      stage_phase(p);
      bf16x8 a[2][2];
#pragma unroll
      for (int fm = 0; fm < 2; ++fm)
#pragma unroll
        for (int kc = 0; kc < 2; ++kc) {
          const int ra = wm * 128 + (p * 2 + fm) * 16 + (lane & 15);
          const int ks = kc * 32 + (lane >> 4) * 8;
          a[fm][kc] =
              *(const bf16x8*)&curA[ra * G256_BK + swz256(ra, ks)];
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
    }
    // drain the prefetch, then one barrier before buffers swap roles
    asm volatile("s_waitcnt vmcnt(0)" ::: "memory");
    __syncthreads();
    cur ^= 1;
  }

  // epilogue (C/D map: col=lane&15, row=(lane>>4)*4+r)
#pragma unroll
  for (int fm = 0; fm < 8; ++fm) {
#pragma unroll
    for (int fn = 0; fn < 4; ++fn) {
      const int col = N0 + wn * 64 + fn * 16 + (lane & 15);
      const float bv = bias ? bias[col] : 0.f;
#pragma unroll
      for (int r = 0; r < 4; ++r) {
        const int row = M0 + wm * 128 + fm * 16 + (lane >> 4) * 4 + r;
        const float v = act_apply256(acc[fm][fn][r] + bv, act);
        C[(size_t)row * N + col] = v;
        if (Cbf) Cbf[(size_t)row * N + col] = (__bf16)v;
      }
    }
  }
}

bool gemm256_eligible(int M, int N, int K, int transA, int transB) {
  // K%8 keeps the glds row stride 16B-aligned for the full tiles; the
  // K%64 tail is handled by scalar staging
  return transA == 0 && transB == 0 && M % 256 == 0 && N % 256 == 0 &&
         K % 8 == 0 && K >= 32;
}

void gemm256_bf16_launch(const void* A, const void* Bst, const float* bias,
                         float* C, void* Cbf, int M, int N, int K, int act,
                         hipStream_t stream) {
  dim3 block(512);
  dim3 grid(N / 256, M / 256);
  static const bool burst = [] {
    const char* e = getenv("LCTR_GEMM_BURST");
    return e && e[0] == '1';
  }();
  // LCTR_GEMM_APIPE=0|1|2 selects the per-phase A-read schedule
  // (2 = pipelined + branch-free staging; re-read per launch so
  // in-process A/B sweeps can toggle it). Default 2 for full tiles:
  // within-process A/B (profiles/r2_09_gemm_apipe.txt) shows +2.3% at
  // 8192^3 (1233 vs 1208 TF, 8/8 passes ahead) and no regression at
  // 4096^3. TAIL shapes default to the round-1 schedule (rarely hit —
  // K%64!=0 dispatches to the p8 kernel upstream).
  const char* ep = getenv("LCTR_GEMM_APIPE");
  const int pipe = ep ? (ep[0] == '2' ? 2 : (ep[0] == '1' ? 1 : 0)) : 2;
#define G256_LAUNCH(T, B, P)                                               \
  hipLaunchKernelGGL((gemm256_bf16_kernel<T, B, P>), grid, block, 0,       \
                     stream, (const __bf16*)A, (const __bf16*)Bst, bias,   \
                     C, (__bf16*)Cbf, M, N, K, act)
  if (K % G256_BK == 0) {
    if (burst) {
      if (pipe == 2) G256_LAUNCH(false, true, 2);
      else if (pipe == 1) G256_LAUNCH(false, true, 1);
      else G256_LAUNCH(false, true, 0);
    } else {
      if (pipe == 2) G256_LAUNCH(false, false, 2);
      else if (pipe == 1) G256_LAUNCH(false, false, 1);
      else G256_LAUNCH(false, false, 0);
    }
  } else {
    const bool tail_pipe = ep && pipe;  // only when explicitly requested
    if (burst) {
      if (tail_pipe) G256_LAUNCH(true, true, 1);
      else G256_LAUNCH(true, true, 0);
    } else {
      if (tail_pipe) G256_LAUNCH(true, false, 1);
      else G256_LAUNCH(true, false, 0);
    }
  }
#undef G256_LAUNCH
}

}  // namespace lightctr
