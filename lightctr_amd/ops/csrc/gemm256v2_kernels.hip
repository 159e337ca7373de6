// 256x256-tile counted-vmcnt MFMA bf16 GEMM (v2) for MI355X/gfx950.
//
// The k-chunk-granular pipeline: K is consumed in 32-deep chunks (one
// MFMA K-step). Four LDS slots rotate (chunk c lives in slot c%4); each
// iteration prefetches chunk c+2 with global_load_lds, then waits a
// COUNTED `s_waitcnt vmcnt(8)` (= the 2 younger chunks' 8 glds stay in
// flight) before a RAW `s_barrier` — the guide's T3+T4 pattern: loads span
// barriers instead of draining per tile. Safe because every wave runs the
// identical schedule, so after the barrier the waited-for chunk is
// globally complete, and a slot is rewritten only 4 chunks (>= 2 barriers)
// after its last read.
//
// Same wave geometry as gemm256 v1 (8 waves 2Mx4N, 128x64 per wave,
// fragment C/D map col=lane&15,row=(lane>>4)*4+r); per chunk each wave
// reads 8 A-frags + 4 B-frags and issues 32 MFMAs in an s_setprio cluster.
// Dispatched for TA=TB=0, M%256==0, N%256==0, K%64==0, K>=192.
#include "common.h"

namespace lightctr {

typedef __attribute__((ext_vector_type(8))) __bf16 bf16x8;
typedef __attribute__((ext_vector_type(4))) float f32x4;

#define V2_CK 32  // k-chunk depth (one mfma_f32_16x16x32 step)

// LDS image for a [256 rows][32 k] chunk: row pairs packed into 128 B
// image rows with a (row>>2)-keyed 16 B-granule XOR —
//   ipos(row, ke) = (row>>1)*64 + (row&1)*32 + (ke ^ (((row>>2)&3)<<3))
// Bank check for the fragment read (16 lanes = rows ra..ra+15, same ks):
// dword = (row>>1)*32 + (row&1)*16 + swz'd/2 -> row&3 picks one of 4 bank
// groups {0,16,32,48}, and within a group rows differ by 4 so the
// (row>>2)&3 XOR spreads them over the 4 granule slots: 16 distinct banks.
__device__ __forceinline__ int ipos_v2(int row, int ke) {
  return (row >> 1) * 64 + (row & 1) * 32 + (ke ^ (((row >> 2) & 3) << 3));
}

// stage one [256 rows][32 k] bf16 chunk with 512 threads (2 glds each).
// glds writes lane-linear (p = seg*512 + lane*8 elems); invert ipos to get
// the per-lane global source (rule 21: swizzle the SOURCE, keep LDS linear)
__device__ __forceinline__ void stage_chunk_512(
    const __bf16* __restrict__ gbase, long stride, __bf16* dst) {
  const int tid = threadIdx.x;
  const int lane = tid & 63;
  const int wave = tid >> 6;
#pragma unroll
  for (int it = 0; it < 2; ++it) {
    const int seg = wave * 2 + it;  // 16 segments of 512 elements
    const int p = seg * 512 + lane * 8;
    const int imgrow = p >> 6;
    const int within = p & 63;
    const int row = imgrow * 2 + (within >> 5);
    const int ke = (within & 31) ^ (((row >> 2) & 3) << 3);
    const __bf16* src = gbase + (long)row * stride + ke;
    __builtin_amdgcn_global_load_lds(
        (const __attribute__((address_space(1))) void*)src,
        (__attribute__((address_space(3))) void*)(dst + seg * 512), 16, 0,
        0);
  }
}

__device__ __forceinline__ float act_v2(float v, int act) {
  if (act == 1) return fmaxf(v, 0.f);
  if (act == 2) return sigmoidf_clamped(v);
  return v;
}

__global__ __launch_bounds__(512, 1) void gemm256v2_bf16_kernel(
    const __bf16* __restrict__ A, const __bf16* __restrict__ Bst,
    const float* __restrict__ bias, float* __restrict__ C,
    __bf16* __restrict__ Cbf, int M, int N, int K, int act) {
  // one __shared__ object: 4 slots x (A[256][32] + B[256][32])
  __shared__ __bf16 smem[4 * 2 * 256 * V2_CK];
  const int lane = threadIdx.x & 63;
  const int wave = threadIdx.x >> 6;
  const int wm = wave >> 2;
  const int wn = wave & 3;
  const int M0 = blockIdx.y * 256;
  const int N0 = blockIdx.x * 256;

  f32x4 acc[8][4] = {};
  const int NC = K / V2_CK;

  // prologue: chunks 0 and 1 (8 glds per thread)
#pragma unroll
  for (int c = 0; c < 2; ++c) {
    __bf16* slotA = smem + (size_t)(c & 3) * 2 * 256 * V2_CK;
    stage_chunk_512(A + (long)M0 * K + c * V2_CK, K, slotA);
    stage_chunk_512(Bst + (long)N0 * K + c * V2_CK, K,
                    slotA + 256 * V2_CK);
  }

  for (int c = 0; c < NC; ++c) {
    if (c + 2 < NC) {  // prefetch chunk c+2 into slot (c+2)%4
      __bf16* slotA = smem + (size_t)((c + 2) & 3) * 2 * 256 * V2_CK;
      stage_chunk_512(A + (long)M0 * K + (c + 2) * V2_CK, K, slotA);
      stage_chunk_512(Bst + (long)N0 * K + (c + 2) * V2_CK, K,
                      slotA + 256 * V2_CK);
      // chunks c+1 and c+2 in flight (8 glds) while we wait for chunk c
      asm volatile("s_waitcnt vmcnt(8)" ::: "memory");
    } else if (c + 1 < NC) {
      asm volatile("s_waitcnt vmcnt(4)" ::: "memory");
    } else {
      asm volatile("s_waitcnt vmcnt(0)" ::: "memory");
    }
    __builtin_amdgcn_s_barrier();

    const __bf16* slotA = smem + (size_t)(c & 3) * 2 * 256 * V2_CK;
    const __bf16* slotB = slotA + 256 * V2_CK;
    const int ks = (lane >> 4) * 8;  // this lane's 8-deep k slice of 32
    bf16x8 b[4];
#pragma unroll
    for (int fn = 0; fn < 4; ++fn) {
      const int rb = wn * 64 + fn * 16 + (lane & 15);
      b[fn] = *(const bf16x8*)&slotB[ipos_v2(rb, ks)];
    }
    bf16x8 a[8];
#pragma unroll
    for (int fm = 0; fm < 8; ++fm) {
      const int ra = wm * 128 + fm * 16 + (lane & 15);
      a[fm] = *(const bf16x8*)&slotA[ipos_v2(ra, ks)];
    }
    __builtin_amdgcn_s_setprio(1);
#pragma unroll
    for (int fm = 0; fm < 8; ++fm)
#pragma unroll
      for (int fn = 0; fn < 4; ++fn)
        acc[fm][fn] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
            a[fm], b[fn], acc[fm][fn], 0, 0, 0);
    __builtin_amdgcn_s_setprio(0);
  }

  // all LDS reads done; no barrier needed before epilogue (no LDS writes
  // follow)
#pragma unroll
  for (int fm = 0; fm < 8; ++fm) {
#pragma unroll
    for (int fn = 0; fn < 4; ++fn) {
      const int col = N0 + wn * 64 + fn * 16 + (lane & 15);
      const float bv = bias ? bias[col] : 0.f;
#pragma unroll
      for (int r = 0; r < 4; ++r) {
        const int row = M0 + wm * 128 + fm * 16 + (lane >> 4) * 4 + r;
        const float v = act_v2(acc[fm][fn][r] + bv, act);
        C[(size_t)row * N + col] = v;
        if (Cbf) Cbf[(size_t)row * N + col] = (__bf16)v;
      }
    }
  }
}

bool gemm256v2_eligible(int M, int N, int K, int transA, int transB) {
  return transA == 0 && transB == 0 && M % 256 == 0 && N % 256 == 0 &&
         K % 64 == 0 && K >= 192;
}

void gemm256v2_bf16_launch(const void* A, const void* Bst, const float* bias,
                           float* C, void* Cbf, int M, int N, int K, int act,
                           hipStream_t stream) {
  dim3 block(512);
  dim3 grid(N / 256, M / 256);
  hipLaunchKernelGGL(gemm256v2_bf16_kernel, grid, block, 0, stream,
                     (const __bf16*)A, (const __bf16*)Bst, bias, C,
                     (__bf16*)Cbf, M, N, K, act);
}

}  // namespace lightctr
