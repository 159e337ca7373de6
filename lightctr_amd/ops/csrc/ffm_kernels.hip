// Fused FFM (field-aware factorization machine) kernels for MI355X/gfx950.
//
// Capability parity with the reference Train_FFM_Algo
// (/root/reference/LightCTR/train/train_ffm_algo.cpp:51-118: explicit O(n^2)
// pairwise dot(V[fi,field_j], V[fj,field_i])*x_i*x_j forward and symmetric
// gradient accumulation) — redesigned for CDNA4:
//   * one 64-lane wavefront per CSR row; lane = (pair_group, factor k) so
//     G = 64/K pairs are evaluated in parallel, each pair's K-dot reduced
//     with __shfl_xor inside its aligned K-lane group
//   * pair index -> (i, j) decoded with a sqrt + fixup (no tables, any row
//     length), V slices streamed from HBM (each V[fi,field_j] is used by
//     exactly one pair: no reuse => no LDS staging, stay bandwidth-bound)
//   * backward scatters symmetric grads with fp32 atomics + touched bitmap
//     (optimizer = generic sparse fused kernels in misc_kernels.hip).
#include <hip/hip_fp16.h>
#include <cstdlib>

#include "common.h"

namespace lightctr {

typedef __attribute__((ext_vector_type(4))) __bf16 ffm_bf16x4;

// V-element abstraction: the FFM kernels run on fp32 V (master weights)
// or on a bf16 V mirror (BASELINE config #3 "FFM bf16": fp32 master +
// optimizer, bf16 compute reads — halves every coalesced V byte).
template <typename VT>
struct VQuad;
template <>
struct VQuad<float> {
  // 4 consecutive elements as float4 (16 B load)
  static __device__ __forceinline__ float4 load(const float* p) {
    return *(const float4*)p;
  }
};
template <>
struct VQuad<__bf16> {
  // 4 consecutive elements in 8 B, widened to fp32
  static __device__ __forceinline__ float4 load(const __bf16* p) {
    const ffm_bf16x4 h = *(const ffm_bf16x4*)p;
    return make_float4((float)h.x, (float)h.y, (float)h.z, (float)h.w);
  }
};

// Per-V-type staging codec for the row_emit kernel: fp32 V stages as fp16
// (halved LDS), bf16 V stages raw (no conversion at all).
template <typename VT>
struct FfmStage;
template <>
struct FfmStage<float> {
  using ST = _Float16;
  static __device__ __forceinline__ uint2 pack(const float* src) {
    const float4 v = *(const float4*)src;
    union {
      __half2 h[2];
      uint2 u;
    } pk;
    pk.h[0] = __floats2half2_rn(v.x, v.y);
    pk.h[1] = __floats2half2_rn(v.z, v.w);
    return pk.u;
  }
  static __device__ __forceinline__ float4 unpack(uint2 u) {
    union {
      __half2 h[2];
      uint2 u2;
    } in;
    in.u2 = u;
    const float2 a = __half22float2(in.h[0]);
    const float2 b = __half22float2(in.h[1]);
    return make_float4(a.x, a.y, b.x, b.y);
  }
};
template <>
struct FfmStage<__bf16> {
  using ST = __bf16;
  static __device__ __forceinline__ uint2 pack(const __bf16* src) {
    return *(const uint2*)src;  // raw copy, 4 elements
  }
  static __device__ __forceinline__ float4 unpack(uint2 u) {
    union {
      ffm_bf16x4 h;
      uint2 u2;
    } in;
    in.u2 = u;
    return make_float4((float)in.h.x, (float)in.h.y, (float)in.h.z,
                       (float)in.h.w);
  }
};

// start offset of pair-row i in the flattened (i<j) enumeration of n items
__device__ __forceinline__ int tri_start(int i, int n) {
  return i * (n - 1) - (i * (i - 1)) / 2;
}

// decode flat pair index p -> (i, j), 0 <= i < j < n
__device__ __forceinline__ void tri_decode(int p, int n, int* pi, int* pj) {
  const float nf = (float)n - 0.5f;
  int i = (int)floorf(nf - sqrtf(fmaxf(nf * nf - 2.0f * (float)p, 0.f)));
  i = max(0, min(i, n - 2));
  while (i + 1 <= n - 2 && tri_start(i + 1, n) <= p) ++i;
  while (i > 0 && tri_start(i, n) > p) --i;
  *pi = i;
  *pj = i + 1 + (p - tri_start(i, n));
}

template <int K>
__global__ void ffm_forward_kernel(const int* __restrict__ row_ptr,
                                   const int* __restrict__ fields,
                                   const int* __restrict__ fids,
                                   const float* __restrict__ vals,
                                   const float* __restrict__ W,
                                   const float* __restrict__ V,  // [F,nf,K]
                                   float* __restrict__ pred, int nfields,
                                   int B) {
  constexpr int G = LCTR_WAVE / K;  // pairs in parallel
  const int lane = threadIdx.x & (LCTR_WAVE - 1);
  const int row = blockIdx.x * (blockDim.x / LCTR_WAVE) + (threadIdx.x >> 6);
  if (row >= B) return;
  const int g = lane / K;
  const int k = lane % K;
  const int beg = row_ptr[row], end = row_ptr[row + 1];
  const int n = end - beg;

  float lin = 0.f;
  for (int j = beg + lane; j < end; j += LCTR_WAVE) lin += W[fids[j]] * vals[j];

  float acc = 0.f;  // replicated K times within each pair group
  const int npairs = n * (n - 1) / 2;
  for (int p = g; p < npairs; p += G) {
    int i, j;
    tri_decode(p, n, &i, &j);
    const int fi = fids[beg + i], fj = fids[beg + j];
    const int Fi = fields[beg + i], Fj = fields[beg + j];
    float t = V[((size_t)fi * nfields + Fj) * K + k] *
              V[((size_t)fj * nfields + Fi) * K + k];
#pragma unroll
    for (int s = 1; s < K; s <<= 1) t += __shfl_xor(t, s);
    acc += t * vals[beg + i] * vals[beg + j];
  }
  const float tot_pair = wave_reduce_sum(acc) * (1.f / K);
  const float tot_lin = wave_reduce_sum(lin);
  if (lane == 0) pred[row] = tot_lin + tot_pair;
}

// Lane-per-pair forward variant: each lane owns a whole pair, reading
// both K-slices itself (2xK*4 B via float4) and reducing the K-dot in
// registers — no cross-lane shuffles, 8x fewer wave-iterations than the
// (pair-group, k) layout, and 4 KB of independent loads in flight per
// wave-iteration instead of 512 B. Same math, same HBM byte count.
template <int K, typename VT>
__global__ void ffm_forward_pp_kernel(const int* __restrict__ row_ptr,
                                      const int* __restrict__ fields,
                                      const int* __restrict__ fids,
                                      const float* __restrict__ vals,
                                      const float* __restrict__ W,
                                      const VT* __restrict__ V,
                                      float* __restrict__ pred, int nfields,
                                      int B) {
  const int lane = threadIdx.x & (LCTR_WAVE - 1);
  const int row = blockIdx.x * (blockDim.x / LCTR_WAVE) + (threadIdx.x >> 6);
  if (row >= B) return;
  const int beg = row_ptr[row], end = row_ptr[row + 1];
  const int n = end - beg;

  float lin = 0.f;
  for (int j = beg + lane; j < end; j += LCTR_WAVE)
    lin += W[fids[j]] * vals[j];

  float acc = 0.f;
  const int npairs = n * (n - 1) / 2;
  // 2-deep pair pipeline: both pairs' slices issued before either dot is
  // consumed (2x the independent bytes in flight per lane)
  int p = lane;
  for (; p + LCTR_WAVE < npairs; p += 2 * LCTR_WAVE) {
    int i0, j0, i1, j1;
    tri_decode(p, n, &i0, &j0);
    tri_decode(p + LCTR_WAVE, n, &i1, &j1);
    const VT* va0 =
        &V[((size_t)fids[beg + i0] * nfields + fields[beg + j0]) * K];
    const VT* vb0 =
        &V[((size_t)fids[beg + j0] * nfields + fields[beg + i0]) * K];
    const VT* va1 =
        &V[((size_t)fids[beg + i1] * nfields + fields[beg + j1]) * K];
    const VT* vb1 =
        &V[((size_t)fids[beg + j1] * nfields + fields[beg + i1]) * K];
    float t0 = 0.f, t1 = 0.f;
#pragma unroll
    for (int q = 0; q < K / 4; ++q) {
      const float4 a0 = VQuad<VT>::load(va0 + 4 * q);
      const float4 b0 = VQuad<VT>::load(vb0 + 4 * q);
      const float4 a1 = VQuad<VT>::load(va1 + 4 * q);
      const float4 b1 = VQuad<VT>::load(vb1 + 4 * q);
      t0 += a0.x * b0.x + a0.y * b0.y + a0.z * b0.z + a0.w * b0.w;
      t1 += a1.x * b1.x + a1.y * b1.y + a1.z * b1.z + a1.w * b1.w;
    }
    acc += t0 * vals[beg + i0] * vals[beg + j0];
    acc += t1 * vals[beg + i1] * vals[beg + j1];
  }
  for (; p < npairs; p += LCTR_WAVE) {
    int i, j;
    tri_decode(p, n, &i, &j);
    const VT* va =
        &V[((size_t)fids[beg + i] * nfields + fields[beg + j]) * K];
    const VT* vb =
        &V[((size_t)fids[beg + j] * nfields + fields[beg + i]) * K];
    float t = 0.f;
#pragma unroll
    for (int q = 0; q < K / 4; ++q) {
      const float4 a = VQuad<VT>::load(va + 4 * q);
      const float4 b = VQuad<VT>::load(vb + 4 * q);
      t += a.x * b.x + a.y * b.y + a.z * b.z + a.w * b.w;
    }
    acc += t * vals[beg + i] * vals[beg + j];
  }
  const float tot = wave_reduce_sum(acc) + wave_reduce_sum(lin);
  if (lane == 0) pred[row] = tot;
}

template <int K>
__global__ void ffm_backward_kernel(
    const int* __restrict__ row_ptr, const int* __restrict__ fields,
    const int* __restrict__ fids, const float* __restrict__ vals,
    const float* __restrict__ V, const float* __restrict__ dpred,
    float* __restrict__ gradW, float* __restrict__ gradV,
    unsigned long long* __restrict__ touched, int nfields, int B) {
  constexpr int G = LCTR_WAVE / K;
  const int lane = threadIdx.x & (LCTR_WAVE - 1);
  const int row = blockIdx.x * (blockDim.x / LCTR_WAVE) + (threadIdx.x >> 6);
  if (row >= B) return;
  const int g = lane / K;
  const int k = lane % K;
  const int beg = row_ptr[row], end = row_ptr[row + 1];
  const int n = end - beg;
  const float d = dpred[row];

  // linear grads + touched marks (feature-strided over the wave)
  for (int j = beg + lane; j < end; j += LCTR_WAVE) {
    const int fid = fids[j];
    atomicAdd(&gradW[fid], d * vals[j]);
    atomicOr(&touched[fid >> 6], 1ull << (fid & 63));
  }

  const int npairs = n * (n - 1) / 2;
  for (int p = g; p < npairs; p += G) {
    int i, j;
    tri_decode(p, n, &i, &j);
    const int fi = fids[beg + i], fj = fids[beg + j];
    const int Fi = fields[beg + i], Fj = fields[beg + j];
    const size_t oi = ((size_t)fi * nfields + Fj) * K + k;
    const size_t oj = ((size_t)fj * nfields + Fi) * K + k;
    const float s = d * vals[beg + i] * vals[beg + j];
    const float a = V[oi], b = V[oj];
    atomicAdd(&gradV[oi], s * b);
    atomicAdd(&gradV[oj], s * a);
  }
}

// ---------------------------------------------------------------------------
// Sorted FFM backward: entries sorted by fid; each 64-lane wave walks a
// contiguous chunk, accumulating the CURRENT feature's full [nfields, K]
// gradient block in LDS (recomputed from the entry's row: for each other
// feature j, block[Fj,:] += d*x_i*x_j*V[fid_j, F_i, :]) and flushing with
// one atomicAdd sweep per run. Removes the hot-feature global-atomic
// serialization of the naive scatter (the FM lesson, applied field-aware).
// LDS atomics absorb duplicate fields within a row.
// ---------------------------------------------------------------------------
template <int K>
__global__ void ffm_sorted_backward_kernel(
    const int* __restrict__ sorted_fids, const int* __restrict__ perm,
    const int* __restrict__ row_of_entry, const int* __restrict__ row_ptr,
    const int* __restrict__ fields, const int* __restrict__ fids,
    const float* __restrict__ vals, const float* __restrict__ V,
    const float* __restrict__ dpred, float* __restrict__ gradW,
    float* __restrict__ gradV, unsigned long long* __restrict__ touched,
    int nfields, int nnz, int chunk) {
  extern __shared__ float lds_acc[];  // [waves_per_block][nfields*K]
  constexpr int G = LCTR_WAVE / K;
  const int lane = threadIdx.x & 63;
  const int wave_in_blk = threadIdx.x >> 6;
  const int wave = blockIdx.x * (blockDim.x >> 6) + wave_in_blk;
  const int jg = lane / K;
  const int k = lane % K;
  const int base = wave * chunk;
  if (base >= nnz) return;
  const int end = min(base + chunk, nnz);
  // padded field stride (K+1): field rows land on coprime-64 bank offsets,
  // so the 64-lane (field-group, k) atomicAdd pattern is conflict-free
  const int fstride = K + 1;
  float* acc = &lds_acc[wave_in_blk * nfields * fstride];
  const int blk = nfields * K;
  for (int i = lane; i < nfields * fstride; i += LCTR_WAVE) acc[i] = 0.f;
  // no __syncthreads needed: each wave owns its LDS slice; lanes of one
  // wave execute in lockstep

  int cur = -1;
  bool head_ok = false;  // this wave saw the run's global start
  float accw = 0.f;
  // flush: a run whose global head AND tail lie in this chunk owns its
  // feature exclusively -> plain stores into the zeroed slab instead of
  // 312 atomics (the FM walk's interior-store trick; ~95% of runs)
  auto flush_run = [&](bool tail_ok) {
    if (cur < 0) return;
    if (head_ok && tail_ok) {
      for (int i = lane; i < blk; i += LCTR_WAVE) {
        const int src = (i / K) * fstride + (i % K);
        if (acc[src] != 0.f) {
          gradV[(size_t)cur * blk + i] = acc[src];
          acc[src] = 0.f;
        }
      }
      if (lane == 0) gradW[cur] = accw;
    } else {
      for (int i = lane; i < blk; i += LCTR_WAVE) {
        const int src = (i / K) * fstride + (i % K);
        if (acc[src] != 0.f) {
          atomicAdd(&gradV[(size_t)cur * blk + i], acc[src]);
          acc[src] = 0.f;
        }
      }
      if (lane == 0) atomicAdd(&gradW[cur], accw);
    }
  };
  for (int e = base; e < end; ++e) {
    const int fid = sorted_fids[e];
    if (fid != cur) {
      flush_run(true);  // next entry differs -> the run ends here
      cur = fid;
      accw = 0.f;
      head_ok = (e == 0 || sorted_fids[e - 1] != fid);
      if (lane == 0 && head_ok) {
        atomicOr(&touched[fid >> 6], 1ull << (fid & 63));
      }
    }
    const int p = perm[e];
    const int r = row_of_entry[p];
    const int Fi = fields[p];
    const float xi = vals[p];
    const float d = dpred[r];
    const int beg = row_ptr[r], rend = row_ptr[r + 1];
    // lane-per-partner (vs (partner-group, k)): each lane pulls its
    // partner's whole K-slice with float4 loads — 8x fewer iterations,
    // 4 KB of independent loads in flight per wave (same win as the
    // lane-per-pair forward, tools/ab_ffm_fwd.py)
    for (int j = beg + lane; j < rend; j += LCTR_WAVE) {
      if (j == p) continue;
      const float4* vb =
          (const float4*)&V[((size_t)fids[j] * nfields + Fi) * K];
      const float sxl = d * xi * vals[j];
      const int fb = fields[j] * fstride;
#pragma unroll
      for (int q = 0; q < K / 4; ++q) {
        const float4 b = vb[q];
        atomicAdd(&acc[fb + 4 * q + 0], sxl * b.x);
        atomicAdd(&acc[fb + 4 * q + 1], sxl * b.y);
        atomicAdd(&acc[fb + 4 * q + 2], sxl * b.z);
        atomicAdd(&acc[fb + 4 * q + 3], sxl * b.w);
      }
    }
    if (lane == 0) accw += d * xi;
  }
  flush_run(end >= nnz || sorted_fids[end] != cur);
}

// ---------------------------------------------------------------------------
// Block-emit FFM backward, phase 1: ONE WAVE PER ENTRY (maximal
// parallelism — no serial chunk walk), computing the entry's full
// [nfields, K] gradient block in its private LDS slice and writing it
// coalesced to gblocks[p*D..]. Phase 2 (ffm_blocks_apply_kernel) walks the
// fid-sorted order gathering CONTIGUOUS blocks (sequential 1.2 KB reads,
// unlike the recompute path's dependent random gathers) and
// segment-reduces into the slabs.
// ---------------------------------------------------------------------------
template <int K>
__global__ void ffm_block_emit_kernel(
    const int* __restrict__ row_of_entry, const int* __restrict__ row_ptr,
    const int* __restrict__ fields, const int* __restrict__ fids,
    const float* __restrict__ vals, const float* __restrict__ V,
    const float* __restrict__ dpred, float* __restrict__ gblocks,
    float* __restrict__ gw, int nfields, int nnz) {
  extern __shared__ float lds_acc[];
  constexpr int G = LCTR_WAVE / K;
  const int lane = threadIdx.x & 63;
  const int wave_in_blk = threadIdx.x >> 6;
  const int p = blockIdx.x * (blockDim.x >> 6) + wave_in_blk;
  if (p >= nnz) return;
  const int jg = lane / K;
  const int k = lane % K;
  const int fstride = K + 1;
  float* acc = &lds_acc[wave_in_blk * nfields * fstride];
  for (int i = lane; i < nfields * fstride; i += LCTR_WAVE) acc[i] = 0.f;

  const int r = row_of_entry[p];
  const int Fi = fields[p];
  const float xi = vals[p];
  const float d = dpred[r];
  const int beg = row_ptr[r], rend = row_ptr[r + 1];
  for (int j = beg + jg; j < rend; j += G) {
    if (j == p) continue;
    const float v = V[((size_t)fids[j] * nfields + Fi) * K + k];
    atomicAdd(&acc[fields[j] * fstride + k], d * xi * vals[j] * v);
  }
  const int D = nfields * K;
  for (int i = lane; i < D; i += LCTR_WAVE)
    gblocks[(size_t)p * D + i] = acc[(i / K) * fstride + (i % K)];
  if (lane == 0) gw[p] = d * xi;
}

// Phase 2: segment-reduce fid-sorted blocks into the grad slabs.
__global__ void ffm_blocks_apply_kernel(
    const int* __restrict__ sorted_fids, const int* __restrict__ perm,
    const float* __restrict__ gblocks, const float* __restrict__ gw,
    float* __restrict__ gradW, float* __restrict__ gradV,
    unsigned long long* __restrict__ touched, int D, int nnz, int chunk) {
  extern __shared__ float lds_acc[];
  const int lane = threadIdx.x & 63;
  const int wave_in_blk = threadIdx.x >> 6;
  const int wave = blockIdx.x * (blockDim.x >> 6) + wave_in_blk;
  const int base = wave * chunk;
  if (base >= nnz) return;
  const int end = min(base + chunk, nnz);
  float* acc = &lds_acc[wave_in_blk * D];
  for (int i = lane; i < D; i += LCTR_WAVE) acc[i] = 0.f;

  int cur = -1;
  float accw = 0.f;
  for (int e = base; e < end; ++e) {
    const int fid = sorted_fids[e];
    if (fid != cur) {
      if (cur >= 0) {
        for (int i = lane; i < D; i += LCTR_WAVE) {
          if (acc[i] != 0.f) {
            atomicAdd(&gradV[(size_t)cur * D + i], acc[i]);
            acc[i] = 0.f;
          }
        }
        if (lane == 0) atomicAdd(&gradW[cur], accw);
      }
      cur = fid;
      accw = 0.f;
      if (lane == 0 && (e == 0 || sorted_fids[e - 1] != fid)) {
        atomicOr(&touched[fid >> 6], 1ull << (fid & 63));
      }
    }
    const long p = (long)perm[e];
    for (int i = lane; i < D; i += LCTR_WAVE)
      acc[i] += gblocks[(size_t)p * D + i];
    if (lane == 0) accw += gw[p];
  }
  if (cur >= 0) {
    for (int i = lane; i < D; i += LCTR_WAVE) {
      if (acc[i] != 0.f) atomicAdd(&gradV[(size_t)cur * D + i], acc[i]);
    }
    if (lane == 0) atomicAdd(&gradW[cur], accw);
  }
}

// ---------------------------------------------------------------------------
// Per-ROW LDS-staged FFM kernels. The direct pair-loop kernels read two
// scattered 32 B V slices per pair; at K=8 that wastes half of every 64 B
// HBM line (measured forward 1.2 ms vs a ~0.4 ms once-read floor). Here a
// WORKGROUP owns one CSR row and first stages every entry's full
// [nfields,K] V row into LDS with fully-coalesced float4 loads (each V
// byte crosses the bus exactly once), then runs the pair loop against
// LDS. Rows longer than `maxn` (duplicate-field outliers) fall back to
// the direct-HBM loop inside the same kernel. Eligibility (launcher):
// staged LDS footprint <= 64 KB.
// ---------------------------------------------------------------------------
template <int K>
__global__ void ffm_fwd_staged_kernel(
    const int* __restrict__ row_ptr, const int* __restrict__ fields,
    const int* __restrict__ fids, const float* __restrict__ vals,
    const float* __restrict__ W, const float* __restrict__ V,
    float* __restrict__ pred, int nfields, int B, int maxn) {
  extern __shared__ float lds[];  // [maxn * nfields * K] staged V rows
  __shared__ float red[8][2];
  constexpr int G = LCTR_WAVE / K;
  const int row = blockIdx.x;
  if (row >= B) return;
  const int tid = threadIdx.x;
  const int lane = tid & 63;
  const int wv = tid >> 6;
  const int nw = blockDim.x >> 6;
  const int g = lane / K;
  const int k = lane % K;
  const int beg = row_ptr[row], end = row_ptr[row + 1];
  const int n = end - beg;
  const int D = nfields * K;
  const int npairs = n * (n - 1) / 2;

  float lin = 0.f;
  for (int j = beg + tid; j < end; j += blockDim.x)
    lin += W[fids[j]] * vals[j];

  float acc = 0.f;
  const bool staged = n <= maxn;
  if (staged) {
    const int nf4 = D / 4;  // K >= 4 -> D % 4 == 0
    float4* st4 = (float4*)lds;
    for (int idx = tid; idx < n * nf4; idx += blockDim.x) {
      const int e = idx / nf4, q = idx - e * nf4;
      st4[idx] = ((const float4*)(V + (size_t)fids[beg + e] * D))[q];
    }
  }
  __syncthreads();
  for (int p = wv * G + g; p < npairs; p += nw * G) {
    int i, j;
    tri_decode(p, n, &i, &j);
    const int Fi = fields[beg + i], Fj = fields[beg + j];
    float t;
    if (staged) {
      t = lds[(i * nfields + Fj) * K + k] * lds[(j * nfields + Fi) * K + k];
    } else {
      t = V[((size_t)fids[beg + i] * nfields + Fj) * K + k] *
          V[((size_t)fids[beg + j] * nfields + Fi) * K + k];
    }
#pragma unroll
    for (int sft = 1; sft < K; sft <<= 1) t += __shfl_xor(t, sft);
    acc += t * vals[beg + i] * vals[beg + j];
  }
  acc = wave_reduce_sum(acc) * (1.f / K);
  lin = wave_reduce_sum(lin);
  if (lane == 0) {
    red[wv][0] = acc;
    red[wv][1] = lin;
  }
  __syncthreads();
  if (tid == 0) {
    float a = 0.f, l = 0.f;
    for (int w = 0; w < nw; ++w) {
      a += red[w][0];
      l += red[w][1];
    }
    pred[row] = l + a;
  }
}

// Backward phase 1, per-row staged: emit each entry's [nfields,K]
// gradient block in HALF precision (the block tensor is the dominant
// traffic: fp16 halves both the emit write and the apply read; gradient
// noise >> fp16 rounding).
//
// Round-2 rewrite (docs/STATUS.md lever 1, ISA-audit driven): the round-1
// version compiled to fully scalar staging (one global_load_dword +
// ds_write_b16 per element), scalar ds_read_b16 partner reads behind LDS
// atomics, and 2-byte global_store_short emits. This version:
//   * stages with float4 loads packed to half2 pairs -> one ds_write_b64
//     per 4 elements (4x fewer instructions, full 16 B coalescing)
//   * builds a field -> entry map for the row; on the (overwhelmingly
//     common) duplicate-free row the emit needs NO accumulation at all:
//     output quad (f, 4k) is  d*x_i*x_j * V_h[fid_j, F_i, 4k]  for the
//     single partner j in field f — read with one ds_read_b64
//   * stores the block as uint2 (dwordx2, 512 B/wave/inst) instead of
//     global_store_short (128 B/wave/inst)
// Rows with duplicate fields (or n > maxn) fall back to the round-1
// atomic-accumulate loop, which stays bit-correct for that case.
template <int K, typename VT>
__global__ void ffm_row_emit_kernel(
    const int* __restrict__ row_ptr, const int* __restrict__ fields,
    const int* __restrict__ fids, const float* __restrict__ vals,
    const VT* __restrict__ V, const float* __restrict__ dpred,
    _Float16* __restrict__ gblocks, float* __restrict__ gw, int nfields,
    int B, int maxn, float scale) {
  // LDS layout: acc[nw][nf*(K+1)] fp32 (dup fallback only) | stage
  // [maxn*D] (fp16 for fp32 V, raw bf16 for bf16 V) | vals_s[maxn] fp32 |
  // fmap[nfields] i32 | dup flag. `scale` (a power of two) is folded into
  // the emitted fp16 blocks so B=65536-scaled logloss gradients (~1e-8)
  // stay inside fp16 normal range; the apply kernel divides it back out.
  using ST = typename FfmStage<VT>::ST;
  extern __shared__ float lds[];
  constexpr int G = LCTR_WAVE / K;
  const int row = blockIdx.x;
  if (row >= B) return;
  const int tid = threadIdx.x;
  const int lane = tid & 63;
  const int wv = tid >> 6;
  const int nw = blockDim.x >> 6;
  const int g = lane / K;
  const int k = lane % K;
  const int beg = row_ptr[row], end = row_ptr[row + 1];
  const int n = end - beg;
  const int D = nfields * K;
  const int fstride = K + 1;
  float* acc = lds + (size_t)wv * nfields * fstride;
  ST* stage = (ST*)(lds + (size_t)nw * nfields * fstride);
  float* vals_s = (float*)(stage + (size_t)maxn * D);
  int* fmap = (int*)(vals_s + maxn);
  int* dupflag = fmap + nfields;
  const float d = dpred[row];

  const bool staged = n <= maxn;
  if (staged) {
    // vectorized stage: 16 B (fp32) or 8 B (bf16) V quad -> one 8 B LDS
    // store
    const int nq = D / 4;  // K % 4 == 0 always (K in {4,8,16,32,64})
    uint2* st2 = (uint2*)stage;
    for (int idx = tid; idx < n * nq; idx += blockDim.x) {
      const int e = idx / nq, q = idx - e * nq;
      st2[idx] = FfmStage<VT>::pack(V + (size_t)fids[beg + e] * D + 4 * q);
    }
  }
  for (int f = tid; f < nfields; f += blockDim.x) fmap[f] = -1;
  if (tid == 0) *dupflag = 0;
  __syncthreads();
  for (int e = tid; e < n; e += blockDim.x) {
    vals_s[e] = vals[beg + e];
    const int old = atomicExch(&fmap[fields[beg + e]], e);
    if (old >= 0) atomicOr(dupflag, 1);
  }
  __syncthreads();

  if (staged && !*dupflag) {
    // fast path: every field has at most one entry -> direct quad emit
    const int nq = D / 4;
    for (int i = wv; i < n; i += nw) {
      const int p = beg + i;
      const int Fi = fields[p];
      const float dxi = d * vals_s[i];
      uint2* out2 = (uint2*)(gblocks + (size_t)p * D);
      for (int u = lane; u < nq; u += LCTR_WAVE) {
        const int f = (u * 4) / K;
        const int j = fmap[f];
        uint2 o = make_uint2(0u, 0u);
        if (j >= 0 && j != i) {
          union {
            __half2 h[2];
            uint2 u2;
          } ov;
          const float4 v = FfmStage<VT>::unpack(
              *(const uint2*)&stage[(size_t)(j * nfields + Fi) * K +
                                    ((u * 4) & (K - 1))]);
          const float s = dxi * vals_s[j] * scale;
          ov.h[0] = __floats2half2_rn(v.x * s, v.y * s);
          ov.h[1] = __floats2half2_rn(v.z * s, v.w * s);
          o = ov.u2;
        }
        out2[u] = o;
      }
      if (lane == 0) gw[p] = dxi;
    }
    return;
  }

  // fallback: duplicate fields in the row (or row longer than the stage)
  for (int i = wv; i < n; i += nw) {
    for (int t = lane; t < nfields * fstride; t += LCTR_WAVE) acc[t] = 0.f;
    const int p = beg + i;
    const int Fi = fields[p];
    const float xi = vals[p];
    for (int j = g; j < n; j += G) {
      if (j == i) continue;
      const float v =
          staged ? (float)stage[(j * nfields + Fi) * K + k]
                 : (float)V[((size_t)fids[beg + j] * nfields + Fi) * K + k];
      atomicAdd(&acc[fields[beg + j] * fstride + k],
                d * xi * vals[beg + j] * v);
    }
    for (int t = lane; t < D; t += LCTR_WAVE)
      gblocks[(size_t)p * D + t] =
          (_Float16)(acc[(t / K) * fstride + (t % K)] * scale);
    if (lane == 0) gw[p] = d * xi;
  }
}

// Backward phase 2 for fp16 blocks: fid-sorted chunked segment reduce with
// the FM interior-store trick — a run whose global head AND tail fall in
// this wave's chunk owns its feature exclusively, so the flush is a plain
// coalesced store sweep (slabs are zeroed by the optimizer pass); only
// chunk-spanning runs use atomicAdd.
// Optional fused optimizer for interior runs (opt_mode 0 = off, 1 =
// Adagrad W+V, 3 = FTRL W + Adagrad V — the FFM production pairings):
// a run whose head AND tail lie in this chunk holds the feature's
// TOTAL batch gradient at flush time, so the update applies in place
// (V/nV [+W z/n state] RMW, optional bf16 mirror refresh) and the fid
// SKIPS the gradV slab, the touched bitmap, and the separate sparse
// optimizer pass entirely. Spanning runs keep the slab+bitmap path.
// FTRL-proximal update (reference gradientUpdater.h:235-278), local copy
// (the misc/fm TU versions are TU-local inlines).
__device__ __forceinline__ void ffm_ftrl_update(float* w, float* z, float* n,
                                                float g, float alpha,
                                                float beta, float l1,
                                                float l2) {
  const float g2 = g * g;
  const float nold = *n;
  const float sigma = (sqrtf(nold + g2) - sqrtf(nold)) / alpha;
  const float znew = *z + g - sigma * (*w);
  const float nnew = nold + g2;
  *z = znew;
  *n = nnew;
  if (fabsf(znew) <= l1) {
    *w = 0.f;
  } else {
    *w = -(znew - copysignf(l1, znew)) / ((beta + sqrtf(nnew)) / alpha + l2);
  }
}

struct FfmOptArgs {
  float* W;
  float* nW;
  float* zW;   // ftrl-W state (mode 3)
  float* nV;
  __bf16* Vh;  // optional bf16 mirror (config #3)
  float p0, p1, p2, p3;  // W: adagrad lr,eps,l2 | ftrl alpha,beta,l1,l2
  float q0, q1, q2;      // V adagrad: lr, eps, l2
};

template <bool FUSED, bool PREF = true, int MAXQ = 4, int WPE = 1>
__global__ __launch_bounds__(256, WPE) void ffm_blocks_apply_f16_kernel(
    const int* __restrict__ sorted_fids, const int* __restrict__ perm,
    const _Float16* __restrict__ gblocks, const float* __restrict__ gw,
    float* __restrict__ gradW, float* __restrict__ gradV,
    unsigned long long* __restrict__ touched, int D, int nnz, int chunk,
    float inv_scale, int opt_mode, float* __restrict__ V, FfmOptArgs oa) {
  // REGISTER accumulator (round 2): each lane owns quads {lane, lane+64,
  // ...} of the [D] block — at most 4 float4 for D <= 1024 (the staged
  // rowemit caps D at ~819). The round-1 LDS accumulator cost a
  // ds_read_b128 + ds_write_b128 round trip per quad per entry, the
  // dominant dependency chain of this latency-bound walk (PMC: VALUBusy
  // 20%, MemUnitStalled 0.4%).
  // MAXQ = quads per lane (template; launcher picks ceil(D/256)): the
  // register footprint scales with it — acc + the 2-deep entry pipeline
  // are 3*MAXQ vec-registers, and at the flagship D=312 only 2 of the
  // round-2 static 4 were ever populated. MAXQ=2 drops the kernel 92 ->
  // ~68 VGPR, 5 -> 7 waves/SIMD on this latency-bound walk.
  const int lane = threadIdx.x & 63;
  const int wave = blockIdx.x * (blockDim.x >> 6) + (threadIdx.x >> 6);
  const int base = wave * chunk;
  if (base >= nnz) return;
  const int end = min(base + chunk, nnz);
  const int nq = D >> 2;  // launcher guarantees D % 4 == 0

  float4 acc[MAXQ];
#pragma unroll
  for (int t = 0; t < MAXQ; ++t) acc[t] = make_float4(0.f, 0.f, 0.f, 0.f);

  int cur = -1;
  bool head_ok = false;
  float accw = 0.f;
  // optimizer-state prefetch (FUSED): the V/nV quad lines of the run
  // being accumulated, issued at run-HEAD detection so their ~600-cycle
  // scattered-load latency overlaps the run's gblock accumulation
  // instead of serializing the flush (the in-flush RMW chain was the
  // diagnosed residual of this latency-bound walk — docs/STATUS.md
  // open lever #2).
  float4 vPre[MAXQ], nPre[MAXQ];
  auto flush = [&](int tail_e) {
    if (cur < 0) return;
    const bool tail_ok = tail_e >= nnz || sorted_fids[tail_e] != cur;
    if (FUSED && head_ok && tail_ok) {
      // exclusive owner: fused V-Adagrad (+W update) in place
      float4* V4 = (float4*)&V[(size_t)cur * D];
      float4* nV4 = (float4*)&oa.nV[(size_t)cur * D];
      ffm_bf16x4* Vh4 =
          oa.Vh ? (ffm_bf16x4*)&oa.Vh[(size_t)cur * D] : nullptr;
#pragma unroll
      for (int t = 0; t < MAXQ; ++t) {
        const int q = t * LCTR_WAVE + lane;
        if (q >= nq) break;
        const float4 ac = acc[t];
        float4 v = PREF ? vPre[t] : V4[q];
        float4 a = PREF ? nPre[t] : nV4[q];
        const float gx = ac.x * inv_scale + oa.q2 * v.x;
        const float gy = ac.y * inv_scale + oa.q2 * v.y;
        const float gz = ac.z * inv_scale + oa.q2 * v.z;
        const float gw2 = ac.w * inv_scale + oa.q2 * v.w;
        a.x += gx * gx;
        a.y += gy * gy;
        a.z += gz * gz;
        a.w += gw2 * gw2;
        v.x -= oa.q0 * gx * __frsqrt_rn(a.x + oa.q1);
        v.y -= oa.q0 * gy * __frsqrt_rn(a.y + oa.q1);
        v.z -= oa.q0 * gz * __frsqrt_rn(a.z + oa.q1);
        v.w -= oa.q0 * gw2 * __frsqrt_rn(a.w + oa.q1);
        nV4[q] = a;
        V4[q] = v;
        if (Vh4) {
          ffm_bf16x4 h;
          h.x = (__bf16)v.x;
          h.y = (__bf16)v.y;
          h.z = (__bf16)v.z;
          h.w = (__bf16)v.w;
          Vh4[q] = h;
        }
        acc[t] = make_float4(0.f, 0.f, 0.f, 0.f);
      }
      if (lane == 0) {
        if (opt_mode == 3) {
          ffm_ftrl_update(&oa.W[cur], &oa.zW[cur], &oa.nW[cur], accw,
                          oa.p0, oa.p1, oa.p2, oa.p3);
        } else {
          const float gw_ = accw + oa.p2 * oa.W[cur];
          const float aw = oa.nW[cur] + gw_ * gw_;
          oa.nW[cur] = aw;
          oa.W[cur] -= oa.p0 * gw_ * __frsqrt_rn(aw + oa.p1);
        }
      }
    } else if (head_ok && tail_ok) {
      float4* gV4 = (float4*)&gradV[(size_t)cur * D];
#pragma unroll
      for (int t = 0; t < MAXQ; ++t) {
        const int q = t * LCTR_WAVE + lane;
        if (q >= nq) break;
        const float4 ac = acc[t];
        gV4[q] = make_float4(ac.x * inv_scale, ac.y * inv_scale,
                             ac.z * inv_scale, ac.w * inv_scale);
        acc[t] = make_float4(0.f, 0.f, 0.f, 0.f);
      }
      if (lane == 0) gradW[cur] = accw;
    } else {
#pragma unroll
      for (int t = 0; t < MAXQ; ++t) {
        const int q = t * LCTR_WAVE + lane;
        if (q >= nq) break;
        const float4 ac = acc[t];
        if (ac.x != 0.f || ac.y != 0.f || ac.z != 0.f || ac.w != 0.f) {
          atomicAdd(&gradV[(size_t)cur * D + 4 * q + 0], ac.x * inv_scale);
          atomicAdd(&gradV[(size_t)cur * D + 4 * q + 1], ac.y * inv_scale);
          atomicAdd(&gradV[(size_t)cur * D + 4 * q + 2], ac.z * inv_scale);
          atomicAdd(&gradV[(size_t)cur * D + 4 * q + 3], ac.w * inv_scale);
        }
        acc[t] = make_float4(0.f, 0.f, 0.f, 0.f);
      }
      if (lane == 0) {
        atomicAdd(&gradW[cur], accw);
        if (FUSED) atomicOr(&touched[cur >> 6], 1ull << (cur & 63));
      }
    }
  };
  // 2-deep entry pipeline: entry e+1's quad loads (scattered ~624 B)
  // are issued before entry e is consumed, so each wave keeps one
  // entry's lines in flight across the run-detection/flush serial
  // chain. (The FM walk is already 8-entry batched; this kernel had no
  // lookahead at all.) Named ping-pong registers.
  uint2 ldA[MAXQ], ldB[MAXQ];
  float gwA = 0.f, gwB = 0.f;
#define FFM_APPLY_LOAD(e0, LD, GW)                                         \
  do {                                                                     \
    const long p_ = (long)perm[e0];                                        \
    const uint2* gb4_ = (const uint2*)&gblocks[(size_t)p_ * D];            \
    _Pragma("unroll") for (int t = 0; t < MAXQ; ++t) {                     \
      const int q_ = t * LCTR_WAVE + lane;                                 \
      if (q_ >= nq) break;                                                 \
      LD[t] = gb4_[q_];                                                    \
    }                                                                      \
    GW = (lane == 0) ? gw[p_] : 0.f;                                       \
  } while (0)
#define FFM_APPLY_PROC(e0, LD, GW)                                         \
  do {                                                                     \
    const int fid_ = sorted_fids[e0];                                      \
    if (fid_ != cur) {                                                     \
      flush(e0);                                                           \
      cur = fid_;                                                          \
      accw = 0.f;                                                          \
      head_ok = ((e0) == 0 || sorted_fids[(e0) - 1] != fid_);              \
      if (lane == 0 && head_ok && !FUSED)                                  \
        atomicOr(&touched[fid_ >> 6], 1ull << (fid_ & 63));                \
      if (FUSED && PREF && head_ok) { /* state prefetch: new run */        \
        const float4* V4_ = (const float4*)&V[(size_t)fid_ * D];           \
        const float4* nV4_ = (const float4*)&oa.nV[(size_t)fid_ * D];      \
        _Pragma("unroll") for (int t = 0; t < MAXQ; ++t) {                 \
          const int q_ = t * LCTR_WAVE + lane;                             \
          if (q_ >= nq) break;                                             \
          vPre[t] = V4_[q_];                                               \
          nPre[t] = nV4_[q_];                                              \
        }                                                                  \
      }                                                                    \
    }                                                                      \
    _Pragma("unroll") for (int t = 0; t < MAXQ; ++t) {                     \
      const int q_ = t * LCTR_WAVE + lane;                                 \
      if (q_ >= nq) break;                                                 \
      union {                                                              \
        uint2 u2;                                                          \
        __half2 h[2];                                                      \
      } in_;                                                               \
      in_.u2 = LD[t];                                                      \
      const float2 a_ = __half22float2(in_.h[0]);                          \
      const float2 b_ = __half22float2(in_.h[1]);                          \
      acc[t].x += a_.x;                                                    \
      acc[t].y += a_.y;                                                    \
      acc[t].z += b_.x;                                                    \
      acc[t].w += b_.y;                                                    \
    }                                                                      \
    accw += GW;                                                            \
  } while (0)
  if (base < end) FFM_APPLY_LOAD(base, ldA, gwA);
  for (int e = base; e < end; e += 2) {
    if (e + 1 < end) FFM_APPLY_LOAD(e + 1, ldB, gwB);
    FFM_APPLY_PROC(e, ldA, gwA);
    if (e + 1 >= end) break;
    if (e + 2 < end) FFM_APPLY_LOAD(e + 2, ldA, gwA);
    FFM_APPLY_PROC(e + 1, ldB, gwB);
  }
#undef FFM_APPLY_LOAD
#undef FFM_APPLY_PROC
  flush(end);
}

static size_t ffm_row_emit_lds_bytes(int nfields, int K, int maxn,
                                     int nwaves) {
  // acc (per wave, dup fallback) | fp16 stage | vals_s | fmap | dup flag
  return (size_t)nwaves * nfields * (K + 1) * sizeof(float) +
         (size_t)maxn * nfields * K * sizeof(_Float16) +
         (size_t)maxn * sizeof(float) + (size_t)nfields * sizeof(int) +
         sizeof(int);
}

bool ffm_staged_eligible(int nfields, int K, int maxn) {
  // sized for the largest selectable block (8 waves)
  return ffm_row_emit_lds_bytes(nfields, K, maxn, 8) <= (64 << 10);
}

void ffm_fwd_staged_launch(const int* row_ptr, const int* fields,
                           const int* fids, const float* vals, const float* W,
                           const float* V, float* pred, int nfields, int B,
                           int maxn, int K, hipStream_t stream) {
  if (B <= 0) return;
  dim3 block(512);
  dim3 grid(B);
  const size_t lds = (size_t)maxn * nfields * K * sizeof(float);
  DISPATCH_FFM_K(K, hipLaunchKernelGGL((ffm_fwd_staged_kernel<KC>), grid,
                                       block, lds, stream, row_ptr, fields,
                                       fids, vals, W, V, pred, nfields, B,
                                       maxn));
}

void ffm_row_emit_launch(const int* row_ptr, const int* fields,
                         const int* fids, const float* vals, const void* V,
                         int v_bf16, const float* dpred, void* gblocks,
                         float* gw, int nfields, int B, int maxn, int K,
                         float scale, hipStream_t stream) {
  if (B <= 0) return;
  // block sweep (B=65536, bf16): 512 -> 0.98 ms emit, 256 -> 1.21,
  // 128 -> worse (more waves per row hide the stage+emit latency)
  static const int bs = [] {
    const char* e = getenv("LCTR_FFM_EMIT_BLOCK");
    const int v = e ? atoi(e) : 512;
    return (v == 128 || v == 256 || v == 512) ? v : 512;
  }();
  dim3 block(bs);
  dim3 grid(B);
  const size_t lds = ffm_row_emit_lds_bytes(nfields, K, maxn, bs / 64);
  if (v_bf16) {
    DISPATCH_FFM_K(
        K, hipLaunchKernelGGL((ffm_row_emit_kernel<KC, __bf16>), grid, block,
                              lds, stream, row_ptr, fields, fids, vals,
                              (const __bf16*)V, dpred, (_Float16*)gblocks, gw,
                              nfields, B, maxn, scale));
  } else {
    DISPATCH_FFM_K(
        K, hipLaunchKernelGGL((ffm_row_emit_kernel<KC, float>), grid, block,
                              lds, stream, row_ptr, fields, fids, vals,
                              (const float*)V, dpred, (_Float16*)gblocks, gw,
                              nfields, B, maxn, scale));
  }
}

void ffm_blocks_apply_f16_launch(const int* sorted_fids, const int* perm,
                                 const void* gblocks, const float* gw,
                                 float* gradW, float* gradV,
                                 unsigned long long* touched, int D, int nnz,
                                 float inv_scale, int opt_mode, float* V,
                                 float* W, float* nW, float* zW, float* nV,
                                 void* Vh, float p0, float p1, float p2,
                                 float p3, float q0, float q1, float q2,
                                 hipStream_t stream) {
  if (nnz <= 0) return;
  // chunk sweep (B=65536, bf16): 64 -> 1.335 ms, 96 -> 1.356, 128 ->
  // 1.445, 192 -> 1.514
  static const int chunk = [] {
    const char* e = getenv("LCTR_FFM_APPLY_CHUNK");
    return e ? atoi(e) : 64;
  }();
  if (D % 4 != 0 || D > 1024) {
    fprintf(stderr, "ffm_blocks_apply_f16: D=%d unsupported\n", D);
    abort();
  }
  const int wpb = 4;
  const int nwaves = (nnz + chunk - 1) / chunk;
  dim3 block(wpb * LCTR_WAVE);
  dim3 grid((nwaves + wpb - 1) / wpb);
  FfmOptArgs oa{W, nW, zW, nV, (__bf16*)Vh, p0, p1, p2, p3, q0, q1, q2};
  // LCTR_FFM_PREF=1 enables the run-head optimizer-state prefetch —
  // measured NEGATIVE (1527 vs 1431 us isolated fused apply, 8/8
  // alternating passes, profiles/r2_11_ffm_pref_negative.txt): the +32
  // VGPRs (92 -> 124) cost a wave of occupancy (5 -> 4 per SIMD),
  // which this latency-bound walk needs more than the overlap. Kept
  // selectable; default off.
  const char* epf = getenv("LCTR_FFM_PREF");
  const bool pref = epf && epf[0] == '1';
  // MAXQ sized to D (LCTR_FFM_MAXQ=4 forces the round-2 static size
  // for A/B); nq = D/4 quads, lanes stride 64
  const int nq_ = D >> 2;
  const char* emq = getenv("LCTR_FFM_MAXQ");
  const int maxq = (emq && emq[0] == '4') ? 4
                   : (nq_ <= 64 ? 1 : (nq_ <= 128 ? 2 : 4));
#define FFM_AP_LAUNCH(F_, P_, Q_)                                        \
  hipLaunchKernelGGL((ffm_blocks_apply_f16_kernel<F_, P_, Q_>), grid,    \
                     block, 0, stream, sorted_fids, perm,                \
                     (const _Float16*)gblocks, gw, gradW, gradV,         \
                     touched, D, nnz, chunk, inv_scale, opt_mode, V, oa)
  // 8-wave cap on the fused MAXQ=2 kernel (72 -> 63 VGPR + 36 B
  // spill): measured 1316.4 vs 1327.9 us (+0.9%, 8/8 alternating
  // passes) — the extra wave of latency cover beats the spill on this
  // gather-bound walk. Default on; LCTR_FFM_WPE=0 reverts.
  const char* ewp = getenv("LCTR_FFM_WPE");
  if (opt_mode != 0 && !pref && maxq == 2 && !(ewp && ewp[0] == '0')) {
    hipLaunchKernelGGL((ffm_blocks_apply_f16_kernel<true, false, 2, 8>),
                       grid, block, 0, stream, sorted_fids, perm,
                       (const _Float16*)gblocks, gw, gradW, gradV, touched,
                       D, nnz, chunk, inv_scale, opt_mode, V, oa);
    return;
  }
  if (opt_mode != 0 && pref) {
    if (maxq == 1) FFM_AP_LAUNCH(true, true, 1);
    else if (maxq == 2) FFM_AP_LAUNCH(true, true, 2);
    else FFM_AP_LAUNCH(true, true, 4);
  } else if (opt_mode != 0) {
    if (maxq == 1) FFM_AP_LAUNCH(true, false, 1);
    else if (maxq == 2) FFM_AP_LAUNCH(true, false, 2);
    else FFM_AP_LAUNCH(true, false, 4);
  } else {
    if (maxq == 1) FFM_AP_LAUNCH(false, true, 1);
    else if (maxq == 2) FFM_AP_LAUNCH(false, true, 2);
    else FFM_AP_LAUNCH(false, true, 4);
  }
#undef FFM_AP_LAUNCH
}

void ffm_forward_pp_launch(const int* row_ptr, const int* fields,
                           const int* fids, const float* vals,
                           const float* W, const void* V, int v_bf16,
                           float* pred, int nfields, int B, int K,
                           hipStream_t stream) {
  if (B <= 0) return;
  const int wpb = 4;
  dim3 block(wpb * LCTR_WAVE);
  dim3 grid((B + wpb - 1) / wpb);
  if (v_bf16) {
    DISPATCH_FFM_K(K, hipLaunchKernelGGL((ffm_forward_pp_kernel<KC, __bf16>),
                                         grid, block, 0, stream, row_ptr,
                                         fields, fids, vals, W,
                                         (const __bf16*)V, pred, nfields, B));
  } else {
    DISPATCH_FFM_K(K, hipLaunchKernelGGL((ffm_forward_pp_kernel<KC, float>),
                                         grid, block, 0, stream, row_ptr,
                                         fields, fids, vals, W,
                                         (const float*)V, pred, nfields, B));
  }
}

void ffm_block_emit_launch(const int* row_of_entry, const int* row_ptr,
                           const int* fields, const int* fids,
                           const float* vals, const float* V,
                           const float* dpred, float* gblocks, float* gw,
                           int nfields, int nnz, int K, hipStream_t stream) {
  if (nnz <= 0) return;
  const int wpb = 4;
  dim3 block(wpb * LCTR_WAVE);
  dim3 grid((nnz + wpb - 1) / wpb);
  const size_t lds = (size_t)wpb * nfields * (K + 1) * sizeof(float);
  DISPATCH_FFM_K(K, hipLaunchKernelGGL((ffm_block_emit_kernel<KC>), grid,
                                       block, lds, stream, row_of_entry,
                                       row_ptr, fields, fids, vals, V, dpred,
                                       gblocks, gw, nfields, nnz));
}

void ffm_blocks_apply_launch(const int* sorted_fids, const int* perm,
                             const float* gblocks, const float* gw,
                             float* gradW, float* gradV,
                             unsigned long long* touched, int D, int nnz,
                             hipStream_t stream) {
  if (nnz <= 0) return;
  const int chunk = 64;
  const int wpb = 4;
  const int nwaves = (nnz + chunk - 1) / chunk;
  dim3 block(wpb * LCTR_WAVE);
  dim3 grid((nwaves + wpb - 1) / wpb);
  const size_t lds = (size_t)wpb * D * sizeof(float);
  hipLaunchKernelGGL(ffm_blocks_apply_kernel, grid, block, lds, stream,
                     sorted_fids, perm, gblocks, gw, gradW, gradV, touched,
                     D, nnz, chunk);
}

// entry -> row index materialization (one wave per row)
__global__ void row_index_kernel(const int* __restrict__ row_ptr,
                                 int* __restrict__ row_idx, int B) {
  const int lane = threadIdx.x & 63;
  const int row = blockIdx.x * (blockDim.x >> 6) + (threadIdx.x >> 6);
  if (row >= B) return;
  for (int j = row_ptr[row] + lane; j < row_ptr[row + 1]; j += 64)
    row_idx[j] = row;
}

void ffm_sorted_backward_launch(const int* sorted_fids, const int* perm,
                                const int* row_of_entry, const int* row_ptr,
                                const int* fields, const int* fids,
                                const float* vals, const float* V,
                                const float* dpred, float* gradW,
                                float* gradV, unsigned long long* touched,
                                int nfields, int nnz, int K,
                                hipStream_t stream) {
  if (nnz <= 0) return;
  const int chunk = 32;  // short chunks: more waves in flight to hide the
                         // per-entry dependent V-gather latency
  const int wpb = 4;
  const int nwaves = (nnz + chunk - 1) / chunk;
  dim3 block(wpb * LCTR_WAVE);
  dim3 grid((nwaves + wpb - 1) / wpb);
  const size_t lds = (size_t)wpb * nfields * (K + 1) * sizeof(float);
  DISPATCH_FFM_K(K, hipLaunchKernelGGL((ffm_sorted_backward_kernel<KC>),
                                       grid, block, lds, stream, sorted_fids,
                                       perm, row_of_entry, row_ptr, fields,
                                       fids, vals, V, dpred, gradW, gradV,
                                       touched, nfields, nnz, chunk));
}

void row_index_launch(const int* row_ptr, int* row_idx, int B,
                      hipStream_t stream) {
  if (B <= 0) return;
  dim3 block(256);
  dim3 grid((B + 3) / 4);
  hipLaunchKernelGGL(row_index_kernel, grid, block, 0, stream, row_ptr,
                     row_idx, B);
}

void ffm_forward_launch(const int* row_ptr, const int* fields, const int* fids,
                        const float* vals, const float* W, const float* V,
                        float* pred, int nfields, int B, int K,
                        hipStream_t stream) {
  if (B <= 0) return;
  dim3 block(256);
  dim3 grid((B + 3) / 4);
  DISPATCH_FFM_K(K, hipLaunchKernelGGL((ffm_forward_kernel<KC>), grid, block,
                                       0, stream, row_ptr, fields, fids, vals,
                                       W, V, pred, nfields, B));
}

void ffm_backward_launch(const int* row_ptr, const int* fields,
                         const int* fids, const float* vals, const float* V,
                         const float* dpred, float* gradW, float* gradV,
                         unsigned long long* touched, int nfields, int B,
                         int K, hipStream_t stream) {
  if (B <= 0) return;
  dim3 block(256);
  dim3 grid((B + 3) / 4);
  DISPATCH_FFM_K(K, hipLaunchKernelGGL((ffm_backward_kernel<KC>), grid, block,
                                       0, stream, row_ptr, fields, fids, vals,
                                       V, dpred, gradW, gradV, touched,
                                       nfields, B));
}

}  // namespace lightctr
