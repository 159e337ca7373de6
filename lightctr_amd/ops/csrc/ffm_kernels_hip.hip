#include "hip/hip_runtime.h"
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
#include "common.h"

namespace lightctr {

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

void ffm_forward_launch(const int* row_ptr, const int* fields, const int* fids,
                        const float* vals, const float* W, const float* V,
                        float* pred, int nfields, int B, int K,
                        hipStream_t stream) {
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
  dim3 block(256);
  dim3 grid((B + 3) / 4);
  DISPATCH_FFM_K(K, hipLaunchKernelGGL((ffm_backward_kernel<KC>), grid, block,
                                       0, stream, row_ptr, fields, fids, vals,
                                       V, dpred, gradW, gradV, touched,
                                       nfields, B));
}

}  // namespace lightctr
