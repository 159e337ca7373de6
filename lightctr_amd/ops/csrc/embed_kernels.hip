// Embedding-table kernels for MI355X/gfx950: fused gather (+bf16 emit) for
// Wide&Deep-style concat inputs, NFM bi-interaction forward/backward, and
// the per-entry backward emits that feed the sorted segment-reduce apply
// (fm_kernels.hip::fm_sorted_apply_kernel).
//
// Parity targets: reference Train_NFM_Algo bi-interaction
// (/root/reference/LightCTR/train/train_nfm_algo.cpp:56-159) and the
// Wide&Deep distributed trainer's embedding+MLP input assembly
// (/root/reference/LightCTR/distributed_algo_abst.h:105-233).
#include "common.h"

namespace lightctr {

// Gather embeddings concatenated by position-in-row (fixed nnz/row layouts,
// e.g. Criteo 39 fields): out[row, pos*K+k] = E[fid]*x, emitted bf16 for the
// MLP GEMM (+optional fp32 copy). Wave per row, (pos-group, k) lanes.
// 8-wave/SIMD cap: the gather allocated 66 VGPR, two registers over
// the 8-wave boundary (512/8 = 64); the cap re-allocates to 64 with no
// spill (same lever as the FM walk's WPE=6 — profiles/r2_13_fm_wpe.txt)
template <int K>
__global__ __launch_bounds__(256, 8) void embed_gather_kernel(
    const int* __restrict__ row_ptr,
                                    const int* __restrict__ fids,
                                    const float* __restrict__ vals,
                                    const float* __restrict__ E,
                                    __bf16* __restrict__ out_bf, int nf,
                                    int B) {
  constexpr int G = LCTR_WAVE / K;
  const int lane = threadIdx.x & 63;
  const int row = blockIdx.x * (blockDim.x >> 6) + (threadIdx.x >> 6);
  if (row >= B) return;
  const int g = lane / K;
  const int k = lane % K;
  const int beg = row_ptr[row], end = row_ptr[row + 1];
  // 4-deep pipelined gathers (latency-bound random E rows; same pattern
  // as fm_forward_kernel)
  const int lim = min(end, beg + nf);
  int j = beg + g;
  for (; j + 3 * G < lim; j += 4 * G) {
    int f[4];
    float x[4], e[4];
#pragma unroll
    for (int u = 0; u < 4; ++u) {
      f[u] = fids[j + u * G];
      x[u] = vals[j + u * G];
    }
#pragma unroll
    for (int u = 0; u < 4; ++u) e[u] = E[(size_t)f[u] * K + k];
#pragma unroll
    for (int u = 0; u < 4; ++u) {
      const int pos = j + u * G - beg;
      out_bf[((size_t)row * nf + pos) * K + k] = (__bf16)(e[u] * x[u]);
    }
  }
  for (; j < lim; j += G) {
    const int pos = j - beg;
    const float v = E[(size_t)fids[j] * K + k] * vals[j];
    out_bf[((size_t)row * nf + pos) * K + k] = (__bf16)v;
  }
}

// Backward emit for the gather: gv[j,k] = dOut[row, pos*K+k] * x_j and
// gw[j] = dwide[row] * x_j (wide/LR part), feeding the sorted apply.
template <int K>
__global__ void embed_backward_emit_kernel(const int* __restrict__ row_ptr,
                                           const float* __restrict__ vals,
                                           const float* __restrict__ dOut,
                                           const float* __restrict__ dwide,
                                           float* __restrict__ gv,
                                           float* __restrict__ gw, int nf,
                                           int B) {
  constexpr int G = LCTR_WAVE / K;
  const int lane = threadIdx.x & 63;
  const int row = blockIdx.x * (blockDim.x >> 6) + (threadIdx.x >> 6);
  if (row >= B) return;
  const int g = lane / K;
  const int k = lane % K;
  const int beg = row_ptr[row], end = row_ptr[row + 1];
  const float dw = dwide ? dwide[row] : 0.f;
  for (int j = beg + g; j < end; j += G) {
    const int pos = j - beg;
    const float d = (pos < nf) ? dOut[((size_t)row * nf + pos) * K + k] : 0.f;
    gv[(size_t)j * K + k] = d * vals[j];
    if (k == 0 && gw) gw[j] = dw * vals[j];
  }
}

// NFM bi-interaction forward: wide[row] = sum w*x;
// vec[row,k] = 0.5*(sumVX[k]^2 - sum (v_k x)^2); emits vec in fp32 + bf16.
template <int K>
__global__ void nfm_forward_kernel(const int* __restrict__ row_ptr,
                                   const int* __restrict__ fids,
                                   const float* __restrict__ vals,
                                   const float* __restrict__ W,
                                   const float* __restrict__ V,
                                   float* __restrict__ wide,
                                   float* __restrict__ sumVX,
                                   float* __restrict__ vec,
                                   __bf16* __restrict__ vec_bf, int B) {
  constexpr int G = LCTR_WAVE / K;
  const int lane = threadIdx.x & 63;
  const int row = blockIdx.x * (blockDim.x >> 6) + (threadIdx.x >> 6);
  if (row >= B) return;
  const int g = lane / K;
  const int k = lane % K;
  const int beg = row_ptr[row], end = row_ptr[row + 1];
  float sVX = 0.f, sV2X2 = 0.f, lin = 0.f;
  int j = beg + g;
  for (; j + 3 * G < end; j += 4 * G) {
    int f[4];
    float x[4], v[4], w[4];
#pragma unroll
    for (int u = 0; u < 4; ++u) {
      f[u] = fids[j + u * G];
      x[u] = vals[j + u * G];
    }
#pragma unroll
    for (int u = 0; u < 4; ++u) {
      v[u] = V[(size_t)f[u] * K + k];
      w[u] = (k == 0) ? W[f[u]] : 0.f;
    }
#pragma unroll
    for (int u = 0; u < 4; ++u) {
      const float vx = v[u] * x[u];
      sVX += vx;
      sV2X2 += vx * vx;
      lin += w[u] * x[u];
    }
  }
  for (; j < end; j += G) {
    const int fid = fids[j];
    const float x = vals[j];
    const float vx = V[(size_t)fid * K + k] * x;
    sVX += vx;
    sV2X2 += vx * vx;
    if (k == 0) lin += W[fid] * x;
  }
  sVX = group_reduce_sum<K>(sVX);
#pragma unroll
  for (int s = K; s < LCTR_WAVE; s <<= 1) sV2X2 += __shfl_xor(sV2X2, s);
  const float tot_lin = wave_reduce_sum(lin);
  if (lane < K) {
    sumVX[(size_t)row * K + lane] = sVX;
    const float v = 0.5f * (sVX * sVX - sV2X2);
    vec[(size_t)row * K + lane] = v;
    if (vec_bf) vec_bf[(size_t)row * K + lane] = (__bf16)v;
  }
  if (lane == 0) wide[row] = tot_lin;
}

// NFM backward emit: per entry j,
//   gv[j,k] = dvec[row,k] * (sumVX[k] - V[fid,k] x) * x
//   gw[j]   = dwide[row] * x
template <int K>
__global__ void nfm_backward_emit_kernel(
    const int* __restrict__ row_ptr, const int* __restrict__ fids,
    const float* __restrict__ vals, const float* __restrict__ V,
    const float* __restrict__ sumVX, const float* __restrict__ dvec,
    const float* __restrict__ dwide, float* __restrict__ gw,
    float* __restrict__ gv, int B) {
  constexpr int G = LCTR_WAVE / K;
  const int lane = threadIdx.x & 63;
  const int row = blockIdx.x * (blockDim.x >> 6) + (threadIdx.x >> 6);
  if (row >= B) return;
  const int g = lane / K;
  const int k = lane % K;
  const float dv = dvec[(size_t)row * K + k];
  const float dw = dwide[row];
  const float sv = sumVX[(size_t)row * K + k];
  const int beg = row_ptr[row], end = row_ptr[row + 1];
  int j = beg + g;
  for (; j + 3 * G < end; j += 4 * G) {
    int f[4];
    float x[4], v[4];
#pragma unroll
    for (int u = 0; u < 4; ++u) {
      f[u] = fids[j + u * G];
      x[u] = vals[j + u * G];
    }
#pragma unroll
    for (int u = 0; u < 4; ++u) v[u] = V[(size_t)f[u] * K + k];
#pragma unroll
    for (int u = 0; u < 4; ++u) {
      gv[(size_t)(j + u * G) * K + k] = dv * (sv - v[u] * x[u]) * x[u];
      if (k == 0) gw[j + u * G] = dw * x[u];
    }
  }
  for (; j < end; j += G) {
    const int fid = fids[j];
    const float x = vals[j];
    gv[(size_t)j * K + k] = dv * (sv - V[(size_t)fid * K + k] * x) * x;
    if (k == 0) gw[j] = dw * x;
  }
}

void embed_gather_launch(const int* row_ptr, const int* fids,
                         const float* vals, const float* E, void* out_bf,
                         int nf, int B, int K, hipStream_t stream) {
  if (B <= 0) return;
  dim3 block(256);
  dim3 grid((B + 3) / 4);
  DISPATCH_K(K, hipLaunchKernelGGL((embed_gather_kernel<KC>), grid, block, 0,
                                   stream, row_ptr, fids, vals, E,
                                   (__bf16*)out_bf, nf, B));
}

void embed_backward_emit_launch(const int* row_ptr, const float* vals,
                                const float* dOut, const float* dwide,
                                float* gv, float* gw, int nf, int B, int K,
                                hipStream_t stream) {
  if (B <= 0) return;
  dim3 block(256);
  dim3 grid((B + 3) / 4);
  DISPATCH_K(K, hipLaunchKernelGGL((embed_backward_emit_kernel<KC>), grid,
                                   block, 0, stream, row_ptr, vals, dOut,
                                   dwide, gv, gw, nf, B));
}

// ---------------------------------------------------------------------------
// Fused CBOW negative-sampling step (word2vec). One wave per example,
// lane = embedding dim (D <= 64; strided loop above):
//   x = mean(E[ctx]); for t in {center} + negs: p = sigmoid(x . O[t]),
//   g = (p - y)*scale; O[t] -= lr*g*x (atomic); gx += g*O[t];
//   E[c] -= lr*gx/C for each ctx word (atomic).
// Updates are Hogwild-style across examples — the reference's own
// semantics ("unsafe multi-thread update", train_embed_algo.cpp:195-200).
// Replaces ~6 torch kernels per step with one fused launch.
// ---------------------------------------------------------------------------
__global__ void w2v_negsample_kernel(
    float* __restrict__ E, float* __restrict__ O,
    const long* __restrict__ centers, const long* __restrict__ ctx,
    const long* __restrict__ negs, float* __restrict__ loss, int B, int C,
    int N, int D, float lr, float scale) {
  const int lane = threadIdx.x & 63;
  const int ex = blockIdx.x * (blockDim.x >> 6) + (threadIdx.x >> 6);
  if (ex >= B) return;
  // D <= 64 fast path: lane == dim
  if (lane >= D) {
    // still participate in shuffles with zeros
  }
  const float inv_c = 1.f / C;
  float x = 0.f;
  for (int c = 0; c < C; ++c) {
    const long w = ctx[(size_t)ex * C + c];
    if (lane < D) x += E[(size_t)w * D + lane];
  }
  x *= inv_c;
  float gx = 0.f;
  float lsum = 0.f;
  for (int t = 0; t < N + 1; ++t) {
    const long tgt = (t == 0) ? centers[ex]
                              : negs[(size_t)ex * N + (t - 1)];
    const float y = (t == 0) ? 1.f : 0.f;
    const float o = (lane < D) ? O[(size_t)tgt * D + lane] : 0.f;
    float dot = wave_reduce_sum(x * o);
    dot = fminf(16.f, fmaxf(-16.f, dot));
    const float p = 1.f / (1.f + __expf(-dot));
    if (lane == 0) {
      lsum += -(y * __logf(fmaxf(p, 1e-7f))
                + (1.f - y) * __logf(fmaxf(1.f - p, 1e-7f)));
    }
    const float g = (p - y) * scale;
    gx += g * o;
    if (lane < D) atomicAdd(&O[(size_t)tgt * D + lane], -lr * g * x);
  }
  const float ge = lr * gx * inv_c;
  for (int c = 0; c < C; ++c) {
    const long w = ctx[(size_t)ex * C + c];
    if (lane < D) atomicAdd(&E[(size_t)w * D + lane], -ge);
  }
  if (lane == 0) loss[ex] = lsum;
}

void w2v_negsample_launch(float* E, float* O, const long* centers,
                          const long* ctx, const long* negs, float* loss,
                          int B, int C, int N, int D, float lr, float scale,
                          hipStream_t stream) {
  if (B <= 0) return;
  dim3 block(256);
  dim3 grid((B + 3) / 4);
  hipLaunchKernelGGL(w2v_negsample_kernel, grid, block, 0, stream, E, O,
                     centers, ctx, negs, loss, B, C, N, D, lr, scale);
}

// wide (LR) forward: wide[row] = sum_j W[fid_j] * x_j
__global__ void wide_forward_kernel(const int* __restrict__ row_ptr,
                                    const int* __restrict__ fids,
                                    const float* __restrict__ vals,
                                    const float* __restrict__ W,
                                    float* __restrict__ wide, int B) {
  const int lane = threadIdx.x & 63;
  const int row = blockIdx.x * (blockDim.x >> 6) + (threadIdx.x >> 6);
  if (row >= B) return;
  const int beg = row_ptr[row], end = row_ptr[row + 1];
  float lin = 0.f;
  for (int j = beg + lane; j < end; j += 64) lin += W[fids[j]] * vals[j];
  lin = wave_reduce_sum(lin);
  if (lane == 0) wide[row] = lin;
}

void wide_forward_launch(const int* row_ptr, const int* fids,
                         const float* vals, const float* W, float* wide,
                         int B, hipStream_t stream) {
  if (B <= 0) return;
  dim3 block(256);
  dim3 grid((B + 3) / 4);
  hipLaunchKernelGGL(wide_forward_kernel, grid, block, 0, stream, row_ptr,
                     fids, vals, W, wide, B);
}

void nfm_forward_launch(const int* row_ptr, const int* fids, const float* vals,
                        const float* W, const float* V, float* wide,
                        float* sumVX, float* vec, void* vec_bf, int B, int K,
                        hipStream_t stream) {
  if (B <= 0) return;
  dim3 block(256);
  dim3 grid((B + 3) / 4);
  DISPATCH_K(K, hipLaunchKernelGGL((nfm_forward_kernel<KC>), grid, block, 0,
                                   stream, row_ptr, fids, vals, W, V, wide,
                                   sumVX, vec, (__bf16*)vec_bf, B));
}

void nfm_backward_emit_launch(const int* row_ptr, const int* fids,
                              const float* vals, const float* V,
                              const float* sumVX, const float* dvec,
                              const float* dwide, float* gw, float* gv, int B,
                              int K, hipStream_t stream) {
  if (B <= 0) return;
  dim3 block(256);
  dim3 grid((B + 3) / 4);
  DISPATCH_K(K, hipLaunchKernelGGL((nfm_backward_emit_kernel<KC>), grid,
                                   block, 0, stream, row_ptr, fids, vals, V,
                                   sumVX, dvec, dwide, gw, gv, B));
}

}  // namespace lightctr
