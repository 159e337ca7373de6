// Generic fused sparse optimizers (runtime latent width D) + small utility
// kernels for MI355X/gfx950. Used by FFM (D = nfields*K) and any model with
// per-feature parameter blocks; semantics follow the reference updaters
// (/root/reference/LightCTR/util/gradientUpdater.h: Adagrad :128-154,
// FTRL-proximal :235-278).
#include "common.h"

namespace lightctr {

typedef __attribute__((ext_vector_type(4))) __bf16 misc_bf16x4;

// optional bf16 mirror write-back (FFM bf16 mode keeps V fp32 master +
// a bf16 compute mirror refreshed for exactly the touched features)
__device__ __forceinline__ void mirror4(__bf16* Vh, size_t off, float4 v) {
  misc_bf16x4 h = {(__bf16)v.x, (__bf16)v.y, (__bf16)v.z, (__bf16)v.w};
  *(misc_bf16x4*)&Vh[off] = h;
}

// declared in fm_kernels.hip
__device__ __forceinline__ void ftrl_update_g(float* w, float* z, float* n,
                                              float g, float alpha, float beta,
                                              float l1, float l2) {
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

// One wavefront per unique feature; lanes stride the D latent params
// (coalesced 64-wide), lane 0 also updates the linear weight W.
__global__ void sparse_adagrad_apply_g_kernel(
    const int* __restrict__ uniq, const int* __restrict__ count,
    float* __restrict__ W, float* __restrict__ V, float* __restrict__ nW,
    float* __restrict__ nV, float* __restrict__ gradW,
    float* __restrict__ gradV, float lr, float eps, float l2, int capacity,
    int D, __bf16* __restrict__ Vh) {
  const int i = blockIdx.x * (blockDim.x >> 6) + (threadIdx.x >> 6);
  const int lane = threadIdx.x & (LCTR_WAVE - 1);
  if (i >= capacity || i >= *count) return;
  const int fid = uniq[i];
  const size_t base = (size_t)fid * D;
  // float4 RMWs: 4x the bytes per instruction on this latency-bound
  // state sweep (D=312 at nf=39: 2 vector iterations/wave instead of 5
  // scalar ones)
  if ((D & 3) == 0) {
    const int nf4 = D >> 2;
    float4* V4 = (float4*)&V[base];
    float4* nV4 = (float4*)&nV[base];
    float4* gV4 = (float4*)&gradV[base];
    for (int d4 = lane; d4 < nf4; d4 += LCTR_WAVE) {
      const float4 gv = gV4[d4];
      float4 v = V4[d4];
      float4 a = nV4[d4];
      const float gx = gv.x + l2 * v.x, gy = gv.y + l2 * v.y,
                  gz = gv.z + l2 * v.z, gw = gv.w + l2 * v.w;
      a.x += gx * gx; a.y += gy * gy; a.z += gz * gz; a.w += gw * gw;
      v.x -= lr * gx * __frsqrt_rn(a.x + eps);
      v.y -= lr * gy * __frsqrt_rn(a.y + eps);
      v.z -= lr * gz * __frsqrt_rn(a.z + eps);
      v.w -= lr * gw * __frsqrt_rn(a.w + eps);
      nV4[d4] = a;
      V4[d4] = v;
      gV4[d4] = make_float4(0.f, 0.f, 0.f, 0.f);
      if (Vh) mirror4(Vh, base + 4 * (size_t)d4, v);
    }
  } else {
    for (int d = lane; d < D; d += LCTR_WAVE) {
      const float g = gradV[base + d] + l2 * V[base + d];
      const float acc = nV[base + d] + g * g;
      nV[base + d] = acc;
      V[base + d] -= lr * g * __frsqrt_rn(acc + eps);
      gradV[base + d] = 0.f;
      if (Vh) Vh[base + d] = (__bf16)V[base + d];
    }
  }
  if (lane == 0 && W != nullptr) {
    const float gw = gradW[fid] + l2 * W[fid];
    const float a = nW[fid] + gw * gw;
    nW[fid] = a;
    W[fid] -= lr * gw * __frsqrt_rn(a + eps);
    gradW[fid] = 0.f;
  }
}

__global__ void sparse_ftrl_apply_g_kernel(
    const int* __restrict__ uniq, const int* __restrict__ count,
    float* __restrict__ W, float* __restrict__ V, float* __restrict__ zW,
    float* __restrict__ nW, float* __restrict__ zV, float* __restrict__ nV,
    float* __restrict__ gradW, float* __restrict__ gradV, float alpha,
    float beta, float l1, float l2, int capacity, int D, int v_adagrad,
    float v_lr, float v_eps, float v_l2, __bf16* __restrict__ Vh) {
  const int i = blockIdx.x * (blockDim.x >> 6) + (threadIdx.x >> 6);
  const int lane = threadIdx.x & (LCTR_WAVE - 1);
  if (i >= capacity || i >= *count) return;
  const int fid = uniq[i];
  const size_t base = (size_t)fid * D;
  if (v_adagrad && (D & 3) == 0) {
    // FTRL on W, Adagrad on the latent block (see fm_kernels.hip
    // rationale); float4 RMWs as in the adagrad kernel above
    const int nf4 = D >> 2;
    float4* V4 = (float4*)&V[base];
    float4* nV4 = (float4*)&nV[base];
    float4* gV4 = (float4*)&gradV[base];
    for (int d4 = lane; d4 < nf4; d4 += LCTR_WAVE) {
      const float4 gv = gV4[d4];
      float4 v = V4[d4];
      float4 a = nV4[d4];
      const float gx = gv.x + v_l2 * v.x, gy = gv.y + v_l2 * v.y,
                  gz = gv.z + v_l2 * v.z, gw = gv.w + v_l2 * v.w;
      a.x += gx * gx; a.y += gy * gy; a.z += gz * gz; a.w += gw * gw;
      v.x -= v_lr * gx * __frsqrt_rn(a.x + v_eps);
      v.y -= v_lr * gy * __frsqrt_rn(a.y + v_eps);
      v.z -= v_lr * gz * __frsqrt_rn(a.z + v_eps);
      v.w -= v_lr * gw * __frsqrt_rn(a.w + v_eps);
      nV4[d4] = a;
      V4[d4] = v;
      gV4[d4] = make_float4(0.f, 0.f, 0.f, 0.f);
      if (Vh) mirror4(Vh, base + 4 * (size_t)d4, v);
    }
  } else {
    for (int d = lane; d < D; d += LCTR_WAVE) {
      if (v_adagrad) {
        const float g = gradV[base + d] + v_l2 * V[base + d];
        const float acc = nV[base + d] + g * g;
        nV[base + d] = acc;
        V[base + d] -= v_lr * g * __frsqrt_rn(acc + v_eps);
      } else {
        ftrl_update_g(&V[base + d], &zV[base + d], &nV[base + d],
                      gradV[base + d], alpha, beta, l1, l2);
      }
      gradV[base + d] = 0.f;
      if (Vh) Vh[base + d] = (__bf16)V[base + d];
    }
  }
  if (lane == 0 && W != nullptr) {
    ftrl_update_g(&W[fid], &zW[fid], &nW[fid], gradW[fid], alpha, beta, l1,
                  l2);
    gradW[fid] = 0.f;
  }
}

// ---------------------------------------------------------------------------
// PS-side fused updaters (reference paramserver.h:217-306): applied to the
// shard tables for one push's key list. updater: 0=sgd, 1=adagrad,
// 2=dcasgd, 3=dcasgda. DCASGD keeps a per-worker shadow copy of each
// pulled weight and compensates w -= lr*(g + lambda*g^2*(w - shadow));
// DCASGDA uses lambda/sqrt(eps + EMA(g^2)). One wave per key, lanes
// stride the K latent params; lane 0 handles the scalar W.
// ---------------------------------------------------------------------------
__global__ void ps_apply_kernel(const long* __restrict__ lidx, int n,
                                const float* __restrict__ gW,
                                const float* __restrict__ gV,
                                float* __restrict__ W, float* __restrict__ V,
                                float* __restrict__ nW,
                                float* __restrict__ nV,
                                float* __restrict__ shadowW,
                                float* __restrict__ shadowV, int K,
                                int updater, float lr, float lam,
                                float eps) {
  const int i = blockIdx.x * (blockDim.x >> 6) + (threadIdx.x >> 6);
  const int lane = threadIdx.x & 63;
  if (i >= n) return;
  const long f = lidx[i];
  for (int k = lane; k < K + 1; k += LCTR_WAVE) {
    const bool is_w = (k == K);
    float* p = is_w ? &W[f] : &V[f * K + k];
    float* st = is_w ? (nW ? &nW[f] : nullptr)
                     : (nV ? &nV[f * K + k] : nullptr);
    float* sh = is_w ? (shadowW ? &shadowW[f] : nullptr)
                     : (shadowV ? &shadowV[f * K + k] : nullptr);
    const float g = is_w ? gW[i] : gV[(size_t)i * K + k];
    if (updater == 0) {  // sgd
      *p -= lr * g;
    } else if (updater == 1) {  // adagrad
      const float acc = *st + g * g;
      *st = acc;
      *p -= lr * g * __frsqrt_rn(acc + eps);
    } else {  // dcasgd / dcasgda
      float l = lam;
      if (updater == 3) {
        const float acc = 0.95f * (*st) + 0.05f * g * g;
        *st = acc;
        l = lam * __frsqrt_rn(acc + eps);
      }
      const float w = *p;
      const float wn = w - lr * (g + l * g * g * (w - *sh));
      *p = wn;
      *sh = wn;
    }
  }
}

// ---------------------------------------------------------------------------
// Histogram AUC (reference util/evaluator.h:61-94: 2^24-bucket pos/neg
// histogram + trapezoid integration) as device kernels: an atomic
// histogram accumulate and a single-workgroup device scan that computes
// correct-pair mass + totals. Semantics match utils/metrics.HistAUC.
// ---------------------------------------------------------------------------

__global__ void auc_hist_add_kernel(const float* __restrict__ pred,
                                    const float* __restrict__ label, long n,
                                    unsigned int* __restrict__ hist,
                                    int buckets) {
  const long i = (long)blockIdx.x * blockDim.x + threadIdx.x;
  const long stride = (long)gridDim.x * blockDim.x;
  for (long j = i; j < n; j += stride) {
    float p = pred[j];
    p = fminf(1.f, fmaxf(0.f, p));
    // floor, matching the host path's `(pred * (buckets-1)).long()`
    int b = (int)(p * (buckets - 1));
    b = min(buckets - 1, max(0, b));
    const int pos = label[j] > 0.5f ? 1 : 0;
    atomicAdd(&hist[(size_t)pos * buckets + b], 1u);
  }
}

// One workgroup (1024 threads = 16 waves): two passes over the buckets.
// Pass 1 reduces P and N; pass 2 runs a block-wide inclusive scan of the
// positive counts chunk by chunk and accumulates
//   correct = sum_b neg_b * (P - cumpos_incl_b) + 0.5 * neg_b * pos_b
// out = {correct, P, N} (fp64).
__global__ void auc_scan_kernel(const unsigned int* __restrict__ hist,
                                int buckets, double* __restrict__ out) {
  __shared__ long wave_tot[16];
  __shared__ double wave_red[16];
  __shared__ long running_s;
  const int tid = threadIdx.x;
  const int lane = tid & 63;
  const int wv = tid >> 6;

  // pass 1: totals
  long pos_t = 0, neg_t = 0;
  for (int b = tid; b < buckets; b += blockDim.x) {
    neg_t += hist[b];
    pos_t += hist[(size_t)buckets + b];
  }
  double pt = (double)pos_t, nt = (double)neg_t;
#pragma unroll
  for (int s = 1; s < LCTR_WAVE; s <<= 1) {
    pt += __shfl_xor(pt, s);
    nt += __shfl_xor(nt, s);
  }
  if (lane == 0) wave_red[wv] = pt;
  __syncthreads();
  double P = 0;
  for (int w = 0; w < 16; ++w) P += wave_red[w];
  __syncthreads();
  if (lane == 0) wave_red[wv] = nt;
  __syncthreads();
  double N = 0;
  for (int w = 0; w < 16; ++w) N += wave_red[w];
  __syncthreads();

  // pass 2: chunked inclusive scan of pos + correct-mass accumulate
  if (tid == 0) running_s = 0;
  __syncthreads();
  double corr = 0;
  for (int base = 0; base < buckets; base += blockDim.x) {
    const int b = base + tid;
    const long pos = b < buckets ? hist[(size_t)buckets + b] : 0;
    const long neg = b < buckets ? hist[b] : 0;
    // block inclusive scan of pos
    long scan = pos;
#pragma unroll
    for (int s = 1; s < LCTR_WAVE; s <<= 1) {
      const long t = __shfl_up(scan, s);
      if (lane >= s) scan += t;
    }
    if (lane == LCTR_WAVE - 1) wave_tot[wv] = scan;
    __syncthreads();
    long wbase = running_s;
    for (int w = 0; w < wv; ++w) wbase += wave_tot[w];
    const long cum_incl = wbase + scan;
    corr += (double)neg * (P - (double)cum_incl) +
            0.5 * (double)neg * (double)pos;
    __syncthreads();
    if (tid == 0) {
      long tot = 0;
      for (int w = 0; w < 16; ++w) tot += wave_tot[w];
      running_s += tot;
    }
    __syncthreads();
  }
#pragma unroll
  for (int s = 1; s < LCTR_WAVE; s <<= 1) corr += __shfl_xor(corr, s);
  if (lane == 0) wave_red[wv] = corr;
  __syncthreads();
  if (tid == 0) {
    double c = 0;
    for (int w = 0; w < 16; ++w) c += wave_red[w];
    out[0] = c;
    out[1] = P;
    out[2] = N;
  }
}

void auc_hist_add_launch(const float* pred, const float* label, long n,
                         unsigned int* hist, int buckets,
                         hipStream_t stream) {
  if (n <= 0) return;
  dim3 block(256);
  dim3 grid((int)min((long)4096, (n + 255) / 256));
  hipLaunchKernelGGL(auc_hist_add_kernel, grid, block, 0, stream, pred,
                     label, n, hist, buckets);
}

void auc_scan_launch(const unsigned int* hist, int buckets, double* out,
                     hipStream_t stream) {
  hipLaunchKernelGGL(auc_scan_kernel, dim3(1), dim3(1024), 0, stream, hist,
                     buckets, out);
}

void ps_apply_launch(const long* lidx, int n, const float* gW,
                     const float* gV, float* W, float* V, float* nW,
                     float* nV, float* shadowW, float* shadowV, int K,
                     int updater, float lr, float lam, float eps,
                     hipStream_t stream) {
  if (n <= 0) return;
  dim3 block(256);
  dim3 grid((n + 3) / 4);
  hipLaunchKernelGGL(ps_apply_kernel, grid, block, 0, stream, lidx, n, gW,
                     gV, W, V, nW, nV, shadowW, shadowV, K, updater, lr,
                     lam, eps);
}

void sparse_adagrad_apply_launch(const int* uniq, const int* count, float* W,
                                 float* V, float* nW, float* nV, float* gradW,
                                 float* gradV, float lr, float eps, float l2,
                                 int capacity, int D, void* Vh,
                                 hipStream_t stream) {
  dim3 block(256);
  dim3 grid((capacity + 3) / 4);
  hipLaunchKernelGGL(sparse_adagrad_apply_g_kernel, grid, block, 0, stream,
                     uniq, count, W, V, nW, nV, gradW, gradV, lr, eps, l2,
                     capacity, D, (__bf16*)Vh);
}

void sparse_ftrl_apply_launch(const int* uniq, const int* count, float* W,
                              float* V, float* zW, float* nW, float* zV,
                              float* nV, float* gradW, float* gradV,
                              float alpha, float beta, float l1, float l2,
                              int capacity, int D, int v_adagrad, float v_lr,
                              float v_eps, float v_l2, void* Vh,
                              hipStream_t stream) {
  dim3 block(256);
  dim3 grid((capacity + 3) / 4);
  hipLaunchKernelGGL(sparse_ftrl_apply_g_kernel, grid, block, 0, stream, uniq,
                     count, W, V, zW, nW, zV, nV, gradW, gradV, alpha, beta,
                     l1, l2, capacity, D, v_adagrad, v_lr, v_eps, v_l2,
                     (__bf16*)Vh);
}

}  // namespace lightctr
