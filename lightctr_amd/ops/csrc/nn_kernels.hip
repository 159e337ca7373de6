// Dense-NN support kernels for MI355X/gfx950: activation backward, bias
// gradient (column sum), dense fused optimizers that keep fp32 master
// weights and maintain the bf16 + bf16-transposed GEMM operand mirrors.
//
// Together with gemm_kernels.hip these replace the reference's
// Fully_Conn_Layer forward/backward + per-layer updaters
// (/root/reference/LightCTR/train/layer/fullyconnLayer.h:95-237,
// util/momentumUpdater.h Adam :113-215) in CDNA4 form.
#include "common.h"

namespace lightctr {

// dZ = dY * act'(Y) where Y is the post-activation output; also emits a
// bf16 mirror of dZ (GEMM operand for dgrad/wgrad). act: 0 none, 1 relu,
// 2 sigmoid.
__global__ void act_backward_kernel(const float* __restrict__ dY,
                                    const float* __restrict__ Y,
                                    float* __restrict__ dZ,
                                    __bf16* __restrict__ dZbf, long n,
                                    int act) {
  const long i = (long)blockIdx.x * blockDim.x + threadIdx.x;
  if (i >= n) return;
  float g = dY[i];
  if (act == 1) {
    g = Y[i] > 0.f ? g : 0.f;
  } else if (act == 2) {
    const float y = Y[i];
    g *= y * (1.f - y);
  }
  dZ[i] = g;
  if (dZbf) dZbf[i] = (__bf16)g;
}

// db[n] = sum_m dZ[m,n]. Column-tiled: block (x = column tile, y = row
// chunk); thread owns one column, reads coalesced across the 256-thread
// tile, keeps a register partial over its row chunk, one global atomic per
// (block, column). N==1 collapses to a flat wave-reduced sum.
__global__ void colsum_kernel(const float* __restrict__ dZ,
                              float* __restrict__ db, int M, int N) {
  if (N == 1) {
    const long total = M;
    float s = 0.f;
    for (long i = (long)blockIdx.x * blockDim.x + threadIdx.x; i < total;
         i += (long)gridDim.x * blockDim.x)
      s += dZ[i];
    s = wave_reduce_sum(s);
    if ((threadIdx.x & 63) == 0 && s != 0.f) atomicAdd(&db[0], s);
    return;
  }
  const int col = blockIdx.x * blockDim.x + threadIdx.x;
  if (col >= N) return;
  const int chunk = (M + gridDim.y - 1) / gridDim.y;
  const int m0 = blockIdx.y * chunk;
  const int m1 = min(M, m0 + chunk);
  // 4 independent partials keep 4 loads in flight per lane (the single-
  // accumulator loop left one outstanding load per wave: 59 us for a
  // 67 MB pass ~ 14x off the HBM floor; profiles/prof_wd)
  float s0 = 0.f, s1 = 0.f, s2 = 0.f, s3 = 0.f;
  int m = m0;
  for (; m + 3 < m1; m += 4) {
    s0 += dZ[(size_t)m * N + col];
    s1 += dZ[(size_t)(m + 1) * N + col];
    s2 += dZ[(size_t)(m + 2) * N + col];
    s3 += dZ[(size_t)(m + 3) * N + col];
  }
  for (; m < m1; ++m) s0 += dZ[(size_t)m * N + col];
  const float s = (s0 + s1) + (s2 + s3);
  if (s != 0.f) atomicAdd(&db[col], s);
}

// f32 -> bf16 convert (vectorized x4)
__global__ void to_bf16_kernel(const float* __restrict__ x,
                               __bf16* __restrict__ y, long n) {
  const long i = ((long)blockIdx.x * blockDim.x + threadIdx.x) * 4;
  if (i + 3 < n) {
    const float4 v = *(const float4*)&x[i];
    y[i] = (__bf16)v.x;
    y[i + 1] = (__bf16)v.y;
    y[i + 2] = (__bf16)v.z;
    y[i + 3] = (__bf16)v.w;
  } else {
    for (long j = i; j < n; ++j) y[j] = (__bf16)x[j];
  }
}

// Dense Adam on fp32 master W[out,in]; refreshes bf16 mirror Wbf[out,in]
// and transposed mirror Wtbf[in,out] (both GEMM operands).
__global__ void dense_adam_kernel(float* __restrict__ W,
                                  const float* __restrict__ grad,
                                  float* __restrict__ m_t,
                                  float* __restrict__ v_t,
                                  __bf16* __restrict__ Wbf,
                                  __bf16* __restrict__ Wtbf, int rows,
                                  int cols, float lr, float beta1, float beta2,
                                  float eps, float bc1, float bc2, float l2) {
  const long i = (long)blockIdx.x * blockDim.x + threadIdx.x;
  const long n = (long)rows * cols;
  if (i >= n) return;
  const float g = grad[i] + l2 * W[i];
  const float m = beta1 * m_t[i] + (1.f - beta1) * g;
  const float v = beta2 * v_t[i] + (1.f - beta2) * g * g;
  m_t[i] = m;
  v_t[i] = v;
  const float w = W[i] - lr * (m / bc1) / (sqrtf(v / bc2) + eps);
  W[i] = w;
  if (Wbf) Wbf[i] = (__bf16)w;
  if (Wtbf) {
    const long r = i / cols, c = i % cols;
    Wtbf[c * rows + r] = (__bf16)w;
  }
}

// Dense Adagrad (reference default updater), same mirror maintenance.
__global__ void dense_adagrad_kernel(float* __restrict__ W,
                                     const float* __restrict__ grad,
                                     float* __restrict__ n_t,
                                     __bf16* __restrict__ Wbf,
                                     __bf16* __restrict__ Wtbf, int rows,
                                     int cols, float lr, float eps, float l2) {
  const long i = (long)blockIdx.x * blockDim.x + threadIdx.x;
  const long n = (long)rows * cols;
  if (i >= n) return;
  const float g = grad[i] + l2 * W[i];
  const float acc = n_t[i] + g * g;
  n_t[i] = acc;
  const float w = W[i] - lr * g * __frsqrt_rn(acc + eps);
  W[i] = w;
  if (Wbf) Wbf[i] = (__bf16)w;
  if (Wtbf) {
    const long r = i / cols, c = i % cols;
    Wtbf[c * rows + r] = (__bf16)w;
  }
}

void act_backward_launch(const float* dY, const float* Y, float* dZ,
                         void* dZbf, long n, int act, hipStream_t stream) {
  if (n <= 0) return;
  dim3 block(256);
  dim3 grid((unsigned)((n + 255) / 256));
  hipLaunchKernelGGL(act_backward_kernel, grid, block, 0, stream, dY, Y, dZ,
                     (__bf16*)dZbf, n, act);
}

void colsum_launch(const float* dZ, float* db, int M, int N,
                   hipStream_t stream) {
  LCTR_CHECK_HIP(hipMemsetAsync(db, 0, (size_t)N * sizeof(float), stream));
  dim3 block(256);
  if (N == 1) {
    dim3 grid(512);
    hipLaunchKernelGGL(colsum_kernel, grid, block, 0, stream, dZ, db, M, N);
    return;
  }
  const int xtiles = (N + 255) / 256;
  const int ychunks = max(1, min(2048 / xtiles, (M + 127) / 128));
  dim3 grid(xtiles * 256 / 256, ychunks);
  grid.x = xtiles;
  hipLaunchKernelGGL(colsum_kernel, grid, block, 0, stream, dZ, db, M, N);
}

void to_bf16_launch(const float* x, void* y, long n, hipStream_t stream) {
  if (n <= 0) return;
  dim3 block(256);
  dim3 grid((unsigned)((n / 4 + 256) / 256) + 1);
  hipLaunchKernelGGL(to_bf16_kernel, grid, block, 0, stream, x, (__bf16*)y,
                     n);
}

void dense_adam_launch(float* W, const float* grad, float* m_t, float* v_t,
                       void* Wbf, void* Wtbf, int rows, int cols, float lr,
                       float beta1, float beta2, float eps, float bc1,
                       float bc2, float l2, hipStream_t stream) {
  const long n = (long)rows * cols;
  dim3 block(256);
  dim3 grid((unsigned)((n + 255) / 256));
  hipLaunchKernelGGL(dense_adam_kernel, grid, block, 0, stream, W, grad, m_t,
                     v_t, (__bf16*)Wbf, (__bf16*)Wtbf, rows, cols, lr, beta1,
                     beta2, eps, bc1, bc2, l2);
}

void dense_adagrad_launch(float* W, const float* grad, float* n_t, void* Wbf,
                          void* Wtbf, int rows, int cols, float lr, float eps,
                          float l2, hipStream_t stream) {
  const long n = (long)rows * cols;
  dim3 block(256);
  dim3 grid((unsigned)((n + 255) / 256));
  hipLaunchKernelGGL(dense_adagrad_kernel, grid, block, 0, stream, W, grad,
                     n_t, (__bf16*)Wbf, (__bf16*)Wtbf, rows, cols, lr, eps,
                     l2);
}

// ---------------------------------------------------------------------------
// Degenerate-GEMM fast paths. The MLP's last layer (out_dim 1) pushes
// three shapes through the 128x128 tile kernel at ~99% tile waste:
// forward [M,1,K] (a GEMV), wgrad [1,N,K] (a weighted row-sum), and
// dgrad [M,N,1] (an outer product). Each is bandwidth-trivial when
// expressed directly. Routed from gemm_bf16_launch (gemm_kernels.hip).
// ---------------------------------------------------------------------------
__device__ __forceinline__ float gemm_act(float v, int act) {
  if (act == 1) return fmaxf(v, 0.f);
  if (act == 2) return sigmoidf_clamped(v);
  return v;
}

// C[m,0] = sum_k A[m,k]*b[k] : one wave per row, lanes stride K.
__global__ void gemv_n1_kernel(const __bf16* __restrict__ A,
                               const __bf16* __restrict__ b,
                               const float* __restrict__ bias,
                               float* __restrict__ C,
                               __bf16* __restrict__ Cbf, int M, int K,
                               int act) {
  const int lane = threadIdx.x & 63;
  const int row = blockIdx.x * (blockDim.x >> 6) + (threadIdx.x >> 6);
  if (row >= M) return;
  float s = 0.f;
  for (int k = lane; k < K; k += 64)
    s += (float)A[(size_t)row * K + k] * (float)b[k];
  s = wave_reduce_sum(s);
  if (lane == 0) {
    const float v = gemm_act(s + (bias ? bias[0] : 0.f), act);
    C[row] = v;
    if (Cbf) Cbf[row] = (__bf16)v;
  }
}

// C[0,n] = sum_k a[k]*Bst[k,n] (wgrad for a 1-wide layer): column-tiled
// weighted sum, 4-deep unrolled partials, atomic per (block,column).
// C must be zeroed by the launcher; no bias/act/Cbf (wgrad has none).
__global__ void wrowsum_m1_kernel(const __bf16* __restrict__ a,
                                  const __bf16* __restrict__ Bst,
                                  float* __restrict__ C, int K, int N) {
  const int col = blockIdx.x * blockDim.x + threadIdx.x;
  if (col >= N) return;
  const int chunk = (K + gridDim.y - 1) / gridDim.y;
  const int k0 = blockIdx.y * chunk;
  const int k1 = min(K, k0 + chunk);
  float s0 = 0.f, s1 = 0.f, s2 = 0.f, s3 = 0.f;
  int k = k0;
  for (; k + 3 < k1; k += 4) {
    s0 += (float)a[k] * (float)Bst[(size_t)k * N + col];
    s1 += (float)a[k + 1] * (float)Bst[(size_t)(k + 1) * N + col];
    s2 += (float)a[k + 2] * (float)Bst[(size_t)(k + 2) * N + col];
    s3 += (float)a[k + 3] * (float)Bst[(size_t)(k + 3) * N + col];
  }
  for (; k < k1; ++k) s0 += (float)a[k] * (float)Bst[(size_t)k * N + col];
  const float s = (s0 + s1) + (s2 + s3);
  if (s != 0.f) atomicAdd(&C[col], s);
}

// C[m,n] = A[m,0]*Bst[n,0] (+bias[n], act): outer product, K == 1.
__global__ void outer_k1_kernel(const __bf16* __restrict__ A,
                                const __bf16* __restrict__ Bst,
                                const float* __restrict__ bias,
                                float* __restrict__ C,
                                __bf16* __restrict__ Cbf, long M, long N,
                                int act) {
  const long i = (long)blockIdx.x * blockDim.x + threadIdx.x;
  if (i >= M * N) return;
  const long m = i / N, n = i - m * N;
  const float v = gemm_act((float)A[m] * (float)Bst[n] +
                               (bias ? bias[n] : 0.f),
                           act);
  C[i] = v;
  if (Cbf) Cbf[i] = (__bf16)v;
}

// C[M,N] = sum_k Ast[k,m]*Bst[k,n] for TINY outputs (M*N <= 4096) and
// deep K (the small MLP's wgrad, e.g. NFM [64,16,65536]): each thread
// owns ceil(M*N/256) outputs in registers, blocks split K, one atomic
// per (block, output) at the end. The tile kernels waste >90% of their
// 128x128 output on these shapes.
__global__ void small_wgrad_kernel(const __bf16* __restrict__ Ast,
                                   const __bf16* __restrict__ Bst,
                                   float* __restrict__ C, int M, int N,
                                   long K, int kchunk) {
  const int tid = threadIdx.x;
  const int MN = M * N;
  const long kbeg = (long)blockIdx.x * kchunk;
  const long kend = min(K, kbeg + kchunk);
  float acc[4];
  int m[4], n[4];
  int no = 0;
  for (int o = tid; o < MN && no < 4; o += 1024, ++no) {
    m[no] = o / N;
    n[no] = o - m[no] * N;
    acc[no] = 0.f;
  }
  for (long k = kbeg; k < kend; ++k) {
    const __bf16* ar = Ast + k * M;
    const __bf16* br = Bst + k * N;
    for (int i = 0; i < no; ++i)
      acc[i] += (float)ar[m[i]] * (float)br[n[i]];
  }
  for (int i = 0; i < no; ++i)
    if (acc[i] != 0.f) atomicAdd(&C[(size_t)m[i] * N + n[i]], acc[i]);
}

void small_wgrad_launch(const void* Ast, const void* Bst, float* C, int M,
                        int N, int K, hipStream_t stream) {
  LCTR_CHECK_HIP(
      hipMemsetAsync(C, 0, (size_t)M * N * sizeof(float), stream));
  const int blocks = 512;
  const int kchunk = (K + blocks - 1) / blocks;
  hipLaunchKernelGGL(small_wgrad_kernel, dim3(blocks), dim3(1024), 0,
                     stream, (const __bf16*)Ast, (const __bf16*)Bst, C, M,
                     N, (long)K, kchunk);
}

void gemv_n1_launch(const void* A, const void* b, const float* bias,
                    float* C, void* Cbf, int M, int K, int act,
                    hipStream_t stream) {
  const int wpb = 4;
  dim3 block(wpb * 64);
  dim3 grid((M + wpb - 1) / wpb);
  hipLaunchKernelGGL(gemv_n1_kernel, grid, block, 0, stream,
                     (const __bf16*)A, (const __bf16*)b, bias, C,
                     (__bf16*)Cbf, M, K, act);
}

void wrowsum_m1_launch(const void* a, const void* Bst, float* C, int K,
                       int N, hipStream_t stream) {
  LCTR_CHECK_HIP(hipMemsetAsync(C, 0, (size_t)N * sizeof(float), stream));
  const int xtiles = (N + 255) / 256;
  const int ychunks = max(1, min(2048 / xtiles, (K + 127) / 128));
  dim3 block(256);
  dim3 grid(xtiles, ychunks);
  hipLaunchKernelGGL(wrowsum_m1_kernel, grid, block, 0, stream,
                     (const __bf16*)a, (const __bf16*)Bst, C, K, N);
}

void outer_k1_launch(const void* A, const void* Bst, const float* bias,
                     float* C, void* Cbf, long M, long N, int act,
                     hipStream_t stream) {
  const long total = M * N;
  dim3 block(256);
  dim3 grid((unsigned)((total + 255) / 256));
  hipLaunchKernelGGL(outer_k1_kernel, grid, block, 0, stream,
                     (const __bf16*)A, (const __bf16*)Bst, bias, C,
                     (__bf16*)Cbf, M, N, act);
}

// ---------------------------------------------------------------------------
// In-tree convolution im2col / col2im (reference util/matrix.h:237-319
// Matrix::convolution/deconvolution, SURVEY §2.4 "im2col+MFMA or direct
// conv" row). im2col gathers patches STRAIGHT into the bf16 GEMM operand
// layout [B*L, C*k*k] (one fused pass instead of F.unfold fp32 +
// transpose + contiguous + to_bf16); col2im is the gather-form data-grad
// (each input pixel sums the k*k output taps that touched it — no
// atomics).
// ---------------------------------------------------------------------------

__global__ void im2col_bf16_kernel(const float* __restrict__ x,
                                   __bf16* __restrict__ col, int B, int C,
                                   int H, int W, int k, int stride, int pad,
                                   int OH, int OW) {
  const long L = (long)OH * OW;
  const long total = (long)B * L * C * k * k;
  const long i0 = (long)blockIdx.x * blockDim.x + threadIdx.x;
  const long gstride = (long)gridDim.x * blockDim.x;
  const int kk = k * k;
  for (long i = i0; i < total; i += gstride) {
    // i = ((b*L + l) * C*k*k) + (c*k + kh)*k + kw
    const long row = i / (C * kk);
    const int cc = (int)(i - row * (C * kk));
    const int c = cc / kk;
    const int kh = (cc % kk) / k;
    const int kw = cc % k;
    const long b = row / L;
    const int l = (int)(row - b * L);
    const int oh = l / OW, ow = l - (l / OW) * OW;
    const int h = oh * stride - pad + kh;
    const int w = ow * stride - pad + kw;
    float v = 0.f;
    if (h >= 0 && h < H && w >= 0 && w < W)
      v = x[((b * C + c) * (long)H + h) * W + w];
    col[i] = (__bf16)v;
  }
}

__global__ void col2im_kernel(const float* __restrict__ dcol,
                              float* __restrict__ dx, int B, int C, int H,
                              int W, int k, int stride, int pad, int OH,
                              int OW) {
  const long total = (long)B * C * H * W;
  const long i0 = (long)blockIdx.x * blockDim.x + threadIdx.x;
  const long gstride = (long)gridDim.x * blockDim.x;
  const long L = (long)OH * OW;
  const int kk = k * k;
  for (long i = i0; i < total; i += gstride) {
    const int w = (int)(i % W);
    const int h = (int)((i / W) % H);
    const int c = (int)((i / ((long)W * H)) % C);
    const long b = i / ((long)W * H * C);
    float acc = 0.f;
    for (int kh = 0; kh < k; ++kh) {
      const int oh_num = h + pad - kh;
      if (oh_num < 0 || oh_num % stride) continue;
      const int oh = oh_num / stride;
      if (oh >= OH) continue;
      for (int kw = 0; kw < k; ++kw) {
        const int ow_num = w + pad - kw;
        if (ow_num < 0 || ow_num % stride) continue;
        const int ow = ow_num / stride;
        if (ow >= OW) continue;
        const long row = b * L + (long)oh * OW + ow;
        acc += dcol[row * (C * kk) + ((long)c * k + kh) * k + kw];
      }
    }
    dx[i] = acc;
  }
}

void im2col_bf16_launch(const float* x, void* col, int B, int C, int H,
                        int W, int k, int stride, int pad, int OH, int OW,
                        hipStream_t stream) {
  const long total = (long)B * OH * OW * C * k * k;
  dim3 block(256);
  dim3 grid((unsigned)min((long)8192, (total + 255) / 256));
  hipLaunchKernelGGL(im2col_bf16_kernel, grid, block, 0, stream, x,
                     (__bf16*)col, B, C, H, W, k, stride, pad, OH, OW);
}

void col2im_launch(const float* dcol, float* dx, int B, int C, int H, int W,
                   int k, int stride, int pad, int OH, int OW,
                   hipStream_t stream) {
  const long total = (long)B * C * H * W;
  dim3 block(256);
  dim3 grid((unsigned)min((long)8192, (total + 255) / 256));
  hipLaunchKernelGGL(col2im_kernel, grid, block, 0, stream, dcol, dx, B, C,
                     H, W, k, stride, pad, OH, OW);
}

}  // namespace lightctr
