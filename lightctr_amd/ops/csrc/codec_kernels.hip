// Gradient/parameter compression codecs for MI355X/gfx950.
//
// Parity targets (redesigned for GPU):
//  * int8/any-width quantile codec — reference util/quantile_compress.h
//    (uniform / log / normal-CDF tables :71-107, binary-search encode
//    :137-148); here: device binary search per element, table in LDS
//  * 1/2-bit quantization — reference util/product_quantizer.h:24-45
//  * fp16 wire codec (reference common/float16.h) is covered by packed
//    v_cvt conversions via torch .half()/.bfloat16() — no custom kernel
//    needed on CDNA4 (hardware packed converts).
#include "common.h"

namespace lightctr {

// Encode: code[i] = argmin_j |x[i] - table[j]| implemented as branchless
// binary search over the sorted boundaries mid[j] = (table[j]+table[j+1])/2.
// Table (<= 256 entries) is staged in LDS once per workgroup.
__global__ void quantile_encode_kernel(const float* __restrict__ x,
                                       unsigned char* __restrict__ code,
                                       const float* __restrict__ table,
                                       int levels, long n) {
  extern __shared__ float lds_tab[];
  for (int i = threadIdx.x; i < levels; i += blockDim.x)
    lds_tab[i] = table[i];
  __syncthreads();
  const long i = (long)blockIdx.x * blockDim.x + threadIdx.x;
  if (i >= n) return;
  const float v = x[i];
  int lo = 0, hi = levels - 1;
  while (lo < hi) {  // find first level with midpoint >= v
    const int mid = (lo + hi) >> 1;
    const float boundary = 0.5f * (lds_tab[mid] + lds_tab[mid + 1]);
    if (v <= boundary)
      hi = mid;
    else
      lo = mid + 1;
  }
  code[i] = (unsigned char)lo;
}

__global__ void quantile_decode_kernel(const unsigned char* __restrict__ code,
                                       float* __restrict__ x,
                                       const float* __restrict__ table,
                                       int levels, long n) {
  extern __shared__ float lds_tab[];
  for (int i = threadIdx.x; i < levels; i += blockDim.x)
    lds_tab[i] = table[i];
  __syncthreads();
  const long i = (long)blockIdx.x * blockDim.x + threadIdx.x;
  if (i >= n) return;
  x[i] = lds_tab[code[i]];
}

// 1-bit / 2-bit sign-magnitude quantization (reference lowbit_quantize):
// 1-bit: sign only, decode to +-scale (scale = mean |x|)
// 2-bit: sign + magnitude-above-threshold, decode to {+-lo, +-hi}
__global__ void lowbit_encode_kernel(const float* __restrict__ x,
                                     unsigned int* __restrict__ words,
                                     float thresh, int bits, long n) {
  const long i = (long)blockIdx.x * blockDim.x + threadIdx.x;
  const long nw = (n * bits + 31) / 32;
  if (i >= nw) return;
  unsigned int w = 0;
  const int per = 32 / bits;
  for (int e = 0; e < per; ++e) {
    const long j = i * per + e;
    if (j >= n) break;
    const float v = x[j];
    unsigned int c = (v >= 0.f) ? 1u : 0u;
    if (bits == 2) c |= (fabsf(v) >= thresh ? 2u : 0u);
    w |= c << (e * bits);
  }
  words[i] = w;
}

__global__ void lowbit_decode_kernel(const unsigned int* __restrict__ words,
                                     float* __restrict__ x, float lo,
                                     float hi, int bits, long n) {
  const long i = (long)blockIdx.x * blockDim.x + threadIdx.x;
  if (i >= n) return;
  const int per = 32 / bits;
  const unsigned int w = words[i / per];
  const unsigned int c = (w >> ((i % per) * bits)) & ((1u << bits) - 1u);
  const float sign = (c & 1u) ? 1.f : -1.f;
  const float mag = (bits == 2 && (c & 2u)) ? hi : lo;
  x[i] = sign * mag;
}

void quantile_encode_launch(const float* x, unsigned char* code,
                            const float* table, int levels, long n,
                            hipStream_t stream) {
  dim3 block(256);
  dim3 grid((unsigned)((n + 255) / 256));
  hipLaunchKernelGGL(quantile_encode_kernel, grid, block,
                     levels * sizeof(float), stream, x, code, table, levels,
                     n);
}

void quantile_decode_launch(const unsigned char* code, float* x,
                            const float* table, int levels, long n,
                            hipStream_t stream) {
  dim3 block(256);
  dim3 grid((unsigned)((n + 255) / 256));
  hipLaunchKernelGGL(quantile_decode_kernel, grid, block,
                     levels * sizeof(float), stream, code, x, table, levels,
                     n);
}

void lowbit_encode_launch(const float* x, unsigned int* words, float thresh,
                          int bits, long n, hipStream_t stream) {
  const long nw = (n * bits + 31) / 32;
  dim3 block(256);
  dim3 grid((unsigned)((nw + 255) / 256));
  hipLaunchKernelGGL(lowbit_encode_kernel, grid, block, 0, stream, x, words,
                     thresh, bits, n);
}

void lowbit_decode_launch(const unsigned int* words, float* x, float lo,
                          float hi, int bits, long n, hipStream_t stream) {
  dim3 block(256);
  dim3 grid((unsigned)((n + 255) / 256));
  hipLaunchKernelGGL(lowbit_decode_kernel, grid, block, 0, stream, words, x,
                     lo, hi, bits, n);
}

}  // namespace lightctr
