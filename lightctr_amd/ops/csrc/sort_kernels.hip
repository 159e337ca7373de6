// Bit-range radix sort for feature-id streams (MI355X / gfx950).
//
// The sorted segment-reduce backward (fm_kernels.hip) needs the batch's
// fids sorted with an index permutation every step. torch.sort runs a
// full 32-bit radix (4 passes over keys+values); feature ids are bounded
// by the table size (<= 2^24 for the Criteo-shaped config, <= U for
// localized ids), so rocPRIM's device radix sort with an explicit
// [0, end_bit) range drops a pass — pure HBM-bytes savings on a
// bandwidth-bound primitive. rocPRIM's LSD sort is stable, which the
// segment-reduce does not even require (any order within an equal-key
// run accumulates the same), so correctness is a strict superset.
//
// Values are int32 (nnz < 2^31): half the payload torch.sort's int64
// indices move through the 2-3 radix passes. Host-side wrappers live
// here because rocPRIM instantiates device kernels (must be compiled by
// hipcc); tensor plumbing stays in bindings.cpp.
#include <rocprim/device/device_radix_sort.hpp>

#include "common.h"

namespace lightctr {

__global__ void iota_i32_kernel(int* __restrict__ out, int n) {
  int i = blockIdx.x * blockDim.x + threadIdx.x;
  if (i < n) out[i] = i;
}

void iota_i32_launch(int* out, int n, hipStream_t stream) {
  if (n <= 0) return;
  int threads = 256;
  int blocks = (n + threads - 1) / threads;
  hipLaunchKernelGGL(iota_i32_kernel, dim3(blocks), dim3(threads), 0, stream,
                     out, n);
}

size_t radix_sort_pairs_i32_temp_bytes(int n, int end_bit) {
  size_t bytes = 0;
  (void)rocprim::radix_sort_pairs(nullptr, bytes, (const int*)nullptr,
                                  (int*)nullptr, (const int*)nullptr,
                                  (int*)nullptr, (size_t)n, 0u,
                                  (unsigned)end_bit, (hipStream_t)0, false);
  return bytes;
}

void radix_sort_pairs_i32_launch(void* temp, size_t temp_bytes,
                                 const int* keys_in, int* keys_out,
                                 const int* vals_in, int* vals_out, int n,
                                 int end_bit, hipStream_t stream) {
  LCTR_CHECK_HIP(rocprim::radix_sort_pairs(temp, temp_bytes, keys_in,
                                           keys_out, vals_in, vals_out,
                                           (size_t)n, 0u, (unsigned)end_bit,
                                           stream, false));
}

}  // namespace lightctr
