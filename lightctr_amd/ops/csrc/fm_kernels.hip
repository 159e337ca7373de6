// Fused FM (factorization machine) kernels for MI355X / gfx950.
//
// Capability parity with the reference FM trainer
// (/root/reference/LightCTR/train/train_fm_algo.cpp:63-127: forward via the
// O(nk) sumVX trick, backward gradV = d*(sumVX - v*x)*x, Adagrad apply;
// FTRL per reference util/gradientUpdater.h:235-278) — redesigned for CDNA4:
//   * one 64-lane wavefront per CSR row; lane = (feature_group, factor k)
//     so a wave gathers G=64/K embedding rows in parallel, K floats each
//   * cross-lane combine via __shfl_xor (no LDS round trip needed at K<=64)
//   * backward scatters into dense per-feature gradient slabs with fp32
//     atomics + a touched-bitmap; a compaction kernel turns the bitmap into
//     a unique-fid list; the optimizer (Adagrad / FTRL) is a separate fused
//     kernel over that list only (sparse apply, zero host round trips).
//
// All buffers live in HBM3E; the whole step is 5 kernel launches and is
// hipGraph-capturable (no device->host syncs).
#include <cstdio>
#include "common.h"

namespace lightctr {

// FTRL-proximal (per-coordinate), semantics of the reference FTRL updater
// (gradientUpdater.h:235-278): z,n state per parameter.
__device__ __forceinline__ void ftrl_update(float* __restrict__ w,
                                            float* __restrict__ z,
                                            float* __restrict__ n, float g,
                                            float alpha, float beta, float l1,
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


// ---------------------------------------------------------------------------
// Forward: pred[row] = sum_j w[fid]*x + 0.5*(||sumVX||^2 - sum_j ||v*x||^2)
// Also writes sumVX[row*K+k] (needed by backward).
// ---------------------------------------------------------------------------
template <int K>
__global__ void fm_forward_kernel(const int* __restrict__ row_ptr,
                                  const int* __restrict__ fids,
                                  const float* __restrict__ vals,
                                  const float* __restrict__ W,
                                  const float* __restrict__ V,
                                  float* __restrict__ pred,
                                  float* __restrict__ sumVX, int B) {
  constexpr int G = LCTR_WAVE / K;  // features processed in parallel
  const int lane = threadIdx.x & (LCTR_WAVE - 1);
  const int row = blockIdx.x * (blockDim.x / LCTR_WAVE) + (threadIdx.x >> 6);
  if (row >= B) return;
  const int g = lane / K;
  const int k = lane % K;
  const int beg = row_ptr[row], end = row_ptr[row + 1];

  float sVX = 0.f;    // partial sum_j V[fid,k]*x over this lane's features
  float sV2X2 = 0.f;  // partial sum_j (V[fid,k]*x)^2
  float lin = 0.f;    // partial sum_j W[fid]*x (k==0 lanes only)
  // 4-deep pipelined gathers (same latency argument as the emit kernel)
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
  // combine feature groups: lanes {k, K+k, 2K+k, ...} sum -> every lane
  // holds the row's total sumVX for its k (replicated G times).
  sVX = group_reduce_sum<K>(sVX);
  const float t = sVX * sVX;  // replicated G times per k
  const float tot_t = wave_reduce_sum(t) * (1.f / G);
  const float tot_v2 = wave_reduce_sum(sV2X2);
  const float tot_lin = wave_reduce_sum(lin);
  if (lane < K) sumVX[(size_t)row * K + lane] = sVX;
  if (lane == 0) pred[row] = tot_lin + 0.5f * (tot_t - tot_v2);
}

// ---------------------------------------------------------------------------
// Loss: numerically stable logloss + dpred = (sigmoid(pred) - y) * scale.
// Writes per-row loss (host reduces with a library sum) and the upstream
// gradient used by the backward kernel. Also emits correct-count bits for
// accuracy if acc != nullptr.
// ---------------------------------------------------------------------------
__global__ void logloss_grad_kernel(const float* __restrict__ pred,
                                    const float* __restrict__ label,
                                    float* __restrict__ loss,
                                    float* __restrict__ dpred, float scale,
                                    int B) {
  const int i = blockIdx.x * blockDim.x + threadIdx.x;
  if (i >= B) return;
  const float z = pred[i];
  const float y = label[i];
  // loss = max(z,0) - z*y + log(1+exp(-|z|))  (stable logloss)
  loss[i] = fmaxf(z, 0.f) - z * y + __logf(1.f + __expf(-fabsf(z)));
  dpred[i] = (sigmoidf_clamped(z) - y) * scale;
}

// ---------------------------------------------------------------------------
// Backward: scatter-accumulate per-feature grads with fp32 atomics.
//   gradW[fid]    += d * x
//   gradV[fid,k]  += d * (sumVX[k]*x - V[fid,k]*x^2)
// Marks fid in a 64-bit touched bitmap for the sparse optimizer pass.
// ---------------------------------------------------------------------------
template <int K>
__global__ void fm_backward_kernel(
    const int* __restrict__ row_ptr, const int* __restrict__ fids,
    const float* __restrict__ vals, const float* __restrict__ V,
    const float* __restrict__ sumVX, const float* __restrict__ dpred,
    float* __restrict__ gradW, float* __restrict__ gradV,
    unsigned long long* __restrict__ touched, int B) {
  constexpr int G = LCTR_WAVE / K;
  const int lane = threadIdx.x & (LCTR_WAVE - 1);
  const int row = blockIdx.x * (blockDim.x / LCTR_WAVE) + (threadIdx.x >> 6);
  if (row >= B) return;
  const int g = lane / K;
  const int k = lane % K;
  const float d = dpred[row];
  const float sv = sumVX[(size_t)row * K + k];
  const int beg = row_ptr[row], end = row_ptr[row + 1];
  for (int j = beg + g; j < end; j += G) {
    const int fid = fids[j];
    const float x = vals[j];
    const float v = V[(size_t)fid * K + k];
    atomicAdd(&gradV[(size_t)fid * K + k], d * (sv - v * x) * x);
    if (k == 0) {
      atomicAdd(&gradW[fid], d * x);
      atomicOr(&touched[fid >> 6], 1ull << (fid & 63));
    }
  }
}

// ---------------------------------------------------------------------------
// Sorted backward, phase 1 (emit): write per-entry gradient contributions
// coalesced, no atomics:
//   gw[j]   = d * x_j
//   gv[j,k] = d * (sumVX[k] - V[fid_j,k]*x_j) * x_j
// The batch's fids are then radix-sorted (rocPRIM bit-range sort,
// sort_kernels.hip) and phase 2 (fm_sorted_apply_kernel) segment-reduces
// them into the dense grad slabs
// with at most a handful of atomics per unique feature — this removes the
// hot-feature atomic serialization that dominates the naive scatter backward
// (measured 2.76 ms/step vs ~0.1 ms total for everything else; see
// profiles/r01_fm_atomic_backward.txt).
// ---------------------------------------------------------------------------
// `pos` (optional): per-entry write slot. When given, entry j's gradient
// lands at gv[pos[j]] / gw[pos[j]] — i.e. directly in SORTED order (pos =
// inverse of the sort permutation). That moves the randomness of the
// sorted pipeline from the apply kernel's dependent 64 B GATHERS (latency
// bound, measured 285 us/step) onto this kernel's independent STORES
// (fire-and-forget through L2), and lets the apply read gv sequentially.
template <int K>
__global__ void fm_backward_emit_kernel(
    const int* __restrict__ row_ptr, const int* __restrict__ fids,
    const float* __restrict__ vals, const float* __restrict__ V,
    const float* __restrict__ sumVX, const float* __restrict__ dpred,
    float* __restrict__ gw, float* __restrict__ gv, int B,
    const int* __restrict__ pos) {
  constexpr int G = LCTR_WAVE / K;
  const int lane = threadIdx.x & (LCTR_WAVE - 1);
  const int row = blockIdx.x * (blockDim.x / LCTR_WAVE) + (threadIdx.x >> 6);
  if (row >= B) return;
  const int g = lane / K;
  const int k = lane % K;
  const float d = dpred[row];
  const float sv = sumVX[(size_t)row * K + k];
  const int beg = row_ptr[row], end = row_ptr[row + 1];
  // 4-deep manual pipeline: the V row gathers are the latency bound
  // (119 us vs a ~42 us traffic floor); issuing 4 independent gathers
  // per lane before consuming them quadruples the bytes in flight
  int j = beg + g;
  for (; j + 3 * G < end; j += 4 * G) {
    int f[4], sl[4];
    float x[4], v[4];
#pragma unroll
    for (int u = 0; u < 4; ++u) {
      const int jj = j + u * G;
      f[u] = fids[jj];
      x[u] = vals[jj];
      sl[u] = pos ? pos[jj] : jj;
    }
#pragma unroll
    for (int u = 0; u < 4; ++u) v[u] = V[(size_t)f[u] * K + k];
#pragma unroll
    for (int u = 0; u < 4; ++u) {
      gv[(size_t)sl[u] * K + k] = d * (sv - v[u] * x[u]) * x[u];
      if (k == 0) gw[sl[u]] = d * x[u];
    }
  }
  for (; j < end; j += G) {
    const int fid = fids[j];
    const float x = vals[j];
    const int slot = pos ? pos[j] : j;
    gv[(size_t)slot * K + k] = d * (sv - V[(size_t)fid * K + k] * x) * x;
    if (k == 0) gw[slot] = d * x;
  }
}

// inv[perm[i]] = i : build the emit kernel's write-slot array from the
// sort permutation (sequential reads of perm, scattered int32 stores).
__global__ void inv_perm_kernel(const int* __restrict__ perm,
                                int* __restrict__ inv, int n) {
  const int i = blockIdx.x * blockDim.x + threadIdx.x;
  if (i < n) inv[perm[i]] = i;
}

// Sorted backward, phase 2: segment-reduce sorted per-entry grads into the
// dense slabs. Each 64-lane wave owns a contiguous chunk of sorted entries;
// K-lane subgroups walk their stride, accumulating runs of equal fid in
// registers and flushing with one atomicAdd per run. Segment heads also set
// the touched bitmap (one atomicOr per unique fid).
// opt_mode: 0 = slab-only (accumulate into gradW/gradV + touched bitmap);
//           1 = fused Adagrad, 2 = fused FTRL. In fused modes, any feature
//           whose sorted segment lies ENTIRELY within one subgroup's range
//           (checked via the global neighbors sorted_fids[first-1] /
//           sorted_fids[last+1]) holds its TOTAL batch gradient at flush
//           time, so the optimizer update is applied in place with no
//           atomics and the slab/bitmap/compaction/apply pass only handles
//           the ~2 boundary-spanning features per subgroup.
struct FmOptArgs {
  float* W;
  float* nW;
  float* zW;
  float* nV;
  float* zV;
  float p0, p1, p2, p3;  // adagrad: lr, eps, l2 | ftrl: alpha, beta, l1, l2
  // opt_mode 3 (FTRL on W + Adagrad on V — the default FM optimizer
  // pairing, models/fm.py ftrl_v="adagrad"): q* are the V-side knobs
  float q0, q1, q2;  // v_lr, v_eps, v_l2
};

// WPE: min waves/SIMD handed to the allocator (HIP __launch_bounds__
// 2nd arg). The K=16 walk allocates 86 VGPR — one register over the
// 6-wave boundary (512/6 = 85.3); WPE=6 caps it at 85 for +20%
// occupancy on this latency-bound kernel (LCTR_FM_APPLY_WPE=6 A/B).
template <int K, bool PP = true, int WPE = 1>
__global__ __launch_bounds__(256, WPE) void fm_sorted_apply_kernel(
    const int* __restrict__ sorted_fids, const int* __restrict__ perm,
    const float* __restrict__ gw, const float* __restrict__ gv,
    float* __restrict__ gradW, float* __restrict__ gradV,
    unsigned long long* __restrict__ touched, int nnz, int chunk,
    int opt_mode, float* __restrict__ V, FmOptArgs oa) {
  constexpr int G = LCTR_WAVE / K;
  const int lane = threadIdx.x & (LCTR_WAVE - 1);
  const int wave = blockIdx.x * (blockDim.x / LCTR_WAVE) + (threadIdx.x >> 6);
  const int g = lane / K;
  const int k = lane % K;
  const int base = wave * chunk;
  if (base >= nnz) return;
  const int end = min(base + chunk, nnz);
  // each K-lane subgroup owns a CONTIGUOUS sub-chunk (runs stay contiguous
  // -> one flush per run) and walks it 4 entries at a time with the gathers
  // issued up-front (4 independent 64B loads in flight per subgroup: the
  // random gv gather is latency-bound, unrolling feeds the memory system)
  const int sub = (chunk + G - 1) / G;
  const int sb = min(base + g * sub, end);
  const int se = min(sb + sub, end);

  int cur_fid = -1;
  bool head_ok = false;  // this subgroup saw the segment's global start
  float acc = 0.f, accw = 0.f;

  auto flush = [&](int tail_e) {
    if (cur_fid < 0) return;
    // tail_e = index of the first entry AFTER the run
    const bool tail_ok =
        (tail_e >= nnz) || (sorted_fids[tail_e] != cur_fid);
    if (opt_mode != 0 && head_ok && tail_ok) {
      // exclusive owner of this feature's total gradient: fused update
      const size_t off = (size_t)cur_fid * K + k;
      if (opt_mode == 1) {  // adagrad
        const float gg = acc + oa.p2 * V[off];
        const float a = oa.nV[off] + gg * gg;
        oa.nV[off] = a;
        V[off] -= oa.p0 * gg * __frsqrt_rn(a + oa.p1);
        if (k == 0) {
          const float gwv = accw + oa.p2 * oa.W[cur_fid];
          const float aw = oa.nW[cur_fid] + gwv * gwv;
          oa.nW[cur_fid] = aw;
          oa.W[cur_fid] -= oa.p0 * gwv * __frsqrt_rn(aw + oa.p1);
        }
      } else if (opt_mode == 3) {  // FTRL on W, Adagrad on V
        const float gg = acc + oa.q2 * V[off];
        const float a = oa.nV[off] + gg * gg;
        oa.nV[off] = a;
        V[off] -= oa.q0 * gg * __frsqrt_rn(a + oa.q1);
        if (k == 0) {
          ftrl_update(&oa.W[cur_fid], &oa.zW[cur_fid], &oa.nW[cur_fid],
                      accw, oa.p0, oa.p1, oa.p2, oa.p3);
        }
      } else {  // ftrl
        ftrl_update(&V[off], &oa.zV[off], &oa.nV[off], acc, oa.p0, oa.p1,
                    oa.p2, oa.p3);
        if (k == 0) {
          ftrl_update(&oa.W[cur_fid], &oa.zW[cur_fid], &oa.nW[cur_fid],
                      accw, oa.p0, oa.p1, oa.p2, oa.p3);
        }
      }
    } else if (head_ok && tail_ok) {
      // exclusive owner of the whole segment: the slabs are zero for every
      // untouched fid (the optimizer pass zeroes what it consumes), so a
      // plain coalesced store replaces 17 atomics. Measured: the flush
      // atomics, not the gv reads, bound this kernel (tools/bench_apply.py).
      gradV[(size_t)cur_fid * K + k] = acc;
      if (k == 0) {
        gradW[cur_fid] = accw;
        atomicOr(&touched[cur_fid >> 6], 1ull << (cur_fid & 63));
      }
    } else {
      atomicAdd(&gradV[(size_t)cur_fid * K + k], acc);
      if (k == 0) {
        atomicAdd(&gradW[cur_fid], accw);
        atomicOr(&touched[cur_fid >> 6], 1ull << (cur_fid & 63));
      }
    }
  };

  // 2-deep batch pipeline — MEASURED NEGATIVE within-probe (210 vs
  // 220 us at chunk 384): despite the PMC profile (VALUBusy 9.6%,
  // MemUnitStalled 0.6% — latency idling), prefetching the next 8-entry
  // batch costs more in register pressure / issue placement than the
  // extra lines in flight buy at ~6.5 waves/SIMD. Kept selectable
  // (chunk=-4) as the documented experiment; the single-buffer walk
  // stays default. Named ping-pong buffers (NOT runtime-indexed arrays
  // — those allocate in scratch, guide common-mistake 20).
  float vA[8], vwA[8], vB[8], vwB[8];
  int fA[8], fB[8];
#define FM_WALK_LOAD(e0, VV, VW, FF)                                       \
  do {                                                                     \
    const int nv_ = min(8, se - (e0));                                     \
    _Pragma("unroll") for (int u = 0; u < 8; ++u) {                        \
      if (u < nv_) {                                                       \
        FF[u] = sorted_fids[(e0) + u];                                     \
        const long p_ = perm ? (long)perm[(e0) + u] : (long)((e0) + u);    \
        VV[u] = gv[(size_t)p_ * K + k];                                    \
        VW[u] = (k == 0) ? gw[p_] : 0.f;                                   \
      }                                                                    \
    }                                                                      \
  } while (0)
#define FM_WALK_PROC(e0, VV, VW, FF)                                       \
  do {                                                                     \
    const int nv_ = min(8, se - (e0));                                     \
    _Pragma("unroll") for (int u = 0; u < 8; ++u) {                        \
      if (u >= nv_) break;                                                 \
      if (FF[u] != cur_fid) {                                              \
        flush((e0) + u);                                                   \
        cur_fid = FF[u];                                                   \
        acc = 0.f;                                                         \
        accw = 0.f;                                                        \
        head_ok = ((e0) + u == 0 || sorted_fids[(e0) + u - 1] != FF[u]);   \
      }                                                                    \
      acc += VV[u];                                                        \
      accw += VW[u];                                                       \
    }                                                                      \
  } while (0)
  if (PP) {
    if (sb < se) FM_WALK_LOAD(sb, vA, vwA, fA);
    for (int e = sb; e < se; e += 16) {
      if (e + 8 < se) FM_WALK_LOAD(e + 8, vB, vwB, fB);
      FM_WALK_PROC(e, vA, vwA, fA);
      if (e + 8 >= se) break;
      if (e + 16 < se) FM_WALK_LOAD(e + 16, vA, vwA, fA);
      FM_WALK_PROC(e + 8, vB, vwB, fB);
    }
  } else {  // round-1 single-buffer walk (within-probe A/B baseline)
    for (int e = sb; e < se; e += 8) {
      FM_WALK_LOAD(e, vA, vwA, fA);
      FM_WALK_PROC(e, vA, vwA, fA);
    }
  }
#undef FM_WALK_LOAD
#undef FM_WALK_PROC
  flush(se);
}

// ---------------------------------------------------------------------------
// Sorted backward, phase 2, segmented-scan variant (K <= 16): one LANE per
// sorted entry, the entry's whole [K] gradient row + gw held in registers.
// A wave covers 64 consecutive sorted entries; run sums are produced by a
// 6-step intra-wave SEGMENTED inclusive scan (shfl_up with head-flag
// propagation) instead of the walk kernel's serial per-subgroup loop —
// the measured bound of the walk was exactly that serial run-detection
// chain + flush divergence (tools/bench_apply.py), not the gv gathers.
// Segment tails write their run's total: a plain 64 B row store when the
// segment's head is in-wave (exclusive owner — slabs are zeroed by the
// optimizer pass), atomicAdd only for the <=2 wave-spanning segments.
// ---------------------------------------------------------------------------
template <int K>
__global__ void fm_segscan_apply_kernel(
    const int* __restrict__ sorted_fids, const int* __restrict__ perm,
    const float* __restrict__ gw, const float* __restrict__ gv,
    float* __restrict__ gradW, float* __restrict__ gradV,
    unsigned long long* __restrict__ touched, int nnz) {
  static_assert(K <= 16, "register-vector variant holds the row per lane");
  const int lane = threadIdx.x & (LCTR_WAVE - 1);
  const int wave = blockIdx.x * (blockDim.x >> 6) + (threadIdx.x >> 6);
  const long base = (long)wave * LCTR_WAVE;
  if (base >= nnz) return;
  const long e = base + lane;
  const bool valid = e < nnz;

  const int fid = valid ? sorted_fids[e] : -1;
  // original head flag: first entry of a run (lane 0 checks its global
  // predecessor — one extra cached read per wave)
  bool head0 = false;
  if (valid) head0 = (e == 0) || (sorted_fids[e - 1] != fid);

  float v[K + 1];  // [0..K) = gv row, [K] = gw
  if (valid) {
    const long p = perm ? (long)perm[e] : e;
    const float4* src = (const float4*)&gv[(size_t)p * K];
#pragma unroll
    for (int q = 0; q < K / 4; ++q) {
      const float4 t = src[q];
      v[q * 4 + 0] = t.x;
      v[q * 4 + 1] = t.y;
      v[q * 4 + 2] = t.z;
      v[q * 4 + 3] = t.w;
    }
    v[K] = gw[p];
  } else {
#pragma unroll
    for (int i = 0; i <= K; ++i) v[i] = 0.f;
  }

  // segmented inclusive scan: f = "a head lies within my covered window";
  // add the incoming prefix only while my window is still head-free.
  unsigned f = head0 ? 1u : 0u;
#pragma unroll
  for (int s = 1; s < LCTR_WAVE; s <<= 1) {
    const unsigned tf = __shfl_up(f, s);
    const bool take = (lane >= s) && (f == 0u);
    float tv[K + 1];
#pragma unroll
    for (int i = 0; i <= K; ++i) tv[i] = __shfl_up(v[i], s);
    if (take) {
#pragma unroll
      for (int i = 0; i <= K; ++i) v[i] += tv[i];
    }
    if (lane >= s) f |= tf;
  }

  // tail = last entry of a run: the next entry starts a new run (its
  // original head flag), or the stream ends here.
  const bool nh = __shfl_down((int)head0, 1);
  bool tail = false;
  if (valid) {
    if (e + 1 >= nnz) {
      tail = true;
    } else if (lane == LCTR_WAVE - 1) {
      tail = sorted_fids[e + 1] != fid;
    } else {
      tail = nh;
    }
  }

  if (tail) {
    if (f) {
      // head in-wave -> this lane holds the run's TOTAL and owns it
      float4* dst = (float4*)&gradV[(size_t)fid * K];
#pragma unroll
      for (int q = 0; q < K / 4; ++q) {
        float4 t;
        t.x = v[q * 4 + 0];
        t.y = v[q * 4 + 1];
        t.z = v[q * 4 + 2];
        t.w = v[q * 4 + 3];
        dst[q] = t;
      }
      gradW[fid] = v[K];
    } else {
#pragma unroll
      for (int i = 0; i < K; ++i)
        atomicAdd(&gradV[(size_t)fid * K + i], v[i]);
      atomicAdd(&gradW[fid], v[K]);
    }
    atomicOr(&touched[fid >> 6], 1ull << (fid & 63));
  } else if (valid && lane == LCTR_WAVE - 1) {
    // run continues into the next wave: flush this wave's partial sum
#pragma unroll
    for (int i = 0; i < K; ++i)
      atomicAdd(&gradV[(size_t)fid * K + i], v[i]);
    atomicAdd(&gradW[fid], v[K]);
    atomicOr(&touched[fid >> 6], 1ull << (fid & 63));
  }
}

// ---------------------------------------------------------------------------
// Sorted backward, phase 2, QUAD-PER-ENTRY segmented scan (round 2,
// docs/STATUS.md lever 3): K/4 lanes per entry, each holding one float4
// quad of the gv row (+ gw on the first lane) — 5 live registers instead
// of the 17 that made the lane-per-entry segscan lose to the walk. The
// wave covers 64/(K/4) consecutive sorted entries; run totals come from
// a log2-step segmented inclusive scan over entry slots (shfl_up at
// entry stride), tails flush with one dwordx4 store per lane (plain for
// in-wave-head runs — slabs zeroed by the optimizer pass — atomics only
// for wave-spanning runs).
// ---------------------------------------------------------------------------
template <int K>
__global__ void fm_segscan4_apply_kernel(
    const int* __restrict__ sorted_fids, const int* __restrict__ perm,
    const float* __restrict__ gw, const float* __restrict__ gv,
    float* __restrict__ gradW, float* __restrict__ gradV,
    unsigned long long* __restrict__ touched, int nnz) {
  static_assert(K % 4 == 0 && K >= 4 && K <= 64, "quad layout needs K%4==0");
  constexpr int L = K / 4;            // lanes per entry
  constexpr int EPW = LCTR_WAVE / L;  // entries per wave
  const int lane = threadIdx.x & (LCTR_WAVE - 1);
  const int wave = blockIdx.x * (blockDim.x >> 6) + (threadIdx.x >> 6);
  const long base = (long)wave * EPW;
  if (base >= nnz) return;
  const int ei = lane / L;
  const int q = lane % L;
  const long e = base + ei;
  const bool valid = e < nnz;

  const int fid = valid ? sorted_fids[e] : -1;
  const bool head0 = valid && ((e == 0) || (sorted_fids[e - 1] != fid));

  float4 v = make_float4(0.f, 0.f, 0.f, 0.f);
  float w = 0.f;
  if (valid) {
    const long p = perm ? (long)perm[e] : e;
    v = ((const float4*)&gv[(size_t)p * K])[q];
    if (q == 0) w = gw[p];
  }

  unsigned f = head0 ? 1u : 0u;
#pragma unroll
  for (int s = L; s < LCTR_WAVE; s <<= 1) {
    const unsigned tf = __shfl_up(f, s);
    float4 tv;
    tv.x = __shfl_up(v.x, s);
    tv.y = __shfl_up(v.y, s);
    tv.z = __shfl_up(v.z, s);
    tv.w = __shfl_up(v.w, s);
    const float tw = __shfl_up(w, s);
    if ((lane >= s) && (f == 0u)) {
      v.x += tv.x;
      v.y += tv.y;
      v.z += tv.z;
      v.w += tv.w;
      w += tw;
    }
    if (lane >= s) f |= tf;
  }

  const bool tail = valid && ((e + 1 >= nnz) || (sorted_fids[e + 1] != fid));
  if (tail && f) {
    // run's head is in-wave -> this entry's lanes hold the TOTAL and own
    // the feature exclusively: plain coalesced row store
    ((float4*)&gradV[(size_t)fid * K])[q] = v;
    if (q == 0) {
      gradW[fid] = w;
      atomicOr(&touched[fid >> 6], 1ull << (fid & 63));
    }
  } else if (tail || (valid && ei == EPW - 1)) {
    // spanning run (tail without in-wave head, or wave's last entry with
    // the run continuing): atomic partial flush
    atomicAdd(&gradV[(size_t)fid * K + q * 4 + 0], v.x);
    atomicAdd(&gradV[(size_t)fid * K + q * 4 + 1], v.y);
    atomicAdd(&gradV[(size_t)fid * K + q * 4 + 2], v.z);
    atomicAdd(&gradV[(size_t)fid * K + q * 4 + 3], v.w);
    if (q == 0) {
      atomicAdd(&gradW[fid], w);
      atomicOr(&touched[fid >> 6], 1ull << (fid & 63));
    }
  }
}

// ---------------------------------------------------------------------------
// Bitmap -> unique-fid list compaction. One thread per 64-feature word;
// clears the word as it goes so the bitmap is reusable next step.
// ---------------------------------------------------------------------------
__global__ void bitmap_compact_kernel(unsigned long long* __restrict__ bitmap,
                                      int nwords, int* __restrict__ out_fids,
                                      int* __restrict__ out_count, int cap) {
  // BLOCK-aggregated append (1024 threads = 16 waves): waves prefix-scan
  // locally, wave totals are scanned in LDS, and only thread 0 touches
  // the global counter — one atomic per 65536 features instead of per
  // wave (the single-counter atomics serialized the old kernel: 57 us
  // for a 2 MB pass).
  __shared__ int wave_tot[16];
  __shared__ int block_base;
  const int w = blockIdx.x * blockDim.x + threadIdx.x;
  const int lane = threadIdx.x & 63;
  const int wv = threadIdx.x >> 6;
  unsigned long long m = (w < nwords) ? bitmap[w] : 0ull;
  if (m) bitmap[w] = 0ull;
  const int n = __popcll(m);
  int scan = n;
#pragma unroll
  for (int s = 1; s < LCTR_WAVE; s <<= 1) {
    const int t = __shfl_up(scan, s);
    if (lane >= s) scan += t;
  }
  if (lane == LCTR_WAVE - 1) wave_tot[wv] = scan;
  __syncthreads();
  if (threadIdx.x == 0) {
    int run = 0;
#pragma unroll
    for (int i = 0; i < 16; ++i) {
      const int t = wave_tot[i];
      wave_tot[i] = run;
      run += t;
    }
    block_base = run > 0 ? atomicAdd(out_count, run) : 0;
  }
  __syncthreads();
  int off = block_base + wave_tot[wv] + scan - n;
  while (m) {
    const int b = __ffsll((long long)m) - 1;
    m &= m - 1;
    if (off < cap) out_fids[off] = w * 64 + b;  // clamp: never overrun uniq
    ++off;
  }
}

// ---------------------------------------------------------------------------
// Sparse fused optimizers over the unique-fid list. Thread (i,k) owns
// V[fid_i, k]; lane k==0 additionally owns W[fid_i]. Reads the accumulated
// gradient, applies the update + state, and zeroes the gradient slot so the
// slabs stay clean for the next step. Launched over `capacity` (max possible
// uniques) and guarded by *count so no host sync is needed.
// ---------------------------------------------------------------------------
template <int K>
__global__ void fm_adagrad_apply_kernel(
    const int* __restrict__ uniq, const int* __restrict__ count,
    float* __restrict__ W, float* __restrict__ V, float* __restrict__ nW,
    float* __restrict__ nV, float* __restrict__ gradW,
    float* __restrict__ gradV, float lr, float eps, float l2, int capacity) {
  const long long idx = (long long)blockIdx.x * blockDim.x + threadIdx.x;
  const long long i = idx / K;
  const int k = (int)(idx % K);
  if (i >= capacity || i >= *count) return;
  const int fid = uniq[i];
  const size_t off = (size_t)fid * K + k;
  const float g = gradV[off] + l2 * V[off];
  const float acc = nV[off] + g * g;
  nV[off] = acc;
  V[off] -= lr * g * __frsqrt_rn(acc + eps);
  gradV[off] = 0.f;
  if (k == 0) {
    const float gw = gradW[fid] + l2 * W[fid];
    const float a = nW[fid] + gw * gw;
    nW[fid] = a;
    W[fid] -= lr * gw * __frsqrt_rn(a + eps);
    gradW[fid] = 0.f;
  }
}

// v_adagrad: the classic CTR split — FTRL-proximal on the sparse linear
// weights W, Adagrad on the latent factors V (FTRL's L1 shrinkage starves
// the interaction terms; measured 0.66 vs 0.79 held-out AUC uniform-FTRL
// vs mixed on the synthetic Criteo stream).
template <int K>
__global__ void fm_ftrl_apply_kernel(
    const int* __restrict__ uniq, const int* __restrict__ count,
    float* __restrict__ W, float* __restrict__ V, float* __restrict__ zW,
    float* __restrict__ nW, float* __restrict__ zV, float* __restrict__ nV,
    float* __restrict__ gradW, float* __restrict__ gradV, float alpha,
    float beta, float l1, float l2, int capacity, int v_adagrad, float v_lr,
    float v_eps, float v_l2) {
  const long long idx = (long long)blockIdx.x * blockDim.x + threadIdx.x;
  const long long i = idx / K;
  const int k = (int)(idx % K);
  if (i >= capacity || i >= *count) return;
  const int fid = uniq[i];
  const size_t off = (size_t)fid * K + k;
  if (v_adagrad) {
    const float g = gradV[off] + v_l2 * V[off];
    const float acc = nV[off] + g * g;
    nV[off] = acc;
    V[off] -= v_lr * g * __frsqrt_rn(acc + v_eps);
  } else {
    ftrl_update(&V[off], &zV[off], &nV[off], gradV[off], alpha, beta, l1,
                l2);
  }
  gradV[off] = 0.f;
  if (k == 0) {
    ftrl_update(&W[fid], &zW[fid], &nW[fid], gradW[fid], alpha, beta, l1, l2);
    gradW[fid] = 0.f;
  }
}

// ---------------------------------------------------------------------------
// Host-side launchers (thin; stream comes from the caller / PyTorch).
// ---------------------------------------------------------------------------
static inline int waves_per_block() { return 4; }  // 256 threads

void fm_forward_launch(const int* row_ptr, const int* fids, const float* vals,
                       const float* W, const float* V, float* pred,
                       float* sumVX, int B, int K, hipStream_t stream) {
  if (B <= 0) return;
  const int wpb = waves_per_block();
  dim3 block(wpb * LCTR_WAVE);
  dim3 grid((B + wpb - 1) / wpb);
  DISPATCH_K(K, hipLaunchKernelGGL((fm_forward_kernel<KC>), grid, block, 0,
                                   stream, row_ptr, fids, vals, W, V, pred,
                                   sumVX, B));
}

void logloss_grad_launch(const float* pred, const float* label, float* loss,
                         float* dpred, float scale, int B,
                         hipStream_t stream) {
  if (B <= 0) return;
  dim3 block(256);
  dim3 grid((B + 255) / 256);
  hipLaunchKernelGGL(logloss_grad_kernel, grid, block, 0, stream, pred, label,
                     loss, dpred, scale, B);
}

void fm_backward_launch(const int* row_ptr, const int* fids, const float* vals,
                        const float* V, const float* sumVX, const float* dpred,
                        float* gradW, float* gradV, unsigned long long* touched,
                        int B, int K, hipStream_t stream) {
  if (B <= 0) return;
  const int wpb = waves_per_block();
  dim3 block(wpb * LCTR_WAVE);
  dim3 grid((B + wpb - 1) / wpb);
  DISPATCH_K(K, hipLaunchKernelGGL((fm_backward_kernel<KC>), grid, block, 0,
                                   stream, row_ptr, fids, vals, V, sumVX,
                                   dpred, gradW, gradV, touched, B));
}

void fm_backward_emit_launch(const int* row_ptr, const int* fids,
                             const float* vals, const float* V,
                             const float* sumVX, const float* dpred, float* gw,
                             float* gv, int B, int K, const int* pos,
                             hipStream_t stream) {
  if (B <= 0) return;
  const int wpb = waves_per_block();
  dim3 block(wpb * LCTR_WAVE);
  dim3 grid((B + wpb - 1) / wpb);
  DISPATCH_K(K, hipLaunchKernelGGL((fm_backward_emit_kernel<KC>), grid, block,
                                   0, stream, row_ptr, fids, vals, V, sumVX,
                                   dpred, gw, gv, B, pos));
}

void inv_perm_launch(const int* perm, int* inv, int n, hipStream_t stream) {
  if (n <= 0) return;
  dim3 block(256);
  dim3 grid((n + 255) / 256);
  hipLaunchKernelGGL(inv_perm_kernel, grid, block, 0, stream, perm, inv, n);
}

void fm_sorted_apply_launch(const int* sorted_fids, const int* perm,
                            const float* gw, const float* gv, float* gradW,
                            float* gradV, unsigned long long* touched, int nnz,
                            int K, int opt_mode, float* V, float* W,
                            float* nW, float* zW, float* nV, float* zV,
                            float p0, float p1, float p2, float p3,
                            float q0, float q1, float q2,
                            int chunk, hipStream_t stream) {
  if (nnz <= 0) return;
  // chunk 0 = auto (walk kernel, chunk 384 — measured best), -1 = the
  // lane-per-entry segmented-scan kernel (correct, measured 275 vs 218
  // us: the 6-step 17-register shfl_up chain costs more than the serial
  // walk saves; kept selectable + parity-tested as a documented negative
  // result), -2 = the round-2 quad-per-entry segscan (K/4 lanes per
  // entry, 5 live registers).
  if (chunk == -2 && (K % 4 == 0) && K <= 64 && opt_mode == 0) {
    const int wpb = waves_per_block();
    const int epw = LCTR_WAVE / (K / 4);
    const int nwaves = (nnz + epw - 1) / epw;
    dim3 block(wpb * LCTR_WAVE);
    dim3 grid((nwaves + wpb - 1) / wpb);
    DISPATCH_K(K, hipLaunchKernelGGL((fm_segscan4_apply_kernel<KC>), grid,
                                     block, 0, stream, sorted_fids, perm,
                                     gw, gv, gradW, gradV, touched, nnz));
    return;
  }
  const bool can_scan =
      (K == 4 || K == 8 || K == 16) && opt_mode == 0;
  if (chunk == -1 && can_scan) {
    const int wpb = waves_per_block();
    const int nwaves = (nnz + LCTR_WAVE - 1) / LCTR_WAVE;
    dim3 block(wpb * LCTR_WAVE);
    dim3 grid((nwaves + wpb - 1) / wpb);
    switch (K) {
      case 4:
        hipLaunchKernelGGL((fm_segscan_apply_kernel<4>), grid, block, 0,
                           stream, sorted_fids, perm, gw, gv, gradW, gradV,
                           touched, nnz);
        return;
      case 8:
        hipLaunchKernelGGL((fm_segscan_apply_kernel<8>), grid, block, 0,
                           stream, sorted_fids, perm, gw, gv, gradW, gradV,
                           touched, nnz);
        return;
      default:
        hipLaunchKernelGGL((fm_segscan_apply_kernel<16>), grid, block, 0,
                           stream, sorted_fids, perm, gw, gv, gradW, gradV,
                           touched, nnz);
        return;
    }
  }
  // chunk == -4 selects the 2-deep pipelined walk (measured negative;
  // kept for A/B); default = single-buffer walk
  const bool pp = chunk == -4;
  if (chunk == -3 || chunk == -4) chunk = 0;
  if (chunk <= 0) chunk = 384;  // measured optimum, tools/bench_apply.py
  const int wpb = waves_per_block();
  const int nwaves = (nnz + chunk - 1) / chunk;
  dim3 block(wpb * LCTR_WAVE);
  dim3 grid((nwaves + wpb - 1) / wpb);
  FmOptArgs oa{W, nW, zW, nV, zV, p0, p1, p2, p3, q0, q1, q2};
  if (pp) {
    DISPATCH_K(K, hipLaunchKernelGGL((fm_sorted_apply_kernel<KC, true>),
                                     grid, block, 0, stream, sorted_fids,
                                     perm, gw, gv, gradW, gradV, touched,
                                     nnz, chunk, opt_mode, V, oa));
  } else {
    // default for the flagship K=16: the 6-wave cap trims 86 -> 80
    // VGPR (no spill) and measured 213.4 vs 222.5 us (+4.3%, 8/8
    // alternating passes, profiles/r2_13_fm_wpe.txt). '0' reverts.
    const char* ew = getenv("LCTR_FM_APPLY_WPE");
    if (K == 16 && !(ew && ew[0] == '0')) {
      hipLaunchKernelGGL((fm_sorted_apply_kernel<16, false, 6>), grid,
                         block, 0, stream, sorted_fids, perm, gw, gv,
                         gradW, gradV, touched, nnz, chunk, opt_mode, V,
                         oa);
      return;
    }
    DISPATCH_K(K, hipLaunchKernelGGL((fm_sorted_apply_kernel<KC, false>),
                                     grid, block, 0, stream, sorted_fids,
                                     perm, gw, gv, gradW, gradV, touched,
                                     nnz, chunk, opt_mode, V, oa));
  }
}

void bitmap_compact_launch(unsigned long long* bitmap, int nwords,
                           int* out_fids, int* out_count, int cap,
                           hipStream_t stream) {
  dim3 block(1024);
  dim3 grid((nwords + 1023) / 1024);
  hipLaunchKernelGGL(bitmap_compact_kernel, grid, block, 0, stream, bitmap,
                     nwords, out_fids, out_count, cap);
}

void fm_adagrad_apply_launch(const int* uniq, const int* count, float* W,
                             float* V, float* nW, float* nV, float* gradW,
                             float* gradV, float lr, float eps, float l2,
                             int capacity, int K, hipStream_t stream) {
  const long long total = (long long)capacity * K;
  dim3 block(256);
  dim3 grid((unsigned)((total + 255) / 256));
  DISPATCH_K(K, hipLaunchKernelGGL((fm_adagrad_apply_kernel<KC>), grid, block,
                                   0, stream, uniq, count, W, V, nW, nV, gradW,
                                   gradV, lr, eps, l2, capacity));
}

void fm_ftrl_apply_launch(const int* uniq, const int* count, float* W,
                          float* V, float* zW, float* nW, float* zV, float* nV,
                          float* gradW, float* gradV, float alpha, float beta,
                          float l1, float l2, int capacity, int K,
                          int v_adagrad, float v_lr, float v_eps, float v_l2,
                          hipStream_t stream) {
  const long long total = (long long)capacity * K;
  dim3 block(256);
  dim3 grid((unsigned)((total + 255) / 256));
  DISPATCH_K(K, hipLaunchKernelGGL((fm_ftrl_apply_kernel<KC>), grid, block, 0,
                                   stream, uniq, count, W, V, zW, nW, zV, nV,
                                   gradW, gradV, alpha, beta, l1, l2,
                                   capacity, v_adagrad, v_lr, v_eps, v_l2));
}

}  // namespace lightctr
