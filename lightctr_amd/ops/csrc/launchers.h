// Declarations of the HIP kernel launchers, shared between the device
// translation units (*.hip, compiled by hipcc for gfx950) and the host
// bindings (bindings.cpp). hipStream_t == ihipStream_t* on ROCm.
#pragma once

struct ihipStream_t;

namespace lightctr {

// --- fm_kernels.hip ---
void fm_forward_launch(const int* row_ptr, const int* fids, const float* vals,
                       const float* W, const float* V, float* pred,
                       float* sumVX, int B, int K, ihipStream_t* stream);
void logloss_grad_launch(const float* pred, const float* label, float* loss,
                         float* dpred, float scale, int B,
                         ihipStream_t* stream);
void fm_backward_launch(const int* row_ptr, const int* fids, const float* vals,
                        const float* V, const float* sumVX, const float* dpred,
                        float* gradW, float* gradV, unsigned long long* touched,
                        int B, int K, ihipStream_t* stream);
void fm_backward_emit_launch(const int* row_ptr, const int* fids,
                             const float* vals, const float* V,
                             const float* sumVX, const float* dpred, float* gw,
                             float* gv, int B, int K, ihipStream_t* stream);
void fm_sorted_apply_launch(const int* sorted_fids, const long* perm,
                            const float* gw, const float* gv, float* gradW,
                            float* gradV, unsigned long long* touched, int nnz,
                            int K, ihipStream_t* stream);
void bitmap_compact_launch(unsigned long long* bitmap, int nwords,
                           int* out_fids, int* out_count,
                           ihipStream_t* stream);

// --- ffm_kernels.hip ---
void ffm_forward_launch(const int* row_ptr, const int* fields, const int* fids,
                        const float* vals, const float* W, const float* V,
                        float* pred, int nfields, int B, int K,
                        ihipStream_t* stream);
void ffm_backward_launch(const int* row_ptr, const int* fields,
                         const int* fids, const float* vals, const float* V,
                         const float* dpred, float* gradW, float* gradV,
                         unsigned long long* touched, int nfields, int B,
                         int K, ihipStream_t* stream);

// --- misc_kernels.hip (generic sparse fused optimizers; D = per-feature
// latent size, runtime) ---
void sparse_adagrad_apply_launch(const int* uniq, const int* count, float* W,
                                 float* V, float* nW, float* nV, float* gradW,
                                 float* gradV, float lr, float eps, float l2,
                                 int capacity, int D, ihipStream_t* stream);
void sparse_ftrl_apply_launch(const int* uniq, const int* count, float* W,
                              float* V, float* zW, float* nW, float* zV,
                              float* nV, float* gradW, float* gradV,
                              float alpha, float beta, float l1, float l2,
                              int capacity, int D, ihipStream_t* stream);
void fm_adagrad_apply_launch(const int* uniq, const int* count, float* W,
                             float* V, float* nW, float* nV, float* gradW,
                             float* gradV, float lr, float eps, float l2,
                             int capacity, int K, ihipStream_t* stream);
void fm_ftrl_apply_launch(const int* uniq, const int* count, float* W,
                          float* V, float* zW, float* nW, float* zV, float* nV,
                          float* gradW, float* gradV, float alpha, float beta,
                          float l1, float l2, int capacity, int K,
                          ihipStream_t* stream);

}  // namespace lightctr
