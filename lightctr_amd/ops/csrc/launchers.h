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
                             float* gv, int B, int K, const int* pos,
                             ihipStream_t* stream);
void inv_perm_launch(const int* perm, int* inv, int n, ihipStream_t* stream);
void fm_sorted_apply_launch(const int* sorted_fids, const int* perm,
                            const float* gw, const float* gv, float* gradW,
                            float* gradV, unsigned long long* touched, int nnz,
                            int K, int opt_mode, float* V, float* W,
                            float* nW, float* zW, float* nV, float* zV,
                            float p0, float p1, float p2, float p3,
                            float q0, float q1, float q2,
                            int chunk, ihipStream_t* stream);
void bitmap_compact_launch(unsigned long long* bitmap, int nwords,
                           int* out_fids, int* out_count, int cap,
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

void ffm_sorted_backward_launch(const int* sorted_fids, const int* perm,
                                const int* row_of_entry, const int* row_ptr,
                                const int* fields, const int* fids,
                                const float* vals, const float* V,
                                const float* dpred, float* gradW,
                                float* gradV, unsigned long long* touched,
                                int nfields, int nnz, int K,
                                ihipStream_t* stream);
void row_index_launch(const int* row_ptr, int* row_idx, int B,
                      ihipStream_t* stream);
void ffm_block_emit_launch(const int* row_of_entry, const int* row_ptr,
                           const int* fields, const int* fids,
                           const float* vals, const float* V,
                           const float* dpred, float* gblocks, float* gw,
                           int nfields, int nnz, int K, ihipStream_t* stream);
void ffm_forward_pp_launch(const int* row_ptr, const int* fields,
                           const int* fids, const float* vals,
                           const float* W, const void* V, int v_bf16,
                           float* pred, int nfields, int B, int K,
                           ihipStream_t* stream);
bool ffm_staged_eligible(int nfields, int K, int maxn);
void ffm_fwd_staged_launch(const int* row_ptr, const int* fields,
                           const int* fids, const float* vals, const float* W,
                           const float* V, float* pred, int nfields, int B,
                           int maxn, int K, ihipStream_t* stream);
void ffm_row_emit_launch(const int* row_ptr, const int* fields,
                         const int* fids, const float* vals, const void* V,
                         int v_bf16, const float* dpred, void* gblocks,
                         float* gw, int nfields, int B, int maxn, int K,
                         float scale, ihipStream_t* stream);
void ffm_blocks_apply_f16_launch(const int* sorted_fids, const int* perm,
                                 const void* gblocks, const float* gw,
                                 float* gradW, float* gradV,
                                 unsigned long long* touched, int D, int nnz,
                                 float inv_scale, int opt_mode, float* V,
                                 float* W, float* nW, float* zW, float* nV,
                                 void* Vh, float p0, float p1, float p2,
                                 float p3, float q0, float q1, float q2,
                                 ihipStream_t* stream);
void ffm_blocks_apply_launch(const int* sorted_fids, const int* perm,
                             const float* gblocks, const float* gw,
                             float* gradW, float* gradV,
                             unsigned long long* touched, int D, int nnz,
                             ihipStream_t* stream);

// --- misc_kernels.hip (generic sparse fused optimizers; D = per-feature
// latent size, runtime) ---
void ps_apply_launch(const long* lidx, int n, const float* gW,
                     const float* gV, float* W, float* V, float* nW,
                     float* nV, float* shadowW, float* shadowV, int K,
                     int updater, float lr, float lam, float eps,
                     ihipStream_t* stream);
void auc_hist_add_launch(const float* pred, const float* label, long n,
                         unsigned int* hist, int buckets,
                         ihipStream_t* stream);
void auc_scan_launch(const unsigned int* hist, int buckets, double* out,
                     ihipStream_t* stream);
void sparse_adagrad_apply_launch(const int* uniq, const int* count, float* W,
                                 float* V, float* nW, float* nV, float* gradW,
                                 float* gradV, float lr, float eps, float l2,
                                 int capacity, int D, void* Vh,
                                 ihipStream_t* stream);
void sparse_ftrl_apply_launch(const int* uniq, const int* count, float* W,
                              float* V, float* zW, float* nW, float* zV,
                              float* nV, float* gradW, float* gradV,
                              float alpha, float beta, float l1, float l2,
                              int capacity, int D, int v_adagrad, float v_lr,
                              float v_eps, float v_l2, void* Vh,
                              ihipStream_t* stream);
void fm_adagrad_apply_launch(const int* uniq, const int* count, float* W,
                             float* V, float* nW, float* nV, float* gradW,
                             float* gradV, float lr, float eps, float l2,
                             int capacity, int K, ihipStream_t* stream);
void fm_ftrl_apply_launch(const int* uniq, const int* count, float* W,
                          float* V, float* zW, float* nW, float* zV, float* nV,
                          float* gradW, float* gradV, float alpha, float beta,
                          float l1, float l2, int capacity, int K,
                          int v_adagrad, float v_lr, float v_eps, float v_l2,
                          ihipStream_t* stream);

// --- gemm_wgrad_kernels.hip ---
bool gemm_wgrad_eligible(int M, int N, int K, int transA, int transB);
void gemm_wgrad_bf16_launch(const void* At, const void* Bt, float* C, int M,
                            int N, int K, ihipStream_t* stream);
void wg_probe_launch(const void* g, int ld, void* out_lds, float* out_frag,
                     int mode, ihipStream_t* stream);

// --- gemm_kernels.hip ---
void gemm_bf16_launch(const void* A, const void* Bst, const float* bias,
                      float* C, void* Cbf, int M, int N, int K, int transA,
                      int transB, int act, ihipStream_t* stream);

bool gemm256_eligible(int M, int N, int K, int transA, int transB);
bool gemm256p8_eligible(int M, int N, int K, int transA, int transB);
void gemm256p8_bf16_launch(const void* A, const void* Bst, const float* bias,
                           float* C, void* Cbf, int M, int N, int K, int act,
                           ihipStream_t* stream);
bool gemm256n128_eligible(int M, int N, int K, int transA, int transB);
void gemm256n128_bf16_launch(const void* A, const void* Bst,
                             const float* bias, float* C, void* Cbf, int M,
                             int N, int K, int act, ihipStream_t* stream);
void gemm256_bf16_launch(const void* A, const void* Bst, const float* bias,
                         float* C, void* Cbf, int M, int N, int K, int act,
                         ihipStream_t* stream);

// --- nn_kernels.hip ---
void small_wgrad_launch(const void* Ast, const void* Bst, float* C, int M,
                        int N, int K, ihipStream_t* stream);
void im2col_bf16_launch(const float* x, void* col, int B, int C, int H,
                        int W, int k, int stride, int pad, int OH, int OW,
                        ihipStream_t* stream);
void col2im_launch(const float* dcol, float* dx, int B, int C, int H, int W,
                   int k, int stride, int pad, int OH, int OW,
                   ihipStream_t* stream);
void gemv_n1_launch(const void* A, const void* b, const float* bias,
                    float* C, void* Cbf, int M, int K, int act,
                    ihipStream_t* stream);
void wrowsum_m1_launch(const void* a, const void* Bst, float* C, int K,
                       int N, ihipStream_t* stream);
void outer_k1_launch(const void* A, const void* Bst, const float* bias,
                     float* C, void* Cbf, long M, long N, int act,
                     ihipStream_t* stream);
void act_backward_launch(const float* dY, const float* Y, float* dZ,
                         void* dZbf, long n, int act, ihipStream_t* stream);
void colsum_launch(const float* dZ, float* db, int M, int N,
                   ihipStream_t* stream);
void to_bf16_launch(const float* x, void* y, long n, ihipStream_t* stream);
void dense_adam_launch(float* W, const float* grad, float* m_t, float* v_t,
                       void* Wbf, void* Wtbf, int rows, int cols, float lr,
                       float beta1, float beta2, float eps, float bc1,
                       float bc2, float l2, ihipStream_t* stream);
void dense_adagrad_launch(float* W, const float* grad, float* n_t, void* Wbf,
                          void* Wtbf, int rows, int cols, float lr, float eps,
                          float l2, ihipStream_t* stream);

// --- embed_kernels.hip ---
void embed_gather_launch(const int* row_ptr, const int* fids,
                         const float* vals, const float* E, void* out_bf,
                         int nf, int B, int K, ihipStream_t* stream);
void embed_backward_emit_launch(const int* row_ptr, const float* vals,
                                const float* dOut, const float* dwide,
                                float* gv, float* gw, int nf, int B, int K,
                                ihipStream_t* stream);
void wide_forward_launch(const int* row_ptr, const int* fids,
                         const float* vals, const float* W, float* wide,
                         int B, ihipStream_t* stream);
void nfm_forward_launch(const int* row_ptr, const int* fids, const float* vals,
                        const float* W, const float* V, float* wide,
                        float* sumVX, float* vec, void* vec_bf, int B, int K,
                        ihipStream_t* stream);
void w2v_negsample_launch(float* E, float* O, const long* centers,
                          const long* ctx, const long* negs, float* loss,
                          int B, int C, int N, int D, float lr, float scale,
                          ihipStream_t* stream);
void nfm_backward_emit_launch(const int* row_ptr, const int* fids,
                              const float* vals, const float* V,
                              const float* sumVX, const float* dvec,
                              const float* dwide, float* gw, float* gv, int B,
                              int K, ihipStream_t* stream);

// --- codec_kernels.hip ---
void quantile_encode_launch(const float* x, unsigned char* code,
                            const float* table, int levels, long n,
                            ihipStream_t* stream);
void quantile_decode_launch(const unsigned char* code, float* x,
                            const float* table, int levels, long n,
                            ihipStream_t* stream);
void lowbit_encode_launch(const float* x, unsigned int* words, float thresh,
                          int bits, long n, ihipStream_t* stream);
void lowbit_decode_launch(const unsigned int* words, float* x, float lo,
                          float hi, int bits, long n, ihipStream_t* stream);

// --- sort_kernels.hip ---
void iota_i32_launch(int* out, int n, ihipStream_t* stream);
unsigned long radix_sort_pairs_i32_temp_bytes(int n, int end_bit);
void radix_sort_pairs_i32_launch(void* temp, unsigned long temp_bytes,
                                 const int* keys_in, int* keys_out,
                                 const int* vals_in, int* vals_out, int n,
                                 int end_bit, ihipStream_t* stream);

// --- gbm_kernels.hip ---
void gbm_hist_launch(const unsigned char* bins, const float* grad,
                     const float* hess, const int* node_of_row, float* hist,
                     int N, int D, int n_nodes, ihipStream_t* stream);

}  // namespace lightctr
