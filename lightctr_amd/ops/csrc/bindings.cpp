// PyTorch bindings for the lightctr_amd HIP kernels (MI355X / gfx950) +
// the native data loader. Host-only translation unit.
#include <torch/extension.h>
#include <ATen/cuda/CUDAContext.h>

#include <cstdio>
#include <cstdlib>
#include <cstring>
#include <stdexcept>
#include <vector>

#include "launchers.h"

namespace {

#define CHK(x, ...) TORCH_CHECK(x, __VA_ARGS__)

void check_cuda_f32(const at::Tensor& t, const char* name) {
  CHK(t.is_cuda(), name, " must be on GPU");
  CHK(t.scalar_type() == at::kFloat, name, " must be fp32");
  CHK(t.is_contiguous(), name, " must be contiguous");
}

void check_cuda_i32(const at::Tensor& t, const char* name) {
  CHK(t.is_cuda(), name, " must be on GPU");
  CHK(t.scalar_type() == at::kInt, name, " must be int32");
  CHK(t.is_contiguous(), name, " must be contiguous");
}

ihipStream_t* cur_stream() {
  return (ihipStream_t*)at::cuda::getCurrentCUDAStream().stream();
}

// sort permutations are int32 end-to-end (half the radix payload of
// torch.sort's int64 indices); accept int64 for compatibility (e.g.
// torch.sort callers in tests) by converting.
at::Tensor perm32(const at::Tensor& perm) {
  if (perm.scalar_type() == at::kInt) return perm;
  CHK(perm.scalar_type() == at::kLong, "perm must be int32 or int64");
  return perm.to(at::kInt);
}

// ---- FM ----

std::vector<at::Tensor> fm_forward(at::Tensor row_ptr, at::Tensor fids,
                                   at::Tensor vals, at::Tensor W,
                                   at::Tensor V) {
  check_cuda_i32(row_ptr, "row_ptr");
  check_cuda_i32(fids, "fids");
  check_cuda_f32(vals, "vals");
  check_cuda_f32(W, "W");
  check_cuda_f32(V, "V");
  const int B = (int)row_ptr.numel() - 1;
  const int K = (int)V.size(1);
  auto pred = at::empty({B}, W.options());
  auto sumVX = at::empty({B, K}, W.options());
  lightctr::fm_forward_launch(row_ptr.data_ptr<int>(), fids.data_ptr<int>(),
                              vals.data_ptr<float>(), W.data_ptr<float>(),
                              V.data_ptr<float>(), pred.data_ptr<float>(),
                              sumVX.data_ptr<float>(), B, K, cur_stream());
  return {pred, sumVX};
}

std::vector<at::Tensor> logloss_grad(at::Tensor pred, at::Tensor label,
                                     double scale) {
  check_cuda_f32(pred, "pred");
  check_cuda_f32(label, "label");
  const int B = (int)pred.numel();
  auto loss = at::empty_like(pred);
  auto dpred = at::empty_like(pred);
  lightctr::logloss_grad_launch(pred.data_ptr<float>(),
                                label.data_ptr<float>(),
                                loss.data_ptr<float>(), dpred.data_ptr<float>(),
                                (float)scale, B, cur_stream());
  return {loss, dpred};
}

void fm_backward(at::Tensor row_ptr, at::Tensor fids, at::Tensor vals,
                 at::Tensor V, at::Tensor sumVX, at::Tensor dpred,
                 at::Tensor gradW, at::Tensor gradV, at::Tensor touched) {
  check_cuda_i32(row_ptr, "row_ptr");
  check_cuda_i32(fids, "fids");
  check_cuda_f32(vals, "vals");
  check_cuda_f32(V, "V");
  check_cuda_f32(sumVX, "sumVX");
  check_cuda_f32(dpred, "dpred");
  check_cuda_f32(gradW, "gradW");
  check_cuda_f32(gradV, "gradV");
  CHK(touched.scalar_type() == at::kLong || touched.scalar_type() == at::kUInt64,
      "touched must be 64-bit");
  const int B = (int)row_ptr.numel() - 1;
  const int K = (int)V.size(1);
  lightctr::fm_backward_launch(
      row_ptr.data_ptr<int>(), fids.data_ptr<int>(), vals.data_ptr<float>(),
      V.data_ptr<float>(), sumVX.data_ptr<float>(), dpred.data_ptr<float>(),
      gradW.data_ptr<float>(), gradV.data_ptr<float>(),
      (unsigned long long*)touched.data_ptr(), B, K, cur_stream());
}

std::vector<at::Tensor> fm_backward_emit(at::Tensor row_ptr, at::Tensor fids,
                                         at::Tensor vals, at::Tensor V,
                                         at::Tensor sumVX, at::Tensor dpred,
                                         c10::optional<at::Tensor> pos) {
  check_cuda_i32(row_ptr, "row_ptr");
  check_cuda_i32(fids, "fids");
  const int B = (int)row_ptr.numel() - 1;
  const int K = (int)V.size(1);
  const auto nnz = fids.numel();
  auto gw = at::empty({nnz}, V.options());
  auto gv = at::empty({nnz, K}, V.options());
  const int* pos_ptr = nullptr;
  if (pos.has_value()) {
    check_cuda_i32(*pos, "pos");
    CHK(pos->numel() == nnz, "pos must have one slot per entry");
    pos_ptr = pos->data_ptr<int>();
  }
  lightctr::fm_backward_emit_launch(
      row_ptr.data_ptr<int>(), fids.data_ptr<int>(), vals.data_ptr<float>(),
      V.data_ptr<float>(), sumVX.data_ptr<float>(), dpred.data_ptr<float>(),
      gw.data_ptr<float>(), gv.data_ptr<float>(), B, K, pos_ptr,
      cur_stream());
  return {gw, gv};
}

// inverse permutation (write slots for scatter-emit): inv[perm[i]] = i
at::Tensor inv_perm_i32(at::Tensor perm_in) {
  auto perm = perm32(perm_in).contiguous();
  CHK(perm.is_cuda(), "perm must be on GPU");
  const int n = (int)perm.numel();
  auto inv = at::empty({n}, perm.options());
  lightctr::inv_perm_launch(perm.data_ptr<int>(), inv.data_ptr<int>(), n,
                            cur_stream());
  return inv;
}

void fm_sorted_apply(at::Tensor sorted_fids, c10::optional<at::Tensor> perm,
                     at::Tensor gw, at::Tensor gv, at::Tensor gradW,
                     at::Tensor gradV, at::Tensor touched, int64_t chunk) {
  check_cuda_i32(sorted_fids, "sorted_fids");
  const int* perm_ptr = nullptr;
  at::Tensor perm_i;
  if (perm.has_value()) {
    perm_i = perm32(*perm).contiguous();
    perm_ptr = perm_i.data_ptr<int>();
  }
  const int K = (int)gradV.size(1);
  lightctr::fm_sorted_apply_launch(
      sorted_fids.data_ptr<int>(), perm_ptr,
      gw.data_ptr<float>(), gv.data_ptr<float>(), gradW.data_ptr<float>(),
      gradV.data_ptr<float>(), (unsigned long long*)touched.data_ptr(),
      (int)sorted_fids.numel(), K, 0, nullptr, nullptr, nullptr, nullptr,
      nullptr, nullptr, 0.f, 0.f, 0.f, 0.f, 0.f, 0.f, 0.f, (int)chunk,
      cur_stream());
}

// Fused variant: interior feature segments get their optimizer update
// applied during the segment reduction; only boundary-spanning features go
// through the slab + bitmap (+ the follow-up sparse apply).
void fm_sorted_apply_fused(at::Tensor sorted_fids,
                           c10::optional<at::Tensor> perm,
                           at::Tensor gw, at::Tensor gv, at::Tensor gradW,
                           at::Tensor gradV, at::Tensor touched,
                           at::Tensor W, at::Tensor V, at::Tensor nW,
                           at::Tensor nV, c10::optional<at::Tensor> zW,
                           c10::optional<at::Tensor> zV, int64_t opt_mode,
                           double p0, double p1, double p2, double p3,
                           double v_lr, double v_eps, double v_l2) {
  check_cuda_i32(sorted_fids, "sorted_fids");
  const int* perm_ptr = nullptr;
  at::Tensor perm_i;
  if (perm.has_value()) {
    perm_i = perm32(*perm).contiguous();
    perm_ptr = perm_i.data_ptr<int>();
  }
  CHK(opt_mode == 1 || opt_mode == 2 || opt_mode == 3,
      "opt_mode 1=adagrad 2=ftrl 3=ftrlW+adagradV");
  if (opt_mode == 3) CHK(zW.has_value(), "opt_mode 3 needs zW");
  const int K = (int)gradV.size(1);
  lightctr::fm_sorted_apply_launch(
      sorted_fids.data_ptr<int>(), perm_ptr,
      gw.data_ptr<float>(), gv.data_ptr<float>(), gradW.data_ptr<float>(),
      gradV.data_ptr<float>(), (unsigned long long*)touched.data_ptr(),
      (int)sorted_fids.numel(), K, (int)opt_mode, V.data_ptr<float>(),
      W.data_ptr<float>(), nW.data_ptr<float>(),
      zW.has_value() ? zW->data_ptr<float>() : nullptr,
      nV.data_ptr<float>(),
      zV.has_value() ? zV->data_ptr<float>() : nullptr, (float)p0,
      (float)p1, (float)p2, (float)p3, (float)v_lr, (float)v_eps,
      (float)v_l2, 384, cur_stream());
}

// ---- FFM ----

at::Tensor ffm_forward(at::Tensor row_ptr, at::Tensor fields, at::Tensor fids,
                       at::Tensor vals, at::Tensor W, at::Tensor V) {
  check_cuda_i32(row_ptr, "row_ptr");
  check_cuda_i32(fields, "fields");
  check_cuda_i32(fids, "fids");
  const bool v_bf16 = V.scalar_type() == at::kBFloat16;
  CHK(v_bf16 || V.scalar_type() == at::kFloat, "V must be fp32 or bf16");
  const int B = (int)row_ptr.numel() - 1;
  const int nfields = (int)V.size(1);
  const int K = (int)V.size(2);
  auto pred = at::empty({B}, W.options());
  // lane-per-pair layout: 458 vs 1174 us for the (pair-group, k) layout
  // at B=65536/nf=39/K=8 (tools/ab_ffm_fwd.py; bit-identical output).
  // The group kernel and a per-row LDS-staged variant (1611 us — LDS
  // capacity caps occupancy) stay in ffm_kernels.hip as documented
  // experiments.
  lightctr::ffm_forward_pp_launch(
      row_ptr.data_ptr<int>(), fields.data_ptr<int>(), fids.data_ptr<int>(),
      vals.data_ptr<float>(), W.data_ptr<float>(), V.data_ptr(),
      v_bf16 ? 1 : 0, pred.data_ptr<float>(), nfields, B, K, cur_stream());
  return pred;
}

// lane-per-pair forward variant (A/B against the group layout)
at::Tensor ffm_forward_pp(at::Tensor row_ptr, at::Tensor fields,
                          at::Tensor fids, at::Tensor vals, at::Tensor W,
                          at::Tensor V) {
  check_cuda_i32(row_ptr, "row_ptr");
  const int B = (int)row_ptr.numel() - 1;
  const int nfields = (int)V.size(1);
  const int K = (int)V.size(2);
  auto pred = at::empty({B}, W.options());
  lightctr::ffm_forward_pp_launch(
      row_ptr.data_ptr<int>(), fields.data_ptr<int>(), fids.data_ptr<int>(),
      vals.data_ptr<float>(), W.data_ptr<float>(), V.data_ptr(),
      V.scalar_type() == at::kBFloat16 ? 1 : 0, pred.data_ptr<float>(),
      nfields, B, K, cur_stream());
  return pred;
}

// per-row staged fp16 block emit -> (gw fp32, gblocks fp16 [nnz, nf*K])
std::vector<at::Tensor> ffm_row_emit(at::Tensor row_ptr, at::Tensor fields,
                                     at::Tensor fids, at::Tensor vals,
                                     at::Tensor V, at::Tensor dpred,
                                     double scale) {
  check_cuda_i32(row_ptr, "row_ptr");
  check_cuda_i32(fields, "fields");
  check_cuda_i32(fids, "fids");
  const bool v_bf16 = V.scalar_type() == at::kBFloat16;
  CHK(v_bf16 || V.scalar_type() == at::kFloat, "V must be fp32 or bf16");
  const int B = (int)row_ptr.numel() - 1;
  const int nfields = (int)V.size(1);
  const int K = (int)V.size(2);
  const int maxn = 40;
  CHK(lightctr::ffm_staged_eligible(nfields, K, maxn),
      "nfields*K too large for the staged row emit");
  const auto nnz = fids.numel();
  auto gw = at::empty({nnz}, vals.options());
  auto gblocks = at::empty({nnz, (long)nfields * K},
                           vals.options().dtype(at::kHalf));
  lightctr::ffm_row_emit_launch(
      row_ptr.data_ptr<int>(), fields.data_ptr<int>(), fids.data_ptr<int>(),
      vals.data_ptr<float>(), V.data_ptr(), v_bf16 ? 1 : 0,
      dpred.data_ptr<float>(), gblocks.data_ptr(), gw.data_ptr<float>(),
      nfields, B, maxn, K, (float)scale, cur_stream());
  return {gw, gblocks};
}

void ffm_blocks_apply_f16(at::Tensor sorted_fids, at::Tensor perm,
                          at::Tensor gblocks, at::Tensor gw, at::Tensor gradW,
                          at::Tensor gradV, at::Tensor touched,
                          double inv_scale, int64_t opt_mode,
                          c10::optional<at::Tensor> V,
                          c10::optional<at::Tensor> W,
                          c10::optional<at::Tensor> nW,
                          c10::optional<at::Tensor> zW,
                          c10::optional<at::Tensor> nV,
                          c10::optional<at::Tensor> Vh,
                          double p0, double p1, double p2, double p3,
                          double q0, double q1, double q2) {
  check_cuda_i32(sorted_fids, "sorted_fids");
  auto perm_i = perm32(perm).contiguous();
  CHK(gblocks.scalar_type() == at::kHalf, "gblocks must be fp16");
  CHK(opt_mode == 0 || opt_mode == 1 || opt_mode == 3,
      "opt_mode 0=slab 1=adagrad 3=ftrlW+adagradV");
  if (opt_mode != 0) {
    CHK(V && W && nW && nV, "fused apply needs V/W/nW/nV");
    if (opt_mode == 3) CHK(zW.has_value(), "opt_mode 3 needs zW");
  }
  const int D = (int)gradV.size(1);
  lightctr::ffm_blocks_apply_f16_launch(
      sorted_fids.data_ptr<int>(), perm_i.data_ptr<int>(), gblocks.data_ptr(),
      gw.data_ptr<float>(), gradW.data_ptr<float>(), gradV.data_ptr<float>(),
      (unsigned long long*)touched.data_ptr(), D, (int)sorted_fids.numel(),
      (float)inv_scale, (int)opt_mode,
      V ? V->data_ptr<float>() : nullptr,
      W ? W->data_ptr<float>() : nullptr,
      nW ? nW->data_ptr<float>() : nullptr,
      zW ? zW->data_ptr<float>() : nullptr,
      nV ? nV->data_ptr<float>() : nullptr,
      Vh ? Vh->data_ptr() : nullptr, (float)p0, (float)p1, (float)p2,
      (float)p3, (float)q0, (float)q1, (float)q2, cur_stream());
}

void ffm_backward(at::Tensor row_ptr, at::Tensor fields, at::Tensor fids,
                  at::Tensor vals, at::Tensor V, at::Tensor dpred,
                  at::Tensor gradW, at::Tensor gradV, at::Tensor touched) {
  check_cuda_i32(row_ptr, "row_ptr");
  const int B = (int)row_ptr.numel() - 1;
  const int nfields = (int)V.size(1);
  const int K = (int)V.size(2);
  lightctr::ffm_backward_launch(
      row_ptr.data_ptr<int>(), fields.data_ptr<int>(), fids.data_ptr<int>(),
      vals.data_ptr<float>(), V.data_ptr<float>(), dpred.data_ptr<float>(),
      gradW.data_ptr<float>(), gradV.data_ptr<float>(),
      (unsigned long long*)touched.data_ptr(), nfields, B, K, cur_stream());
}

void ffm_sorted_backward(at::Tensor sorted_fids, at::Tensor perm,
                         at::Tensor row_of_entry, at::Tensor row_ptr,
                         at::Tensor fields, at::Tensor fids, at::Tensor vals,
                         at::Tensor V, at::Tensor dpred, at::Tensor gradW,
                         at::Tensor gradV, at::Tensor touched) {
  check_cuda_i32(sorted_fids, "sorted_fids");
  auto perm_i = perm32(perm).contiguous();
  check_cuda_i32(row_of_entry, "row_of_entry");
  const int nfields = (int)V.size(1);
  const int K = (int)V.size(2);
  lightctr::ffm_sorted_backward_launch(
      sorted_fids.data_ptr<int>(), perm_i.data_ptr<int>(),
      row_of_entry.data_ptr<int>(), row_ptr.data_ptr<int>(),
      fields.data_ptr<int>(), fids.data_ptr<int>(), vals.data_ptr<float>(),
      V.data_ptr<float>(), dpred.data_ptr<float>(), gradW.data_ptr<float>(),
      gradV.data_ptr<float>(), (unsigned long long*)touched.data_ptr(),
      nfields, (int)sorted_fids.numel(), K, cur_stream());
}

std::vector<at::Tensor> ffm_block_emit(at::Tensor row_of_entry,
                                       at::Tensor row_ptr, at::Tensor fields,
                                       at::Tensor fids, at::Tensor vals,
                                       at::Tensor V, at::Tensor dpred) {
  check_cuda_i32(row_of_entry, "row_of_entry");
  check_cuda_f32(V, "V");
  const int nfields = (int)V.size(1);
  const int K = (int)V.size(2);
  const auto nnz = fids.numel();
  auto gblocks = at::empty({nnz, (long)nfields * K}, V.options());
  auto gw = at::empty({nnz}, V.options());
  lightctr::ffm_block_emit_launch(
      row_of_entry.data_ptr<int>(), row_ptr.data_ptr<int>(),
      fields.data_ptr<int>(), fids.data_ptr<int>(), vals.data_ptr<float>(),
      V.data_ptr<float>(), dpred.data_ptr<float>(),
      gblocks.data_ptr<float>(), gw.data_ptr<float>(), nfields, (int)nnz, K,
      cur_stream());
  return {gw, gblocks};
}

void ffm_blocks_apply(at::Tensor sorted_fids, at::Tensor perm,
                      at::Tensor gblocks, at::Tensor gw, at::Tensor gradW,
                      at::Tensor gradV, at::Tensor touched) {
  check_cuda_i32(sorted_fids, "sorted_fids");
  auto perm_i = perm32(perm).contiguous();
  const int D = (int)gblocks.size(1);
  lightctr::ffm_blocks_apply_launch(
      sorted_fids.data_ptr<int>(), perm_i.data_ptr<int>(),
      gblocks.data_ptr<float>(), gw.data_ptr<float>(),
      gradW.data_ptr<float>(), gradV.data_ptr<float>(),
      (unsigned long long*)touched.data_ptr(), D,
      (int)sorted_fids.numel(), cur_stream());
}

at::Tensor row_index(at::Tensor row_ptr, int64_t nnz) {
  check_cuda_i32(row_ptr, "row_ptr");
  auto out = at::empty({nnz}, row_ptr.options());
  lightctr::row_index_launch(row_ptr.data_ptr<int>(), out.data_ptr<int>(),
                             (int)row_ptr.numel() - 1, cur_stream());
  return out;
}

void ps_apply(at::Tensor lidx, at::Tensor gW, at::Tensor gV, at::Tensor W,
              at::Tensor V, c10::optional<at::Tensor> nW,
              c10::optional<at::Tensor> nV,
              c10::optional<at::Tensor> shadowW,
              c10::optional<at::Tensor> shadowV, int64_t updater, double lr,
              double lam, double eps) {
  CHK(lidx.is_cuda() && lidx.scalar_type() == at::kLong, "lidx i64 GPU");
  const int K = (int)V.size(1);
  lightctr::ps_apply_launch(
      lidx.data_ptr<long>(), (int)lidx.numel(), gW.data_ptr<float>(),
      gV.data_ptr<float>(), W.data_ptr<float>(), V.data_ptr<float>(),
      nW.has_value() ? nW->data_ptr<float>() : nullptr,
      nV.has_value() ? nV->data_ptr<float>() : nullptr,
      shadowW.has_value() ? shadowW->data_ptr<float>() : nullptr,
      shadowV.has_value() ? shadowV->data_ptr<float>() : nullptr, K,
      (int)updater, (float)lr, (float)lam, (float)eps, cur_stream());
}

// ---- generic sparse optimizers (D = latent block width, runtime) ----

void sparse_adagrad_apply(at::Tensor uniq, at::Tensor count, at::Tensor W,
                          at::Tensor V, at::Tensor nW, at::Tensor nV,
                          at::Tensor gradW, at::Tensor gradV, double lr,
                          double eps, double l2,
                          c10::optional<at::Tensor> Vh) {
  check_cuda_i32(uniq, "uniq");
  const int D = (int)(V.numel() / V.size(0));
  if (Vh) CHK(Vh->scalar_type() == at::kBFloat16, "Vh mirror must be bf16");
  lightctr::sparse_adagrad_apply_launch(
      uniq.data_ptr<int>(), count.data_ptr<int>(), W.data_ptr<float>(),
      V.data_ptr<float>(), nW.data_ptr<float>(), nV.data_ptr<float>(),
      gradW.data_ptr<float>(), gradV.data_ptr<float>(), (float)lr, (float)eps,
      (float)l2, (int)uniq.numel(), D,
      Vh ? Vh->data_ptr() : nullptr, cur_stream());
}

void sparse_ftrl_apply(at::Tensor uniq, at::Tensor count, at::Tensor W,
                       at::Tensor V, at::Tensor zW, at::Tensor nW,
                       at::Tensor zV, at::Tensor nV, at::Tensor gradW,
                       at::Tensor gradV, double alpha, double beta, double l1,
                       double l2, int64_t v_adagrad, double v_lr,
                       double v_eps, double v_l2,
                       c10::optional<at::Tensor> Vh) {
  check_cuda_i32(uniq, "uniq");
  const int D = (int)(V.numel() / V.size(0));
  if (Vh) CHK(Vh->scalar_type() == at::kBFloat16, "Vh mirror must be bf16");
  lightctr::sparse_ftrl_apply_launch(
      uniq.data_ptr<int>(), count.data_ptr<int>(), W.data_ptr<float>(),
      V.data_ptr<float>(), zW.data_ptr<float>(), nW.data_ptr<float>(),
      zV.data_ptr<float>(), nV.data_ptr<float>(), gradW.data_ptr<float>(),
      gradV.data_ptr<float>(), (float)alpha, (float)beta, (float)l1,
      (float)l2, (int)uniq.numel(), D, (int)v_adagrad, (float)v_lr,
      (float)v_eps, (float)v_l2, Vh ? Vh->data_ptr() : nullptr,
      cur_stream());
}

at::Tensor gemm_wgrad_bf16(at::Tensor At, at::Tensor Bt, int64_t M,
                           int64_t N, int64_t K) {
  CHK(At.is_cuda() && At.scalar_type() == at::kBFloat16 &&
          At.is_contiguous(), "At must be cuda bf16 [K,M]");
  CHK(Bt.is_cuda() && Bt.scalar_type() == at::kBFloat16 &&
          Bt.is_contiguous(), "Bt must be cuda bf16 [K,N]");
  CHK(lightctr::gemm_wgrad_eligible((int)M, (int)N, (int)K, 1, 1),
      "shape not eligible for the wgrad kernel");
  auto C = at::empty({M, N}, At.options().dtype(at::kFloat));
  lightctr::gemm_wgrad_bf16_launch(At.data_ptr(), Bt.data_ptr(),
                                   C.data_ptr<float>(), (int)M, (int)N,
                                   (int)K, cur_stream());
  return C;
}

std::vector<at::Tensor> wg_probe(at::Tensor g, int64_t ld, int64_t mode) {
  CHK(g.is_cuda() && g.scalar_type() == at::kBFloat16, "g bf16");
  auto out_lds = at::zeros({64 * 128}, g.options());
  auto out_frag = at::zeros({2 * 2 * 64 * 8}, g.options().dtype(at::kFloat));
  lightctr::wg_probe_launch(g.data_ptr(), (int)ld, out_lds.data_ptr(),
                            out_frag.data_ptr<float>(), (int)mode,
                            cur_stream());
  return {out_lds, out_frag};
}

at::Tensor im2col_bf16(at::Tensor x, int64_t k, int64_t stride,
                       int64_t pad) {
  check_cuda_f32(x, "x");
  CHK(x.dim() == 4, "x must be [B,C,H,W]");
  const int B = (int)x.size(0), C = (int)x.size(1), H = (int)x.size(2),
            W = (int)x.size(3);
  const int OH = (int)((H + 2 * pad - k) / stride + 1);
  const int OW = (int)((W + 2 * pad - k) / stride + 1);
  auto col = at::empty({(long)B * OH * OW, (long)C * k * k},
                       x.options().dtype(at::kBFloat16));
  lightctr::im2col_bf16_launch(x.data_ptr<float>(), col.data_ptr(), B, C, H,
                               W, (int)k, (int)stride, (int)pad, OH, OW,
                               cur_stream());
  return col;
}

at::Tensor col2im(at::Tensor dcol, int64_t B, int64_t C, int64_t H,
                  int64_t W, int64_t k, int64_t stride, int64_t pad) {
  check_cuda_f32(dcol, "dcol");
  const int OH = (int)((H + 2 * pad - k) / stride + 1);
  const int OW = (int)((W + 2 * pad - k) / stride + 1);
  CHK(dcol.size(0) == (long)B * OH * OW && dcol.size(1) == C * k * k,
      "dcol shape mismatch");
  auto dx = at::empty({B, C, H, W}, dcol.options());
  lightctr::col2im_launch(dcol.data_ptr<float>(), dx.data_ptr<float>(),
                          (int)B, (int)C, (int)H, (int)W, (int)k,
                          (int)stride, (int)pad, OH, OW, cur_stream());
  return dx;
}

void auc_hist_add(at::Tensor pred, at::Tensor label, at::Tensor hist) {
  check_cuda_f32(pred, "pred");
  check_cuda_f32(label, "label");
  CHK(hist.is_cuda() && hist.scalar_type() == at::kInt &&
          hist.is_contiguous(),
      "hist must be cuda int32 [2*buckets]");
  const int buckets = (int)(hist.numel() / 2);
  lightctr::auc_hist_add_launch(pred.data_ptr<float>(),
                                label.data_ptr<float>(),
                                (long)pred.numel(),
                                (unsigned int*)hist.data_ptr(), buckets,
                                cur_stream());
}

at::Tensor auc_scan(at::Tensor hist) {
  CHK(hist.is_cuda() && hist.scalar_type() == at::kInt &&
          hist.is_contiguous(),
      "hist must be cuda int32 [2*buckets]");
  const int buckets = (int)(hist.numel() / 2);
  auto out = at::zeros({3}, hist.options().dtype(at::kDouble));
  lightctr::auc_scan_launch((const unsigned int*)hist.data_ptr(), buckets,
                            out.data_ptr<double>(), cur_stream());
  return out;  // {correct-pair mass, P, N}
}

at::Tensor bitmap_compact(at::Tensor bitmap, at::Tensor out_fids,
                          at::Tensor out_count) {
  CHK(bitmap.is_cuda() && bitmap.is_contiguous(), "bitmap");
  check_cuda_i32(out_fids, "out_fids");
  check_cuda_i32(out_count, "out_count");
  lightctr::bitmap_compact_launch((unsigned long long*)bitmap.data_ptr(),
                                  (int)bitmap.numel(),
                                  out_fids.data_ptr<int>(),
                                  out_count.data_ptr<int>(),
                                  (int)out_fids.numel(), cur_stream());
  return out_count;
}

void fm_adagrad_apply(at::Tensor uniq, at::Tensor count, at::Tensor W,
                      at::Tensor V, at::Tensor nW, at::Tensor nV,
                      at::Tensor gradW, at::Tensor gradV, double lr,
                      double eps, double l2) {
  check_cuda_i32(uniq, "uniq");
  check_cuda_i32(count, "count");
  const int K = (int)V.size(1);
  lightctr::fm_adagrad_apply_launch(
      uniq.data_ptr<int>(), count.data_ptr<int>(), W.data_ptr<float>(),
      V.data_ptr<float>(), nW.data_ptr<float>(), nV.data_ptr<float>(),
      gradW.data_ptr<float>(), gradV.data_ptr<float>(), (float)lr, (float)eps,
      (float)l2, (int)uniq.numel(), K, cur_stream());
}

void fm_ftrl_apply(at::Tensor uniq, at::Tensor count, at::Tensor W,
                   at::Tensor V, at::Tensor zW, at::Tensor nW, at::Tensor zV,
                   at::Tensor nV, at::Tensor gradW, at::Tensor gradV,
                   double alpha, double beta, double l1, double l2,
                   int64_t v_adagrad, double v_lr, double v_eps,
                   double v_l2) {
  check_cuda_i32(uniq, "uniq");
  check_cuda_i32(count, "count");
  const int K = (int)V.size(1);
  lightctr::fm_ftrl_apply_launch(
      uniq.data_ptr<int>(), count.data_ptr<int>(), W.data_ptr<float>(),
      V.data_ptr<float>(), zW.data_ptr<float>(), nW.data_ptr<float>(),
      zV.data_ptr<float>(), nV.data_ptr<float>(), gradW.data_ptr<float>(),
      gradV.data_ptr<float>(), (float)alpha, (float)beta, (float)l1,
      (float)l2, (int)uniq.numel(), K, (int)v_adagrad, (float)v_lr,
      (float)v_eps, (float)v_l2, cur_stream());
}

// ---- dense NN ops ----

at::Tensor gemm_bf16(at::Tensor A, at::Tensor Bst,
                     c10::optional<at::Tensor> bias, int64_t M, int64_t N,
                     int64_t K, int64_t transA, int64_t transB, int64_t act,
                     bool emit_bf16) {
  CHK(A.is_cuda() && A.scalar_type() == at::kBFloat16, "A must be bf16 GPU");
  CHK(Bst.is_cuda() && Bst.scalar_type() == at::kBFloat16, "Bst bf16");
  auto C = at::empty({M, N}, A.options().dtype(at::kFloat));
  at::Tensor Cbf;
  void* cbf_ptr = nullptr;
  if (emit_bf16) {
    Cbf = at::empty({M, N}, A.options());
    cbf_ptr = Cbf.data_ptr();
  }
  const float* bias_ptr =
      bias.has_value() ? bias->data_ptr<float>() : nullptr;
  lightctr::gemm_bf16_launch(A.data_ptr(), Bst.data_ptr(), bias_ptr,
                             C.data_ptr<float>(), cbf_ptr, (int)M, (int)N,
                             (int)K, (int)transA, (int)transB, (int)act,
                             cur_stream());
  return C;
}

std::vector<at::Tensor> gemm_bf16_full(at::Tensor A, at::Tensor Bst,
                                       c10::optional<at::Tensor> bias,
                                       int64_t M, int64_t N, int64_t K,
                                       int64_t transA, int64_t transB,
                                       int64_t act) {
  CHK(A.is_cuda() && A.scalar_type() == at::kBFloat16, "A must be bf16 GPU");
  CHK(Bst.is_cuda() && Bst.scalar_type() == at::kBFloat16, "Bst bf16");
  auto C = at::empty({M, N}, A.options().dtype(at::kFloat));
  auto Cbf = at::empty({M, N}, A.options());
  const float* bias_ptr =
      bias.has_value() ? bias->data_ptr<float>() : nullptr;
  lightctr::gemm_bf16_launch(A.data_ptr(), Bst.data_ptr(), bias_ptr,
                             C.data_ptr<float>(), Cbf.data_ptr(), (int)M,
                             (int)N, (int)K, (int)transA, (int)transB,
                             (int)act, cur_stream());
  return {C, Cbf};
}

std::vector<at::Tensor> act_backward(at::Tensor dY, at::Tensor Y,
                                     int64_t act) {
  check_cuda_f32(dY, "dY");
  auto dZ = at::empty_like(dY);
  auto dZbf = at::empty_like(dY, dY.options().dtype(at::kBFloat16));
  lightctr::act_backward_launch(dY.data_ptr<float>(), Y.data_ptr<float>(),
                                dZ.data_ptr<float>(), dZbf.data_ptr(),
                                dY.numel(), (int)act, cur_stream());
  return {dZ, dZbf};
}

// Bit-range radix sort: drop-in for `torch.sort(fids)` on nonnegative
// int32 keys < 2^end_bit. Returns (sorted_keys int32, perm int64).
std::vector<at::Tensor> radix_sort_index(at::Tensor keys, int64_t end_bit) {
  check_cuda_i32(keys, "keys");
  CHK(end_bit >= 1 && end_bit <= 32, "end_bit in [1,32]");
  const int n = (int)keys.numel();
  auto sorted = at::empty_like(keys);
  auto perm = at::empty({n}, keys.options());
  auto iota = at::empty({n}, keys.options());
  lightctr::iota_i32_launch(iota.data_ptr<int>(), n, cur_stream());
  const unsigned long bytes =
      lightctr::radix_sort_pairs_i32_temp_bytes(n, (int)end_bit);
  auto temp = at::empty({(long)bytes}, keys.options().dtype(at::kByte));
  lightctr::radix_sort_pairs_i32_launch(
      temp.data_ptr(), bytes, keys.data_ptr<int>(), sorted.data_ptr<int>(),
      iota.data_ptr<int>(), perm.data_ptr<int>(), n, (int)end_bit,
      cur_stream());
  return {sorted, perm};
}

at::Tensor colsum(at::Tensor dZ) {
  check_cuda_f32(dZ, "dZ");
  const int M = (int)dZ.size(0), N = (int)dZ.size(1);
  auto db = at::empty({N}, dZ.options());
  lightctr::colsum_launch(dZ.data_ptr<float>(), db.data_ptr<float>(), M, N,
                          cur_stream());
  return db;
}

at::Tensor to_bf16(at::Tensor x) {
  check_cuda_f32(x, "x");
  auto y = at::empty_like(x, x.options().dtype(at::kBFloat16));
  lightctr::to_bf16_launch(x.data_ptr<float>(), y.data_ptr(), x.numel(),
                           cur_stream());
  return y;
}

void dense_adam(at::Tensor W, at::Tensor grad, at::Tensor m, at::Tensor v,
                c10::optional<at::Tensor> Wbf, c10::optional<at::Tensor> Wtbf,
                double lr, double beta1, double beta2, double eps,
                int64_t step, double l2) {
  check_cuda_f32(W, "W");
  const int rows = (int)W.size(0);
  const int cols = (int)(W.numel() / W.size(0));
  const float bc1 = 1.f - powf((float)beta1, (float)step);
  const float bc2 = 1.f - powf((float)beta2, (float)step);
  lightctr::dense_adam_launch(
      W.data_ptr<float>(), grad.data_ptr<float>(), m.data_ptr<float>(),
      v.data_ptr<float>(), Wbf.has_value() ? Wbf->data_ptr() : nullptr,
      Wtbf.has_value() ? Wtbf->data_ptr() : nullptr, rows, cols, (float)lr,
      (float)beta1, (float)beta2, (float)eps, bc1, bc2, (float)l2,
      cur_stream());
}

void dense_adagrad(at::Tensor W, at::Tensor grad, at::Tensor n,
                   c10::optional<at::Tensor> Wbf,
                   c10::optional<at::Tensor> Wtbf, double lr, double eps,
                   double l2) {
  check_cuda_f32(W, "W");
  const int rows = (int)W.size(0);
  const int cols = (int)(W.numel() / W.size(0));
  lightctr::dense_adagrad_launch(
      W.data_ptr<float>(), grad.data_ptr<float>(), n.data_ptr<float>(),
      Wbf.has_value() ? Wbf->data_ptr() : nullptr,
      Wtbf.has_value() ? Wtbf->data_ptr() : nullptr, rows, cols, (float)lr,
      (float)eps, (float)l2, cur_stream());
}

// ---- embedding / NFM ----

at::Tensor embed_gather(at::Tensor row_ptr, at::Tensor fids, at::Tensor vals,
                        at::Tensor E, int64_t nf) {
  check_cuda_i32(row_ptr, "row_ptr");
  check_cuda_f32(E, "E");
  const int B = (int)row_ptr.numel() - 1;
  const int K = (int)E.size(1);
  auto out = at::empty({B, nf * K}, E.options().dtype(at::kBFloat16));
  lightctr::embed_gather_launch(row_ptr.data_ptr<int>(), fids.data_ptr<int>(),
                                vals.data_ptr<float>(), E.data_ptr<float>(),
                                out.data_ptr(), (int)nf, B, K, cur_stream());
  return out;
}

std::vector<at::Tensor> embed_backward_emit(at::Tensor row_ptr,
                                            at::Tensor vals, at::Tensor dOut,
                                            at::Tensor dwide, int64_t nf,
                                            int64_t K) {
  check_cuda_i32(row_ptr, "row_ptr");
  check_cuda_f32(dOut, "dOut");
  const int B = (int)row_ptr.numel() - 1;
  const auto nnz = vals.numel();
  auto gv = at::empty({nnz, K}, dOut.options());
  auto gw = at::empty({nnz}, dOut.options());
  lightctr::embed_backward_emit_launch(
      row_ptr.data_ptr<int>(), vals.data_ptr<float>(), dOut.data_ptr<float>(),
      dwide.data_ptr<float>(), gv.data_ptr<float>(), gw.data_ptr<float>(),
      (int)nf, B, (int)K, cur_stream());
  return {gw, gv};
}

at::Tensor w2v_negsample_step(at::Tensor E, at::Tensor O,
                              at::Tensor centers, at::Tensor ctx,
                              at::Tensor negs, double lr, double scale) {
  check_cuda_f32(E, "E");
  check_cuda_f32(O, "O");
  CHK(centers.scalar_type() == at::kLong, "centers i64");
  const int B = (int)centers.numel();
  const int C = (int)ctx.size(1);
  const int N = (int)negs.size(1);
  const int D = (int)E.size(1);
  CHK(D <= 64, "w2v kernel supports dim <= 64");
  auto loss = at::empty({B}, E.options());
  lightctr::w2v_negsample_launch(
      E.data_ptr<float>(), O.data_ptr<float>(), centers.data_ptr<long>(),
      ctx.data_ptr<long>(), negs.data_ptr<long>(), loss.data_ptr<float>(),
      B, C, N, D, (float)lr, (float)scale, cur_stream());
  return loss;
}

at::Tensor wide_forward(at::Tensor row_ptr, at::Tensor fids, at::Tensor vals,
                        at::Tensor W) {
  check_cuda_i32(row_ptr, "row_ptr");
  check_cuda_f32(W, "W");
  const int B = (int)row_ptr.numel() - 1;
  auto wide = at::empty({B}, W.options());
  lightctr::wide_forward_launch(row_ptr.data_ptr<int>(), fids.data_ptr<int>(),
                                vals.data_ptr<float>(), W.data_ptr<float>(),
                                wide.data_ptr<float>(), B, cur_stream());
  return wide;
}

std::vector<at::Tensor> nfm_forward(at::Tensor row_ptr, at::Tensor fids,
                                    at::Tensor vals, at::Tensor W,
                                    at::Tensor V) {
  check_cuda_i32(row_ptr, "row_ptr");
  check_cuda_f32(V, "V");
  const int B = (int)row_ptr.numel() - 1;
  const int K = (int)V.size(1);
  auto wide = at::empty({B}, W.options());
  auto sumVX = at::empty({B, K}, V.options());
  auto vec = at::empty({B, K}, V.options());
  auto vec_bf = at::empty({B, K}, V.options().dtype(at::kBFloat16));
  lightctr::nfm_forward_launch(row_ptr.data_ptr<int>(), fids.data_ptr<int>(),
                               vals.data_ptr<float>(), W.data_ptr<float>(),
                               V.data_ptr<float>(), wide.data_ptr<float>(),
                               sumVX.data_ptr<float>(), vec.data_ptr<float>(),
                               vec_bf.data_ptr(), B, K, cur_stream());
  return {wide, sumVX, vec, vec_bf};
}

std::vector<at::Tensor> nfm_backward_emit(at::Tensor row_ptr, at::Tensor fids,
                                          at::Tensor vals, at::Tensor V,
                                          at::Tensor sumVX, at::Tensor dvec,
                                          at::Tensor dwide) {
  check_cuda_i32(row_ptr, "row_ptr");
  const int B = (int)row_ptr.numel() - 1;
  const int K = (int)V.size(1);
  const auto nnz = fids.numel();
  auto gw = at::empty({nnz}, V.options());
  auto gv = at::empty({nnz, K}, V.options());
  lightctr::nfm_backward_emit_launch(
      row_ptr.data_ptr<int>(), fids.data_ptr<int>(), vals.data_ptr<float>(),
      V.data_ptr<float>(), sumVX.data_ptr<float>(), dvec.data_ptr<float>(),
      dwide.data_ptr<float>(), gw.data_ptr<float>(), gv.data_ptr<float>(), B,
      K, cur_stream());
  return {gw, gv};
}

// ---- native data loader ----
// Fast single-pass libffm parser ("label field:fid:val" per line) —
// native equivalent of the reference's C++ loader
// (/root/reference/LightCTR/fm_algo_abst.h:70-107), ~20x the Python
// line-split loader. Returns (row_ptr i32, fields i32, fids i32,
// vals f32, labels f32) CPU tensors.
std::vector<at::Tensor> parse_libffm(const std::string& path,
                                     int64_t max_rows) {
  FILE* f = fopen(path.c_str(), "rb");
  TORCH_CHECK(f != nullptr, "cannot open ", path);
  fseek(f, 0, SEEK_END);
  const long size = ftell(f);
  fseek(f, 0, SEEK_SET);
  std::vector<char> buf(size + 1);
  const size_t rd = fread(buf.data(), 1, size, f);
  fclose(f);
  TORCH_CHECK((long)rd == size, "short read on ", path);
  buf[size] = '\0';

  std::vector<int> row_ptr{0};
  std::vector<int> fields, fids;
  std::vector<float> vals, labels;
  char* p = buf.data();
  char* end = buf.data() + size;
  while (p < end) {
    // label
    while (p < end && (*p == ' ' || *p == '\n' || *p == '\r')) ++p;
    if (p >= end) break;
    char* q;
    const float label = strtof(p, &q);
    TORCH_CHECK(q != p, "bad label near byte ", (long)(p - buf.data()));
    p = q;
    // features until newline
    while (p < end && *p != '\n') {
      while (p < end && *p == ' ') ++p;
      if (p >= end || *p == '\n') break;
      const long fld = strtol(p, &q, 10);
      TORCH_CHECK(q != p && *q == ':', "bad field token");
      p = q + 1;
      const long fid = strtol(p, &q, 10);
      TORCH_CHECK(q != p && *q == ':', "bad fid token");
      p = q + 1;
      const float v = strtof(p, &q);
      TORCH_CHECK(q != p, "bad value token");
      p = q;
      fields.push_back((int)fld);
      fids.push_back((int)fid);
      vals.push_back(v);
    }
    labels.push_back(label);
    row_ptr.push_back((int)fids.size());
    if (max_rows > 0 && (int64_t)labels.size() >= max_rows) break;
  }
  auto opts_i = at::TensorOptions().dtype(at::kInt);
  auto opts_f = at::TensorOptions().dtype(at::kFloat);
  auto t_rp = at::empty({(long)row_ptr.size()}, opts_i);
  memcpy(t_rp.data_ptr<int>(), row_ptr.data(), row_ptr.size() * 4);
  auto t_fl = at::empty({(long)fields.size()}, opts_i);
  memcpy(t_fl.data_ptr<int>(), fields.data(), fields.size() * 4);
  auto t_fi = at::empty({(long)fids.size()}, opts_i);
  memcpy(t_fi.data_ptr<int>(), fids.data(), fids.size() * 4);
  auto t_v = at::empty({(long)vals.size()}, opts_f);
  memcpy(t_v.data_ptr<float>(), vals.data(), vals.size() * 4);
  auto t_l = at::empty({(long)labels.size()}, opts_f);
  memcpy(t_l.data_ptr<float>(), labels.data(), labels.size() * 4);
  return {t_rp, t_fl, t_fi, t_v, t_l};
}

// ---- codecs ----

at::Tensor quantile_encode(at::Tensor x, at::Tensor table) {
  check_cuda_f32(x, "x");
  check_cuda_f32(table, "table");
  auto code = at::empty(x.sizes(), x.options().dtype(at::kByte));
  lightctr::quantile_encode_launch(x.data_ptr<float>(),
                                   code.data_ptr<unsigned char>(),
                                   table.data_ptr<float>(),
                                   (int)table.numel(), x.numel(),
                                   cur_stream());
  return code;
}

at::Tensor quantile_decode(at::Tensor code, at::Tensor table) {
  check_cuda_f32(table, "table");
  auto x = at::empty(code.sizes(), table.options());
  lightctr::quantile_decode_launch(code.data_ptr<unsigned char>(),
                                   x.data_ptr<float>(),
                                   table.data_ptr<float>(),
                                   (int)table.numel(), code.numel(),
                                   cur_stream());
  return x;
}

at::Tensor lowbit_encode(at::Tensor x, double thresh, int64_t bits) {
  check_cuda_f32(x, "x");
  const long n = x.numel();
  const long nw = (n * bits + 31) / 32;
  auto words = at::empty({nw}, x.options().dtype(at::kInt));
  lightctr::lowbit_encode_launch(x.data_ptr<float>(),
                                 (unsigned int*)words.data_ptr(),
                                 (float)thresh, (int)bits, n, cur_stream());
  return words;
}

at::Tensor lowbit_decode(at::Tensor words, double lo, double hi, int64_t bits,
                         int64_t n) {
  auto x = at::empty({n}, words.options().dtype(at::kFloat));
  lightctr::lowbit_decode_launch((const unsigned int*)words.data_ptr(),
                                 x.data_ptr<float>(), (float)lo, (float)hi,
                                 (int)bits, n, cur_stream());
  return x;
}

at::Tensor gbm_hist(at::Tensor bins, at::Tensor grad, at::Tensor hess,
                    at::Tensor node_of_row, int64_t n_nodes) {
  CHK(bins.is_cuda() && bins.scalar_type() == at::kByte, "bins u8");
  check_cuda_f32(grad, "grad");
  check_cuda_f32(hess, "hess");
  check_cuda_i32(node_of_row, "node_of_row");
  const int N = (int)bins.size(0), D = (int)bins.size(1);
  auto hist = at::zeros({n_nodes, D, 256, 2}, grad.options());
  lightctr::gbm_hist_launch(bins.data_ptr<unsigned char>(),
                            grad.data_ptr<float>(), hess.data_ptr<float>(),
                            node_of_row.data_ptr<int>(),
                            hist.data_ptr<float>(), N, D, (int)n_nodes,
                            cur_stream());
  return hist;
}

}  // namespace

PYBIND11_MODULE(TORCH_EXTENSION_NAME, m) {
  m.doc() = "lightctr_amd HIP kernels (gfx950)";
  m.def("fm_forward", &fm_forward, "FM fused forward (pred, sumVX)");
  m.def("logloss_grad", &logloss_grad, "stable logloss + dpred");
  m.def("fm_backward", &fm_backward, "FM fused backward scatter");
  m.def("fm_backward_emit", &fm_backward_emit,
        "FM backward phase 1: per-entry grads (no atomics); pos=write slots",
        py::arg("row_ptr"), py::arg("fids"), py::arg("vals"), py::arg("V"),
        py::arg("sumVX"), py::arg("dpred"), py::arg("pos") = py::none());
  m.def("fm_sorted_apply", &fm_sorted_apply,
        "FM backward phase 2: segment-reduce sorted grads into slabs "
        "(perm=None -> grads already in sorted order)",
        py::arg("sorted_fids"), py::arg("perm"), py::arg("gw"),
        py::arg("gv"), py::arg("gradW"), py::arg("gradV"),
        py::arg("touched"), py::arg("chunk") = 0);
  m.def("fm_sorted_apply_fused", &fm_sorted_apply_fused,
        "segment-reduce + fused optimizer for interior segments",
        py::arg("sorted_fids"), py::arg("perm"), py::arg("gw"),
        py::arg("gv"), py::arg("gradW"), py::arg("gradV"),
        py::arg("touched"), py::arg("W"), py::arg("V"), py::arg("nW"),
        py::arg("nV"), py::arg("zW"), py::arg("zV"), py::arg("opt_mode"),
        py::arg("p0"), py::arg("p1"), py::arg("p2"), py::arg("p3"),
        py::arg("v_lr") = 0.05, py::arg("v_eps") = 1e-8,
        py::arg("v_l2") = 1e-5);
  m.def("ffm_forward", &ffm_forward, "FFM pairwise forward (LDS-staged)");
  m.def("ffm_forward_pp", &ffm_forward_pp, "lane-per-pair FFM forward");
  m.def("ffm_row_emit", &ffm_row_emit,
        "FFM per-row staged fp16 block emit -> (gw, gblocks)",
        py::arg("row_ptr"), py::arg("fields"), py::arg("fids"),
        py::arg("vals"), py::arg("V"), py::arg("dpred"),
        py::arg("scale") = 1.0);
  m.def("ffm_blocks_apply_f16", &ffm_blocks_apply_f16,
        "segment-reduce fp16 blocks into slabs (interior stores), "
        "optionally with the optimizer fused for interior runs",
        py::arg("sorted_fids"), py::arg("perm"), py::arg("gblocks"),
        py::arg("gw"), py::arg("gradW"), py::arg("gradV"),
        py::arg("touched"), py::arg("inv_scale") = 1.0,
        py::arg("opt_mode") = 0, py::arg("V") = py::none(),
        py::arg("W") = py::none(), py::arg("nW") = py::none(),
        py::arg("zW") = py::none(), py::arg("nV") = py::none(),
        py::arg("Vh") = py::none(), py::arg("p0") = 0.0,
        py::arg("p1") = 0.0, py::arg("p2") = 0.0, py::arg("p3") = 0.0,
        py::arg("q0") = 0.0, py::arg("q1") = 0.0, py::arg("q2") = 0.0);
  m.def("ffm_backward", &ffm_backward, "FFM fused pairwise backward scatter");
  m.def("ffm_sorted_backward", &ffm_sorted_backward,
        "FFM sorted segment-reduce backward (LDS block accumulate)");
  m.def("row_index", &row_index, "entry -> row index from row_ptr");
  m.def("ffm_block_emit", &ffm_block_emit,
        "FFM per-entry [nf,K] gradient blocks (wave/entry)");
  m.def("ffm_blocks_apply", &ffm_blocks_apply,
        "segment-reduce sorted FFM blocks into slabs");
  m.def("parse_libffm", &parse_libffm, "native libffm text parser",
        py::arg("path"), py::arg("max_rows") = -1);
  m.def("sparse_adagrad_apply", &sparse_adagrad_apply,
        "generic sparse fused Adagrad (runtime D)",
        py::arg("uniq"), py::arg("count"), py::arg("W"), py::arg("V"),
        py::arg("nW"), py::arg("nV"), py::arg("gradW"), py::arg("gradV"),
        py::arg("lr"), py::arg("eps"), py::arg("l2"),
        py::arg("Vh") = py::none());
  m.def("ps_apply", &ps_apply,
        "PS-side fused updater (sgd/adagrad/dcasgd/dcasgda)");
  m.def("sparse_ftrl_apply", &sparse_ftrl_apply,
        "generic sparse fused FTRL (runtime D; optional Adagrad on V)",
        py::arg("uniq"), py::arg("count"), py::arg("W"), py::arg("V"),
        py::arg("zW"), py::arg("nW"), py::arg("zV"), py::arg("nV"),
        py::arg("gradW"), py::arg("gradV"), py::arg("alpha"),
        py::arg("beta"), py::arg("l1"), py::arg("l2"),
        py::arg("v_adagrad") = 0, py::arg("v_lr") = 0.05,
        py::arg("v_eps") = 1e-8, py::arg("v_l2") = 1e-5,
        py::arg("Vh") = py::none());
  m.def("gemm_bf16", &gemm_bf16, "MFMA bf16 GEMM (C fp32), fused bias+act");
  m.def("gemm_bf16_full", &gemm_bf16_full,
        "MFMA bf16 GEMM returning (C fp32, C bf16)");
  m.def("act_backward", &act_backward, "dZ = dY*act'(Y), + bf16 mirror");
  m.def("inv_perm_i32", &inv_perm_i32, "inverse permutation (int32)");
  m.def("radix_sort_index", &radix_sort_index,
        "bit-range radix sort -> (sorted, perm)");
  m.def("colsum", &colsum, "bias gradient column sum");
  m.def("to_bf16", &to_bf16, "f32 -> bf16 convert");
  m.def("dense_adam", &dense_adam, "dense Adam + bf16 mirror refresh");
  m.def("dense_adagrad", &dense_adagrad,
        "dense Adagrad + bf16 mirror refresh");
  m.def("embed_gather", &embed_gather, "concat embedding gather -> bf16");
  m.def("embed_backward_emit", &embed_backward_emit,
        "per-entry embedding grads (gw, gv) for sorted apply");
  m.def("wide_forward", &wide_forward, "LR wide forward sum");
  m.def("w2v_negsample_step", &w2v_negsample_step,
        "fused CBOW negative-sampling train step (Hogwild)");
  m.def("quantile_encode", &quantile_encode, "int8 quantile encode");
  m.def("quantile_decode", &quantile_decode, "int8 quantile decode");
  m.def("lowbit_encode", &lowbit_encode, "1/2-bit sign quantize");
  m.def("lowbit_decode", &lowbit_decode, "1/2-bit dequantize");
  m.def("gbm_hist", &gbm_hist, "GBM per-node (grad,hess) histograms");
  m.def("nfm_forward", &nfm_forward,
        "NFM bi-interaction fwd (wide, sumVX, vec, vec_bf16)");
  m.def("nfm_backward_emit", &nfm_backward_emit,
        "NFM per-entry grads for sorted apply");
  m.def("bitmap_compact", &bitmap_compact, "touched bitmap -> fid list");
  m.def("wg_probe", &wg_probe, "wgrad staging/tr-read debug probe");
  m.def("gemm_wgrad_bf16", &gemm_wgrad_bf16,
        "K-major x K-major wgrad GEMM (tr16 fragment reads)");
  m.def("im2col_bf16", &im2col_bf16,
        "fused unfold -> bf16 GEMM operand [B*L, C*k*k]");
  m.def("col2im", &col2im, "conv data-grad fold (gather form)");
  m.def("auc_hist_add", &auc_hist_add,
        "atomic pos/neg histogram accumulate for AUC");
  m.def("auc_scan", &auc_scan,
        "device scan of the AUC histogram -> {correct, P, N}");
  m.def("fm_adagrad_apply", &fm_adagrad_apply, "sparse fused Adagrad");
  m.def("fm_ftrl_apply", &fm_ftrl_apply,
        "sparse fused FTRL-proximal (optionally Adagrad on V)",
        py::arg("uniq"), py::arg("count"), py::arg("W"), py::arg("V"),
        py::arg("zW"), py::arg("nW"), py::arg("zV"), py::arg("nV"),
        py::arg("gradW"), py::arg("gradV"), py::arg("alpha"),
        py::arg("beta"), py::arg("l1"), py::arg("l2"),
        py::arg("v_adagrad") = 0, py::arg("v_lr") = 0.05,
        py::arg("v_eps") = 1e-8, py::arg("v_l2") = 1e-5);
}
