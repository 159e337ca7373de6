// PyTorch bindings for the lightctr_amd HIP kernels (MI355X / gfx950).
// Host-only translation unit: tensor checks + launcher calls.
#include <torch/extension.h>
#include <ATen/cuda/CUDAContext.h>

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
                                         at::Tensor sumVX, at::Tensor dpred) {
  check_cuda_i32(row_ptr, "row_ptr");
  check_cuda_i32(fids, "fids");
  const int B = (int)row_ptr.numel() - 1;
  const int K = (int)V.size(1);
  const auto nnz = fids.numel();
  auto gw = at::empty({nnz}, V.options());
  auto gv = at::empty({nnz, K}, V.options());
  lightctr::fm_backward_emit_launch(
      row_ptr.data_ptr<int>(), fids.data_ptr<int>(), vals.data_ptr<float>(),
      V.data_ptr<float>(), sumVX.data_ptr<float>(), dpred.data_ptr<float>(),
      gw.data_ptr<float>(), gv.data_ptr<float>(), B, K, cur_stream());
  return {gw, gv};
}

void fm_sorted_apply(at::Tensor sorted_fids, at::Tensor perm, at::Tensor gw,
                     at::Tensor gv, at::Tensor gradW, at::Tensor gradV,
                     at::Tensor touched) {
  check_cuda_i32(sorted_fids, "sorted_fids");
  CHK(perm.scalar_type() == at::kLong, "perm must be int64");
  const int K = (int)gradV.size(1);
  lightctr::fm_sorted_apply_launch(
      sorted_fids.data_ptr<int>(), perm.data_ptr<long>(),
      gw.data_ptr<float>(), gv.data_ptr<float>(), gradW.data_ptr<float>(),
      gradV.data_ptr<float>(), (unsigned long long*)touched.data_ptr(),
      (int)sorted_fids.numel(), K, cur_stream());
}

// ---- FFM ----

at::Tensor ffm_forward(at::Tensor row_ptr, at::Tensor fields, at::Tensor fids,
                       at::Tensor vals, at::Tensor W, at::Tensor V) {
  check_cuda_i32(row_ptr, "row_ptr");
  check_cuda_i32(fields, "fields");
  check_cuda_i32(fids, "fids");
  check_cuda_f32(V, "V");
  const int B = (int)row_ptr.numel() - 1;
  const int nfields = (int)V.size(1);
  const int K = (int)V.size(2);
  auto pred = at::empty({B}, W.options());
  lightctr::ffm_forward_launch(row_ptr.data_ptr<int>(),
                               fields.data_ptr<int>(), fids.data_ptr<int>(),
                               vals.data_ptr<float>(), W.data_ptr<float>(),
                               V.data_ptr<float>(), pred.data_ptr<float>(),
                               nfields, B, K, cur_stream());
  return pred;
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

// ---- generic sparse optimizers (D = latent block width, runtime) ----

void sparse_adagrad_apply(at::Tensor uniq, at::Tensor count, at::Tensor W,
                          at::Tensor V, at::Tensor nW, at::Tensor nV,
                          at::Tensor gradW, at::Tensor gradV, double lr,
                          double eps, double l2) {
  check_cuda_i32(uniq, "uniq");
  const int D = (int)(V.numel() / V.size(0));
  lightctr::sparse_adagrad_apply_launch(
      uniq.data_ptr<int>(), count.data_ptr<int>(), W.data_ptr<float>(),
      V.data_ptr<float>(), nW.data_ptr<float>(), nV.data_ptr<float>(),
      gradW.data_ptr<float>(), gradV.data_ptr<float>(), (float)lr, (float)eps,
      (float)l2, (int)uniq.numel(), D, cur_stream());
}

void sparse_ftrl_apply(at::Tensor uniq, at::Tensor count, at::Tensor W,
                       at::Tensor V, at::Tensor zW, at::Tensor nW,
                       at::Tensor zV, at::Tensor nV, at::Tensor gradW,
                       at::Tensor gradV, double alpha, double beta, double l1,
                       double l2) {
  check_cuda_i32(uniq, "uniq");
  const int D = (int)(V.numel() / V.size(0));
  lightctr::sparse_ftrl_apply_launch(
      uniq.data_ptr<int>(), count.data_ptr<int>(), W.data_ptr<float>(),
      V.data_ptr<float>(), zW.data_ptr<float>(), nW.data_ptr<float>(),
      zV.data_ptr<float>(), nV.data_ptr<float>(), gradW.data_ptr<float>(),
      gradV.data_ptr<float>(), (float)alpha, (float)beta, (float)l1,
      (float)l2, (int)uniq.numel(), D, cur_stream());
}

at::Tensor bitmap_compact(at::Tensor bitmap, at::Tensor out_fids,
                          at::Tensor out_count) {
  CHK(bitmap.is_cuda() && bitmap.is_contiguous(), "bitmap");
  check_cuda_i32(out_fids, "out_fids");
  check_cuda_i32(out_count, "out_count");
  lightctr::bitmap_compact_launch((unsigned long long*)bitmap.data_ptr(),
                                  (int)bitmap.numel(),
                                  out_fids.data_ptr<int>(),
                                  out_count.data_ptr<int>(), cur_stream());
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
                   double alpha, double beta, double l1, double l2) {
  check_cuda_i32(uniq, "uniq");
  check_cuda_i32(count, "count");
  const int K = (int)V.size(1);
  lightctr::fm_ftrl_apply_launch(
      uniq.data_ptr<int>(), count.data_ptr<int>(), W.data_ptr<float>(),
      V.data_ptr<float>(), zW.data_ptr<float>(), nW.data_ptr<float>(),
      zV.data_ptr<float>(), nV.data_ptr<float>(), gradW.data_ptr<float>(),
      gradV.data_ptr<float>(), (float)alpha, (float)beta, (float)l1,
      (float)l2, (int)uniq.numel(), K, cur_stream());
}

}  // namespace

PYBIND11_MODULE(TORCH_EXTENSION_NAME, m) {
  m.doc() = "lightctr_amd HIP kernels (gfx950)";
  m.def("fm_forward", &fm_forward, "FM fused forward (pred, sumVX)");
  m.def("logloss_grad", &logloss_grad, "stable logloss + dpred");
  m.def("fm_backward", &fm_backward, "FM fused backward scatter");
  m.def("fm_backward_emit", &fm_backward_emit,
        "FM backward phase 1: per-entry grads (no atomics)");
  m.def("fm_sorted_apply", &fm_sorted_apply,
        "FM backward phase 2: segment-reduce sorted grads into slabs");
  m.def("ffm_forward", &ffm_forward, "FFM fused pairwise forward");
  m.def("ffm_backward", &ffm_backward, "FFM fused pairwise backward scatter");
  m.def("sparse_adagrad_apply", &sparse_adagrad_apply,
        "generic sparse fused Adagrad (runtime D)");
  m.def("sparse_ftrl_apply", &sparse_ftrl_apply,
        "generic sparse fused FTRL (runtime D)");
  m.def("bitmap_compact", &bitmap_compact, "touched bitmap -> fid list");
  m.def("fm_adagrad_apply", &fm_adagrad_apply, "sparse fused Adagrad");
  m.def("fm_ftrl_apply", &fm_ftrl_apply, "sparse fused FTRL-proximal");
}
