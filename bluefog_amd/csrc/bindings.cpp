// Copyright 2026. Licensed under the Apache License, Version 2.0.
//
// pybind11/torch bindings for the bluefog_amd native kernels
// (reference analog: the pybind layer of bluefog/torch/mpi_ops.cc:572-690,
// minus the op queueing — scheduling lives in Python over stream-ordered
// RCCL).

#include <torch/extension.h>

#include <ATen/hip/HIPContext.h>
#include <c10/hip/HIPStream.h>
#include <hip/hip_runtime.h>

#include <string>
#include <vector>

extern "C" {
hipError_t bf_weighted_combine(void* out, const void* self, double self_w,
                               const void* gathered, const double* w, int n_nbr,
                               long numel, int dtype, hipStream_t stream);
hipError_t bf_scale_put(void* dst, const void* src, double w, long numel,
                        int dtype, bool accum, hipStream_t stream);
hipError_t bf_scale_inplace(void* buf, double f, long numel, int dtype,
                            hipStream_t stream);
hipError_t bf_combine_sgd(void* p, double self_w, const void* gathered,
                          const double* w, int n_nbr, const void* grad,
                          void* mom, double lr, double mu, double wd,
                          double dampening, int nesterov, long numel, int dtype,
                          hipStream_t stream);
hipError_t bf_combine_adam(void* p, double self_w, const void* gathered,
                           const double* w, int n_nbr, const void* grad,
                           float* exp_avg, float* exp_avg_sq, double lr,
                           double beta1, double beta2, double eps, double wd,
                           double bias1, double bias2, long numel, int dtype,
                           hipStream_t stream);
hipError_t bf_add_relu_fwd(void* out, const void* a, const void* b, long numel,
                           int dtype, hipStream_t stream);
hipError_t bf_relu_bwd_mask(void* gin, const void* g, const void* out,
                            long numel, int dtype, hipStream_t stream);
hipError_t bf_ln_add_fwd(void* y, const void* x, const void* r,
                         const float* gamma, const float* beta, float* mean,
                         float* rstd, int H, long nrows, double eps, int dtype,
                         hipStream_t stream);
hipError_t bf_ln_add_bwd(void* dx, const void* x, const void* r,
                         const void* dy, const float* gamma, const float* mean,
                         const float* rstd, float* dgamma, float* dbeta,
                         float* scratch, int H, long nrows, int dtype,
                         hipStream_t stream);
int bf_ln_bwd_scratch_rows(long nrows);
}

namespace {

int dtype_code(const at::Tensor& t) {
  switch (t.scalar_type()) {
    case at::kFloat:
      return 0;
    case at::kDouble:
      return 1;
    case at::kHalf:
      return 2;
    case at::kBFloat16:
      return 3;
    default:
      TORCH_CHECK(false, "bluefog_amd kernels support f32/f64/f16/bf16, got ",
                  t.scalar_type());
  }
}

hipStream_t current_stream() {
  return at::hip::getCurrentHIPStream().stream();
}

void check_hip(hipError_t e, const char* what) {
  TORCH_CHECK(e == hipSuccess, "bluefog_amd ", what, ": ", hipGetErrorString(e));
}

void weighted_combine(at::Tensor output, at::Tensor self, double self_weight,
                      at::Tensor gathered, std::vector<double> weights) {
  TORCH_CHECK(output.is_contiguous() && self.is_contiguous(),
              "weighted_combine needs contiguous tensors");
  TORCH_CHECK(output.numel() == self.numel(), "output/self numel mismatch");
  const int n = static_cast<int>(weights.size());
  if (n > 0) {
    TORCH_CHECK(gathered.is_contiguous(), "gathered must be contiguous");
    TORCH_CHECK(gathered.numel() == static_cast<long>(n) * self.numel(),
                "gathered must hold one self-shaped slice per weight");
    TORCH_CHECK(gathered.scalar_type() == self.scalar_type(), "dtype mismatch");
  }
  check_hip(bf_weighted_combine(output.data_ptr(), self.data_ptr(),
                                self_weight,
                                n ? gathered.data_ptr() : self.data_ptr(),
                                weights.data(), n, self.numel(),
                                dtype_code(self), current_stream()),
            "weighted_combine");
}

void scale_put(at::Tensor dst, at::Tensor src, double weight) {
  TORCH_CHECK(dst.is_contiguous() && src.is_contiguous(), "contiguous only");
  TORCH_CHECK(dst.numel() == src.numel(), "numel mismatch");
  TORCH_CHECK(dst.scalar_type() == src.scalar_type(), "dtype mismatch");
  check_hip(bf_scale_put(dst.data_ptr(), src.data_ptr(), weight, src.numel(),
                         dtype_code(src), /*accum=*/false, current_stream()),
            "scale_put");
}

void accum_put(at::Tensor dst, at::Tensor src, double weight) {
  TORCH_CHECK(dst.is_contiguous() && src.is_contiguous(), "contiguous only");
  TORCH_CHECK(dst.numel() == src.numel(), "numel mismatch");
  TORCH_CHECK(dst.scalar_type() == src.scalar_type(), "dtype mismatch");
  check_hip(bf_scale_put(dst.data_ptr(), src.data_ptr(), weight, src.numel(),
                         dtype_code(src), /*accum=*/true, current_stream()),
            "accum_put");
}

void scale_inplace(at::Tensor buf, double factor) {
  TORCH_CHECK(buf.is_contiguous(), "contiguous only");
  check_hip(bf_scale_inplace(buf.data_ptr(), factor, buf.numel(),
                             dtype_code(buf), current_stream()),
            "scale_inplace");
}

void weighted_combine_sgd(at::Tensor param, double self_weight,
                          at::Tensor gathered, std::vector<double> weights,
                          at::Tensor grad, at::Tensor momentum_buf, double lr,
                          double momentum, double weight_decay,
                          double dampening, bool nesterov) {
  TORCH_CHECK(param.is_contiguous() && grad.is_contiguous(), "contiguous only");
  TORCH_CHECK(param.numel() == grad.numel(), "param/grad numel mismatch");
  const int n = static_cast<int>(weights.size());
  void* mom = nullptr;
  if (momentum_buf.defined() && momentum_buf.numel() > 0) {
    TORCH_CHECK(momentum_buf.numel() == param.numel(), "momentum numel");
    mom = momentum_buf.data_ptr();
  }
  check_hip(
      bf_combine_sgd(param.data_ptr(), self_weight,
                     n ? gathered.data_ptr() : param.data_ptr(),
                     weights.data(), n, grad.data_ptr(), mom, lr, momentum,
                     weight_decay, dampening, nesterov ? 1 : 0, param.numel(),
                     dtype_code(param), current_stream()),
      "weighted_combine_sgd");
}

void weighted_combine_adam(at::Tensor param, double self_weight,
                           at::Tensor gathered, std::vector<double> weights,
                           at::Tensor grad, at::Tensor exp_avg,
                           at::Tensor exp_avg_sq, double lr, double beta1,
                           double beta2, double eps, double weight_decay,
                           long step) {
  TORCH_CHECK(param.is_contiguous() && grad.is_contiguous(), "contiguous only");
  TORCH_CHECK(exp_avg.scalar_type() == at::kFloat &&
                  exp_avg_sq.scalar_type() == at::kFloat,
              "adam state must be fp32");
  const int n = static_cast<int>(weights.size());
  const double bias1 = 1.0 - std::pow(beta1, static_cast<double>(step));
  const double bias2 = 1.0 - std::pow(beta2, static_cast<double>(step));
  check_hip(
      bf_combine_adam(param.data_ptr(), self_weight,
                      n ? gathered.data_ptr() : param.data_ptr(),
                      weights.data(), n, grad.data_ptr(),
                      exp_avg.data_ptr<float>(), exp_avg_sq.data_ptr<float>(),
                      lr, beta1, beta2, eps, weight_decay, bias1, bias2,
                      param.numel(), dtype_code(param), current_stream()),
      "weighted_combine_adam");
}

void add_relu_fwd(at::Tensor out, at::Tensor a, at::Tensor b) {
  // elementwise over identically-laid-out dense tensors (plain contiguous
  // or channels_last both qualify)
  TORCH_CHECK(out.is_non_overlapping_and_dense() &&
                  a.is_non_overlapping_and_dense() &&
                  b.is_non_overlapping_and_dense() &&
                  a.strides() == b.strides() && out.strides() == a.strides(),
              "add_relu needs dense tensors with identical layout");
  TORCH_CHECK(a.numel() == b.numel() && out.numel() == a.numel(), "numel mismatch");
  TORCH_CHECK(a.scalar_type() == b.scalar_type() &&
                  out.scalar_type() == a.scalar_type(),
              "dtype mismatch");
  check_hip(bf_add_relu_fwd(out.data_ptr(), a.data_ptr(), b.data_ptr(),
                            a.numel(), dtype_code(a), current_stream()),
            "add_relu_fwd");
}

void relu_bwd_mask(at::Tensor gin, at::Tensor g, at::Tensor out) {
  TORCH_CHECK(gin.is_non_overlapping_and_dense() &&
                  g.is_non_overlapping_and_dense() &&
                  out.is_non_overlapping_and_dense() &&
                  g.strides() == out.strides() && gin.strides() == g.strides(),
              "relu_bwd_mask needs dense tensors with identical layout");
  TORCH_CHECK(g.numel() == out.numel() && gin.numel() == g.numel(), "numel mismatch");
  check_hip(bf_relu_bwd_mask(gin.data_ptr(), g.data_ptr(), out.data_ptr(),
                             g.numel(), dtype_code(g), current_stream()),
            "relu_bwd_mask");
}

void ln_add_fwd(at::Tensor y, at::Tensor x, at::Tensor r, at::Tensor gamma,
                at::Tensor beta, at::Tensor mean, at::Tensor rstd, double eps) {
  TORCH_CHECK(x.is_contiguous() && r.is_contiguous() && y.is_contiguous(),
              "ln_add_fwd needs contiguous tensors");
  TORCH_CHECK(x.sizes() == r.sizes() && y.sizes() == x.sizes(), "shape mismatch");
  TORCH_CHECK(x.scalar_type() != at::kDouble, "ln_add_fwd: f64 unsupported");
  const int H = static_cast<int>(x.size(-1));
  const long nrows = x.numel() / H;
  TORCH_CHECK(gamma.scalar_type() == at::kFloat && gamma.numel() == H &&
                  beta.scalar_type() == at::kFloat && beta.numel() == H &&
                  gamma.is_contiguous() && beta.is_contiguous(),
              "ln_add_fwd: gamma/beta must be fp32[H]");
  TORCH_CHECK(mean.scalar_type() == at::kFloat && mean.numel() >= nrows &&
                  rstd.scalar_type() == at::kFloat && rstd.numel() >= nrows,
              "ln_add_fwd: mean/rstd must be fp32[rows]");
  check_hip(bf_ln_add_fwd(y.data_ptr(), x.data_ptr(), r.data_ptr(),
                          gamma.data_ptr<float>(), beta.data_ptr<float>(),
                          mean.data_ptr<float>(), rstd.data_ptr<float>(), H,
                          nrows, eps, dtype_code(x), current_stream()),
            "ln_add_fwd");
}

void ln_add_bwd(at::Tensor dx, at::Tensor x, at::Tensor r, at::Tensor dy,
                at::Tensor gamma, at::Tensor mean, at::Tensor rstd,
                at::Tensor dgamma, at::Tensor dbeta, at::Tensor scratch) {
  TORCH_CHECK(x.is_contiguous() && r.is_contiguous() && dy.is_contiguous() &&
                  dx.is_contiguous(),
              "ln_add_bwd needs contiguous tensors");
  TORCH_CHECK(x.sizes() == r.sizes() && dy.sizes() == x.sizes() &&
                  dx.sizes() == x.sizes(),
              "shape mismatch");
  const int H = static_cast<int>(x.size(-1));
  const long nrows = x.numel() / H;
  TORCH_CHECK(dgamma.scalar_type() == at::kFloat && dgamma.numel() == H &&
                  dbeta.scalar_type() == at::kFloat && dbeta.numel() == H,
              "ln_add_bwd: dgamma/dbeta must be fp32[H] (zero-initialized)");
  TORCH_CHECK(scratch.scalar_type() == at::kFloat &&
                  scratch.numel() >= 2L * H * bf_ln_bwd_scratch_rows(nrows),
              "ln_add_bwd: scratch must be fp32[2*H*ln_bwd_scratch_rows]");
  check_hip(bf_ln_add_bwd(dx.data_ptr(), x.data_ptr(), r.data_ptr(),
                          dy.data_ptr(), gamma.data_ptr<float>(),
                          mean.data_ptr<float>(), rstd.data_ptr<float>(),
                          dgamma.data_ptr<float>(), dbeta.data_ptr<float>(),
                          scratch.data_ptr<float>(), H, nrows, dtype_code(x),
                          current_stream()),
            "ln_add_bwd");
}

}  // namespace

PYBIND11_MODULE(TORCH_EXTENSION_NAME, m) {
  m.doc() = "bluefog_amd native CDNA4 kernels (gfx950)";
  m.def("weighted_combine", &weighted_combine,
        "out = self_w*self + sum_k w[k]*gathered[k]");
  m.def("scale_put", &scale_put, "dst = w*src (dst may be xGMI peer memory)");
  m.def("accum_put", &accum_put, "dst += w*src (dst may be xGMI peer memory)");
  m.def("scale_inplace", &scale_inplace, "buf *= f");
  m.def("weighted_combine_sgd", &weighted_combine_sgd,
        "fused neighbor-average + SGD(momentum) over a flat bucket");
  m.def("weighted_combine_adam", &weighted_combine_adam,
        "fused neighbor-average + Adam over a flat bucket");
  m.def("add_relu_fwd", &add_relu_fwd, "out = max(a+b, 0)");
  m.def("relu_bwd_mask", &relu_bwd_mask, "gin = out>0 ? g : 0");
  m.def("ln_add_fwd", &ln_add_fwd,
        "y = LayerNorm(x + r); saves per-row mean/rstd");
  m.def("ln_add_bwd", &ln_add_bwd,
        "dx (shared by both residual branches) + dgamma/dbeta");
  m.def("ln_bwd_scratch_rows",
        [](long nrows) { return bf_ln_bwd_scratch_rows(nrows); },
        "workgroup slots the ln_add_bwd scratch buffer must provide");
}
