// torch.ops bindings for the dlx_hip kernel suite.
//
// Registered under namespace `dlx_hip`; loaded via torch.ops.load_library
// from deeplearning_cfn_amd/ops/__init__.py.  All ops take dense tensors
// (contiguous in their suggested memory format) and run on the current
// HIP stream.
#include <torch/extension.h>
#include <ATen/cuda/CUDAContext.h>

#include "dlx_common.h"

namespace dlx {

// elementwise.hip
template <typename T> void launch_ew1_relu(const T*, T*, long, hipStream_t);
template <typename T> void launch_add_relu(const T*, const T*, T*, long, hipStream_t);
template <typename T> void launch_relu_bwd(const T*, const T*, T*, long, hipStream_t);

// sgd.hip
void launch_sgd_flat(at::Tensor& p, const at::Tensor& g, at::Tensor& m,
                     const at::Tensor* master, double lr, double mu,
                     double wd, bool nesterov, hipStream_t s);
void launch_adamw_flat(at::Tensor& p, const at::Tensor& g, at::Tensor& m,
                       at::Tensor& v, const at::Tensor* master, double lr,
                       double b1, double b2, double eps, double wd,
                       double bc1, double bc2, hipStream_t s);

// softmax_ce.hip
std::tuple<at::Tensor, at::Tensor> softmax_ce_fwd_impl(
    const at::Tensor& logits, const at::Tensor& target);
at::Tensor softmax_ce_bwd_impl(const at::Tensor& probs,
                               const at::Tensor& target,
                               const at::Tensor& grad,
                               at::ScalarType out_dtype);

// gemm.hip
at::Tensor gemm_nt_impl(const at::Tensor& A, const at::Tensor& B,
                        bool out_bf16);

// attention.hip
std::tuple<at::Tensor, at::Tensor> attn_fwd_impl(const at::Tensor& q,
                                                 const at::Tensor& k,
                                                 const at::Tensor& v,
                                                 double scale);
std::tuple<at::Tensor, at::Tensor, at::Tensor> attn_bwd_impl(
    const at::Tensor& grad, const at::Tensor& q, const at::Tensor& k,
    const at::Tensor& v, const at::Tensor& o, const at::Tensor& lse,
    double scale);

// batchnorm.hip
std::tuple<at::Tensor, at::Tensor, at::Tensor, at::Tensor>
bn_fwd_train_impl(
    const at::Tensor& x, const at::Tensor& w, const at::Tensor& b,
    at::Tensor& rmean, at::Tensor& rvar, double momentum, double eps,
    bool relu);
at::Tensor bn_fwd_eval_impl(const at::Tensor& x, const at::Tensor& w,
                            const at::Tensor& b, const at::Tensor& rmean,
                            const at::Tensor& rvar, double eps, bool relu);
std::tuple<at::Tensor, at::Tensor, at::Tensor> bn_bwd_m_impl(
    const at::Tensor& dy, const at::Tensor& x, const at::Tensor& w,
    const at::Tensor& save_mean, const at::Tensor& save_invstd,
    const at::Tensor& mask);
std::tuple<at::Tensor, at::Tensor, at::Tensor, at::Tensor> bn_bwd_res_m_impl(
    const at::Tensor& dy, const at::Tensor& x, const at::Tensor& w,
    const at::Tensor& save_mean, const at::Tensor& save_invstd,
    const at::Tensor& mask);
std::tuple<at::Tensor, at::Tensor, at::Tensor, at::Tensor>
bn_fwd_train_res_impl(
    const at::Tensor& x, const at::Tensor& w, const at::Tensor& b,
    at::Tensor& rmean, at::Tensor& rvar, double momentum, double eps,
    const at::Tensor& res);
at::Tensor bn_fwd_eval_res_impl(const at::Tensor& x, const at::Tensor& w,
                                const at::Tensor& b, const at::Tensor& rmean,
                                const at::Tensor& rvar, double eps,
                                const at::Tensor& res);
std::tuple<at::Tensor, at::Tensor, at::Tensor, at::Tensor> bn_bwd_res_impl(
    const at::Tensor& dy, const at::Tensor& x, const at::Tensor& w,
    const at::Tensor& save_mean, const at::Tensor& save_invstd,
    const at::Tensor& y);
std::tuple<at::Tensor, at::Tensor, at::Tensor> bn_bwd_impl(
    const at::Tensor& dy, const at::Tensor& x, const at::Tensor& w,
    const at::Tensor& save_mean, const at::Tensor& save_invstd,
    const at::Tensor& y, bool relu);

// pooling.hip
std::tuple<at::Tensor, at::Tensor> maxpool3x3s2_fwd_impl(const at::Tensor& x);
at::Tensor maxpool3x3s2_bwd_impl(const at::Tensor& dy, const at::Tensor& argmax,
                                 int64_t H, int64_t W);
at::Tensor gap_fwd_impl(const at::Tensor& x);
at::Tensor gap_bwd_impl(const at::Tensor& dy, int64_t H, int64_t W);

// pack.hip
at::Tensor pack_nchw_to_nhwc_impl(const at::Tensor& x);
at::Tensor pack_nhwc_to_nchw_impl(const at::Tensor& x);
at::Tensor transpose2d_impl(const at::Tensor& x);

// conv.hip
at::Tensor conv2d_fwd_impl(const at::Tensor& x, const at::Tensor& w,
                           int64_t stride, int64_t pad);
at::Tensor conv2d_dgrad_impl(const at::Tensor& g, const at::Tensor& w,
                             int64_t H, int64_t W, int64_t stride,
                             int64_t pad);
at::Tensor conv2d_dgrad_acc_impl(const at::Tensor& g, const at::Tensor& w,
                                 int64_t H, int64_t W, int64_t stride,
                                 int64_t pad, const at::Tensor& accum);

// conv_wrw.hip
at::Tensor conv2d_wrw_impl(const at::Tensor& g, const at::Tensor& x,
                           int64_t Kh, int64_t Kw, int64_t stride,
                           int64_t pad);

// layernorm.hip
std::tuple<at::Tensor, at::Tensor, at::Tensor> ln_fwd_impl(
    const at::Tensor& x, const at::Tensor& w, const at::Tensor& b, double eps);
std::tuple<at::Tensor, at::Tensor, at::Tensor> ln_bwd_impl(
    const at::Tensor& dy, const at::Tensor& x, const at::Tensor& w,
    const at::Tensor& mean, const at::Tensor& invstd);
at::Tensor gelu_fwd_impl(const at::Tensor& x);
at::Tensor gelu_bwd_impl(const at::Tensor& g, const at::Tensor& x);

// conv_wrw.hip — reduce-major GEMM (Linear wgrad)
at::Tensor gemm_tn_mred_impl(const at::Tensor& a, const at::Tensor& b);

// conv fwd with channel-stats epilogue + BN presum forward (fusion seam)
std::tuple<at::Tensor, at::Tensor> conv2d_fwd_stats_impl(
    const at::Tensor& x, const at::Tensor& w, int64_t stride, int64_t pad);
std::tuple<at::Tensor, at::Tensor, at::Tensor, at::Tensor>
bn_fwd_train_presum_impl(
    const at::Tensor& x, const at::Tensor& w, const at::Tensor& b,
    at::Tensor& rmean, at::Tensor& rvar, double momentum, double eps,
    bool relu, const at::Tensor& sums);
std::tuple<at::Tensor, at::Tensor, at::Tensor, at::Tensor>
bn_fwd_train_res_presum_impl(
    const at::Tensor& x, const at::Tensor& w, const at::Tensor& b,
    at::Tensor& rmean, at::Tensor& rvar, double momentum, double eps,
    const at::Tensor& res, const at::Tensor& sums);

namespace {

hipStream_t cur_stream() { return at::cuda::getCurrentCUDAStream(); }

at::Tensor dense(const at::Tensor& t) {
  return t.contiguous(t.suggest_memory_format());
}

template <typename F16, typename F32>
at::Tensor ew_dispatch(const at::Tensor& x, F16 f16, F32 f32) {
  auto xc = dense(x);
  auto out = at::empty_like(xc);
  long n = xc.numel();
  if (xc.scalar_type() == at::kBFloat16) {
    f16(reinterpret_cast<const uint16_t*>(xc.data_ptr()),
        reinterpret_cast<uint16_t*>(out.data_ptr()), n);
  } else if (xc.scalar_type() == at::kFloat) {
    f32(xc.data_ptr<float>(), out.data_ptr<float>(), n);
  } else {
    TORCH_CHECK(false, "dlx_hip elementwise: unsupported dtype ",
                xc.scalar_type());
  }
  return out;
}

at::Tensor relu_fwd(const at::Tensor& x) {
  auto s = cur_stream();
  return ew_dispatch(
      x,
      [&](const uint16_t* a, uint16_t* o, long n) { launch_ew1_relu(a, o, n, s); },
      [&](const float* a, float* o, long n) { launch_ew1_relu(a, o, n, s); });
}

at::Tensor add_relu_fwd(const at::Tensor& a, const at::Tensor& b) {
  auto ac = dense(a);
  auto bc = b.contiguous(a.suggest_memory_format());
  TORCH_CHECK(ac.sizes() == bc.sizes() && ac.scalar_type() == bc.scalar_type(),
              "add_relu: mismatched inputs");
  auto out = at::empty_like(ac);
  long n = ac.numel();
  auto s = cur_stream();
  if (ac.scalar_type() == at::kBFloat16) {
    launch_add_relu(reinterpret_cast<const uint16_t*>(ac.data_ptr()),
                    reinterpret_cast<const uint16_t*>(bc.data_ptr()),
                    reinterpret_cast<uint16_t*>(out.data_ptr()), n, s);
  } else if (ac.scalar_type() == at::kFloat) {
    launch_add_relu(ac.data_ptr<float>(), bc.data_ptr<float>(),
                    out.data_ptr<float>(), n, s);
  } else {
    TORCH_CHECK(false, "add_relu: unsupported dtype ", ac.scalar_type());
  }
  return out;
}

at::Tensor relu_bwd(const at::Tensor& grad, const at::Tensor& out) {
  auto oc = dense(out);
  auto gc = grad.contiguous(out.suggest_memory_format());
  TORCH_CHECK(gc.sizes() == oc.sizes() && gc.scalar_type() == oc.scalar_type(),
              "relu_bwd: mismatched inputs");
  auto dx = at::empty_like(oc);
  long n = oc.numel();
  auto s = cur_stream();
  if (oc.scalar_type() == at::kBFloat16) {
    launch_relu_bwd(reinterpret_cast<const uint16_t*>(gc.data_ptr()),
                    reinterpret_cast<const uint16_t*>(oc.data_ptr()),
                    reinterpret_cast<uint16_t*>(dx.data_ptr()), n, s);
  } else if (oc.scalar_type() == at::kFloat) {
    launch_relu_bwd(gc.data_ptr<float>(), oc.data_ptr<float>(),
                    dx.data_ptr<float>(), n, s);
  } else {
    TORCH_CHECK(false, "relu_bwd: unsupported dtype ", oc.scalar_type());
  }
  return dx;
}

void sgd_momentum_flat(at::Tensor p, at::Tensor g, at::Tensor m,
                       double lr, double momentum, double wd, bool nesterov) {
  launch_sgd_flat(p, g, m, nullptr, lr, momentum, wd, nesterov, cur_stream());
}

void sgd_momentum_flat_master(at::Tensor p, at::Tensor g, at::Tensor m,
                              at::Tensor master, double lr, double momentum,
                              double wd, bool nesterov) {
  launch_sgd_flat(p, g, m, &master, lr, momentum, wd, nesterov, cur_stream());
}

void adamw_flat(at::Tensor p, at::Tensor g, at::Tensor m, at::Tensor v,
                double lr, double b1, double b2, double eps, double wd,
                double bc1, double bc2) {
  launch_adamw_flat(p, g, m, v, nullptr, lr, b1, b2, eps, wd, bc1, bc2,
                    cur_stream());
}

void adamw_flat_master(at::Tensor p, at::Tensor g, at::Tensor m, at::Tensor v,
                       at::Tensor master, double lr, double b1, double b2,
                       double eps, double wd, double bc1, double bc2) {
  launch_adamw_flat(p, g, m, v, &master, lr, b1, b2, eps, wd, bc1, bc2,
                    cur_stream());
}

std::tuple<at::Tensor, at::Tensor> softmax_ce_fwd(const at::Tensor& logits,
                                                  const at::Tensor& target) {
  return softmax_ce_fwd_impl(logits.contiguous(), target.contiguous());
}

at::Tensor softmax_ce_bwd(const at::Tensor& probs, const at::Tensor& target,
                          const at::Tensor& grad, bool out_bf16) {
  return softmax_ce_bwd_impl(probs, target, grad,
                             out_bf16 ? at::kBFloat16 : at::kFloat);
}

// x[M,K] @ W[N,K]^T -> [M,N]  (the natural MFMA form; both K-contiguous)
at::Tensor gemm_nt(const at::Tensor& a, const at::Tensor& b) {
  return gemm_nt_impl(a, b, /*out_bf16=*/true);
}

// g[M,N] @ W[N,K] -> [M,K]  (dgrad; W transposed to K-contiguous form)
at::Tensor gemm_nn(const at::Tensor& g, const at::Tensor& w) {
  return gemm_nt_impl(g, w.t().contiguous(), true);
}

// g[M,N]^T @ x[M,K] -> [N,K]  (wgrad) — reduce-major kernel, no
// transposed copies (conv_wrw.hip)
at::Tensor gemm_tn(const at::Tensor& g, const at::Tensor& x) {
  return gemm_tn_mred_impl(g, x);
}

}  // namespace
}  // namespace dlx

TORCH_LIBRARY(dlx_hip, m) {
  m.def("relu_fwd(Tensor x) -> Tensor");
  m.def("relu_bwd(Tensor grad, Tensor out) -> Tensor");
  m.def("add_relu_fwd(Tensor a, Tensor b) -> Tensor");
  m.def("sgd_momentum_flat(Tensor(a!) p, Tensor g, Tensor(b!) m, float lr, float momentum, float wd, bool nesterov) -> ()");
  m.def("sgd_momentum_flat_master(Tensor(a!) p, Tensor g, Tensor(b!) m, Tensor(c!) master, float lr, float momentum, float wd, bool nesterov) -> ()");
  m.def("adamw_flat(Tensor(a!) p, Tensor g, Tensor(b!) m, Tensor(c!) v, float lr, float b1, float b2, float eps, float wd, float bc1, float bc2) -> ()");
  m.def("adamw_flat_master(Tensor(a!) p, Tensor g, Tensor(b!) m, Tensor(c!) v, Tensor(d!) master, float lr, float b1, float b2, float eps, float wd, float bc1, float bc2) -> ()");
  m.def("softmax_ce_fwd(Tensor logits, Tensor target) -> (Tensor, Tensor)");
  m.def("softmax_ce_bwd(Tensor probs, Tensor target, Tensor grad, bool out_bf16) -> Tensor");
  m.def("bn_fwd_train(Tensor x, Tensor w, Tensor b, Tensor(a!) rmean, Tensor(b!) rvar, float momentum, float eps, bool relu) -> (Tensor, Tensor, Tensor, Tensor)");
  m.def("bn_fwd_eval(Tensor x, Tensor w, Tensor b, Tensor rmean, Tensor rvar, float eps, bool relu) -> Tensor");
  m.def("bn_bwd(Tensor dy, Tensor x, Tensor w, Tensor save_mean, Tensor save_invstd, Tensor y, bool relu) -> (Tensor, Tensor, Tensor)");
  m.def("bn_fwd_train_res(Tensor x, Tensor w, Tensor b, Tensor(a!) rmean, Tensor(b!) rvar, float momentum, float eps, Tensor res) -> (Tensor, Tensor, Tensor, Tensor)");
  m.def("bn_bwd_m(Tensor dy, Tensor x, Tensor w, Tensor save_mean, Tensor save_invstd, Tensor mask) -> (Tensor, Tensor, Tensor)");
  m.def("bn_bwd_res_m(Tensor dy, Tensor x, Tensor w, Tensor save_mean, Tensor save_invstd, Tensor mask) -> (Tensor, Tensor, Tensor, Tensor)");
  m.def("bn_fwd_eval_res(Tensor x, Tensor w, Tensor b, Tensor rmean, Tensor rvar, float eps, Tensor res) -> Tensor");
  m.def("bn_bwd_res(Tensor dy, Tensor x, Tensor w, Tensor save_mean, Tensor save_invstd, Tensor y) -> (Tensor, Tensor, Tensor, Tensor)");
  m.def("gemm_nt(Tensor a, Tensor b) -> Tensor");
  m.def("gemm_nn(Tensor g, Tensor w) -> Tensor");
  m.def("gemm_tn(Tensor g, Tensor x) -> Tensor");
  m.def("maxpool3x3s2_fwd(Tensor x) -> (Tensor, Tensor)");
  m.def("maxpool3x3s2_bwd(Tensor dy, Tensor argmax, int H, int W) -> Tensor");
  m.def("gap_fwd(Tensor x) -> Tensor");
  m.def("gap_bwd(Tensor dy, int H, int W) -> Tensor");
  m.def("pack_nchw_to_nhwc(Tensor x) -> Tensor");
  m.def("pack_nhwc_to_nchw(Tensor x) -> Tensor");
  m.def("transpose2d(Tensor x) -> Tensor");
  m.def("conv2d_fwd(Tensor x, Tensor w, int stride, int pad) -> Tensor");
  m.def("conv2d_dgrad(Tensor g, Tensor w, int H, int W, int stride, int pad) -> Tensor");
  m.def("conv2d_dgrad_acc(Tensor g, Tensor w, int H, int W, int stride, int pad, Tensor accum) -> Tensor");
  m.def("conv2d_wrw(Tensor g, Tensor x, int Kh, int Kw, int stride, int pad) -> Tensor");
  m.def("conv2d_fwd_stats(Tensor x, Tensor w, int stride, int pad) -> (Tensor, Tensor)");
  m.def("bn_fwd_train_presum(Tensor x, Tensor w, Tensor b, Tensor(a!) rmean, Tensor(b!) rvar, float momentum, float eps, bool relu, Tensor sums) -> (Tensor, Tensor, Tensor, Tensor)");
  m.def("bn_fwd_train_res_presum(Tensor x, Tensor w, Tensor b, Tensor(a!) rmean, Tensor(b!) rvar, float momentum, float eps, Tensor res, Tensor sums) -> (Tensor, Tensor, Tensor, Tensor)");
  m.def("ln_fwd(Tensor x, Tensor w, Tensor b, float eps) -> (Tensor, Tensor, Tensor)");
  m.def("ln_bwd(Tensor dy, Tensor x, Tensor w, Tensor mean, Tensor invstd) -> (Tensor, Tensor, Tensor)");
  m.def("gelu_fwd(Tensor x) -> Tensor");
  m.def("gelu_bwd(Tensor g, Tensor x) -> Tensor");
  m.def("attn_fwd(Tensor q, Tensor k, Tensor v, float scale) -> (Tensor, Tensor)");
  m.def("attn_bwd(Tensor grad, Tensor q, Tensor k, Tensor v, Tensor o, Tensor lse, float scale) -> (Tensor, Tensor, Tensor)");
}

TORCH_LIBRARY_IMPL(dlx_hip, CUDA, m) {
  m.impl("relu_fwd", dlx::relu_fwd);
  m.impl("relu_bwd", dlx::relu_bwd);
  m.impl("add_relu_fwd", dlx::add_relu_fwd);
  m.impl("sgd_momentum_flat", dlx::sgd_momentum_flat);
  m.impl("sgd_momentum_flat_master", dlx::sgd_momentum_flat_master);
  m.impl("adamw_flat", dlx::adamw_flat);
  m.impl("adamw_flat_master", dlx::adamw_flat_master);
  m.impl("softmax_ce_fwd", dlx::softmax_ce_fwd);
  m.impl("softmax_ce_bwd", dlx::softmax_ce_bwd);
  m.impl("bn_fwd_train", dlx::bn_fwd_train_impl);
  m.impl("bn_fwd_eval", dlx::bn_fwd_eval_impl);
  m.impl("bn_bwd", dlx::bn_bwd_impl);
  m.impl("bn_fwd_train_res", dlx::bn_fwd_train_res_impl);
  m.impl("bn_fwd_eval_res", dlx::bn_fwd_eval_res_impl);
  m.impl("bn_bwd_res", dlx::bn_bwd_res_impl);
  m.impl("bn_bwd_m", dlx::bn_bwd_m_impl);
  m.impl("bn_bwd_res_m", dlx::bn_bwd_res_m_impl);
  m.impl("gemm_nt", dlx::gemm_nt);
  m.impl("gemm_nn", dlx::gemm_nn);
  m.impl("gemm_tn", dlx::gemm_tn);
  m.impl("maxpool3x3s2_fwd", dlx::maxpool3x3s2_fwd_impl);
  m.impl("maxpool3x3s2_bwd", dlx::maxpool3x3s2_bwd_impl);
  m.impl("gap_fwd", dlx::gap_fwd_impl);
  m.impl("gap_bwd", dlx::gap_bwd_impl);
  m.impl("pack_nchw_to_nhwc", dlx::pack_nchw_to_nhwc_impl);
  m.impl("pack_nhwc_to_nchw", dlx::pack_nhwc_to_nchw_impl);
  m.impl("transpose2d", dlx::transpose2d_impl);
  m.impl("conv2d_fwd", dlx::conv2d_fwd_impl);
  m.impl("conv2d_dgrad", dlx::conv2d_dgrad_impl);
  m.impl("conv2d_dgrad_acc", dlx::conv2d_dgrad_acc_impl);
  m.impl("conv2d_wrw", dlx::conv2d_wrw_impl);
  m.impl("conv2d_fwd_stats", dlx::conv2d_fwd_stats_impl);
  m.impl("bn_fwd_train_presum", dlx::bn_fwd_train_presum_impl);
  m.impl("bn_fwd_train_res_presum", dlx::bn_fwd_train_res_presum_impl);
  m.impl("ln_fwd", dlx::ln_fwd_impl);
  m.impl("ln_bwd", dlx::ln_bwd_impl);
  m.impl("gelu_fwd", dlx::gelu_fwd_impl);
  m.impl("gelu_bwd", dlx::gelu_bwd_impl);
  m.impl("attn_fwd", dlx::attn_fwd_impl);
  m.impl("attn_bwd", dlx::attn_bwd_impl);
}
