// Implicit-GEMM NHWC convolution on MFMA (fwd + dgrad).
//
// Replaces cuDNN convolutions (SURVEY.md §3.2 "cuDNN conv kernels"): the
// im2col gather is FUSED into global->LDS staging via per-lane glds
// source addresses — no im2col tensor is materialized; padding and tail
// taps read a 16-B zero page.  Core: igemm_core.h.
//
//   fwd:   out[(n,ho,wo), co] = sum_k x[(n,ho*s-p+dh, wo*s-p+dw), ci] *
//                               W_cl[co, (dh,dw,ci)]
//   dgrad: dx[(n,hi,wi), ci]  = sum_k g[(n,(hi+p-dh)/s,(wi+p-dw)/s), co] *
//                               W_t[ci, (dh,dw,co)]   (taps where /s exact)
//
// Requires the contiguous k-axis (Cin fwd / Cout dgrad) % 8 == 0; the
// 3-channel stem stays on the library path (functional.py gate).
#include <torch/extension.h>
#include <ATen/cuda/CUDAContext.h>

#include "igemm_core.h"

namespace dlx {

namespace {
at::Tensor zero_page_for(const at::Tensor& like) {
  static at::Tensor zp;
  if (!zp.defined() || zp.device() != like.device())
    zp = at::zeros({16}, like.options().dtype(at::kBFloat16));
  return zp;
}
}  // namespace

// x: [N,Ci,H,W] channels_last bf16; w: [Co,Ci,Kh,Kw] bf16 -> [N,Co,Ho,Wo] CL
at::Tensor conv2d_fwd_impl(const at::Tensor& x_in, const at::Tensor& w_in,
                           int64_t stride, int64_t pad) {
  auto x = x_in.contiguous(at::MemoryFormat::ChannelsLast);
  auto w = w_in.contiguous(at::MemoryFormat::ChannelsLast);
  TORCH_CHECK(x.scalar_type() == at::kBFloat16, "conv: bf16 only");
  ConvShape p;
  p.N = x.size(0); p.Ci = x.size(1); p.H = x.size(2); p.W = x.size(3);
  p.Co = w.size(0); p.Kh = w.size(2); p.Kw = w.size(3);
  p.stride = (int)stride; p.pad = (int)pad;
  p.Ho = (p.H + 2 * p.pad - p.Kh) / p.stride + 1;
  p.Wo = (p.W + 2 * p.pad - p.Kw) / p.stride + 1;
  p.K = (long)p.Kh * p.Kw * p.Ci;
  p.M = (long)p.N * p.Ho * p.Wo;
  p.finalize();
  TORCH_CHECK(p.Ci % 8 == 0, "conv fwd: Cin % 8 != 0 (use library path)");

  auto y = at::empty({p.N, p.Co, p.Ho, p.Wo},
                     x.options().memory_format(at::MemoryFormat::ChannelsLast));
  auto zp = zero_page_for(x);
  auto s = at::cuda::getCurrentCUDAStream();
  ConvFwdGather ag{reinterpret_cast<const uint16_t*>(x.data_ptr()), p};
  RowGather bg{reinterpret_cast<const uint16_t*>(w.data_ptr()), p.Co, p.K};
  igemm_launch<ConvFwdGather, RowGather, true>(
      ag, bg, y.data_ptr(),
      reinterpret_cast<const uint16_t*>(zp.data_ptr()), p.M, p.Co, p.K, s);
  return y;
}

// g: [N,Co,Ho,Wo] CL; w: [Co,Ci,Kh,Kw]; returns dx [N,Ci,H,W] CL
at::Tensor conv2d_dgrad_impl(const at::Tensor& g_in, const at::Tensor& w_in,
                             int64_t H, int64_t W, int64_t stride,
                             int64_t pad) {
  auto g = g_in.contiguous(at::MemoryFormat::ChannelsLast);
  ConvShape p;
  p.N = g.size(0); p.Co = g.size(1); p.Ho = g.size(2); p.Wo = g.size(3);
  p.Ci = w_in.size(1); p.Kh = w_in.size(2); p.Kw = w_in.size(3);
  p.H = (int)H; p.W = (int)W; p.stride = (int)stride; p.pad = (int)pad;
  p.K = (long)p.Kh * p.Kw * p.Co;
  p.M = (long)p.N * p.H * p.W;
  p.finalize();
  TORCH_CHECK(p.Co % 8 == 0, "conv dgrad: Cout % 8 != 0");

  // W_t[ci, (dh,dw,co)] = W[co, ci, dh, dw]
  auto wt = w_in.permute({1, 2, 3, 0}).contiguous()
                 .view({p.Ci, (long)p.Kh * p.Kw * p.Co});
  auto dx = at::empty({p.N, p.Ci, p.H, p.W},
                      g.options().memory_format(at::MemoryFormat::ChannelsLast));
  auto zp = zero_page_for(g);
  auto s = at::cuda::getCurrentCUDAStream();
  ConvDgradGather ag{reinterpret_cast<const uint16_t*>(g.data_ptr()), p};
  RowGather bg{reinterpret_cast<const uint16_t*>(wt.data_ptr()), p.Ci, p.K};
  igemm_launch<ConvDgradGather, RowGather, true>(
      ag, bg, dx.data_ptr(),
      reinterpret_cast<const uint16_t*>(zp.data_ptr()), p.M, p.Ci, p.K, s);
  return dx;
}

}  // namespace dlx
