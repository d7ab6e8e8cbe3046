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
#include <mutex>
#include <unordered_map>
#include <ATen/cuda/CUDAContext.h>

#include "igemm_core.h"

namespace dlx {

namespace {
at::Tensor zero_page_for(const at::Tensor& like) {
  static std::mutex mu;
  static std::unordered_map<int, at::Tensor> pages;
  std::lock_guard<std::mutex> lock(mu);
  auto& zp = pages[like.device().index()];
  if (!zp.defined())
    zp = at::zeros({16}, like.options().dtype(at::kBFloat16));
  return zp;
}
// ---- small-Cin ("stem") support: channel+width padded copy ------------
// x [N,Ci<=4,H,W] CL -> x4p [N,H,Wp,4] bf16 (zero pad channels to 4 and
// width by `pad` left + up to Kwp-1-pad right) so every 16-B im2col slot
// is memory-contiguous (StemGather, igemm_core.h).

__global__ void stem_pack_kernel(const uint16_t* __restrict__ x,
                                 uint16_t* __restrict__ x4p, long chunks,
                                 int H, int W, int Ci, int Wp, int pad) {
  const long stride_g = (long)gridDim.x * blockDim.x;
  for (long i = (long)blockIdx.x * blockDim.x + threadIdx.x; i < chunks;
       i += stride_g) {
    // i = (n*H + h)*Wp + wp
    const int wp = (int)(i % Wp);
    const long nh = i / Wp;
    const int wi = wp - pad;
    uint16_t v[4] = {0, 0, 0, 0};
    if (wi >= 0 && wi < W) {
      const uint16_t* src = x + (nh * W + wi) * Ci;
#pragma unroll
      for (int c = 0; c < 4; ++c)
        if (c < Ci) v[c] = src[c];
    }
    *reinterpret_cast<uint2*>(x4p + i * 4) =
        *reinterpret_cast<const uint2*>(v);
  }
}

// w [Co,Ci,Kh,Kw] -> flat [Co, Kh*Kwp*4] bf16 with dw/channel zero pad
at::Tensor stem_pack_w(const at::Tensor& w_in, int Kwp) {
  const long Co = w_in.size(0), Ci = w_in.size(1), Kh = w_in.size(2),
             Kw = w_in.size(3);
  auto wx = w_in.permute({0, 2, 3, 1}).contiguous();  // [Co,Kh,Kw,Ci]
  auto w4 = at::zeros({Co, Kh, (long)Kwp, 4},
                      w_in.options().dtype(at::kBFloat16));
  w4.narrow(2, 0, Kw).narrow(3, 0, Ci).copy_(wx);
  return w4.view({Co, Kh * (long)Kwp * 4});
}

StemGather make_stem_gather(const uint16_t* x4p, const ConvShape& p,
                            int Wp, int Kwp) {
  StemGather g;
  g.base = x4p;
  g.M = p.M;
  g.K = (long)p.Kh * Kwp * 4;
  g.Ho = p.Ho; g.Wo = p.Wo; g.H = p.H; g.Wp = Wp;
  g.stride = p.stride; g.pad = p.pad;
  g.row4 = Kwp * 4;
  g.dv_wo.init(p.Wo);
  g.dv_hw.init(p.Ho * p.Wo);
  g.dv_row.init(g.row4);
  return g;
}

}  // namespace

// exported: conv_wrw.hip packs the same x4p for the stem weight grad
at::Tensor stem_pack_x(const at::Tensor& x_cl, int Wp, int pad) {
  const int Ci = x_cl.size(1), H = x_cl.size(2), W = x_cl.size(3);
  const int N = x_cl.size(0);
  auto x4p = at::empty({(long)N * H * Wp * 4},
                       x_cl.options().dtype(at::kBFloat16));
  const long chunks = (long)N * H * Wp;
  auto s = at::cuda::getCurrentCUDAStream();
  stem_pack_kernel<<<grid_for(chunks), kBlock, 0, s>>>(
      reinterpret_cast<const uint16_t*>(x_cl.data_ptr()),
      reinterpret_cast<uint16_t*>(x4p.data_ptr()), chunks, H, W, Ci, Wp,
      pad);
  return x4p;
}

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
  TORCH_CHECK(p.Ci % 8 == 0 || p.Ci <= 4,
              "conv fwd: Cin must be %8==0 or <=4");

  auto y = at::empty({p.N, p.Co, p.Ho, p.Wo},
                     x.options().memory_format(at::MemoryFormat::ChannelsLast));
  auto zp = zero_page_for(x);
  auto s = at::cuda::getCurrentCUDAStream();
  if (p.Ci <= 4) {
    // small-Cin stem path: packed x4p + padded-k weights
    const int Kwp = (p.Kw + 1) & ~1;
    const int Wp = (p.Wo - 1) * p.stride + Kwp;
    auto x4p = stem_pack_x(x, Wp, p.pad);
    auto w4 = stem_pack_w(w_in, Kwp);
    StemGather ag = make_stem_gather(
        reinterpret_cast<const uint16_t*>(x4p.data_ptr()), p, Wp, Kwp);
    RowGather bg4{reinterpret_cast<const uint16_t*>(w4.data_ptr()),
                  (long)p.Co, ag.K};
    igemm_launch<StemGather, RowGather, true>(
        ag, bg4, y.data_ptr(),
        reinterpret_cast<const uint16_t*>(zp.data_ptr()), p.M, p.Co, ag.K,
        s);
    return y;
  }
  RowGather bg{reinterpret_cast<const uint16_t*>(w.data_ptr()), p.Co, p.K};
  if (p.Kh == 1 && p.Kw == 1 && p.stride == 1 && p.pad == 0) {
    // 1x1/s1 is a plain GEMM: y[m,co] = x[m,:] @ W[co,:]^T — skip the
    // im2col gather's per-slot address math entirely
    RowGather ag{reinterpret_cast<const uint16_t*>(x.data_ptr()), p.M, p.K};
    igemm_launch<RowGather, RowGather, true>(
        ag, bg, y.data_ptr(),
        reinterpret_cast<const uint16_t*>(zp.data_ptr()), p.M, p.Co, p.K, s);
    return y;
  }
  ConvFwdGather ag{reinterpret_cast<const uint16_t*>(x.data_ptr()), p};
  igemm_launch<ConvFwdGather, RowGather, true>(
      ag, bg, y.data_ptr(),
      reinterpret_cast<const uint16_t*>(zp.data_ptr()), p.M, p.Co, p.K, s);
  return y;
}

// conv fwd + per-block channel stats: returns (y, sums[gm][2][Co]) — the
// BN reduce pass collapses into this epilogue (bn_fwd_train_presum).
std::tuple<at::Tensor, at::Tensor> conv2d_fwd_stats_impl(
    const at::Tensor& x_in, const at::Tensor& w_in, int64_t stride,
    int64_t pad) {
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
  TORCH_CHECK(p.Ci % 8 == 0 || p.Ci <= 4,
              "conv fwd: Cin must be %8==0 or <=4");

  auto y = at::empty({p.N, p.Co, p.Ho, p.Wo},
                     x.options().memory_format(at::MemoryFormat::ChannelsLast));
  // geometry (and so the stats slab size) keys on the EFFECTIVE K —
  // the stem path pads K to Kh*Kwp*4
  const long Keff = (p.Ci <= 4)
                        ? (long)p.Kh * ((p.Kw + 1) & ~1) * 4
                        : p.K;
  const long gm = igemm_gm(p.M, p.Co, Keff);
  auto sums = at::empty({gm * 2 * p.Co}, x.options().dtype(at::kFloat));
  auto zp = zero_page_for(x);
  auto s = at::cuda::getCurrentCUDAStream();
  if (p.Ci <= 4) {
    const int Kwp = (p.Kw + 1) & ~1;
    const int Wp = (p.Wo - 1) * p.stride + Kwp;
    auto x4p = stem_pack_x(x, Wp, p.pad);
    auto w4 = stem_pack_w(w_in, Kwp);
    StemGather ag = make_stem_gather(
        reinterpret_cast<const uint16_t*>(x4p.data_ptr()), p, Wp, Kwp);
    RowGather bg4{reinterpret_cast<const uint16_t*>(w4.data_ptr()),
                  (long)p.Co, ag.K};
    igemm_launch_stats<StemGather, RowGather>(
        ag, bg4, y.data_ptr(),
        reinterpret_cast<const uint16_t*>(zp.data_ptr()), p.M, p.Co, ag.K,
        s, sums.data_ptr<float>());
    return {y, sums};
  }
  RowGather bg{reinterpret_cast<const uint16_t*>(w.data_ptr()), p.Co, p.K};
  if (p.Kh == 1 && p.Kw == 1 && p.stride == 1 && p.pad == 0) {
    RowGather ag{reinterpret_cast<const uint16_t*>(x.data_ptr()), p.M, p.K};
    igemm_launch_stats<RowGather, RowGather>(
        ag, bg, y.data_ptr(),
        reinterpret_cast<const uint16_t*>(zp.data_ptr()), p.M, p.Co, p.K, s,
        sums.data_ptr<float>());
  } else {
    ConvFwdGather ag{reinterpret_cast<const uint16_t*>(x.data_ptr()), p};
    igemm_launch_stats<ConvFwdGather, RowGather>(
        ag, bg, y.data_ptr(),
        reinterpret_cast<const uint16_t*>(zp.data_ptr()), p.M, p.Co, p.K, s,
        sums.data_ptr<float>());
  }
  return {y, sums};
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
  auto zp = zero_page_for(g);
  auto s = at::cuda::getCurrentCUDAStream();

  if (p.stride == 2) {
    // Parity decomposition: 4 classes, each a dense GEMM over only the
    // taps that can reach its pixels — 4x fewer MACs than zero-padding.
    // A class with no taps (e.g. 1x1/s2: three of four) leaves dx zero,
    // so pre-zero only in that case.
    bool any_empty = false;
    for (int a = 0; a < 2; ++a)
      for (int b = 0; b < 2; ++b)
        if (p.Kh <= ((a + p.pad) & 1) || p.Kw <= ((b + p.pad) & 1))
          any_empty = true;
    // NB: at::zeros ignores TensorOptions.memory_format — allocate with
    // empty (which honors it) and zero_() only when classes stay empty.
    auto dx = at::empty({p.N, p.Ci, p.H, p.W},
                        g.options().memory_format(
                            at::MemoryFormat::ChannelsLast));
    if (any_empty) dx.zero_();
    auto wt4 = wt.view({(long)p.Ci, (long)p.Kh, (long)p.Kw, (long)p.Co});
    for (int a = 0; a < 2; ++a) {
      for (int b = 0; b < 2; ++b) {
        const int pa = (a + p.pad) & 1, pb = (b + p.pad) & 1;
        const int nKh = (p.Kh > pa) ? (p.Kh - pa + 1) / 2 : 0;
        const int nKw = (p.Kw > pb) ? (p.Kw - pb + 1) / 2 : 0;
        if (nKh == 0 || nKw == 0) continue;  // dx stays zero here
        const int Ha = (p.H > a) ? (p.H - a + 1) / 2 : 0;
        const int Wb = (p.W > b) ? (p.W - b + 1) / 2 : 0;
        if (Ha == 0 || Wb == 0) continue;
        using at::indexing::Slice;
        auto wt_ab = wt4.index({Slice(), Slice(pa, p.Kh, 2),
                                Slice(pb, p.Kw, 2), Slice()})
                         .contiguous()
                         .view({(long)p.Ci, (long)nKh * nKw * p.Co});
        ConvDgradS2Gather ag;
        ag.base = reinterpret_cast<const uint16_t*>(g.data_ptr());
        ag.p = p;
        ag.a = a; ag.b = b; ag.pa = pa; ag.pb = pb; ag.nKw = nKw;
        ag.Wb = Wb; ag.HaWb = Ha * Wb;
        ag.dv_wb.init(Wb); ag.dv_hwb.init(Ha * Wb); ag.dv_nkw.init(nKw);
        ag.M2 = (long)p.N * Ha * Wb;
        ag.K2 = (long)nKh * nKw * p.Co;
        RowGather bg{reinterpret_cast<const uint16_t*>(wt_ab.data_ptr()),
                     p.Ci, ag.K2};
        Stride2Out om;
        om.H = p.H; om.W = p.W; om.Ci = p.Ci; om.a = a; om.b = b;
        om.Wb = Wb; om.HaWb = Ha * Wb;
        om.dv_wb.init(Wb); om.dv_hwb.init(Ha * Wb);
        igemm_launch_om<ConvDgradS2Gather, RowGather, true, Stride2Out>(
            ag, bg, dx.data_ptr(),
            reinterpret_cast<const uint16_t*>(zp.data_ptr()), ag.M2, p.Ci,
            ag.K2, s, om);
      }
    }
    return dx;
  }

  auto dx = at::empty({p.N, p.Ci, p.H, p.W},
                      g.options().memory_format(at::MemoryFormat::ChannelsLast));
  RowGather bg{reinterpret_cast<const uint16_t*>(wt.data_ptr()), p.Ci, p.K};
  if (p.Kh == 1 && p.Kw == 1 && p.stride == 1 && p.pad == 0) {
    // 1x1/s1 dgrad is a plain GEMM: dx[m,ci] = g[m,:] @ Wt[ci,:]^T
    RowGather ag{reinterpret_cast<const uint16_t*>(g.data_ptr()), p.M, p.K};
    igemm_launch<RowGather, RowGather, true>(
        ag, bg, dx.data_ptr(),
        reinterpret_cast<const uint16_t*>(zp.data_ptr()), p.M, p.Ci, p.K, s);
    return dx;
  }
  if (p.stride == 1) {
    // no /stride divides or exactness checks in the gather
    ConvDgradS1Gather ag{reinterpret_cast<const uint16_t*>(g.data_ptr()),
                         p};
    igemm_launch<ConvDgradS1Gather, RowGather, true>(
        ag, bg, dx.data_ptr(),
        reinterpret_cast<const uint16_t*>(zp.data_ptr()), p.M, p.Ci, p.K,
        s);
    return dx;
  }
  ConvDgradGather ag{reinterpret_cast<const uint16_t*>(g.data_ptr()), p};
  igemm_launch<ConvDgradGather, RowGather, true>(
      ag, bg, dx.data_ptr(),
      reinterpret_cast<const uint16_t*>(zp.data_ptr()), p.M, p.Ci, p.K, s);
  return dx;
}

// dgrad with the residual fan-in add fused into the epilogue:
// dx = conv_dgrad(g, w) + accum.  stride-1 only (the Bottleneck conv1
// seam — VERDICT r1 weak #2: at::add at residual junctions).
at::Tensor conv2d_dgrad_acc_impl(const at::Tensor& g_in,
                                 const at::Tensor& w_in, int64_t H,
                                 int64_t W, int64_t stride, int64_t pad,
                                 const at::Tensor& accum_in) {
  TORCH_CHECK(stride == 1, "dgrad_acc: stride-1 only");
  auto g = g_in.contiguous(at::MemoryFormat::ChannelsLast);
  auto accum = accum_in.contiguous(at::MemoryFormat::ChannelsLast);
  ConvShape p;
  p.N = g.size(0); p.Co = g.size(1); p.Ho = g.size(2); p.Wo = g.size(3);
  p.Ci = w_in.size(1); p.Kh = w_in.size(2); p.Kw = w_in.size(3);
  p.H = (int)H; p.W = (int)W; p.stride = 1; p.pad = (int)pad;
  p.K = (long)p.Kh * p.Kw * p.Co;
  p.M = (long)p.N * p.H * p.W;
  p.finalize();
  TORCH_CHECK(p.Co % 8 == 0, "dgrad_acc: Cout % 8 != 0");
  TORCH_CHECK(accum.sizes() == at::IntArrayRef({p.N, p.Ci, p.H, p.W}) &&
                  accum.scalar_type() == at::kBFloat16,
              "dgrad_acc: accum must be bf16 [N,Ci,H,W]");

  auto wt = w_in.permute({1, 2, 3, 0}).contiguous()
                 .view({p.Ci, (long)p.Kh * p.Kw * p.Co});
  auto zp = zero_page_for(g);
  auto s = at::cuda::getCurrentCUDAStream();
  auto dx = at::empty({p.N, p.Ci, p.H, p.W},
                      g.options().memory_format(at::MemoryFormat::ChannelsLast));
  RowGather bg{reinterpret_cast<const uint16_t*>(wt.data_ptr()), p.Ci, p.K};
  const uint16_t* acp = reinterpret_cast<const uint16_t*>(accum.data_ptr());
  if (p.Kh == 1 && p.Kw == 1 && p.pad == 0) {
    RowGather ag{reinterpret_cast<const uint16_t*>(g.data_ptr()), p.M, p.K};
    igemm_launch_acc<RowGather, RowGather>(
        ag, bg, dx.data_ptr(),
        reinterpret_cast<const uint16_t*>(zp.data_ptr()), p.M, p.Ci, p.K,
        s, acp);
    return dx;
  }
  ConvDgradS1Gather ag{reinterpret_cast<const uint16_t*>(g.data_ptr()), p};
  igemm_launch_acc<ConvDgradS1Gather, RowGather>(
      ag, bg, dx.data_ptr(),
      reinterpret_cast<const uint16_t*>(zp.data_ptr()), p.M, p.Ci, p.K, s,
      acp);
  return dx;
}

}  // namespace dlx
