// Implicit-GEMM NHWC convolution on MFMA (fwd + dgrad).
//
// Replaces cuDNN convolutions (SURVEY.md §3.2 "cuDNN conv kernels"):
// the im2col gather is FUSED into the global->LDS staging (the per-lane
// SOURCE address of global_load_lds is free — guide §5 caveat), so no
// im2col tensor is ever materialized.  Out-of-range taps (padding, row
// tails, K tails) redirect the lane's source to a 16-B zero page.
//
//   fwd:   out[(n,ho,wo), co] = sum_k x[(n,ho*s-p+dh, wo*s-p+dw), ci] *
//                               W_cl[co, (dh,dw,ci)]
//   dgrad: dx[(n,hi,wi), ci]  = sum_k g[(n,(hi+p-dh)/s,(wi+p-dw)/s), co] *
//                               W_t[ci, (dh,dw,co)]   (taps where /s exact)
//
// Same 128x128xBK64 MFMA skeleton as gemm.hip; requires the contiguous
// k-axis (Cin for fwd, Cout for dgrad) to be a multiple of 8 (true for
// every ResNet shape except the 3-channel stem, which stays on the
// library path).
#include <torch/extension.h>
#include <ATen/cuda/CUDAContext.h>

#include "dlx_common.h"

namespace dlx {

namespace {

constexpr int BM = 128, BN = 128, BK = 64;
constexpr int WAVES = 4;
constexpr int THREADS = WAVES * kWave;
constexpr int FRAG = 16;
constexpr int MR = 4, NR = 4;

using bf16x8 = __attribute__((ext_vector_type(8))) short;
using f32x4 = __attribute__((ext_vector_type(4))) float;

constexpr int SLOTS = (BK / 8) * BM;
typedef __attribute__((address_space(3))) uint32_t lds_u32_t;
typedef const __attribute__((address_space(1))) uint32_t glb_u32_t;
constexpr int GLDS_PER_TILE = SLOTS / THREADS;

struct ConvParams {
  int N, H, W, Ci, Ho, Wo, Co, Kh, Kw, stride, pad;
  long K;        // contraction length
  long M;        // output rows
};

// ---- A-gather policies -------------------------------------------------

// fwd: row m = (n,ho,wo); k = (dh,dw,ci)
struct FwdGather {
  DLX_DEV static const uint16_t* src(const uint16_t* x, const ConvParams& p,
                                     long m, long k,
                                     const uint16_t* zero_page) {
    if (m >= p.M || k >= p.K) return zero_page;
    const int hw = p.Ho * p.Wo;
    const int n = (int)(m / hw);
    const int rem = (int)(m - (long)n * hw);
    const int ho = rem / p.Wo, wo = rem - (rem / p.Wo) * p.Wo;
    const int dhw = (int)(k / p.Ci);
    const int ci = (int)(k - (long)dhw * p.Ci);
    const int dh = dhw / p.Kw, dw = dhw - (dhw / p.Kw) * p.Kw;
    const int hi = ho * p.stride - p.pad + dh;
    const int wi = wo * p.stride - p.pad + dw;
    if (hi < 0 || hi >= p.H || wi < 0 || wi >= p.W) return zero_page;
    return x + (((long)n * p.H + hi) * p.W + wi) * p.Ci + ci;
  }
};

// dgrad: row m = (n,hi,wi); k = (dh,dw,co); x here is grad_out [N,Ho,Wo,Co]
struct DgradGather {
  DLX_DEV static const uint16_t* src(const uint16_t* g, const ConvParams& p,
                                     long m, long k,
                                     const uint16_t* zero_page) {
    if (m >= p.M || k >= p.K) return zero_page;
    const int hw = p.H * p.W;      // dgrad rows run over INPUT pixels
    const int n = (int)(m / hw);
    const int rem = (int)(m - (long)n * hw);
    const int hi = rem / p.W, wi = rem - (rem / p.W) * p.W;
    const int dhw = (int)(k / p.Co);
    const int co = (int)(k - (long)dhw * p.Co);
    const int dh = dhw / p.Kw, dw = dhw - (dhw / p.Kw) * p.Kw;
    int ht = hi + p.pad - dh;
    int wt = wi + p.pad - dw;
    if (ht < 0 || wt < 0 || (ht % p.stride) || (wt % p.stride))
      return zero_page;
    const int ho = ht / p.stride, wo = wt / p.stride;
    if (ho >= p.Ho || wo >= p.Wo) return zero_page;
    return g + (((long)n * p.Ho + ho) * p.Wo + wo) * p.Co + co;
  }
};

// weights: plain [rows, K] row-major, zero-page OOB
DLX_DEV const uint16_t* wsrc(const uint16_t* w, long rows, long K,
                             long row, long k, const uint16_t* zero_page) {
  if (row >= rows || k >= K) return zero_page;
  return w + row * K + k;
}

template <class AG, bool OUT_BF16>
__global__ __launch_bounds__(THREADS) void conv_igemm_kernel(
    const uint16_t* __restrict__ A,      // activations / grad_out
    const uint16_t* __restrict__ Bw,     // packed weights [Nrows, K]
    void* __restrict__ Cout_,            // [M, Nc] bf16 or fp32
    const uint16_t* __restrict__ zero_page,
    ConvParams p, long Nc, int grid_n) {
  __shared__ uint16_t lsA[2 * SLOTS * 8];
  __shared__ uint16_t lsB[2 * SLOTS * 8];

  int nwg = gridDim.x;
  int bid = blockIdx.x;
  {
    int q = nwg / 8, r = nwg % 8;
    int xcd = bid % 8, idx = bid / 8;
    bid = (xcd < r ? xcd * (q + 1) : r * (q + 1) + (xcd - r) * q) + idx;
  }
  const int bm = bid / grid_n, bn = bid % grid_n;
  const long m0 = (long)bm * BM, n0 = (long)bn * BN;

  const int lane = threadIdx.x % kWave;
  const int wid = threadIdx.x / kWave;
  const int wr = wid >> 1, wc = wid & 1;
  const int l15 = lane & 15, l4 = lane >> 4;

  f32x4 acc[MR][NR] = {};
  const long ntile = (p.K + BK - 1) / BK;

  auto stage = [&](int buf, long t) {
    const long k0 = t * BK;
    uint16_t* la = lsA + buf * SLOTS * 8;
    uint16_t* lb = lsB + buf * SLOTS * 8;
#pragma unroll
    for (int i = 0; i < GLDS_PER_TILE; ++i) {
      int f = i * THREADS + threadIdx.x;
      int kc = f / BM;
      int m = f - kc * BM;
      const uint16_t* sa = AG::src(A, p, m0 + m, k0 + kc * 8, zero_page);
      __builtin_amdgcn_global_load_lds(
          (glb_u32_t*)sa, (lds_u32_t*)(la + (long)f * 8), 16, 0, 0);
      const uint16_t* sb = wsrc(Bw, Nc, p.K, n0 + m, k0 + kc * 8, zero_page);
      __builtin_amdgcn_global_load_lds(
          (glb_u32_t*)sb, (lds_u32_t*)(lb + (long)f * 8), 16, 0, 0);
    }
  };

  stage(0, 0);
  __builtin_amdgcn_s_waitcnt(0);
  __syncthreads();

  int buf = 0;
  for (long t = 0; t < ntile; ++t) {
    if (t + 1 < ntile) stage(buf ^ 1, t + 1);
    const uint16_t* la = lsA + buf * SLOTS * 8;
    const uint16_t* lb = lsB + buf * SLOTS * 8;
#pragma unroll
    for (int ks = 0; ks < BK / 32; ++ks) {
      bf16x8 af[MR], bfr[NR];
#pragma unroll
      for (int mi = 0; mi < MR; ++mi) {
        int m = wr * 64 + mi * FRAG + l15;
        int kc = ks * 4 + l4;
        af[mi] = *reinterpret_cast<const bf16x8*>(la + ((long)kc * BM + m) * 8);
      }
#pragma unroll
      for (int ni = 0; ni < NR; ++ni) {
        int n = wc * 64 + ni * FRAG + l15;
        int kc = ks * 4 + l4;
        bfr[ni] = *reinterpret_cast<const bf16x8*>(lb + ((long)kc * BM + n) * 8);
      }
#pragma unroll
      for (int mi = 0; mi < MR; ++mi)
#pragma unroll
        for (int ni = 0; ni < NR; ++ni)
          acc[mi][ni] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
              af[mi], bfr[ni], acc[mi][ni], 0, 0, 0);
    }
    __builtin_amdgcn_s_waitcnt(0);
    __syncthreads();
    buf ^= 1;
  }

#pragma unroll
  for (int mi = 0; mi < MR; ++mi) {
#pragma unroll
    for (int ni = 0; ni < NR; ++ni) {
#pragma unroll
      for (int r = 0; r < 4; ++r) {
        long row = m0 + wr * 64 + mi * FRAG + l4 * 4 + r;
        long col = n0 + wc * 64 + ni * FRAG + l15;
        if (row < p.M && col < Nc) {
          if (OUT_BF16)
            reinterpret_cast<uint16_t*>(Cout_)[row * Nc + col] =
                f2bf(acc[mi][ni][r]);
          else
            reinterpret_cast<float*>(Cout_)[row * Nc + col] = acc[mi][ni][r];
        }
      }
    }
  }
}

static at::Tensor zero_page_for(const at::Tensor& like) {
  // per-device 16-B zero page, cached
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
  ConvParams p;
  p.N = x.size(0); p.Ci = x.size(1); p.H = x.size(2); p.W = x.size(3);
  p.Co = w.size(0); p.Kh = w.size(2); p.Kw = w.size(3);
  p.stride = (int)stride; p.pad = (int)pad;
  p.Ho = (p.H + 2 * p.pad - p.Kh) / p.stride + 1;
  p.Wo = (p.W + 2 * p.pad - p.Kw) / p.stride + 1;
  p.K = (long)p.Kh * p.Kw * p.Ci;
  p.M = (long)p.N * p.Ho * p.Wo;
  TORCH_CHECK(p.Ci % 8 == 0, "conv fwd: Cin % 8 != 0 (use library path)");

  auto y = at::empty({p.N, p.Co, p.Ho, p.Wo},
                     x.options().memory_format(at::MemoryFormat::ChannelsLast));
  auto zp = zero_page_for(x);
  auto s = at::cuda::getCurrentCUDAStream();
  const int grid_m = (int)((p.M + BM - 1) / BM);
  const int grid_n = (p.Co + BN - 1) / BN;
  conv_igemm_kernel<FwdGather, true><<<grid_m * grid_n, THREADS, 0, s>>>(
      reinterpret_cast<const uint16_t*>(x.data_ptr()),
      reinterpret_cast<const uint16_t*>(w.data_ptr()),
      y.data_ptr(), reinterpret_cast<const uint16_t*>(zp.data_ptr()),
      p, p.Co, grid_n);
  return y;
}

// g: [N,Co,Ho,Wo] CL; w: [Co,Ci,Kh,Kw]; returns dx [N,Ci,H,W] CL
at::Tensor conv2d_dgrad_impl(const at::Tensor& g_in, const at::Tensor& w_in,
                             int64_t H, int64_t W, int64_t stride,
                             int64_t pad) {
  auto g = g_in.contiguous(at::MemoryFormat::ChannelsLast);
  ConvParams p;
  p.N = g.size(0); p.Co = g.size(1); p.Ho = g.size(2); p.Wo = g.size(3);
  p.Ci = w_in.size(1); p.Kh = w_in.size(2); p.Kw = w_in.size(3);
  p.H = (int)H; p.W = (int)W; p.stride = (int)stride; p.pad = (int)pad;
  p.K = (long)p.Kh * p.Kw * p.Co;
  p.M = (long)p.N * p.H * p.W;
  TORCH_CHECK(p.Co % 8 == 0, "conv dgrad: Cout % 8 != 0");

  // W_t[ci, (dh,dw,co)] = W[co, ci, dh, dw]
  auto wt = w_in.permute({1, 2, 3, 0}).contiguous()
                 .view({p.Ci, (long)p.Kh * p.Kw * p.Co});
  auto dx = at::empty({p.N, p.Ci, p.H, p.W},
                      g.options().memory_format(at::MemoryFormat::ChannelsLast));
  auto zp = zero_page_for(g);
  auto s = at::cuda::getCurrentCUDAStream();
  const int grid_m = (int)((p.M + BM - 1) / BM);
  const int grid_n = (p.Ci + BN - 1) / BN;
  conv_igemm_kernel<DgradGather, true><<<grid_m * grid_n, THREADS, 0, s>>>(
      reinterpret_cast<const uint16_t*>(g.data_ptr()),
      reinterpret_cast<const uint16_t*>(wt.data_ptr()),
      dx.data_ptr(), reinterpret_cast<const uint16_t*>(zp.data_ptr()),
      p, p.Ci, grid_n);
  return dx;
}

}  // namespace dlx
