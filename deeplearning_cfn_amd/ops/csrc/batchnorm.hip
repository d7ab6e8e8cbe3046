// NHWC BatchNorm (training fwd / eval fwd / bwd), fused with ReLU.
//
// Replaces cuDNN BatchNorm + the adjacent framework ReLU (SURVEY.md §3.2
// rows "cuDNN BatchNorm", "framework elementwise").  Design:
//  * x is channels-last ([R=N*H*W, C] with C contiguous); statistics and
//    all per-channel coefficients accumulate in fp32;
//  * every elementwise pass (reduce / normalize / dx) uses ONE geometry:
//    each thread owns a fixed VW-channel chunk and streams a contiguous
//    row-stripe, so per-channel coefficients are loaded ONCE per thread
//    (not per element) and the inner loop is pure 16 B/lane streaming;
//  * per-stripe fp32 partials (deterministic) + a per-channel parallel
//    finalize block (C blocks, LDS tree) — no serial stripe walk;
//  * ReLU is fused: fwd applies max(0,·); bwd masks dy by (y > 0).
#include <torch/extension.h>
#include <ATen/cuda/CUDAContext.h>

#include "dlx_common.h"

namespace dlx {

namespace {

// channel-tile width: largest power of two <= 256 dividing C
static int channel_tile(int C) {
  int ct = 256;
  while (ct > 1 && (C % ct)) ct >>= 1;
  return ct;
}

struct BNShape {
  long R;
  int C;
  int CT;       // chunks per block (vec: VW-channel units; scalar: channels)
  bool vec;
  int stripes;  // contiguous row partitions (deterministic partials)
  long rows_per_stripe;
  dim3 grid;    // (stripes, C-blocks)
  int lds;
};

static BNShape bn_shape(const at::Tensor& x, int vw) {
  BNShape s;
  s.C = x.size(1);
  s.R = x.numel() / s.C;
  s.vec = (s.C % vw) == 0;
  const int units = s.vec ? s.C / vw : s.C;
  s.CT = channel_tile(units);
  const int cblocks = units / s.CT;
  const int rows_grp = kBlock / s.CT;
  // ~1280 blocks fills 256 CUs 5x; finalize parallelizes over C so more
  // stripes only grow the (tiny) partial buffer.
  long max_stripes = std::min<long>(2048, (s.R + rows_grp - 1) / rows_grp);
  long st = std::max<long>(1, 1280 / std::max(1, cblocks));
  if (st > max_stripes) st = max_stripes;
  s.stripes = (int)st;
  s.rows_per_stripe = (s.R + st - 1) / st;
  s.grid = dim3((unsigned)st, (unsigned)cblocks);
  s.lds = 2 * s.CT * (s.vec ? vw : 1) * sizeof(float);
  return s;
}

// ---- reduction: sum(x), sum(x^2) per channel ---------------------------
// One fp32 partial pair per (stripe, channel): sums[stripe][2][C].

template <typename T, bool VEC>
__global__ void bn_reduce_fwd_kernel(const T* __restrict__ x,
                                     float* __restrict__ sums, long R, int C,
                                     int CT, long rows_per_stripe) {
  using VT = VecTraits<T>;
  using Vec = typename VT::Vec;
  constexpr int VW = VEC ? VT::width : 1;
  extern __shared__ float lds[];  // [2*CT*VW]
  for (int i = threadIdx.x; i < 2 * CT * VW; i += blockDim.x) lds[i] = 0.f;
  __syncthreads();

  const int chunk = threadIdx.x % CT;
  const int c0 = (blockIdx.y * CT + chunk) * VW;
  const int sub = threadIdx.x / CT;
  const int rows_grp = blockDim.x / CT;
  const long rbeg = (long)blockIdx.x * rows_per_stripe;
  const long rend = (rbeg + rows_per_stripe < R) ? rbeg + rows_per_stripe : R;
  float s[VW] = {}, sq[VW] = {};
  for (long r = rbeg + sub; r < rend; r += rows_grp) {
    if (VEC) {
      Vec v = *reinterpret_cast<const Vec*>(x + r * C + c0);
#pragma unroll
      for (int j = 0; j < VW; ++j) {
        float f = VT::load(v, j);
        s[j] += f;
        sq[j] += f * f;
      }
    } else {
      float f = VT::load1(x + r * C + c0);
      s[0] += f;
      sq[0] += f * f;
    }
  }
#pragma unroll
  for (int j = 0; j < VW; ++j) {
    atomicAdd(&lds[chunk * VW + j], s[j]);
    atomicAdd(&lds[CT * VW + chunk * VW + j], sq[j]);
  }
  __syncthreads();
  float* part = sums + (long)blockIdx.x * 2 * C;
  const int base = blockIdx.y * CT * VW;
  for (int i = threadIdx.x; i < CT * VW; i += blockDim.x) {
    part[base + i] = lds[i];
    part[C + base + i] = lds[CT * VW + i];
  }
}

// ---- finalize fwd: one block PER CHANNEL reduces its stripe partials ----

__global__ void bn_finalize_fwd_kernel(const float* __restrict__ sums,
                                       const float* __restrict__ w,
                                       const float* __restrict__ b,
                                       float* __restrict__ rmean,
                                       float* __restrict__ rvar,
                                       float* __restrict__ save_mean,
                                       float* __restrict__ save_invstd,
                                       float* __restrict__ scale,
                                       float* __restrict__ shift, long R,
                                       int C, int nstripes, float momentum,
                                       float eps) {
  __shared__ float scratch[kBlock / kWave];
  const int c = blockIdx.x;
  float s1 = 0.f, s2 = 0.f;
  for (int st = threadIdx.x; st < nstripes; st += blockDim.x) {
    s1 += sums[(long)st * 2 * C + c];
    s2 += sums[(long)st * 2 * C + C + c];
  }
  auto add = [] __device__(float a, float b2) { return a + b2; };
  s1 = block_reduce(s1, scratch, 0.f, add);
  __syncthreads();
  s2 = block_reduce(s2, scratch, 0.f, add);
  if (threadIdx.x != 0) return;
  float mean = s1 / R;
  float var = fmaxf(s2 / R - mean * mean, 0.f);  // biased
  float invstd = rsqrtf(var + eps);
  save_mean[c] = mean;
  save_invstd[c] = invstd;
  // torch semantics: running stats use the UNBIASED variance
  float unbiased = (R > 1) ? var * (float)R / (float)(R - 1) : var;
  rmean[c] = (1.f - momentum) * rmean[c] + momentum * mean;
  rvar[c] = (1.f - momentum) * rvar[c] + momentum * unbiased;
  float sc = w[c] * invstd;
  scale[c] = sc;
  shift[c] = b[c] - mean * sc;
}

__global__ void bn_eval_coeffs_kernel(const float* __restrict__ w,
                                      const float* __restrict__ b,
                                      const float* __restrict__ rmean,
                                      const float* __restrict__ rvar,
                                      float* __restrict__ scale,
                                      float* __restrict__ shift, int C,
                                      float eps) {
  int c = blockIdx.x * blockDim.x + threadIdx.x;
  if (c >= C) return;
  float invstd = rsqrtf(rvar[c] + eps);
  float sc = w[c] * invstd;
  scale[c] = sc;
  shift[c] = b[c] - rmean[c] * sc;
}

// ---- normalize: y = x*scale[c] + shift[c] (+ReLU) ----------------------
// Same chunk/stripe geometry; scale/shift hoisted out of the row loop.

template <typename T, bool RELU, bool VEC, bool RES = false,
          bool WMASK = false>
__global__ void bn_norm_kernel(const T* __restrict__ x, T* __restrict__ y,
                               const float* __restrict__ scale,
                               const float* __restrict__ shift, long R, int C,
                               int CT, long rows_per_stripe,
                               const T* __restrict__ res = nullptr,
                               uint8_t* __restrict__ mask = nullptr) {
  using VT = VecTraits<T>;
  using Vec = typename VT::Vec;
  constexpr int VW = VEC ? VT::width : 1;
  const int chunk = threadIdx.x % CT;
  const int c0 = (blockIdx.y * CT + chunk) * VW;
  const int sub = threadIdx.x / CT;
  const int rows_grp = blockDim.x / CT;
  const long rbeg = (long)blockIdx.x * rows_per_stripe;
  const long rend = (rbeg + rows_per_stripe < R) ? rbeg + rows_per_stripe : R;
  float sc[VW], sh[VW];
#pragma unroll
  for (int j = 0; j < VW; ++j) {
    sc[j] = scale[c0 + j];
    sh[j] = shift[c0 + j];
  }
  for (long r = rbeg + sub; r < rend; r += rows_grp) {
    const long i = r * C + c0;
    if (VEC) {
      Vec xv = *reinterpret_cast<const Vec*>(x + i);
      Vec rv;
      if (RES) rv = *reinterpret_cast<const Vec*>(res + i);
      Vec yv;
      uint8_t mb = 0;
#pragma unroll
      for (int j = 0; j < VW; ++j) {
        float v = VT::load(xv, j) * sc[j] + sh[j];
        if (RES) v += VT::load(rv, j);
        if (WMASK && v > 0.f) mb |= (uint8_t)(1u << j);
        VT::store(yv, j, RELU ? fmaxf(v, 0.f) : v);
      }
      *reinterpret_cast<Vec*>(y + i) = yv;
      // relu bit-mask: 1 byte per VW-chunk — lets backward skip re-reading
      // y (saves ~2 full tensor passes per BN backward)
      if (WMASK) mask[r * (C / VW) + blockIdx.y * CT + chunk] = mb;
    } else {
      float v = VT::load1(x + i) * sc[0] + sh[0];
      if (RES) v += VT::load1(res + i);
      VT::store1(y + i, RELU ? fmaxf(v, 0.f) : v);
    }
  }
}

// ---- backward reduce: sum(dy), sum(dy * xhat) per channel --------------

template <typename T, bool RELU, bool VEC, bool MASKED = false>
__global__ void bn_reduce_bwd_kernel(const T* __restrict__ dy,
                                     const T* __restrict__ x,
                                     const T* __restrict__ y,
                                     const float* __restrict__ save_mean,
                                     const float* __restrict__ save_invstd,
                                     float* __restrict__ sums, long R, int C,
                                     int CT, long rows_per_stripe,
                                     const uint8_t* __restrict__ mask =
                                         nullptr) {
  using VT = VecTraits<T>;
  using Vec = typename VT::Vec;
  constexpr int VW = VEC ? VT::width : 1;
  extern __shared__ float lds[];
  for (int i = threadIdx.x; i < 2 * CT * VW; i += blockDim.x) lds[i] = 0.f;
  __syncthreads();

  const int chunk = threadIdx.x % CT;
  const int c0 = (blockIdx.y * CT + chunk) * VW;
  const int sub = threadIdx.x / CT;
  const int rows_grp = blockDim.x / CT;
  const long rbeg = (long)blockIdx.x * rows_per_stripe;
  const long rend = (rbeg + rows_per_stripe < R) ? rbeg + rows_per_stripe : R;
  float mean[VW], invstd[VW];
#pragma unroll
  for (int j = 0; j < VW; ++j) {
    mean[j] = save_mean[c0 + j];
    invstd[j] = save_invstd[c0 + j];
  }
  float a[VW] = {}, bx[VW] = {};
  for (long r = rbeg + sub; r < rend; r += rows_grp) {
    const long i = r * C + c0;
    if (VEC) {
      Vec gv = *reinterpret_cast<const Vec*>(dy + i);
      Vec xv = *reinterpret_cast<const Vec*>(x + i);
      Vec yv;
      uint8_t mb = 0xff;
      if (MASKED) mb = mask[r * (C / VW) + blockIdx.y * CT + chunk];
      else if (RELU) yv = *reinterpret_cast<const Vec*>(y + i);
#pragma unroll
      for (int j = 0; j < VW; ++j) {
        float g = VT::load(gv, j);
        if (MASKED) {
          if (!(mb & (1u << j))) g = 0.f;
        } else if (RELU && VT::load(yv, j) <= 0.f) {
          g = 0.f;
        }
        float xh = (VT::load(xv, j) - mean[j]) * invstd[j];
        a[j] += g;
        bx[j] += g * xh;
      }
    } else {
      float g = VT::load1(dy + i);
      if (RELU && VT::load1(y + i) <= 0.f) g = 0.f;
      float xh = (VT::load1(x + i) - mean[0]) * invstd[0];
      a[0] += g;
      bx[0] += g * xh;
    }
  }
#pragma unroll
  for (int j = 0; j < VW; ++j) {
    atomicAdd(&lds[chunk * VW + j], a[j]);
    atomicAdd(&lds[CT * VW + chunk * VW + j], bx[j]);
  }
  __syncthreads();
  float* part = sums + (long)blockIdx.x * 2 * C;
  const int base = blockIdx.y * CT * VW;
  for (int i = threadIdx.x; i < CT * VW; i += blockDim.x) {
    part[base + i] = lds[i];
    part[C + base + i] = lds[CT * VW + i];
  }
}

// ---- finalize bwd: per-channel block; emits dgamma/dbeta and the three
// dx coefficients:  dx = cA[c]*dy_eff + cB[c]*x + cD[c]
//   (expansion of gi*(dy_eff - k1 - k2*(x-mean)*invstd))

__global__ void bn_finalize_bwd_kernel(const float* __restrict__ sums,
                                       const float* __restrict__ w,
                                       const float* __restrict__ save_mean,
                                       const float* __restrict__ save_invstd,
                                       float* __restrict__ dgamma,
                                       float* __restrict__ dbeta,
                                       float* __restrict__ cA,
                                       float* __restrict__ cB,
                                       float* __restrict__ cD, long R, int C,
                                       int nstripes) {
  __shared__ float scratch[kBlock / kWave];
  const int c = blockIdx.x;
  float s1 = 0.f, s2 = 0.f;
  for (int st = threadIdx.x; st < nstripes; st += blockDim.x) {
    s1 += sums[(long)st * 2 * C + c];
    s2 += sums[(long)st * 2 * C + C + c];
  }
  auto add = [] __device__(float a, float b2) { return a + b2; };
  s1 = block_reduce(s1, scratch, 0.f, add);
  __syncthreads();
  s2 = block_reduce(s2, scratch, 0.f, add);
  if (threadIdx.x != 0) return;
  dbeta[c] = s1;
  dgamma[c] = s2;
  const float invstd = save_invstd[c];
  const float gi = w[c] * invstd;
  const float k1 = s1 / R, k2 = s2 / R;
  cA[c] = gi;
  cB[c] = -gi * k2 * invstd;
  cD[c] = gi * (k2 * invstd * save_mean[c] - k1);
}

// ---- dx pass: dx = cA*dy_eff + cB*x + cD (coefficients hoisted) --------

template <typename T, bool RELU, bool VEC, bool DRES = false,
          bool MASKED = false>
__global__ void bn_dx_kernel(const T* __restrict__ dy, const T* __restrict__ x,
                             const T* __restrict__ y,
                             const float* __restrict__ cA,
                             const float* __restrict__ cB,
                             const float* __restrict__ cD, T* __restrict__ dx,
                             long R, int C, int CT, long rows_per_stripe,
                             T* __restrict__ dres = nullptr,
                             const uint8_t* __restrict__ mask = nullptr) {
  using VT = VecTraits<T>;
  using Vec = typename VT::Vec;
  constexpr int VW = VEC ? VT::width : 1;
  const int chunk = threadIdx.x % CT;
  const int c0 = (blockIdx.y * CT + chunk) * VW;
  const int sub = threadIdx.x / CT;
  const int rows_grp = blockDim.x / CT;
  const long rbeg = (long)blockIdx.x * rows_per_stripe;
  const long rend = (rbeg + rows_per_stripe < R) ? rbeg + rows_per_stripe : R;
  float A[VW], B[VW], D[VW];
#pragma unroll
  for (int j = 0; j < VW; ++j) {
    A[j] = cA[c0 + j];
    B[j] = cB[c0 + j];
    D[j] = cD[c0 + j];
  }
  for (long r = rbeg + sub; r < rend; r += rows_grp) {
    const long i = r * C + c0;
    if (VEC) {
      Vec gv = *reinterpret_cast<const Vec*>(dy + i);
      Vec xv = *reinterpret_cast<const Vec*>(x + i);
      Vec yv;
      uint8_t mb = 0xff;
      if (MASKED) mb = mask[r * (C / VW) + blockIdx.y * CT + chunk];
      else if (RELU) yv = *reinterpret_cast<const Vec*>(y + i);
      Vec ov, rv;
#pragma unroll
      for (int j = 0; j < VW; ++j) {
        float g = VT::load(gv, j);
        if (MASKED) {
          if (!(mb & (1u << j))) g = 0.f;
        } else if (RELU && VT::load(yv, j) <= 0.f) {
          g = 0.f;
        }
        if (DRES) VT::store(rv, j, g);
        VT::store(ov, j, A[j] * g + B[j] * VT::load(xv, j) + D[j]);
      }
      *reinterpret_cast<Vec*>(dx + i) = ov;
      if (DRES) *reinterpret_cast<Vec*>(dres + i) = rv;
    } else {
      float g = VT::load1(dy + i);
      if (RELU && VT::load1(y + i) <= 0.f) g = 0.f;
      if (DRES) VT::store1(dres + i, g);
      VT::store1(dx + i, A[0] * g + B[0] * VT::load1(x + i) + D[0]);
    }
  }
}

// ---- host-side helpers --------------------------------------------------

template <typename T>
static const T* tp(const at::Tensor& t) {
  return reinterpret_cast<const T*>(t.data_ptr());
}
template <typename T>
static T* tpm(at::Tensor& t) {
  return reinterpret_cast<T*>(t.data_ptr());
}

}  // namespace

std::tuple<at::Tensor, at::Tensor, at::Tensor, at::Tensor>
bn_fwd_train_impl(
    const at::Tensor& x_in, const at::Tensor& w, const at::Tensor& b,
    at::Tensor& rmean, at::Tensor& rvar, double momentum, double eps,
    bool relu) {
  auto x = x_in.contiguous(at::MemoryFormat::ChannelsLast);
  TORCH_CHECK(x.dim() == 4, "bn: 4D NHWC expected");
  TORCH_CHECK(w.scalar_type() == at::kFloat, "bn: params must be fp32");
  auto s = at::cuda::getCurrentCUDAStream();
  const bool bf = x.scalar_type() == at::kBFloat16;
  TORCH_CHECK(bf || x.scalar_type() == at::kFloat, "bn: bf16/fp32 only");
  auto sh = bn_shape(x, bf ? 8 : 4);
  auto fopts = x.options().dtype(at::kFloat);
  auto sums = at::empty({(long)sh.stripes * 2 * sh.C}, fopts);
  auto save_mean = at::empty({sh.C}, fopts);
  auto save_invstd = at::empty({sh.C}, fopts);
  auto scale = at::empty({sh.C}, fopts);
  auto shift = at::empty({sh.C}, fopts);
  auto y = at::empty_like(x);
  // relu bit-mask (1 byte per VW-chunk) — backward then never re-reads y
  const bool wmask = relu && sh.vec;
  const int vw = bf ? 8 : 4;
  auto mask = wmask ? at::empty({sh.R * (sh.C / vw)},
                                x.options().dtype(at::kByte))
                    : at::empty({0}, x.options().dtype(at::kByte));

#define DLX_BN_T(T, VEC)                                                     \
  {                                                                          \
    bn_reduce_fwd_kernel<T, VEC><<<sh.grid, kBlock, sh.lds, s>>>(            \
        tp<T>(x), sums.data_ptr<float>(), sh.R, sh.C, sh.CT,                 \
        sh.rows_per_stripe);                                                 \
    bn_finalize_fwd_kernel<<<sh.C, kBlock, 0, s>>>(                          \
        sums.data_ptr<float>(), w.data_ptr<float>(), b.data_ptr<float>(),    \
        rmean.data_ptr<float>(), rvar.data_ptr<float>(),                     \
        save_mean.data_ptr<float>(), save_invstd.data_ptr<float>(),          \
        scale.data_ptr<float>(), shift.data_ptr<float>(), sh.R, sh.C,        \
        sh.stripes, (float)momentum, (float)eps);                            \
    if (relu && VEC)                                                         \
      bn_norm_kernel<T, true, VEC, false, true>                              \
          <<<sh.grid, kBlock, 0, s>>>(                                       \
              tp<T>(x), tpm<T>(y), scale.data_ptr<float>(),                  \
              shift.data_ptr<float>(), sh.R, sh.C, sh.CT,                    \
              sh.rows_per_stripe, nullptr, mask.data_ptr<uint8_t>());        \
    else if (relu)                                                           \
      bn_norm_kernel<T, true, VEC><<<sh.grid, kBlock, 0, s>>>(               \
          tp<T>(x), tpm<T>(y), scale.data_ptr<float>(),                      \
          shift.data_ptr<float>(), sh.R, sh.C, sh.CT, sh.rows_per_stripe);   \
    else                                                                     \
      bn_norm_kernel<T, false, VEC><<<sh.grid, kBlock, 0, s>>>(              \
          tp<T>(x), tpm<T>(y), scale.data_ptr<float>(),                      \
          shift.data_ptr<float>(), sh.R, sh.C, sh.CT, sh.rows_per_stripe);   \
  }

  if (bf && sh.vec) { DLX_BN_T(uint16_t, true) }
  else if (bf) { DLX_BN_T(uint16_t, false) }
  else if (sh.vec) { DLX_BN_T(float, true) }
  else { DLX_BN_T(float, false) }
#undef DLX_BN_T
  return {y, save_mean, save_invstd, mask};
}

at::Tensor bn_fwd_eval_impl(const at::Tensor& x_in, const at::Tensor& w,
                            const at::Tensor& b, const at::Tensor& rmean,
                            const at::Tensor& rvar, double eps, bool relu) {
  auto x = x_in.contiguous(at::MemoryFormat::ChannelsLast);
  auto s = at::cuda::getCurrentCUDAStream();
  const bool bf = x.scalar_type() == at::kBFloat16;
  auto sh = bn_shape(x, bf ? 8 : 4);
  auto fopts = x.options().dtype(at::kFloat);
  auto scale = at::empty({sh.C}, fopts);
  auto shift = at::empty({sh.C}, fopts);
  bn_eval_coeffs_kernel<<<(sh.C + 255) / 256, 256, 0, s>>>(
      w.data_ptr<float>(), b.data_ptr<float>(), rmean.data_ptr<float>(),
      rvar.data_ptr<float>(), scale.data_ptr<float>(), shift.data_ptr<float>(),
      sh.C, (float)eps);
  auto y = at::empty_like(x);

#define DLX_BN_EVAL(T, VEC)                                                  \
  {                                                                          \
    if (relu)                                                                \
      bn_norm_kernel<T, true, VEC><<<sh.grid, kBlock, 0, s>>>(               \
          tp<T>(x), tpm<T>(y), scale.data_ptr<float>(),                      \
          shift.data_ptr<float>(), sh.R, sh.C, sh.CT, sh.rows_per_stripe);   \
    else                                                                     \
      bn_norm_kernel<T, false, VEC><<<sh.grid, kBlock, 0, s>>>(              \
          tp<T>(x), tpm<T>(y), scale.data_ptr<float>(),                      \
          shift.data_ptr<float>(), sh.R, sh.C, sh.CT, sh.rows_per_stripe);   \
  }
  if (bf && sh.vec) { DLX_BN_EVAL(uint16_t, true) }
  else if (bf) { DLX_BN_EVAL(uint16_t, false) }
  else if (sh.vec) { DLX_BN_EVAL(float, true) }
  else { DLX_BN_EVAL(float, false) }
#undef DLX_BN_EVAL
  return y;
}

std::tuple<at::Tensor, at::Tensor, at::Tensor> bn_bwd_impl(
    const at::Tensor& dy_in, const at::Tensor& x_in, const at::Tensor& w,
    const at::Tensor& save_mean, const at::Tensor& save_invstd,
    const at::Tensor& y_in, bool relu) {
  auto x = x_in.contiguous(at::MemoryFormat::ChannelsLast);
  auto dy = dy_in.contiguous(at::MemoryFormat::ChannelsLast);
  auto y = y_in.contiguous(at::MemoryFormat::ChannelsLast);
  auto s = at::cuda::getCurrentCUDAStream();
  const bool bf = x.scalar_type() == at::kBFloat16;
  auto sh = bn_shape(x, bf ? 8 : 4);
  auto fopts = x.options().dtype(at::kFloat);
  auto sums = at::empty({(long)sh.stripes * 2 * sh.C}, fopts);
  auto dgamma = at::empty({sh.C}, fopts);
  auto dbeta = at::empty({sh.C}, fopts);
  auto cA = at::empty({sh.C}, fopts);
  auto cB = at::empty({sh.C}, fopts);
  auto cD = at::empty({sh.C}, fopts);
  auto dx = at::empty_like(x);

#define DLX_BN_BWD(T, RELU, VEC)                                             \
  {                                                                          \
    bn_reduce_bwd_kernel<T, RELU, VEC><<<sh.grid, kBlock, sh.lds, s>>>(      \
        tp<T>(dy), tp<T>(x), tp<T>(y), save_mean.data_ptr<float>(),          \
        save_invstd.data_ptr<float>(), sums.data_ptr<float>(), sh.R, sh.C,   \
        sh.CT, sh.rows_per_stripe);                                          \
    bn_finalize_bwd_kernel<<<sh.C, kBlock, 0, s>>>(                          \
        sums.data_ptr<float>(), w.data_ptr<float>(),                         \
        save_mean.data_ptr<float>(), save_invstd.data_ptr<float>(),          \
        dgamma.data_ptr<float>(), dbeta.data_ptr<float>(),                   \
        cA.data_ptr<float>(), cB.data_ptr<float>(), cD.data_ptr<float>(),    \
        sh.R, sh.C, sh.stripes);                                             \
    bn_dx_kernel<T, RELU, VEC><<<sh.grid, kBlock, 0, s>>>(                   \
        tp<T>(dy), tp<T>(x), tp<T>(y), cA.data_ptr<float>(),                 \
        cB.data_ptr<float>(), cD.data_ptr<float>(), tpm<T>(dx), sh.R, sh.C,  \
        sh.CT, sh.rows_per_stripe);                                          \
  }
  if (bf && sh.vec && relu) { DLX_BN_BWD(uint16_t, true, true) }
  else if (bf && sh.vec) { DLX_BN_BWD(uint16_t, false, true) }
  else if (bf && relu) { DLX_BN_BWD(uint16_t, true, false) }
  else if (bf) { DLX_BN_BWD(uint16_t, false, false) }
  else if (sh.vec && relu) { DLX_BN_BWD(float, true, true) }
  else if (sh.vec) { DLX_BN_BWD(float, false, true) }
  else if (relu) { DLX_BN_BWD(float, true, false) }
  else { DLX_BN_BWD(float, false, false) }
#undef DLX_BN_BWD
  return {dx, dgamma, dbeta};
}

// ---- fused BN + residual add + ReLU (ResNet block tail) -----------------
// fwd: y = relu(x*scale + shift + residual); bwd additionally returns
// dres = relu-masked dy in the same dx pass (one fewer elementwise pass
// and two fewer kernels than bn + add_relu).

std::tuple<at::Tensor, at::Tensor, at::Tensor, at::Tensor>
bn_fwd_train_res_impl(
    const at::Tensor& x_in, const at::Tensor& w, const at::Tensor& b,
    at::Tensor& rmean, at::Tensor& rvar, double momentum, double eps,
    const at::Tensor& res_in) {
  auto x = x_in.contiguous(at::MemoryFormat::ChannelsLast);
  auto res = res_in.contiguous(at::MemoryFormat::ChannelsLast);
  TORCH_CHECK(res.sizes() == x.sizes() &&
              res.scalar_type() == x.scalar_type(),
              "bn_res: residual must match x in shape and dtype");
  auto s = at::cuda::getCurrentCUDAStream();
  const bool bf = x.scalar_type() == at::kBFloat16;
  auto sh = bn_shape(x, bf ? 8 : 4);
  TORCH_CHECK(sh.vec, "bn_res: C must be a multiple of the vector width");
  auto fopts = x.options().dtype(at::kFloat);
  auto sums = at::empty({(long)sh.stripes * 2 * sh.C}, fopts);
  auto save_mean = at::empty({sh.C}, fopts);
  auto save_invstd = at::empty({sh.C}, fopts);
  auto scale = at::empty({sh.C}, fopts);
  auto shift = at::empty({sh.C}, fopts);
  auto y = at::empty_like(x);
  const int vw = bf ? 8 : 4;
  auto mask = at::empty({sh.R * (sh.C / vw)},
                        x.options().dtype(at::kByte));

#define DLX_BN_RES(T)                                                        \
  {                                                                          \
    bn_reduce_fwd_kernel<T, true><<<sh.grid, kBlock, sh.lds, s>>>(           \
        tp<T>(x), sums.data_ptr<float>(), sh.R, sh.C, sh.CT,                 \
        sh.rows_per_stripe);                                                 \
    bn_finalize_fwd_kernel<<<sh.C, kBlock, 0, s>>>(                          \
        sums.data_ptr<float>(), w.data_ptr<float>(), b.data_ptr<float>(),    \
        rmean.data_ptr<float>(), rvar.data_ptr<float>(),                     \
        save_mean.data_ptr<float>(), save_invstd.data_ptr<float>(),          \
        scale.data_ptr<float>(), shift.data_ptr<float>(), sh.R, sh.C,        \
        sh.stripes, (float)momentum, (float)eps);                            \
    bn_norm_kernel<T, true, true, true, true><<<sh.grid, kBlock, 0, s>>>(    \
        tp<T>(x), tpm<T>(y), scale.data_ptr<float>(),                        \
        shift.data_ptr<float>(), sh.R, sh.C, sh.CT, sh.rows_per_stripe,     \
        tp<T>(res), mask.data_ptr<uint8_t>());                               \
  }
  if (bf) { DLX_BN_RES(uint16_t) } else { DLX_BN_RES(float) }
#undef DLX_BN_RES
  return {y, save_mean, save_invstd, mask};
}

at::Tensor bn_fwd_eval_res_impl(const at::Tensor& x_in, const at::Tensor& w,
                                const at::Tensor& b, const at::Tensor& rmean,
                                const at::Tensor& rvar, double eps,
                                const at::Tensor& res_in) {
  auto x = x_in.contiguous(at::MemoryFormat::ChannelsLast);
  auto res = res_in.contiguous(at::MemoryFormat::ChannelsLast);
  TORCH_CHECK(res.sizes() == x.sizes() &&
              res.scalar_type() == x.scalar_type(),
              "bn_res: residual must match x in shape and dtype");
  auto s = at::cuda::getCurrentCUDAStream();
  const bool bf = x.scalar_type() == at::kBFloat16;
  auto sh = bn_shape(x, bf ? 8 : 4);
  TORCH_CHECK(sh.vec, "bn_res: C must be a multiple of the vector width");
  auto fopts = x.options().dtype(at::kFloat);
  auto scale = at::empty({sh.C}, fopts);
  auto shift = at::empty({sh.C}, fopts);
  bn_eval_coeffs_kernel<<<(sh.C + 255) / 256, 256, 0, s>>>(
      w.data_ptr<float>(), b.data_ptr<float>(), rmean.data_ptr<float>(),
      rvar.data_ptr<float>(), scale.data_ptr<float>(), shift.data_ptr<float>(),
      sh.C, (float)eps);
  auto y = at::empty_like(x);
  if (bf)
    bn_norm_kernel<uint16_t, true, true, true><<<sh.grid, kBlock, 0, s>>>(
        tp<uint16_t>(x), tpm<uint16_t>(y), scale.data_ptr<float>(),
        shift.data_ptr<float>(), sh.R, sh.C, sh.CT, sh.rows_per_stripe,
        tp<uint16_t>(res));
  else
    bn_norm_kernel<float, true, true, true><<<sh.grid, kBlock, 0, s>>>(
        tp<float>(x), tpm<float>(y), scale.data_ptr<float>(),
        shift.data_ptr<float>(), sh.R, sh.C, sh.CT, sh.rows_per_stripe,
        tp<float>(res));
  return y;
}

std::tuple<at::Tensor, at::Tensor, at::Tensor, at::Tensor> bn_bwd_res_impl(
    const at::Tensor& dy_in, const at::Tensor& x_in, const at::Tensor& w,
    const at::Tensor& save_mean, const at::Tensor& save_invstd,
    const at::Tensor& y_in) {
  auto x = x_in.contiguous(at::MemoryFormat::ChannelsLast);
  auto dy = dy_in.contiguous(at::MemoryFormat::ChannelsLast);
  auto y = y_in.contiguous(at::MemoryFormat::ChannelsLast);
  auto s = at::cuda::getCurrentCUDAStream();
  const bool bf = x.scalar_type() == at::kBFloat16;
  auto sh = bn_shape(x, bf ? 8 : 4);
  TORCH_CHECK(sh.vec, "bn_res: C must be a multiple of the vector width");
  auto fopts = x.options().dtype(at::kFloat);
  auto sums = at::empty({(long)sh.stripes * 2 * sh.C}, fopts);
  auto dgamma = at::empty({sh.C}, fopts);
  auto dbeta = at::empty({sh.C}, fopts);
  auto cA = at::empty({sh.C}, fopts);
  auto cB = at::empty({sh.C}, fopts);
  auto cD = at::empty({sh.C}, fopts);
  auto dx = at::empty_like(x);
  auto dres = at::empty_like(x);

#define DLX_BN_BWDR(T)                                                       \
  {                                                                          \
    bn_reduce_bwd_kernel<T, true, true><<<sh.grid, kBlock, sh.lds, s>>>(     \
        tp<T>(dy), tp<T>(x), tp<T>(y), save_mean.data_ptr<float>(),          \
        save_invstd.data_ptr<float>(), sums.data_ptr<float>(), sh.R, sh.C,   \
        sh.CT, sh.rows_per_stripe);                                          \
    bn_finalize_bwd_kernel<<<sh.C, kBlock, 0, s>>>(                          \
        sums.data_ptr<float>(), w.data_ptr<float>(),                         \
        save_mean.data_ptr<float>(), save_invstd.data_ptr<float>(),          \
        dgamma.data_ptr<float>(), dbeta.data_ptr<float>(),                   \
        cA.data_ptr<float>(), cB.data_ptr<float>(), cD.data_ptr<float>(),    \
        sh.R, sh.C, sh.stripes);                                             \
    bn_dx_kernel<T, true, true, true><<<sh.grid, kBlock, 0, s>>>(            \
        tp<T>(dy), tp<T>(x), tp<T>(y), cA.data_ptr<float>(),                 \
        cB.data_ptr<float>(), cD.data_ptr<float>(), tpm<T>(dx), sh.R, sh.C,  \
        sh.CT, sh.rows_per_stripe, tpm<T>(dres));                            \
  }
  if (bf) { DLX_BN_BWDR(uint16_t) } else { DLX_BN_BWDR(float) }
#undef DLX_BN_BWDR
  return {dx, dgamma, dbeta, dres};
}

// ---- masked backward (relu mask from the fwd bit-mask; y never read) ---

std::tuple<at::Tensor, at::Tensor, at::Tensor> bn_bwd_m_impl(
    const at::Tensor& dy_in, const at::Tensor& x_in, const at::Tensor& w,
    const at::Tensor& save_mean, const at::Tensor& save_invstd,
    const at::Tensor& mask) {
  auto x = x_in.contiguous(at::MemoryFormat::ChannelsLast);
  auto dy = dy_in.contiguous(at::MemoryFormat::ChannelsLast);
  auto s = at::cuda::getCurrentCUDAStream();
  const bool bf = x.scalar_type() == at::kBFloat16;
  auto sh = bn_shape(x, bf ? 8 : 4);
  TORCH_CHECK(sh.vec && mask.numel() == sh.R * (sh.C / (bf ? 8 : 4)),
              "bn_bwd_m: mask/shape mismatch");
  auto fopts = x.options().dtype(at::kFloat);
  auto sums = at::empty({(long)sh.stripes * 2 * sh.C}, fopts);
  auto dgamma = at::empty({sh.C}, fopts);
  auto dbeta = at::empty({sh.C}, fopts);
  auto cA = at::empty({sh.C}, fopts);
  auto cB = at::empty({sh.C}, fopts);
  auto cD = at::empty({sh.C}, fopts);
  auto dx = at::empty_like(x);

#define DLX_BN_BWDM(T)                                                       \
  {                                                                          \
    bn_reduce_bwd_kernel<T, true, true, true>                                \
        <<<sh.grid, kBlock, sh.lds, s>>>(                                    \
            tp<T>(dy), tp<T>(x), nullptr, save_mean.data_ptr<float>(),       \
            save_invstd.data_ptr<float>(), sums.data_ptr<float>(), sh.R,     \
            sh.C, sh.CT, sh.rows_per_stripe, mask.data_ptr<uint8_t>());      \
    bn_finalize_bwd_kernel<<<sh.C, kBlock, 0, s>>>(                          \
        sums.data_ptr<float>(), w.data_ptr<float>(),                         \
        save_mean.data_ptr<float>(), save_invstd.data_ptr<float>(),          \
        dgamma.data_ptr<float>(), dbeta.data_ptr<float>(),                   \
        cA.data_ptr<float>(), cB.data_ptr<float>(), cD.data_ptr<float>(),    \
        sh.R, sh.C, sh.stripes);                                             \
    bn_dx_kernel<T, true, true, false, true><<<sh.grid, kBlock, 0, s>>>(     \
        tp<T>(dy), tp<T>(x), nullptr, cA.data_ptr<float>(),                  \
        cB.data_ptr<float>(), cD.data_ptr<float>(), tpm<T>(dx), sh.R, sh.C,  \
        sh.CT, sh.rows_per_stripe, nullptr, mask.data_ptr<uint8_t>());       \
  }
  if (bf) { DLX_BN_BWDM(uint16_t) } else { DLX_BN_BWDM(float) }
#undef DLX_BN_BWDM
  return {dx, dgamma, dbeta};
}

std::tuple<at::Tensor, at::Tensor, at::Tensor, at::Tensor> bn_bwd_res_m_impl(
    const at::Tensor& dy_in, const at::Tensor& x_in, const at::Tensor& w,
    const at::Tensor& save_mean, const at::Tensor& save_invstd,
    const at::Tensor& mask) {
  auto x = x_in.contiguous(at::MemoryFormat::ChannelsLast);
  auto dy = dy_in.contiguous(at::MemoryFormat::ChannelsLast);
  auto s = at::cuda::getCurrentCUDAStream();
  const bool bf = x.scalar_type() == at::kBFloat16;
  auto sh = bn_shape(x, bf ? 8 : 4);
  TORCH_CHECK(sh.vec && mask.numel() == sh.R * (sh.C / (bf ? 8 : 4)),
              "bn_bwd_res_m: mask/shape mismatch");
  auto fopts = x.options().dtype(at::kFloat);
  auto sums = at::empty({(long)sh.stripes * 2 * sh.C}, fopts);
  auto dgamma = at::empty({sh.C}, fopts);
  auto dbeta = at::empty({sh.C}, fopts);
  auto cA = at::empty({sh.C}, fopts);
  auto cB = at::empty({sh.C}, fopts);
  auto cD = at::empty({sh.C}, fopts);
  auto dx = at::empty_like(x);
  auto dres = at::empty_like(x);

#define DLX_BN_BWDRM(T)                                                      \
  {                                                                          \
    bn_reduce_bwd_kernel<T, true, true, true>                                \
        <<<sh.grid, kBlock, sh.lds, s>>>(                                    \
            tp<T>(dy), tp<T>(x), nullptr, save_mean.data_ptr<float>(),       \
            save_invstd.data_ptr<float>(), sums.data_ptr<float>(), sh.R,     \
            sh.C, sh.CT, sh.rows_per_stripe, mask.data_ptr<uint8_t>());      \
    bn_finalize_bwd_kernel<<<sh.C, kBlock, 0, s>>>(                          \
        sums.data_ptr<float>(), w.data_ptr<float>(),                         \
        save_mean.data_ptr<float>(), save_invstd.data_ptr<float>(),          \
        dgamma.data_ptr<float>(), dbeta.data_ptr<float>(),                   \
        cA.data_ptr<float>(), cB.data_ptr<float>(), cD.data_ptr<float>(),    \
        sh.R, sh.C, sh.stripes);                                             \
    bn_dx_kernel<T, true, true, true, true><<<sh.grid, kBlock, 0, s>>>(      \
        tp<T>(dy), tp<T>(x), nullptr, cA.data_ptr<float>(),                  \
        cB.data_ptr<float>(), cD.data_ptr<float>(), tpm<T>(dx), sh.R, sh.C,  \
        sh.CT, sh.rows_per_stripe, tpm<T>(dres),                             \
        mask.data_ptr<uint8_t>());                                           \
  }
  if (bf) { DLX_BN_BWDRM(uint16_t) } else { DLX_BN_BWDRM(float) }
#undef DLX_BN_BWDRM
  return {dx, dgamma, dbeta, dres};
}

// ---- presum forward: channel sums come from the conv epilogue ---------
// (conv2d_fwd_stats slab [stripes][2][C]); only finalize + normalize run.

std::tuple<at::Tensor, at::Tensor, at::Tensor, at::Tensor>
bn_fwd_train_presum_impl(
    const at::Tensor& x_in, const at::Tensor& w, const at::Tensor& b,
    at::Tensor& rmean, at::Tensor& rvar, double momentum, double eps,
    bool relu, const at::Tensor& sums) {
  auto x = x_in.contiguous(at::MemoryFormat::ChannelsLast);
  auto s = at::cuda::getCurrentCUDAStream();
  const bool bf = x.scalar_type() == at::kBFloat16;
  auto sh = bn_shape(x, bf ? 8 : 4);
  TORCH_CHECK(sh.vec, "bn presum: C % vector width != 0");
  const int nstripes = (int)(sums.numel() / (2 * sh.C));
  TORCH_CHECK((long)nstripes * 2 * sh.C == sums.numel(),
              "bn presum: sums size mismatch");
  auto fopts = x.options().dtype(at::kFloat);
  auto save_mean = at::empty({sh.C}, fopts);
  auto save_invstd = at::empty({sh.C}, fopts);
  auto scale = at::empty({sh.C}, fopts);
  auto shift = at::empty({sh.C}, fopts);
  auto y = at::empty_like(x);
  const int vw = bf ? 8 : 4;
  auto mask = relu ? at::empty({sh.R * (sh.C / vw)},
                               x.options().dtype(at::kByte))
                   : at::empty({0}, x.options().dtype(at::kByte));

#define DLX_BN_PRE(T)                                                        \
  {                                                                          \
    bn_finalize_fwd_kernel<<<sh.C, kBlock, 0, s>>>(                          \
        sums.data_ptr<float>(), w.data_ptr<float>(), b.data_ptr<float>(),    \
        rmean.data_ptr<float>(), rvar.data_ptr<float>(),                     \
        save_mean.data_ptr<float>(), save_invstd.data_ptr<float>(),          \
        scale.data_ptr<float>(), shift.data_ptr<float>(), sh.R, sh.C,        \
        nstripes, (float)momentum, (float)eps);                              \
    if (relu)                                                                \
      bn_norm_kernel<T, true, true, false, true>                             \
          <<<sh.grid, kBlock, 0, s>>>(                                       \
              tp<T>(x), tpm<T>(y), scale.data_ptr<float>(),                  \
              shift.data_ptr<float>(), sh.R, sh.C, sh.CT,                    \
              sh.rows_per_stripe, nullptr, mask.data_ptr<uint8_t>());        \
    else                                                                     \
      bn_norm_kernel<T, false, true><<<sh.grid, kBlock, 0, s>>>(             \
          tp<T>(x), tpm<T>(y), scale.data_ptr<float>(),                      \
          shift.data_ptr<float>(), sh.R, sh.C, sh.CT, sh.rows_per_stripe);   \
  }
  if (bf) { DLX_BN_PRE(uint16_t) } else { DLX_BN_PRE(float) }
#undef DLX_BN_PRE
  return {y, save_mean, save_invstd, mask};
}

std::tuple<at::Tensor, at::Tensor, at::Tensor, at::Tensor>
bn_fwd_train_res_presum_impl(
    const at::Tensor& x_in, const at::Tensor& w, const at::Tensor& b,
    at::Tensor& rmean, at::Tensor& rvar, double momentum, double eps,
    const at::Tensor& res_in, const at::Tensor& sums) {
  auto x = x_in.contiguous(at::MemoryFormat::ChannelsLast);
  auto res = res_in.contiguous(at::MemoryFormat::ChannelsLast);
  TORCH_CHECK(res.sizes() == x.sizes() &&
              res.scalar_type() == x.scalar_type(),
              "bn res presum: residual must match x in shape and dtype");
  auto s = at::cuda::getCurrentCUDAStream();
  const bool bf = x.scalar_type() == at::kBFloat16;
  auto sh = bn_shape(x, bf ? 8 : 4);
  TORCH_CHECK(sh.vec, "bn presum: C % vector width != 0");
  const int nstripes = (int)(sums.numel() / (2 * sh.C));
  TORCH_CHECK((long)nstripes * 2 * sh.C == sums.numel(),
              "bn presum: sums size mismatch");
  auto fopts = x.options().dtype(at::kFloat);
  auto save_mean = at::empty({sh.C}, fopts);
  auto save_invstd = at::empty({sh.C}, fopts);
  auto scale = at::empty({sh.C}, fopts);
  auto shift = at::empty({sh.C}, fopts);
  auto y = at::empty_like(x);
  const int vw = bf ? 8 : 4;
  auto mask = at::empty({sh.R * (sh.C / vw)}, x.options().dtype(at::kByte));

#define DLX_BN_PRER(T)                                                       \
  {                                                                          \
    bn_finalize_fwd_kernel<<<sh.C, kBlock, 0, s>>>(                          \
        sums.data_ptr<float>(), w.data_ptr<float>(), b.data_ptr<float>(),    \
        rmean.data_ptr<float>(), rvar.data_ptr<float>(),                     \
        save_mean.data_ptr<float>(), save_invstd.data_ptr<float>(),          \
        scale.data_ptr<float>(), shift.data_ptr<float>(), sh.R, sh.C,        \
        nstripes, (float)momentum, (float)eps);                              \
    bn_norm_kernel<T, true, true, true, true><<<sh.grid, kBlock, 0, s>>>(    \
        tp<T>(x), tpm<T>(y), scale.data_ptr<float>(),                        \
        shift.data_ptr<float>(), sh.R, sh.C, sh.CT, sh.rows_per_stripe,     \
        tp<T>(res), mask.data_ptr<uint8_t>());                               \
  }
  if (bf) { DLX_BN_PRER(uint16_t) } else { DLX_BN_PRER(float) }
#undef DLX_BN_PRER
  return {y, save_mean, save_invstd, mask};
}

}  // namespace dlx
