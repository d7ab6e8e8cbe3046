// NHWC BatchNorm (training fwd / eval fwd / bwd), fused with ReLU.
//
// Replaces cuDNN BatchNorm + the adjacent framework ReLU (SURVEY.md §3.2
// rows "cuDNN BatchNorm", "framework elementwise").  Design:
//  * x is channels-last ([R=N*H*W, C] with C contiguous) — the native
//    layout of the conv path; statistics accumulate in fp32;
//  * reduce kernel: each block owns a (channel-tile, row-stripe); LDS
//    per-channel partials, one global fp32 atomicAdd per channel/block;
//  * normalize / dx kernels: vectorized 16 B/lane when C % width == 0;
//  * ReLU is fused: fwd applies max(0,·); bwd masks dy by (y > 0) before
//    the reductions (y is the saved post-activation output).
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

// ---- reduction: sum(x), sum(x^2) per channel ---------------------------

// Vectorized variant (C % VW == 0): each thread owns a VW-channel chunk
// (16-B loads, guide G13); CT counts CHUNKS here.
template <typename T>
__global__ void bn_reduce_fwd_vec_kernel(const T* __restrict__ x,
                                         float* __restrict__ sums,  // [2C]
                                         long R, int C, int CT) {
  using VT = VecTraits<T>;
  using Vec = typename VT::Vec;
  constexpr int VW = VT::width;
  extern __shared__ float lds[];          // [2*CT*VW]
  for (int i = threadIdx.x; i < 2 * CT * VW; i += blockDim.x) lds[i] = 0.f;
  __syncthreads();

  const int chunk = threadIdx.x % CT;
  const int c0 = blockIdx.y * CT * VW + chunk * VW;
  const int sub = threadIdx.x / CT;
  const int rows_grp = blockDim.x / CT;
  float s[VW] = {}, sq[VW] = {};
#pragma unroll 4
  for (long r = (long)blockIdx.x * rows_grp + sub; r < R;
       r += (long)gridDim.x * rows_grp) {
    Vec v = *reinterpret_cast<const Vec*>(x + r * C + c0);
#pragma unroll
    for (int j = 0; j < VW; ++j) {
      float f = VT::load(v, j);
      s[j] += f;
      sq[j] += f * f;
    }
  }
  float* lsum = lds + chunk * VW;
  float* lsq = lds + CT * VW + chunk * VW;
#pragma unroll
  for (int j = 0; j < VW; ++j) {
    atomicAdd(&lsum[j], s[j]);
    atomicAdd(&lsq[j], sq[j]);
  }
  __syncthreads();
  float* part = sums + (long)blockIdx.x * 2 * C;
  for (int i = threadIdx.x; i < CT * VW; i += blockDim.x) {
    part[blockIdx.y * CT * VW + i] = lds[i];
    part[C + blockIdx.y * CT * VW + i] = lds[CT * VW + i];
  }
}

template <typename T>
__global__ void bn_reduce_fwd_kernel(const T* __restrict__ x,
                                     float* __restrict__ sums,  // [2C]
                                     long R, int C, int CT) {
  extern __shared__ float lds[];          // [2*CT]
  float* lsum = lds;
  float* lsq = lds + CT;
  for (int i = threadIdx.x; i < 2 * CT; i += blockDim.x) lds[i] = 0.f;
  __syncthreads();

  const int c = blockIdx.y * CT + (threadIdx.x % CT);
  const int sub = threadIdx.x / CT;
  const int rows_grp = blockDim.x / CT;
  float s = 0.f, sq = 0.f;
  for (long r = (long)blockIdx.x * rows_grp + sub; r < R;
       r += (long)gridDim.x * rows_grp) {
    float v = VecTraits<T>::load1(x + r * C + c);
    s += v;
    sq += v * v;
  }
  atomicAdd(&lsum[threadIdx.x % CT], s);
  atomicAdd(&lsq[threadIdx.x % CT], sq);
  __syncthreads();
  float* part = sums + (long)blockIdx.x * 2 * C;
  if (threadIdx.x < CT) {
    part[blockIdx.y * CT + threadIdx.x] = lsum[threadIdx.x];
    part[C + blockIdx.y * CT + threadIdx.x] = lsq[threadIdx.x];
  }
}

// ---- finalize: stats + running update + scale/shift --------------------

__global__ void bn_finalize_fwd_kernel(const float* __restrict__ sums,
                                       const float* __restrict__ w,
                                       const float* __restrict__ b,
                                       float* __restrict__ rmean,
                                       float* __restrict__ rvar,
                                       float* __restrict__ save_mean,
                                       float* __restrict__ save_invstd,
                                       float* __restrict__ scale,
                                       float* __restrict__ shift,
                                       long R, int C, int nstripes,
                                       float momentum, float eps) {
  __shared__ float l1[256], l2[256];
  const int cl = threadIdx.x & 63;          // channel lane
  const int sl = threadIdx.x >> 6;          // stripe slice (0..3)
  int c = blockIdx.x * 64 + cl;
  float s1 = 0.f, s2 = 0.f;
  if (c < C) {
    for (int st = sl; st < nstripes; st += 4) {
      s1 += sums[(long)st * 2 * C + c];
      s2 += sums[(long)st * 2 * C + C + c];
    }
  }
  l1[threadIdx.x] = s1;
  l2[threadIdx.x] = s2;
  __syncthreads();
  if (sl != 0 || c >= C) return;
  s1 = l1[cl] + l1[cl + 64] + l1[cl + 128] + l1[cl + 192];
  s2 = l2[cl] + l2[cl + 64] + l2[cl + 128] + l2[cl + 192];
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
                                      float* __restrict__ shift,
                                      int C, float eps) {
  int c = blockIdx.x * blockDim.x + threadIdx.x;
  if (c >= C) return;
  float invstd = rsqrtf(rvar[c] + eps);
  float sc = w[c] * invstd;
  scale[c] = sc;
  shift[c] = b[c] - rmean[c] * sc;
}

// ---- normalize: y = x*scale[c] + shift[c] (+ReLU) ----------------------

template <typename T, bool RELU, bool VEC>
__global__ void bn_norm_kernel(const T* __restrict__ x, T* __restrict__ y,
                               const float* __restrict__ scale,
                               const float* __restrict__ shift,
                               long total, int C) {
  using VT = VecTraits<T>;
  using Vec = typename VT::Vec;
  const long stride = (long)gridDim.x * blockDim.x;
  if (VEC) {
    const long nvec = total / VT::width;
    for (long i = blockIdx.x * (long)blockDim.x + threadIdx.x; i < nvec;
         i += stride) {
      long base = i * VT::width;
      int c0 = (int)(base % C);
      Vec xv = reinterpret_cast<const Vec*>(x)[i];
      Vec yv;
#pragma unroll
      for (int j = 0; j < VT::width; ++j) {
        float v = VT::load(xv, j) * scale[c0 + j] + shift[c0 + j];
        VT::store(yv, j, RELU ? fmaxf(v, 0.f) : v);
      }
      reinterpret_cast<Vec*>(y)[i] = yv;
    }
  } else {
    for (long i = blockIdx.x * (long)blockDim.x + threadIdx.x; i < total;
         i += stride) {
      int c = (int)(i % C);
      float v = VT::load1(x + i) * scale[c] + shift[c];
      VT::store1(y + i, RELU ? fmaxf(v, 0.f) : v);
    }
  }
}

// ---- backward reduce: sum(dy), sum(dy * xhat) per channel --------------

template <typename T, bool RELU>
__global__ void bn_reduce_bwd_vec_kernel(const T* __restrict__ dy,
                                         const T* __restrict__ x,
                                         const T* __restrict__ y,
                                         const float* __restrict__ save_mean,
                                         const float* __restrict__ save_invstd,
                                         float* __restrict__ sums,  // [2C]
                                         long R, int C, int CT) {
  using VT = VecTraits<T>;
  using Vec = typename VT::Vec;
  constexpr int VW = VT::width;
  extern __shared__ float lds[];
  for (int i = threadIdx.x; i < 2 * CT * VW; i += blockDim.x) lds[i] = 0.f;
  __syncthreads();

  const int chunk = threadIdx.x % CT;
  const int c0 = blockIdx.y * CT * VW + chunk * VW;
  const int sub = threadIdx.x / CT;
  const int rows_grp = blockDim.x / CT;
  float mean[VW], invstd[VW];
#pragma unroll
  for (int j = 0; j < VW; ++j) {
    mean[j] = save_mean[c0 + j];
    invstd[j] = save_invstd[c0 + j];
  }
  float a[VW] = {}, bx[VW] = {};
#pragma unroll 4
  for (long r = (long)blockIdx.x * rows_grp + sub; r < R;
       r += (long)gridDim.x * rows_grp) {
    const long i = r * C + c0;
    Vec gv = *reinterpret_cast<const Vec*>(dy + i);
    Vec xv = *reinterpret_cast<const Vec*>(x + i);
    Vec yv;
    if (RELU) yv = *reinterpret_cast<const Vec*>(y + i);
#pragma unroll
    for (int j = 0; j < VW; ++j) {
      float g = VT::load(gv, j);
      if (RELU && VT::load(yv, j) <= 0.f) g = 0.f;
      float xh = (VT::load(xv, j) - mean[j]) * invstd[j];
      a[j] += g;
      bx[j] += g * xh;
    }
  }
  float* s_dy = lds + chunk * VW;
  float* s_dyx = lds + CT * VW + chunk * VW;
#pragma unroll
  for (int j = 0; j < VW; ++j) {
    atomicAdd(&s_dy[j], a[j]);
    atomicAdd(&s_dyx[j], bx[j]);
  }
  __syncthreads();
  float* part = sums + (long)blockIdx.x * 2 * C;
  for (int i = threadIdx.x; i < CT * VW; i += blockDim.x) {
    part[blockIdx.y * CT * VW + i] = lds[i];
    part[C + blockIdx.y * CT * VW + i] = lds[CT * VW + i];
  }
}

template <typename T, bool RELU>
__global__ void bn_reduce_bwd_kernel(const T* __restrict__ dy,
                                     const T* __restrict__ x,
                                     const T* __restrict__ y,
                                     const float* __restrict__ save_mean,
                                     const float* __restrict__ save_invstd,
                                     float* __restrict__ sums,  // [2C]
                                     long R, int C, int CT) {
  extern __shared__ float lds[];
  float* s_dy = lds;
  float* s_dyx = lds + CT;
  for (int i = threadIdx.x; i < 2 * CT; i += blockDim.x) lds[i] = 0.f;
  __syncthreads();

  const int c = blockIdx.y * CT + (threadIdx.x % CT);
  const int sub = threadIdx.x / CT;
  const int rows_grp = blockDim.x / CT;
  const float mean = save_mean[c], invstd = save_invstd[c];
  float a = 0.f, bx = 0.f;
  for (long r = (long)blockIdx.x * rows_grp + sub; r < R;
       r += (long)gridDim.x * rows_grp) {
    long i = r * C + c;
    float g = VecTraits<T>::load1(dy + i);
    if (RELU && VecTraits<T>::load1(y + i) <= 0.f) g = 0.f;
    float xh = (VecTraits<T>::load1(x + i) - mean) * invstd;
    a += g;
    bx += g * xh;
  }
  atomicAdd(&s_dy[threadIdx.x % CT], a);
  atomicAdd(&s_dyx[threadIdx.x % CT], bx);
  __syncthreads();
  float* part = sums + (long)blockIdx.x * 2 * C;
  if (threadIdx.x < CT) {
    part[blockIdx.y * CT + threadIdx.x] = s_dy[threadIdx.x];
    part[C + blockIdx.y * CT + threadIdx.x] = s_dyx[threadIdx.x];
  }
}

__global__ void bn_finalize_bwd_kernel(const float* __restrict__ sums,
                                       const float* __restrict__ w,
                                       const float* __restrict__ save_invstd,
                                       float* __restrict__ dgamma,
                                       float* __restrict__ dbeta,
                                       float* __restrict__ k1,  // sum_dy/R
                                       float* __restrict__ k2,  // sum_dyxhat/R
                                       float* __restrict__ gi,  // gamma*invstd
                                       long R, int C, int nstripes) {
  __shared__ float l1[256], l2[256];
  const int cl = threadIdx.x & 63;
  const int sl = threadIdx.x >> 6;
  int c = blockIdx.x * 64 + cl;
  float s1 = 0.f, s2 = 0.f;
  if (c < C) {
    for (int st = sl; st < nstripes; st += 4) {
      s1 += sums[(long)st * 2 * C + c];
      s2 += sums[(long)st * 2 * C + C + c];
    }
  }
  l1[threadIdx.x] = s1;
  l2[threadIdx.x] = s2;
  __syncthreads();
  if (sl != 0 || c >= C) return;
  s1 = l1[cl] + l1[cl + 64] + l1[cl + 128] + l1[cl + 192];
  s2 = l2[cl] + l2[cl + 64] + l2[cl + 128] + l2[cl + 192];
  dbeta[c] = s1;
  dgamma[c] = s2;
  k1[c] = s1 / R;
  k2[c] = s2 / R;
  gi[c] = w[c] * save_invstd[c];
}

// dx = gi[c] * (dy_eff - k1[c] - xhat * k2[c])
template <typename T, bool RELU, bool VEC>
__global__ void bn_dx_kernel(const T* __restrict__ dy,
                             const T* __restrict__ x,
                             const T* __restrict__ y,
                             const float* __restrict__ save_mean,
                             const float* __restrict__ save_invstd,
                             const float* __restrict__ k1,
                             const float* __restrict__ k2,
                             const float* __restrict__ gi,
                             T* __restrict__ dx, long total, int C) {
  using VT = VecTraits<T>;
  using Vec = typename VT::Vec;
  const long stride = (long)gridDim.x * blockDim.x;
  if (VEC) {
    const long nvec = total / VT::width;
    for (long i = blockIdx.x * (long)blockDim.x + threadIdx.x; i < nvec;
         i += stride) {
      long base = i * VT::width;
      int c0 = (int)(base % C);
      Vec gv = reinterpret_cast<const Vec*>(dy)[i];
      Vec xv = reinterpret_cast<const Vec*>(x)[i];
      Vec yv;
      if (RELU) yv = reinterpret_cast<const Vec*>(y)[i];
      Vec ov;
#pragma unroll
      for (int j = 0; j < VT::width; ++j) {
        int c = c0 + j;
        float g = VT::load(gv, j);
        if (RELU && VT::load(yv, j) <= 0.f) g = 0.f;
        float xh = (VT::load(xv, j) - save_mean[c]) * save_invstd[c];
        VT::store(ov, j, gi[c] * (g - k1[c] - xh * k2[c]));
      }
      reinterpret_cast<Vec*>(dx)[i] = ov;
    }
  } else {
    for (long i = blockIdx.x * (long)blockDim.x + threadIdx.x; i < total;
         i += stride) {
      int c = (int)(i % C);
      float g = VT::load1(dy + i);
      if (RELU && VT::load1(y + i) <= 0.f) g = 0.f;
      float xh = (VT::load1(x + i) - save_mean[c]) * save_invstd[c];
      VT::store1(dx + i, gi[c] * (g - k1[c] - xh * k2[c]));
    }
  }
}

// ---- host-side helpers --------------------------------------------------

struct BNShape {
  long R;
  int C;
  int CT;
  bool vec;     // vectorized reduce path (CT counts VW-chunks)
  dim3 rgrid;   // reduce grid
  int lds;      // reduce LDS bytes
};

static BNShape bn_shape(const at::Tensor& x, int vw) {
  BNShape s;
  s.C = x.size(1);
  s.R = x.numel() / s.C;
  s.vec = (s.C % vw) == 0;
  const int units = s.vec ? s.C / vw : s.C;   // per-thread ownership units
  s.CT = channel_tile(units);
  int rows_grp = kBlock / s.CT;
  long stripes = (s.R + rows_grp - 1) / rows_grp;
  // ~512 blocks total fills the 256 CUs ~2x; more stripes only grows the
  // partial buffer and the finalize loop
  int max_stripes = std::max(1, 1024 / std::max(1, units / s.CT));
  if (stripes > max_stripes) stripes = max_stripes;
  if (stripes < 1) stripes = 1;
  s.rgrid = dim3((unsigned)stripes, (unsigned)(units / s.CT));
  s.lds = 2 * s.CT * (s.vec ? vw : 1) * sizeof(float);
  return s;
}

template <typename T>
static const T* tp(const at::Tensor& t) {
  return reinterpret_cast<const T*>(t.data_ptr());
}
template <typename T>
static T* tpm(at::Tensor& t) {
  return reinterpret_cast<T*>(t.data_ptr());
}

}  // namespace

std::tuple<at::Tensor, at::Tensor, at::Tensor> bn_fwd_train_impl(
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
  auto sums = at::empty({(long)sh.rgrid.x * 2 * sh.C}, fopts);
  auto save_mean = at::empty({sh.C}, fopts);
  auto save_invstd = at::empty({sh.C}, fopts);
  auto scale = at::empty({sh.C}, fopts);
  auto shift = at::empty({sh.C}, fopts);
  auto y = at::empty_like(x);

  const long total = x.numel();

#define DLX_BN_T(T)                                                          \
  if (sh.vec)                                                                \
    bn_reduce_fwd_vec_kernel<T><<<sh.rgrid, kBlock, sh.lds, s>>>(            \
        tp<T>(x), sums.data_ptr<float>(), sh.R, sh.C, sh.CT);                \
  else                                                                       \
    bn_reduce_fwd_kernel<T><<<sh.rgrid, kBlock, sh.lds, s>>>(                \
        tp<T>(x), sums.data_ptr<float>(), sh.R, sh.C, sh.CT);                \
  bn_finalize_fwd_kernel<<<(sh.C + 63) / 64, 256, 0, s>>>(                 \
      sums.data_ptr<float>(), w.data_ptr<float>(), b.data_ptr<float>(),      \
      rmean.data_ptr<float>(), rvar.data_ptr<float>(),                       \
      save_mean.data_ptr<float>(), save_invstd.data_ptr<float>(),            \
      scale.data_ptr<float>(), shift.data_ptr<float>(), sh.R, sh.C,          \
      (int)sh.rgrid.x, (float)momentum, (float)eps);                         \
  {                                                                          \
    const bool vec = (sh.C % VecTraits<T>::width) == 0;                      \
    int grid = grid_for(total / (vec ? VecTraits<T>::width : 1) + 1);        \
    if (relu && vec)                                                         \
      bn_norm_kernel<T, true, true><<<grid, kBlock, 0, s>>>(                 \
          tp<T>(x), tpm<T>(y), scale.data_ptr<float>(),                      \
          shift.data_ptr<float>(), total, sh.C);                             \
    else if (relu)                                                           \
      bn_norm_kernel<T, true, false><<<grid, kBlock, 0, s>>>(                \
          tp<T>(x), tpm<T>(y), scale.data_ptr<float>(),                      \
          shift.data_ptr<float>(), total, sh.C);                             \
    else if (vec)                                                            \
      bn_norm_kernel<T, false, true><<<grid, kBlock, 0, s>>>(                \
          tp<T>(x), tpm<T>(y), scale.data_ptr<float>(),                      \
          shift.data_ptr<float>(), total, sh.C);                             \
    else                                                                     \
      bn_norm_kernel<T, false, false><<<grid, kBlock, 0, s>>>(               \
          tp<T>(x), tpm<T>(y), scale.data_ptr<float>(),                      \
          shift.data_ptr<float>(), total, sh.C);                             \
  }

  if (bf) { DLX_BN_T(uint16_t) } else { DLX_BN_T(float) }
#undef DLX_BN_T
  return {y, save_mean, save_invstd};
}

at::Tensor bn_fwd_eval_impl(const at::Tensor& x_in, const at::Tensor& w,
                            const at::Tensor& b, const at::Tensor& rmean,
                            const at::Tensor& rvar, double eps, bool relu) {
  auto x = x_in.contiguous(at::MemoryFormat::ChannelsLast);
  auto s = at::cuda::getCurrentCUDAStream();
  const int C = x.size(1);
  const long total = x.numel();
  auto fopts = x.options().dtype(at::kFloat);
  auto scale = at::empty({C}, fopts);
  auto shift = at::empty({C}, fopts);
  bn_eval_coeffs_kernel<<<(C + 255) / 256, 256, 0, s>>>(
      w.data_ptr<float>(), b.data_ptr<float>(), rmean.data_ptr<float>(),
      rvar.data_ptr<float>(), scale.data_ptr<float>(), shift.data_ptr<float>(),
      C, (float)eps);
  auto y = at::empty_like(x);
  const bool bf = x.scalar_type() == at::kBFloat16;

#define DLX_BN_EVAL(T)                                                       \
  {                                                                          \
    const bool vec = (C % VecTraits<T>::width) == 0;                         \
    int grid = grid_for(total / (vec ? VecTraits<T>::width : 1) + 1);        \
    if (relu && vec)                                                         \
      bn_norm_kernel<T, true, true><<<grid, kBlock, 0, s>>>(                 \
          tp<T>(x), tpm<T>(y), scale.data_ptr<float>(),                      \
          shift.data_ptr<float>(), total, C);                                \
    else if (relu)                                                           \
      bn_norm_kernel<T, true, false><<<grid, kBlock, 0, s>>>(                \
          tp<T>(x), tpm<T>(y), scale.data_ptr<float>(),                      \
          shift.data_ptr<float>(), total, C);                                \
    else if (vec)                                                            \
      bn_norm_kernel<T, false, true><<<grid, kBlock, 0, s>>>(                \
          tp<T>(x), tpm<T>(y), scale.data_ptr<float>(),                      \
          shift.data_ptr<float>(), total, C);                                \
    else                                                                     \
      bn_norm_kernel<T, false, false><<<grid, kBlock, 0, s>>>(               \
          tp<T>(x), tpm<T>(y), scale.data_ptr<float>(),                      \
          shift.data_ptr<float>(), total, C);                                \
  }
  if (bf) { DLX_BN_EVAL(uint16_t) } else { DLX_BN_EVAL(float) }
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
  auto sums = at::empty({(long)sh.rgrid.x * 2 * sh.C}, fopts);
  auto dgamma = at::empty({sh.C}, fopts);
  auto dbeta = at::empty({sh.C}, fopts);
  auto k1 = at::empty({sh.C}, fopts);
  auto k2 = at::empty({sh.C}, fopts);
  auto gi = at::empty({sh.C}, fopts);
  auto dx = at::empty_like(x);
  const long total = x.numel();

#define DLX_BN_BWD(T)                                                        \
  {                                                                          \
    if (sh.vec && relu)                                                      \
      bn_reduce_bwd_vec_kernel<T, true><<<sh.rgrid, kBlock, sh.lds, s>>>(    \
          tp<T>(dy), tp<T>(x), tp<T>(y), save_mean.data_ptr<float>(),        \
          save_invstd.data_ptr<float>(), sums.data_ptr<float>(), sh.R,       \
          sh.C, sh.CT);                                                      \
    else if (sh.vec)                                                         \
      bn_reduce_bwd_vec_kernel<T, false><<<sh.rgrid, kBlock, sh.lds, s>>>(   \
          tp<T>(dy), tp<T>(x), tp<T>(y), save_mean.data_ptr<float>(),        \
          save_invstd.data_ptr<float>(), sums.data_ptr<float>(), sh.R,       \
          sh.C, sh.CT);                                                      \
    else if (relu)                                                           \
      bn_reduce_bwd_kernel<T, true><<<sh.rgrid, kBlock, sh.lds, s>>>(        \
          tp<T>(dy), tp<T>(x), tp<T>(y), save_mean.data_ptr<float>(),        \
          save_invstd.data_ptr<float>(), sums.data_ptr<float>(), sh.R,       \
          sh.C, sh.CT);                                                      \
    else                                                                     \
      bn_reduce_bwd_kernel<T, false><<<sh.rgrid, kBlock, sh.lds, s>>>(       \
          tp<T>(dy), tp<T>(x), tp<T>(y), save_mean.data_ptr<float>(),        \
          save_invstd.data_ptr<float>(), sums.data_ptr<float>(), sh.R,       \
          sh.C, sh.CT);                                                      \
    bn_finalize_bwd_kernel<<<(sh.C + 63) / 64, 256, 0, s>>>(               \
        sums.data_ptr<float>(), w.data_ptr<float>(),                         \
        save_invstd.data_ptr<float>(), dgamma.data_ptr<float>(),             \
        dbeta.data_ptr<float>(), k1.data_ptr<float>(),                       \
        k2.data_ptr<float>(), gi.data_ptr<float>(), sh.R, sh.C,              \
        (int)sh.rgrid.x);                                                    \
    const bool vec = (sh.C % VecTraits<T>::width) == 0;                      \
    int grid = grid_for(total / (vec ? VecTraits<T>::width : 1) + 1);        \
    if (relu && vec)                                                         \
      bn_dx_kernel<T, true, true><<<grid, kBlock, 0, s>>>(                   \
          tp<T>(dy), tp<T>(x), tp<T>(y), save_mean.data_ptr<float>(),        \
          save_invstd.data_ptr<float>(), k1.data_ptr<float>(),               \
          k2.data_ptr<float>(), gi.data_ptr<float>(), tpm<T>(dx), total,     \
          sh.C);                                                             \
    else if (relu)                                                           \
      bn_dx_kernel<T, true, false><<<grid, kBlock, 0, s>>>(                  \
          tp<T>(dy), tp<T>(x), tp<T>(y), save_mean.data_ptr<float>(),        \
          save_invstd.data_ptr<float>(), k1.data_ptr<float>(),               \
          k2.data_ptr<float>(), gi.data_ptr<float>(), tpm<T>(dx), total,     \
          sh.C);                                                             \
    else if (vec)                                                            \
      bn_dx_kernel<T, false, true><<<grid, kBlock, 0, s>>>(                  \
          tp<T>(dy), tp<T>(x), tp<T>(y), save_mean.data_ptr<float>(),        \
          save_invstd.data_ptr<float>(), k1.data_ptr<float>(),               \
          k2.data_ptr<float>(), gi.data_ptr<float>(), tpm<T>(dx), total,     \
          sh.C);                                                             \
    else                                                                     \
      bn_dx_kernel<T, false, false><<<grid, kBlock, 0, s>>>(                 \
          tp<T>(dy), tp<T>(x), tp<T>(y), save_mean.data_ptr<float>(),        \
          save_invstd.data_ptr<float>(), k1.data_ptr<float>(),               \
          k2.data_ptr<float>(), gi.data_ptr<float>(), tpm<T>(dx), total,     \
          sh.C);                                                             \
  }
  if (bf) { DLX_BN_BWD(uint16_t) } else { DLX_BN_BWD(float) }
#undef DLX_BN_BWD
  return {dx, dgamma, dbeta};
}

}  // namespace dlx
