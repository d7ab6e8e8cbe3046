// NHWC pooling: 3x3/s2/p1 maxpool (ResNet stem) + global average pool.
//
// Replaces the reference stack's cuDNN/framework pooling (SURVEY.md §3.2
// "cuDNN/framework pooling").  NHWC: channels contiguous, so each lane
// owns an 8-channel vector (16 B loads) and windows walk rows.
// The argmax is stored as ONE BYTE per element (window-relative tap index
// 0..8), not an int32 linear index — 4x less index traffic on the
// backward pass.  Maxpool backward is GATHER-style (each input position
// scans its <=4 candidate windows) — no atomics.
#include <torch/extension.h>
#include <ATen/cuda/CUDAContext.h>

#include "dlx_common.h"

namespace dlx {

namespace {

template <int W> struct TapVec;
template <> struct alignas(8) TapVec<8> { uint8_t v[8]; };
template <> struct alignas(4) TapVec<4> { uint8_t v[4]; };
template <> struct alignas(1) TapVec<1> { uint8_t v[1]; };

// ---- maxpool 3x3 s2 p1 fwd: y + per-element tap byte -------------------

template <typename T, bool VEC>
__global__ void maxpool_fwd_kernel(const T* __restrict__ x, T* __restrict__ y,
                                   uint8_t* __restrict__ argtap, int Nb,
                                   int H, int W, int C, int Ho, int Wo) {
  using VT = VecTraits<T>;
  using Vec = typename VT::Vec;
  constexpr int VW = VEC ? VT::width : 1;
  const int CU = C / VW;  // channel units
  const long total = (long)Nb * Ho * Wo * CU;
  const long stride = (long)gridDim.x * blockDim.x;
  for (long i = blockIdx.x * (long)blockDim.x + threadIdx.x; i < total;
       i += stride) {
    const int cu = (int)(i % CU);
    long t = i / CU;
    const int wo = (int)(t % Wo);
    t /= Wo;
    const int ho = (int)(t % Ho);
    const int nb = (int)(t / Ho);
    const int c0 = cu * VW;
    float best[VW];
    uint8_t tap[VW];
#pragma unroll
    for (int j = 0; j < VW; ++j) {
      best[j] = -INFINITY;
      tap[j] = 0;
    }
    const T* xin = x + (long)nb * H * W * C + c0;
#pragma unroll
    for (int dh = 0; dh < 3; ++dh) {
      const int hi = ho * 2 - 1 + dh;
      if (hi < 0 || hi >= H) continue;
#pragma unroll
      for (int dw = 0; dw < 3; ++dw) {
        const int wi = wo * 2 - 1 + dw;
        if (wi < 0 || wi >= W) continue;
        const T* p = xin + ((long)hi * W + wi) * C;
        if (VEC) {
          Vec v = *reinterpret_cast<const Vec*>(p);
#pragma unroll
          for (int j = 0; j < VW; ++j) {
            float f = VT::load(v, j);
            if (f > best[j]) {
              best[j] = f;
              tap[j] = (uint8_t)(dh * 3 + dw);
            }
          }
        } else {
          float f = VT::load1(p);
          if (f > best[0]) {
            best[0] = f;
            tap[0] = (uint8_t)(dh * 3 + dw);
          }
        }
      }
    }
    const long o = (((long)nb * Ho + ho) * Wo + wo) * C + c0;
    if (VEC) {
      Vec ov;
      TapVec<VW> tv;
#pragma unroll
      for (int j = 0; j < VW; ++j) {
        VT::store(ov, j, best[j]);
        tv.v[j] = tap[j];
      }
      *reinterpret_cast<Vec*>(y + o) = ov;
      *reinterpret_cast<TapVec<VW>*>(argtap + o) = tv;
    } else {
      VT::store1(y + o, best[0]);
      argtap[o] = tap[0];
    }
  }
}

// ---- maxpool bwd: gather over <=4 candidate windows --------------------

template <typename T, bool VEC>
__global__ void maxpool_bwd_kernel(const T* __restrict__ dy,
                                   const uint8_t* __restrict__ argtap,
                                   T* __restrict__ dx, int Nb, int H, int W,
                                   int C, int Ho, int Wo) {
  using VT = VecTraits<T>;
  using Vec = typename VT::Vec;
  constexpr int VW = VEC ? VT::width : 1;
  const int CU = C / VW;
  const long total = (long)Nb * H * W * CU;
  const long stride = (long)gridDim.x * blockDim.x;
  for (long i = blockIdx.x * (long)blockDim.x + threadIdx.x; i < total;
       i += stride) {
    const int cu = (int)(i % CU);
    long t = i / CU;
    const int wi = (int)(t % W);
    t /= W;
    const int hi = (int)(t % H);
    const int nb = (int)(t / H);
    const int c0 = cu * VW;
    float acc[VW] = {};
    const int ho_lo = hi >> 1, wo_lo = wi >> 1;
    const int ho_hi = (hi + 1) >> 1, wo_hi = (wi + 1) >> 1;
#pragma unroll
    for (int ho = ho_lo; ho <= ho_hi; ++ho) {
      if (ho >= Ho) continue;
      const int dh = hi - (ho * 2 - 1);  // 0..2 by construction
#pragma unroll
      for (int wo = wo_lo; wo <= wo_hi; ++wo) {
        if (wo >= Wo) continue;
        const uint8_t want = (uint8_t)(dh * 3 + (wi - (wo * 2 - 1)));
        const long oi = (((long)nb * Ho + ho) * Wo + wo) * C + c0;
        if (VEC) {
          TapVec<VW> tv = *reinterpret_cast<const TapVec<VW>*>(argtap + oi);
          Vec gv = *reinterpret_cast<const Vec*>(dy + oi);
#pragma unroll
          for (int j = 0; j < VW; ++j)
            if (tv.v[j] == want) acc[j] += VT::load(gv, j);
        } else {
          if (argtap[oi] == want) acc[0] += VT::load1(dy + oi);
        }
      }
    }
    const long o = (((long)nb * H + hi) * W + wi) * C + c0;
    if (VEC) {
      Vec ov;
#pragma unroll
      for (int j = 0; j < VW; ++j) VT::store(ov, j, acc[j]);
      *reinterpret_cast<Vec*>(dx + o) = ov;
    } else {
      VT::store1(dx + o, acc[0]);
    }
  }
}

// ---- global average pool: y[n,c] = mean over H*W -----------------------

template <typename T>
__global__ void gap_fwd_kernel(const T* __restrict__ x, T* __restrict__ y,
                               int Nb, int HW, int C) {
  // one block per image; threads stride channels; loop rows
  const int nb = blockIdx.x;
  const float inv = 1.f / HW;
  for (int c = threadIdx.x; c < C; c += blockDim.x) {
    const T* xin = x + (long)nb * HW * C + c;
    float s = 0.f;
    for (int r = 0; r < HW; ++r) s += VecTraits<T>::load1(xin + (long)r * C);
    VecTraits<T>::store1(y + (long)nb * C + c, s * inv);
  }
}

template <typename T>
__global__ void gap_bwd_kernel(const T* __restrict__ dy, T* __restrict__ dx,
                               int Nb, int HW, int C) {
  const long total = (long)Nb * HW * C;
  const long stride = (long)gridDim.x * blockDim.x;
  const float inv = 1.f / HW;
  for (long i = blockIdx.x * (long)blockDim.x + threadIdx.x; i < total;
       i += stride) {
    int c = (int)(i % C);
    int nb = (int)(i / ((long)HW * C));
    VecTraits<T>::store1(
        dx + i, VecTraits<T>::load1(dy + (long)nb * C + c) * inv);
  }
}

}  // namespace

std::tuple<at::Tensor, at::Tensor> maxpool3x3s2_fwd_impl(const at::Tensor& x_in) {
  auto x = x_in.contiguous(at::MemoryFormat::ChannelsLast);
  const int Nb = x.size(0), C = x.size(1), H = x.size(2), W = x.size(3);
  const int Ho = (H + 1) / 2, Wo = (W + 1) / 2;
  auto y = at::empty({Nb, C, Ho, Wo},
                     x.options().memory_format(at::MemoryFormat::ChannelsLast));
  auto argtap = at::empty({Nb, C, Ho, Wo},
                          x.options().dtype(at::kByte)
                              .memory_format(at::MemoryFormat::ChannelsLast));
  auto s = at::cuda::getCurrentCUDAStream();
  const bool bf = x.scalar_type() == at::kBFloat16;
  const bool vec = (C % (bf ? 8 : 4)) == 0;
  const long total = y.numel() / (vec ? (bf ? 8 : 4) : 1);
  int grid = grid_for(total);
#define DLX_MPF(T, VEC)                                                     \
  maxpool_fwd_kernel<T, VEC><<<grid, kBlock, 0, s>>>(                       \
      reinterpret_cast<const T*>(x.data_ptr()),                             \
      reinterpret_cast<T*>(y.data_ptr()),                                   \
      argtap.data_ptr<uint8_t>(), Nb, H, W, C, Ho, Wo);
  if (bf && vec) { DLX_MPF(uint16_t, true) }
  else if (bf) { DLX_MPF(uint16_t, false) }
  else if (vec) { DLX_MPF(float, true) }
  else { DLX_MPF(float, false) }
#undef DLX_MPF
  return {y, argtap};
}

at::Tensor maxpool3x3s2_bwd_impl(const at::Tensor& dy_in,
                                 const at::Tensor& argtap,
                                 int64_t H, int64_t W) {
  auto dy = dy_in.contiguous(at::MemoryFormat::ChannelsLast);
  const int Nb = dy.size(0), C = dy.size(1), Ho = dy.size(2), Wo = dy.size(3);
  auto dx = at::empty({Nb, C, H, W},
                      dy.options().memory_format(at::MemoryFormat::ChannelsLast));
  auto s = at::cuda::getCurrentCUDAStream();
  const bool bf = dy.scalar_type() == at::kBFloat16;
  const bool vec = (C % (bf ? 8 : 4)) == 0;
  int grid = grid_for(dx.numel() / (vec ? (bf ? 8 : 4) : 1));
#define DLX_MPB(T, VEC)                                                     \
  maxpool_bwd_kernel<T, VEC><<<grid, kBlock, 0, s>>>(                       \
      reinterpret_cast<const T*>(dy.data_ptr()),                            \
      argtap.data_ptr<uint8_t>(),                                           \
      reinterpret_cast<T*>(dx.data_ptr()), Nb, (int)H, (int)W, C, Ho, Wo);
  if (bf && vec) { DLX_MPB(uint16_t, true) }
  else if (bf) { DLX_MPB(uint16_t, false) }
  else if (vec) { DLX_MPB(float, true) }
  else { DLX_MPB(float, false) }
#undef DLX_MPB
  return dx;
}

at::Tensor gap_fwd_impl(const at::Tensor& x_in) {
  auto x = x_in.contiguous(at::MemoryFormat::ChannelsLast);
  const int Nb = x.size(0), C = x.size(1), HW = x.size(2) * x.size(3);
  auto y = at::empty({Nb, C, 1, 1}, x.options());
  auto s = at::cuda::getCurrentCUDAStream();
  if (x.scalar_type() == at::kBFloat16)
    gap_fwd_kernel<uint16_t><<<Nb, kBlock, 0, s>>>(
        reinterpret_cast<const uint16_t*>(x.data_ptr()),
        reinterpret_cast<uint16_t*>(y.data_ptr()), Nb, HW, C);
  else
    gap_fwd_kernel<float><<<Nb, kBlock, 0, s>>>(
        x.data_ptr<float>(), y.data_ptr<float>(), Nb, HW, C);
  return y;
}

at::Tensor gap_bwd_impl(const at::Tensor& dy_in, int64_t H, int64_t W) {
  auto dy = dy_in.contiguous();
  const int Nb = dy.size(0), C = dy.size(1);
  const int HW = (int)(H * W);
  auto dx = at::empty({Nb, C, H, W},
                      dy.options().memory_format(at::MemoryFormat::ChannelsLast));
  auto s = at::cuda::getCurrentCUDAStream();
  int grid = grid_for(dx.numel());
  if (dy.scalar_type() == at::kBFloat16)
    gap_bwd_kernel<uint16_t><<<grid, kBlock, 0, s>>>(
        reinterpret_cast<const uint16_t*>(dy.data_ptr()),
        reinterpret_cast<uint16_t*>(dx.data_ptr()), Nb, HW, C);
  else
    gap_bwd_kernel<float><<<grid, kBlock, 0, s>>>(
        dy.data_ptr<float>(), dx.data_ptr<float>(), Nb, HW, C);
  return dx;
}

}  // namespace dlx
