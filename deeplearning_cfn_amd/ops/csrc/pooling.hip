// NHWC pooling: 3x3/s2/p1 maxpool (ResNet stem) + global average pool.
//
// Replaces the reference stack's cuDNN/framework pooling (SURVEY.md §3.2
// "cuDNN/framework pooling").  NHWC: channels contiguous, so each lane
// owns a channel-vector and windows walk rows — fully coalesced.
// Maxpool backward is GATHER-style (each input position scans its <=4
// candidate windows' argmax) — no atomics.
#include <torch/extension.h>
#include <ATen/cuda/CUDAContext.h>

#include "dlx_common.h"

namespace dlx {

namespace {

// ---- maxpool 3x3 s2 p1 fwd: also records argmax (input h*W+w) ----------

template <typename T>
__global__ void maxpool_fwd_kernel(const T* __restrict__ x,
                                   T* __restrict__ y,
                                   int32_t* __restrict__ argmax,
                                   int Nb, int H, int W, int C,
                                   int Ho, int Wo) {
  const long total = (long)Nb * Ho * Wo * C;
  const long stride = (long)gridDim.x * blockDim.x;
  for (long i = blockIdx.x * (long)blockDim.x + threadIdx.x; i < total;
       i += stride) {
    int c = (int)(i % C);
    long t = i / C;
    int wo = (int)(t % Wo);
    t /= Wo;
    int ho = (int)(t % Ho);
    int nb = (int)(t / Ho);
    float best = -INFINITY;
    int bestidx = 0;
    const T* xin = x + (long)nb * H * W * C + c;
#pragma unroll
    for (int dh = -1; dh <= 1; ++dh) {
      int hi = ho * 2 + dh;
      if (hi < 0 || hi >= H) continue;
#pragma unroll
      for (int dw = -1; dw <= 1; ++dw) {
        int wi = wo * 2 + dw;
        if (wi < 0 || wi >= W) continue;
        float v = VecTraits<T>::load1(xin + ((long)hi * W + wi) * C);
        if (v > best) {
          best = v;
          bestidx = hi * W + wi;
        }
      }
    }
    VecTraits<T>::store1(y + i, best);
    argmax[i] = bestidx;
  }
}

template <typename T>
__global__ void maxpool_bwd_kernel(const T* __restrict__ dy,
                                   const int32_t* __restrict__ argmax,
                                   T* __restrict__ dx,
                                   int Nb, int H, int W, int C,
                                   int Ho, int Wo) {
  const long total = (long)Nb * H * W * C;
  const long stride = (long)gridDim.x * blockDim.x;
  for (long i = blockIdx.x * (long)blockDim.x + threadIdx.x; i < total;
       i += stride) {
    int c = (int)(i % C);
    long t = i / C;
    int wi = (int)(t % W);
    t /= W;
    int hi = (int)(t % H);
    int nb = (int)(t / H);
    const int lin = hi * W + wi;
    float acc = 0.f;
    const int ho_lo = hi >> 1, wo_lo = wi >> 1;       // see gather note
    const int ho_hi = (hi + 1) >> 1, wo_hi = (wi + 1) >> 1;
#pragma unroll
    for (int ho = ho_lo; ho <= ho_hi; ++ho) {
      if (ho >= Ho) continue;
#pragma unroll
      for (int wo = wo_lo; wo <= wo_hi; ++wo) {
        if (wo >= Wo) continue;
        long oi = (((long)nb * Ho + ho) * Wo + wo) * C + c;
        if (argmax[oi] == lin) acc += VecTraits<T>::load1(dy + oi);
      }
    }
    VecTraits<T>::store1(dx + i, acc);
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
  auto argmax = at::empty({Nb, C, Ho, Wo},
                          x.options().dtype(at::kInt)
                              .memory_format(at::MemoryFormat::ChannelsLast));
  auto s = at::cuda::getCurrentCUDAStream();
  const long total = y.numel();
  int grid = grid_for(total);
  if (x.scalar_type() == at::kBFloat16)
    maxpool_fwd_kernel<uint16_t><<<grid, kBlock, 0, s>>>(
        reinterpret_cast<const uint16_t*>(x.data_ptr()),
        reinterpret_cast<uint16_t*>(y.data_ptr()),
        argmax.data_ptr<int32_t>(), Nb, H, W, C, Ho, Wo);
  else
    maxpool_fwd_kernel<float><<<grid, kBlock, 0, s>>>(
        x.data_ptr<float>(), y.data_ptr<float>(),
        argmax.data_ptr<int32_t>(), Nb, H, W, C, Ho, Wo);
  return {y, argmax};
}

at::Tensor maxpool3x3s2_bwd_impl(const at::Tensor& dy_in,
                                 const at::Tensor& argmax,
                                 int64_t H, int64_t W) {
  auto dy = dy_in.contiguous(at::MemoryFormat::ChannelsLast);
  const int Nb = dy.size(0), C = dy.size(1), Ho = dy.size(2), Wo = dy.size(3);
  auto dx = at::empty({Nb, C, H, W},
                      dy.options().memory_format(at::MemoryFormat::ChannelsLast));
  auto s = at::cuda::getCurrentCUDAStream();
  int grid = grid_for(dx.numel());
  if (dy.scalar_type() == at::kBFloat16)
    maxpool_bwd_kernel<uint16_t><<<grid, kBlock, 0, s>>>(
        reinterpret_cast<const uint16_t*>(dy.data_ptr()),
        argmax.data_ptr<int32_t>(),
        reinterpret_cast<uint16_t*>(dx.data_ptr()),
        Nb, (int)H, (int)W, C, Ho, Wo);
  else
    maxpool_bwd_kernel<float><<<grid, kBlock, 0, s>>>(
        dy.data_ptr<float>(), argmax.data_ptr<int32_t>(),
        dx.data_ptr<float>(), Nb, (int)H, (int)W, C, Ho, Wo);
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
