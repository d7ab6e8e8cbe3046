// Layout pack kernels: NCHW <-> NHWC (the explicit north-star item,
// SURVEY.md §3.2 "layout/transform kernels").
//
// Per image: a [C, HW] <-> [HW, C] transpose staged through a padded
// 32x32 LDS tile (guide §6 G4: +1 padding breaks the power-of-2 row
// stride) so both the global read and the global write are coalesced.
#include <torch/extension.h>
#include <ATen/cuda/CUDAContext.h>

#include "dlx_common.h"

namespace dlx {

namespace {

constexpr int TD = 32;   // tile dim
constexpr int TR = 8;    // rows per block-y step (block = 32x8)

// transpose src[R, S] -> dst[S, R]  (per batch image, 2D grid over tiles)
template <typename T>
__global__ void transpose2d_kernel(const T* __restrict__ src,
                                   T* __restrict__ dst,
                                   long R, long S) {
  __shared__ T tile[TD][TD + 1];
  const long nb = blockIdx.z;
  src += nb * R * S;
  dst += nb * R * S;
  const long r0 = (long)blockIdx.y * TD;
  const long s0 = (long)blockIdx.x * TD;
#pragma unroll
  for (int i = 0; i < TD; i += TR) {
    long r = r0 + threadIdx.y + i;
    long s = s0 + threadIdx.x;
    if (r < R && s < S)
      tile[threadIdx.y + i][threadIdx.x] = src[r * S + s];
  }
  __syncthreads();
#pragma unroll
  for (int i = 0; i < TD; i += TR) {
    long s = s0 + threadIdx.y + i;
    long r = r0 + threadIdx.x;
    if (s < S && r < R)
      dst[s * R + r] = tile[threadIdx.x][threadIdx.y + i];
  }
}

template <typename T>
void launch_transpose(const T* src, T* dst, long Nb, long R, long S,
                      hipStream_t st) {
  dim3 grid((unsigned)((S + TD - 1) / TD), (unsigned)((R + TD - 1) / TD),
            (unsigned)Nb);
  transpose2d_kernel<T><<<grid, dim3(TD, TR), 0, st>>>(src, dst, R, S);
}

}  // namespace

// NCHW (contiguous) -> NHWC-dense tensor (torch channels_last)
at::Tensor pack_nchw_to_nhwc_impl(const at::Tensor& x_in) {
  auto x = x_in.contiguous();  // [N,C,H,W] standard
  const long Nb = x.size(0), C = x.size(1), HW = x.size(2) * x.size(3);
  auto y = at::empty(x.sizes(),
                     x.options().memory_format(at::MemoryFormat::ChannelsLast));
  auto s = at::cuda::getCurrentCUDAStream();
  if (x.scalar_type() == at::kBFloat16)
    launch_transpose(reinterpret_cast<const uint16_t*>(x.data_ptr()),
                     reinterpret_cast<uint16_t*>(y.data_ptr()), Nb, C, HW, s);
  else if (x.scalar_type() == at::kFloat)
    launch_transpose(x.data_ptr<float>(), y.data_ptr<float>(), Nb, C, HW, s);
  else
    TORCH_CHECK(false, "pack: unsupported dtype ", x.scalar_type());
  return y;
}

// NHWC-dense (channels_last) -> standard NCHW contiguous
at::Tensor pack_nhwc_to_nchw_impl(const at::Tensor& x_in) {
  auto x = x_in.contiguous(at::MemoryFormat::ChannelsLast);
  const long Nb = x.size(0), C = x.size(1), HW = x.size(2) * x.size(3);
  auto y = at::empty(x.sizes(), x.options());  // contiguous NCHW
  auto s = at::cuda::getCurrentCUDAStream();
  if (x.scalar_type() == at::kBFloat16)
    launch_transpose(reinterpret_cast<const uint16_t*>(x.data_ptr()),
                     reinterpret_cast<uint16_t*>(y.data_ptr()), Nb, HW, C, s);
  else if (x.scalar_type() == at::kFloat)
    launch_transpose(x.data_ptr<float>(), y.data_ptr<float>(), Nb, HW, C, s);
  else
    TORCH_CHECK(false, "pack: unsupported dtype ", x.scalar_type());
  return y;
}

// generic 2D transpose (used by the GEMM nn/tn paths)
at::Tensor transpose2d_impl(const at::Tensor& x_in) {
  auto x = x_in.contiguous();
  TORCH_CHECK(x.dim() == 2, "transpose2d: 2D only");
  const long R = x.size(0), S = x.size(1);
  auto y = at::empty({S, R}, x.options());
  auto s = at::cuda::getCurrentCUDAStream();
  if (x.scalar_type() == at::kBFloat16)
    launch_transpose(reinterpret_cast<const uint16_t*>(x.data_ptr()),
                     reinterpret_cast<uint16_t*>(y.data_ptr()), 1, R, S, s);
  else if (x.scalar_type() == at::kFloat)
    launch_transpose(x.data_ptr<float>(), y.data_ptr<float>(), 1, R, S, s);
  else
    TORCH_CHECK(false, "transpose2d: unsupported dtype ", x.scalar_type());
  return y;
}

}  // namespace dlx
