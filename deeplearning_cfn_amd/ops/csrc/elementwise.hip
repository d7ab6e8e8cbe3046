// Elementwise suite: ReLU fwd/bwd, fused residual add+ReLU.
//
// Replaces the reference stack's framework ReLU kernels (SURVEY.md §3.2
// "framework elementwise" row).  Memory-bound: vectorized 16 B/lane,
// grid-stride, capped grid (guide §B "Element-wise").
#include "dlx_common.h"

namespace dlx {

template <typename T, typename Op>
__global__ void ew1_kernel(const T* __restrict__ a, T* __restrict__ o,
                           long n, Op op) {
  using VT = VecTraits<T>;
  using Vec = typename VT::Vec;
  const long nvec = n / VT::width;
  const long stride = (long)gridDim.x * blockDim.x;
  for (long i = blockIdx.x * (long)blockDim.x + threadIdx.x; i < nvec;
       i += stride) {
    Vec x = reinterpret_cast<const Vec*>(a)[i];
    Vec y;
#pragma unroll
    for (int j = 0; j < VT::width; ++j) VT::store(y, j, op(VT::load(x, j)));
    reinterpret_cast<Vec*>(o)[i] = y;
  }
  // scalar tail
  const long tail0 = nvec * VT::width;
  for (long i = tail0 + blockIdx.x * (long)blockDim.x + threadIdx.x; i < n;
       i += stride)
    VT::store1(o + i, op(VT::load1(a + i)));
}

template <typename T, typename Op>
__global__ void ew2_kernel(const T* __restrict__ a, const T* __restrict__ b,
                           T* __restrict__ o, long n, Op op) {
  using VT = VecTraits<T>;
  using Vec = typename VT::Vec;
  const long nvec = n / VT::width;
  const long stride = (long)gridDim.x * blockDim.x;
  for (long i = blockIdx.x * (long)blockDim.x + threadIdx.x; i < nvec;
       i += stride) {
    Vec x = reinterpret_cast<const Vec*>(a)[i];
    Vec yb = reinterpret_cast<const Vec*>(b)[i];
    Vec y;
#pragma unroll
    for (int j = 0; j < VT::width; ++j)
      VT::store(y, j, op(VT::load(x, j), VT::load(yb, j)));
    reinterpret_cast<Vec*>(o)[i] = y;
  }
  const long tail0 = nvec * VT::width;
  for (long i = tail0 + blockIdx.x * (long)blockDim.x + threadIdx.x; i < n;
       i += stride)
    VT::store1(o + i, op(VT::load1(a + i), VT::load1(b + i)));
}

struct ReluOp {
  DLX_DEV float operator()(float x) const { return fmaxf(x, 0.f); }
};
struct AddReluOp {
  DLX_DEV float operator()(float x, float y) const {
    return fmaxf(x + y, 0.f);
  }
};
// grad * (out > 0): a = grad, b = saved output
struct ReluBwdOp {
  DLX_DEV float operator()(float g, float y) const {
    return y > 0.f ? g : 0.f;
  }
};

template <typename T>
void launch_ew1_relu(const T* a, T* o, long n, hipStream_t s) {
  ew1_kernel<T, ReluOp><<<grid_for(n / VecTraits<T>::width + 1), kBlock, 0, s>>>(
      a, o, n, ReluOp{});
}
template void launch_ew1_relu<uint16_t>(const uint16_t*, uint16_t*, long, hipStream_t);
template void launch_ew1_relu<float>(const float*, float*, long, hipStream_t);

template <typename T>
void launch_add_relu(const T* a, const T* b, T* o, long n, hipStream_t s) {
  ew2_kernel<T, AddReluOp><<<grid_for(n / VecTraits<T>::width + 1), kBlock, 0, s>>>(
      a, b, o, n, AddReluOp{});
}
template void launch_add_relu<uint16_t>(const uint16_t*, const uint16_t*, uint16_t*, long, hipStream_t);
template void launch_add_relu<float>(const float*, const float*, float*, long, hipStream_t);

template <typename T>
void launch_relu_bwd(const T* g, const T* y, T* o, long n, hipStream_t s) {
  ew2_kernel<T, ReluBwdOp><<<grid_for(n / VecTraits<T>::width + 1), kBlock, 0, s>>>(
      g, y, o, n, ReluBwdOp{});
}
template void launch_relu_bwd<uint16_t>(const uint16_t*, const uint16_t*, uint16_t*, long, hipStream_t);
template void launch_relu_bwd<float>(const float*, const float*, float*, long, hipStream_t);

}  // namespace dlx
