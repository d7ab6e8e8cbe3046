// Fused softmax + cross-entropy (classifier head).
//
// Replaces the reference stack's framework softmax+CE:
// forward computes probs (fp32) and mean NLL in one pass over the logits;
// backward is the closed-form (probs - onehot) * grad / N.
#include <torch/extension.h>
#include <ATen/cuda/CUDAContext.h>

#include "dlx_common.h"

namespace dlx {

namespace {

struct MaxR { DLX_DEV float operator()(float a, float b) const { return fmaxf(a, b); } };
struct SumR { DLX_DEV float operator()(float a, float b) const { return a + b; } };

// one block per row; C strided by blockDim
template <typename T>
__global__ void softmax_ce_fwd_kernel(const T* __restrict__ logits,
                                      const int64_t* __restrict__ target,
                                      float* __restrict__ probs,
                                      float* __restrict__ rowloss,
                                      int N, int C) {
  __shared__ float scratch[kBlock / kWave];
  const int row = blockIdx.x;
  if (row >= N) return;
  const T* in = logits + (long)row * C;
  float* out = probs + (long)row * C;

  float mx = -INFINITY;
  for (int c = threadIdx.x; c < C; c += blockDim.x)
    mx = fmaxf(mx, VecTraits<T>::load1(in + c));
  mx = block_reduce(mx, scratch, -INFINITY, MaxR{});

  float sum = 0.f;
  for (int c = threadIdx.x; c < C; c += blockDim.x) {
    float e = __expf(VecTraits<T>::load1(in + c) - mx);
    out[c] = e;
    sum += e;
  }
  sum = block_reduce(sum, scratch, 0.f, SumR{});
  const float inv = 1.f / sum;
  for (int c = threadIdx.x; c < C; c += blockDim.x) out[c] *= inv;
  if (threadIdx.x == 0) {
    int64_t t = target[row];
    rowloss[row] = logf(sum) + mx - VecTraits<T>::load1(in + t);
  }
}

template <typename T>
__global__ void softmax_ce_bwd_kernel(const float* __restrict__ probs,
                                      const int64_t* __restrict__ target,
                                      const float* __restrict__ grad,
                                      T* __restrict__ dlogits,
                                      long total, int C) {
  const float scale = grad[0] / (float)(total / C);
  const long stride = (long)gridDim.x * blockDim.x;
  for (long i = blockIdx.x * (long)blockDim.x + threadIdx.x; i < total;
       i += stride) {
    long row = i / C;
    int c = (int)(i - row * C);
    float p = probs[i];
    if (c == (int)target[row]) p -= 1.f;
    VecTraits<T>::store1(dlogits + i, p * scale);
  }
}

}  // namespace

std::tuple<at::Tensor, at::Tensor> softmax_ce_fwd_impl(
    const at::Tensor& logits, const at::Tensor& target) {
  TORCH_CHECK(logits.dim() == 2, "softmax_ce: logits must be 2D");
  TORCH_CHECK(target.scalar_type() == at::kLong, "softmax_ce: target int64");
  const int N = logits.size(0), C = logits.size(1);
  auto opts = logits.options().dtype(at::kFloat);
  auto probs = at::empty({N, C}, opts);
  auto rowloss = at::empty({N}, opts);
  auto s = at::cuda::getCurrentCUDAStream();
  if (logits.scalar_type() == at::kBFloat16) {
    softmax_ce_fwd_kernel<uint16_t><<<N, kBlock, 0, s>>>(
        reinterpret_cast<const uint16_t*>(logits.data_ptr()),
        target.data_ptr<int64_t>(), probs.data_ptr<float>(),
        rowloss.data_ptr<float>(), N, C);
  } else if (logits.scalar_type() == at::kFloat) {
    softmax_ce_fwd_kernel<float><<<N, kBlock, 0, s>>>(
        logits.data_ptr<float>(), target.data_ptr<int64_t>(),
        probs.data_ptr<float>(), rowloss.data_ptr<float>(), N, C);
  } else {
    TORCH_CHECK(false, "softmax_ce: unsupported dtype");
  }
  return {rowloss.mean(), probs};
}

at::Tensor softmax_ce_bwd_impl(const at::Tensor& probs,
                               const at::Tensor& target,
                               const at::Tensor& grad,
                               at::ScalarType out_dtype) {
  const int N = probs.size(0), C = probs.size(1);
  const long total = (long)N * C;
  auto s = at::cuda::getCurrentCUDAStream();
  auto gradc = grad.to(at::kFloat).contiguous();
  // output dtype follows the logits dtype recorded by the caller
  auto dl = at::empty({N, C}, probs.options().dtype(out_dtype));
  int grid = grid_for(total);
  if (out_dtype == at::kBFloat16) {
    softmax_ce_bwd_kernel<uint16_t><<<grid, kBlock, 0, s>>>(
        probs.data_ptr<float>(), target.data_ptr<int64_t>(),
        gradc.data_ptr<float>(),
        reinterpret_cast<uint16_t*>(dl.data_ptr()), total, C);
  } else {
    softmax_ce_bwd_kernel<float><<<grid, kBlock, 0, s>>>(
        probs.data_ptr<float>(), target.data_ptr<int64_t>(),
        gradc.data_ptr<float>(), dl.data_ptr<float>(), total, C);
  }
  return dl;
}

}  // namespace dlx
