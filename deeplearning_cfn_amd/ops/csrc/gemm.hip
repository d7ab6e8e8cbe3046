// MFMA bf16 GEMM — C[M,N] = A[M,K] · B[N,K]^T  (both operands K-contiguous).
//
// Replaces the reference stack's cuBLAS GEMM: FC head,
// BERT projections, and the engine behind the implicit-GEMM conv path.
// Core: igemm_core.h (unified MFMA kernel, zero-page edge handling,
// 128^2 / 256^2 geometries).
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

// A [M,K] bf16, B [N,K] bf16 -> C [M,N] (bf16 or fp32)
at::Tensor gemm_nt_impl(const at::Tensor& A, const at::Tensor& B,
                        bool out_bf16) {
  TORCH_CHECK(A.dim() == 2 && B.dim() == 2 && A.size(1) == B.size(1),
              "gemm_nt: bad shapes");
  TORCH_CHECK(A.scalar_type() == at::kBFloat16 &&
              B.scalar_type() == at::kBFloat16, "gemm_nt: bf16 only");
  auto Ac = A.contiguous();
  auto Bc = B.contiguous();
  const long M = Ac.size(0), K = Ac.size(1), N = Bc.size(0);
  auto C = at::empty({M, N},
                     Ac.options().dtype(out_bf16 ? at::kBFloat16 : at::kFloat));
  auto zp = zero_page_for(Ac);
  auto s = at::cuda::getCurrentCUDAStream();
  RowGather ag{reinterpret_cast<const uint16_t*>(Ac.data_ptr()), M, K};
  RowGather bg{reinterpret_cast<const uint16_t*>(Bc.data_ptr()), N, K};
  if (out_bf16)
    igemm_launch<RowGather, RowGather, true>(
        ag, bg, C.data_ptr(),
        reinterpret_cast<const uint16_t*>(zp.data_ptr()), M, N, K, s);
  else
    igemm_launch<RowGather, RowGather, false>(
        ag, bg, C.data_ptr(),
        reinterpret_cast<const uint16_t*>(zp.data_ptr()), M, N, K, s);
  return C;
}

}  // namespace dlx
