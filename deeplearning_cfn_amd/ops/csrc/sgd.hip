// Fused optimizers over the engine's flat gradient buckets.
//
// Replaces the reference stack's server-side SGD aggregation / framework
// optimizer kernels (SURVEY.md §3.2 "framework elementwise/optimizer"):
// one launch updates an entire bucket (weight decay + momentum + nesterov
// + lr in a single pass, fp32 momentum and optional fp32 master weights
// for bf16 params).
#include <torch/extension.h>

#include "dlx_common.h"

namespace dlx {

// W elements per thread iteration: bf16 params load 8 B, momentum 16+16 B.
constexpr int kW = 4;

struct alignas(8) BF16x4 { uint16_t v[4]; };

template <typename T> DLX_DEV void ldvec(const T* p, float* out);
template <> DLX_DEV void ldvec<uint16_t>(const uint16_t* p, float* out) {
  BF16x4 x = *reinterpret_cast<const BF16x4*>(p);
#pragma unroll
  for (int i = 0; i < kW; ++i) out[i] = bf2f(x.v[i]);
}
template <> DLX_DEV void ldvec<float>(const float* p, float* out) {
  F32x4 x = *reinterpret_cast<const F32x4*>(p);
#pragma unroll
  for (int i = 0; i < kW; ++i) out[i] = x.v[i];
}
template <typename T> DLX_DEV void stvec(T* p, const float* in);
template <> DLX_DEV void stvec<uint16_t>(uint16_t* p, const float* in) {
  BF16x4 x;
#pragma unroll
  for (int i = 0; i < kW; ++i) x.v[i] = f2bf(in[i]);
  *reinterpret_cast<BF16x4*>(p) = x;
}
template <> DLX_DEV void stvec<float>(float* p, const float* in) {
  F32x4 x;
#pragma unroll
  for (int i = 0; i < kW; ++i) x.v[i] = in[i];
  *reinterpret_cast<F32x4*>(p) = x;
}

template <typename T, bool MASTER, bool NESTEROV>
__global__ void sgd_kernel(T* __restrict__ p, const T* __restrict__ g,
                           float* __restrict__ m, float* __restrict__ master,
                           long n, float lr, float mu, float wd) {
  const long stride = (long)gridDim.x * blockDim.x;
  const long nvec = n / kW;
  for (long i = blockIdx.x * (long)blockDim.x + threadIdx.x; i < nvec;
       i += stride) {
    float pv[kW], gv[kW], mv[kW];
    if (MASTER) ldvec(master + i * kW, pv); else ldvec(p + i * kW, pv);
    ldvec(g + i * kW, gv);
    ldvec(m + i * kW, mv);
#pragma unroll
    for (int j = 0; j < kW; ++j) {
      float gr = gv[j] + wd * pv[j];
      mv[j] = mu * mv[j] + gr;
      float upd = NESTEROV ? gr + mu * mv[j] : mv[j];
      pv[j] -= lr * upd;
    }
    stvec(m + i * kW, mv);
    if (MASTER) stvec(master + i * kW, pv);
    stvec(p + i * kW, pv);
  }
  // scalar tail
  const long t0 = nvec * kW;
  for (long i = t0 + blockIdx.x * (long)blockDim.x + threadIdx.x; i < n;
       i += stride) {
    float pv = MASTER ? master[i] : VecTraits<T>::load1(p + i);
    float gr = VecTraits<T>::load1(g + i) + wd * pv;
    float mv = mu * m[i] + gr;
    float upd = NESTEROV ? gr + mu * mv : mv;
    pv -= lr * upd;
    m[i] = mv;
    if (MASTER) master[i] = pv;
    VecTraits<T>::store1(p + i, pv);
  }
}

template <typename T, bool MASTER>
__global__ void adamw_kernel(T* __restrict__ p, const T* __restrict__ g,
                             float* __restrict__ m, float* __restrict__ v,
                             float* __restrict__ master, long n, float lr,
                             float b1, float b2, float eps, float wd,
                             float inv_bc1, float inv_bc2) {
  const long stride = (long)gridDim.x * blockDim.x;
  const long nvec = n / kW;
  for (long i = blockIdx.x * (long)blockDim.x + threadIdx.x; i < nvec;
       i += stride) {
    float pv[kW], gv[kW], mv[kW], vv[kW];
    if (MASTER) ldvec(master + i * kW, pv); else ldvec(p + i * kW, pv);
    ldvec(g + i * kW, gv);
    ldvec(m + i * kW, mv);
    ldvec(v + i * kW, vv);
#pragma unroll
    for (int j = 0; j < kW; ++j) {
      mv[j] = b1 * mv[j] + (1.f - b1) * gv[j];
      vv[j] = b2 * vv[j] + (1.f - b2) * gv[j] * gv[j];
      float mh = mv[j] * inv_bc1;
      float vh = vv[j] * inv_bc2;
      pv[j] -= lr * (wd * pv[j] + mh / (sqrtf(vh) + eps));
    }
    stvec(m + i * kW, mv);
    stvec(v + i * kW, vv);
    if (MASTER) stvec(master + i * kW, pv);
    stvec(p + i * kW, pv);
  }
  const long t0 = nvec * kW;
  for (long i = t0 + blockIdx.x * (long)blockDim.x + threadIdx.x; i < n;
       i += stride) {
    float pv = MASTER ? master[i] : VecTraits<T>::load1(p + i);
    float gv = VecTraits<T>::load1(g + i);
    float mv = b1 * m[i] + (1.f - b1) * gv;
    float vv = b2 * v[i] + (1.f - b2) * gv * gv;
    pv -= lr * (wd * pv + (mv * inv_bc1) / (sqrtf(vv * inv_bc2) + eps));
    m[i] = mv;
    v[i] = vv;
    if (MASTER) master[i] = pv;
    VecTraits<T>::store1(p + i, pv);
  }
}

template <typename T>
static void sgd_launch_t(at::Tensor& p, const at::Tensor& g, at::Tensor& m,
                         const at::Tensor* master, double lr, double mu,
                         double wd, bool nesterov, hipStream_t s) {
  long n = p.numel();
  int grid = grid_for(n / kW + 1);
  T* pp = reinterpret_cast<T*>(p.data_ptr());
  const T* gp = reinterpret_cast<const T*>(g.data_ptr());
  float* mp = m.data_ptr<float>();
  float* ms = master ? master->data_ptr<float>() : nullptr;
  if (master && nesterov)
    sgd_kernel<T, true, true><<<grid, kBlock, 0, s>>>(pp, gp, mp, ms, n, lr, mu, wd);
  else if (master)
    sgd_kernel<T, true, false><<<grid, kBlock, 0, s>>>(pp, gp, mp, ms, n, lr, mu, wd);
  else if (nesterov)
    sgd_kernel<T, false, true><<<grid, kBlock, 0, s>>>(pp, gp, mp, ms, n, lr, mu, wd);
  else
    sgd_kernel<T, false, false><<<grid, kBlock, 0, s>>>(pp, gp, mp, ms, n, lr, mu, wd);
}

void launch_sgd_flat(at::Tensor& p, const at::Tensor& g, at::Tensor& m,
                     const at::Tensor* master, double lr, double mu,
                     double wd, bool nesterov, hipStream_t s) {
  TORCH_CHECK(m.scalar_type() == at::kFloat, "sgd: momentum must be fp32");
  TORCH_CHECK(p.numel() == g.numel() && p.numel() == m.numel(),
              "sgd: size mismatch");
  if (master)
    TORCH_CHECK(master->scalar_type() == at::kFloat &&
                master->numel() == p.numel(), "sgd: bad master");
  if (p.scalar_type() == at::kBFloat16)
    sgd_launch_t<uint16_t>(p, g, m, master, lr, mu, wd, nesterov, s);
  else if (p.scalar_type() == at::kFloat)
    sgd_launch_t<float>(p, g, m, master, lr, mu, wd, nesterov, s);
  else
    TORCH_CHECK(false, "sgd: unsupported dtype ", p.scalar_type());
}

template <typename T>
static void adamw_launch_t(at::Tensor& p, const at::Tensor& g, at::Tensor& m,
                           at::Tensor& v, const at::Tensor* master, double lr,
                           double b1, double b2, double eps, double wd,
                           double bc1, double bc2, hipStream_t s) {
  long n = p.numel();
  int grid = grid_for(n / kW + 1);
  T* pp = reinterpret_cast<T*>(p.data_ptr());
  const T* gp = reinterpret_cast<const T*>(g.data_ptr());
  float* mp = m.data_ptr<float>();
  float* vp = v.data_ptr<float>();
  float* ms = master ? master->data_ptr<float>() : nullptr;
  float ib1 = 1.0 / bc1, ib2 = 1.0 / bc2;
  if (master)
    adamw_kernel<T, true><<<grid, kBlock, 0, s>>>(pp, gp, mp, vp, ms, n, lr,
                                                  b1, b2, eps, wd, ib1, ib2);
  else
    adamw_kernel<T, false><<<grid, kBlock, 0, s>>>(pp, gp, mp, vp, ms, n, lr,
                                                   b1, b2, eps, wd, ib1, ib2);
}

void launch_adamw_flat(at::Tensor& p, const at::Tensor& g, at::Tensor& m,
                       at::Tensor& v, const at::Tensor* master, double lr,
                       double b1, double b2, double eps, double wd,
                       double bc1, double bc2, hipStream_t s) {
  TORCH_CHECK(m.scalar_type() == at::kFloat && v.scalar_type() == at::kFloat,
              "adamw: m/v must be fp32");
  if (p.scalar_type() == at::kBFloat16)
    adamw_launch_t<uint16_t>(p, g, m, v, master, lr, b1, b2, eps, wd, bc1, bc2, s);
  else if (p.scalar_type() == at::kFloat)
    adamw_launch_t<float>(p, g, m, v, master, lr, b1, b2, eps, wd, bc1, bc2, s);
  else
    TORCH_CHECK(false, "adamw: unsupported dtype ", p.scalar_type());
}

}  // namespace dlx
