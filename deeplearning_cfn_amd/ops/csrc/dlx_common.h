// Common device helpers for the gfx950 (CDNA4) kernels.
//
// Conventions (per the CDNA HIP programming guide):
//  * wavefront = 64 lanes; block sizes are multiples of 64 (256 default);
//  * bf16 traffic is vectorized 8-wide (16 B/lane) — hipcc does not
//    auto-vectorize scalar bf16 loads;
//  * memory-bound grids are capped (~2048 blocks) with grid-stride loops;
//  * fp32 accumulation everywhere stats/reductions appear.
#pragma once

#include <hip/hip_runtime.h>
#include <cstdint>

#define DLX_DEV __device__ __forceinline__

namespace dlx {

constexpr int kWave = 64;
constexpr int kBlock = 256;
constexpr int kMaxGrid = 2048;  // G11: cap + grid-stride for mem-bound ops

static inline int grid_for(long items, int block = kBlock, int cap = kMaxGrid) {
  long b = (items + block - 1) / block;
  if (b < 1) b = 1;
  if (b > cap) b = cap;
  return static_cast<int>(b);
}

// ---- bf16 scalar conversions (bit-exact with torch's bfloat16) ----------
DLX_DEV float bf2f(uint16_t h) {
  union { uint32_t u; float f; } v;
  v.u = static_cast<uint32_t>(h) << 16;
  return v.f;
}

DLX_DEV uint16_t f2bf(float f) {
  union { float f; uint32_t u; } v;
  v.f = f;
  if ((v.u & 0x7fffffffu) > 0x7f800000u) return 0x7fc0u;  // NaN
  uint32_t rounding = 0x7fffu + ((v.u >> 16) & 1u);       // round-to-nearest-even
  return static_cast<uint16_t>((v.u + rounding) >> 16);
}

// ---- 16-byte vector views ----------------------------------------------
// A 16-B chunk is 8 bf16 or 4 f32 — one coalesced load per lane.
struct alignas(16) BF16x8 { uint16_t v[8]; };
struct alignas(16) F32x4  { float v[4]; };

template <typename T> struct VecTraits;
template <> struct VecTraits<uint16_t> {
  using Vec = BF16x8;
  static constexpr int width = 8;
  static DLX_DEV float load(const Vec& x, int i) { return bf2f(x.v[i]); }
  static DLX_DEV void store(Vec& x, int i, float f) { x.v[i] = f2bf(f); }
  static DLX_DEV float load1(const uint16_t* p) { return bf2f(*p); }
  static DLX_DEV void store1(uint16_t* p, float f) { *p = f2bf(f); }
};
template <> struct VecTraits<float> {
  using Vec = F32x4;
  static constexpr int width = 4;
  static DLX_DEV float load(const Vec& x, int i) { return x.v[i]; }
  static DLX_DEV void store(Vec& x, int i, float f) { x.v[i] = f; }
  static DLX_DEV float load1(const float* p) { return *p; }
  static DLX_DEV void store1(float* p, float f) { *p = f; }
};

// ---- wave + block reductions -------------------------------------------
DLX_DEV float wave_sum(float x) {
#pragma unroll
  for (int off = kWave / 2; off > 0; off >>= 1)
    x += __shfl_down(x, off, kWave);
  return x;  // valid in lane 0 of the wave
}

DLX_DEV float wave_max(float x) {
#pragma unroll
  for (int off = kWave / 2; off > 0; off >>= 1)
    x = fmaxf(x, __shfl_down(x, off, kWave));
  return x;
}

// Block reduce via LDS; call with one float per thread, returns total in
// every thread (broadcast). `scratch` must hold blockDim.x/kWave floats.
template <typename Reduce>
DLX_DEV float block_reduce(float x, float* scratch, float init, Reduce r) {
  const int lane = threadIdx.x & (kWave - 1);
  const int wid = threadIdx.x / kWave;
  const int nw = blockDim.x / kWave;
#pragma unroll
  for (int off = kWave / 2; off > 0; off >>= 1)
    x = r(x, __shfl_down(x, off, kWave));
  if (lane == 0) scratch[wid] = x;
  __syncthreads();
  float total = init;
  if (threadIdx.x < nw) total = scratch[threadIdx.x];
#pragma unroll
  for (int off = kWave / 2; off > 0; off >>= 1)
    total = r(total, __shfl_down(total, off, kWave));
  if (threadIdx.x == 0) scratch[0] = total;
  __syncthreads();
  total = scratch[0];
  __syncthreads();
  return total;
}

}  // namespace dlx
