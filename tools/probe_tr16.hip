// Probe: exact semantics of ds_read_b64_tr_b16 on gfx950.
//
// Fills LDS elements with their own index (bf16 of small ints), issues the
// transpose read with a per-lane address, and dumps what each lane holds.
// Build:  hipcc -O1 --offload-arch=gfx950 tools/probe_tr16.hip -o tools/probe_tr16
// Run (GPU box):  ./tools/probe_tr16
#include <hip/hip_runtime.h>
#include <cstdio>

using bf16v4 = __attribute__((ext_vector_type(4))) __bf16;
typedef __attribute__((address_space(3))) bf16v4 lds_bf16v4_t;

__device__ __forceinline__ float bf2f(unsigned short h) {
  union { unsigned u; float f; } v;
  v.u = (unsigned)h << 16;
  return v.f;
}
__device__ __forceinline__ unsigned short f2bf(float f) {
  union { float f; unsigned u; } v;
  v.f = f;
  return (unsigned short)((v.u + 0x8000u) >> 16);
}

// mode 0: per-lane addr = ((l&15) + (l>>4)*64) elements  (guide formula)
// mode 1: uniform addr = 0
// mode 2: per-lane addr = l*4 elements (contiguous 8B per lane)
__global__ void probe(float* out, int mode) {
  __shared__ unsigned short lds[512];
  for (int i = threadIdx.x; i < 512; i += blockDim.x)
    lds[i] = f2bf((float)i);
  __syncthreads();
  const int l = threadIdx.x & 63;
  int e;
  if (mode == 0) e = (l & 15) + (l >> 4) * 64;
  else if (mode == 1) e = 0;
  else e = l * 4;
  bf16v4 v = __builtin_amdgcn_ds_read_tr16_b64_v4bf16(
      (lds_bf16v4_t*)&lds[e]);
  using u16x4 = __attribute__((ext_vector_type(4))) short;
  u16x4 s = __builtin_bit_cast(u16x4, v);
  for (int j = 0; j < 4; ++j)
    out[l * 4 + j] = bf2f((unsigned short)s[j]);
}

int main() {
  float* out;
  (void)hipMalloc(&out, 64 * 4 * sizeof(float));
  float host[256];
  for (int mode = 0; mode < 3; ++mode) {
    hipLaunchKernelGGL(probe, dim3(1), dim3(64), 0, 0, out, mode);
    (void)hipMemcpy(host, out, sizeof(host), hipMemcpyDeviceToHost);
    printf("mode %d:\n", mode);
    for (int l = 0; l < 64; ++l) {
      printf("  lane %2d: %5.0f %5.0f %5.0f %5.0f\n", l, host[l * 4 + 0],
             host[l * 4 + 1], host[l * 4 + 2], host[l * 4 + 3]);
    }
  }
  (void)hipFree(out);
  return 0;
}
