// MFMA bf16 GEMM — C[M,N] = A[M,K] · B[N,K]^T  (both operands K-contiguous).
//
// Replaces the reference stack's cuBLAS GEMM (SURVEY.md §3.2 "cuBLAS
// GEMM" row): FC head, BERT projections; also the inner engine of the
// implicit-GEMM conv path.  Structure: the guide's (§5) canonical CDNA
// anatomy with the "minimum 2-phase" pipeline —
//   * 128x128 block tile, BK=64, 4 waves (2x2), 64x64 per wave,
//     v_mfma_f32_16x16x32_bf16, fp32 accumulators;
//   * LDS k-subtile layout [BK/8][BM][bf16x8]: fragment ds_read_b128 is
//     conflict-free (16-lane groups hit consecutive 16-B slots) and the
//     layout is lane-linear for global_load_lds staging (guide rule 21);
//   * double-buffered LDS, STAGE(t+1) issued before the ds_read/MFMA of
//     tile t, one vmcnt(0)+barrier per K-tile (guide §5.5 T3 minimum);
//   * XCD-aware bijective blockIdx swizzle (guide §5.5 T1);
//   * edge blocks (M/N/K tails) take a guarded register->LDS staging
//     path; interior blocks take the glds fast path.
#include <torch/extension.h>
#include <ATen/cuda/CUDAContext.h>

#include "dlx_common.h"

namespace dlx {

namespace {

constexpr int BM = 128, BN = 128, BK = 64;
constexpr int WAVES = 4;            // 2x2 waves of 64x64
constexpr int THREADS = WAVES * kWave;
constexpr int FRAG = 16;            // mfma tile edge
constexpr int MR = 4, NR = 4;       // 16x16 frags per wave

using bf16x8 = __attribute__((ext_vector_type(8))) short;
using f32x4 = __attribute__((ext_vector_type(4))) float;

// LDS tile: [BK/8][BM][8] bf16 == flat array of BK/8*BM 16-B slots.
constexpr int SLOTS = (BK / 8) * BM;        // 1024 16-B slots per operand
constexpr int TILE_BYTES = SLOTS * 16;      // 16 KiB

typedef __attribute__((address_space(3))) uint32_t lds_u32_t;
typedef const __attribute__((address_space(1))) uint32_t glb_u32_t;

// one glds instruction stages THREADS*16 B; a tile needs SLOTS/THREADS of them
constexpr int GLDS_PER_TILE = SLOTS / THREADS;  // 4

// ---- staging -----------------------------------------------------------

// interior fast path: global_load_lds, lane-linear dest
DLX_DEV void stage_glds(const uint16_t* gbase, long ld, int row0, int k0,
                        uint16_t* lds) {
  // flat slot f = kc*BM + m ; source = gbase + (row0+m)*ld + k0 + kc*8
#pragma unroll
  for (int i = 0; i < GLDS_PER_TILE; ++i) {
    int f = i * THREADS + threadIdx.x;
    int kc = f / BM;
    int m = f - kc * BM;
    const uint16_t* src = gbase + (long)(row0 + m) * ld + k0 + kc * 8;
    __builtin_amdgcn_global_load_lds(
        (glb_u32_t*)src, (lds_u32_t*)(lds + (long)f * 8), 16, 0, 0);
  }
}

// edge path: guarded scalar loads -> ds_write (zero-fill out of range)
DLX_DEV void stage_guarded(const uint16_t* gbase, long ld, int row0, int k0,
                           int rows, long K, uint16_t* lds) {
#pragma unroll
  for (int i = 0; i < GLDS_PER_TILE; ++i) {
    int f = i * THREADS + threadIdx.x;
    int kc = f / BM;
    int m = f - kc * BM;
    bf16x8 val = {};
    if (m < rows) {
      const uint16_t* src = gbase + (long)(row0 + m) * ld + k0 + kc * 8;
#pragma unroll
      for (int j = 0; j < 8; ++j) {
        long k = (long)k0 + kc * 8 + j;
        reinterpret_cast<uint16_t*>(&val)[j] = (k < K) ? src[j] : (uint16_t)0;
      }
    }
    *reinterpret_cast<bf16x8*>(lds + (long)f * 8) = val;
  }
}

// ---- the kernel --------------------------------------------------------

template <bool EDGE>
__global__ __launch_bounds__(THREADS) void gemm_nt_kernel(
    const uint16_t* __restrict__ A,   // [M,K] row-major
    const uint16_t* __restrict__ B,   // [N,K] row-major
    float* __restrict__ C,            // [M,N] fp32 out
    long M, long N, long K, int grid_n) {
  __shared__ uint16_t lsA[2 * SLOTS * 8];
  __shared__ uint16_t lsB[2 * SLOTS * 8];

  // XCD-aware bijective remap (guide T1): contiguous chunk per XCD
  int nwg = gridDim.x;
  int bid = blockIdx.x;
  {
    int q = nwg / 8, r = nwg % 8;
    int xcd = bid % 8, idx = bid / 8;
    bid = (xcd < r ? xcd * (q + 1) : r * (q + 1) + (xcd - r) * q) + idx;
  }
  const int bm = bid / grid_n, bn = bid % grid_n;
  const int m0 = bm * BM, n0 = bn * BN;

  const int wid = threadIdx.x / kWave;
  const int lane = threadIdx.x % kWave;
  const int wr = wid >> 1, wc = wid & 1;          // 2x2 wave grid
  const int l15 = lane & 15, l4 = lane >> 4;

  f32x4 acc[MR][NR] = {};

  const int rowsA = EDGE ? (int)((M - m0 < BM) ? (M - m0) : BM) : BM;
  const int rowsB = EDGE ? (int)((N - n0 < BN) ? (N - n0) : BN) : BN;
  const long ntile = (K + BK - 1) / BK;

  auto stage = [&](int buf, long t) {
    const int k0 = (int)(t * BK);
    uint16_t* la = lsA + buf * SLOTS * 8;
    uint16_t* lb = lsB + buf * SLOTS * 8;
    if (!EDGE && (k0 + BK) <= K) {
      stage_glds(A, K, m0, k0, la);
      stage_glds(B, K, n0, k0, lb);
    } else {
      stage_guarded(A, K, m0, k0, rowsA, K, la);
      stage_guarded(B, K, n0, k0, rowsB, K, lb);
    }
  };

  stage(0, 0);
  __builtin_amdgcn_s_waitcnt(/*vmcnt(0) lgkmcnt(0)*/ 0);
  __syncthreads();

  int buf = 0;
  for (long t = 0; t < ntile; ++t) {
    if (t + 1 < ntile) stage(buf ^ 1, t + 1);
    const uint16_t* la = lsA + buf * SLOTS * 8;
    const uint16_t* lb = lsB + buf * SLOTS * 8;
#pragma unroll
    for (int ks = 0; ks < BK / 32; ++ks) {
      bf16x8 af[MR], bfr[NR];
#pragma unroll
      for (int mi = 0; mi < MR; ++mi) {
        int m = wr * 64 + mi * FRAG + l15;
        int kc = ks * 4 + l4;
        af[mi] = *reinterpret_cast<const bf16x8*>(la + ((long)kc * BM + m) * 8);
      }
#pragma unroll
      for (int ni = 0; ni < NR; ++ni) {
        int n = wc * 64 + ni * FRAG + l15;
        int kc = ks * 4 + l4;
        bfr[ni] = *reinterpret_cast<const bf16x8*>(lb + ((long)kc * BM + n) * 8);
      }
#pragma unroll
      for (int mi = 0; mi < MR; ++mi)
#pragma unroll
        for (int ni = 0; ni < NR; ++ni)
          acc[mi][ni] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
              af[mi], bfr[ni], acc[mi][ni], 0, 0, 0);
    }
    // one vmcnt(0)+barrier per K-tile (stage t+1 must have landed before
    // the next iteration reads buf^1; writes to buf are done)
    __builtin_amdgcn_s_waitcnt(0);
    __syncthreads();
    buf ^= 1;
  }

  // epilogue: C/D map for 16x16x32: col = lane&15, row = (lane>>4)*4 + r
#pragma unroll
  for (int mi = 0; mi < MR; ++mi) {
#pragma unroll
    for (int ni = 0; ni < NR; ++ni) {
#pragma unroll
      for (int r = 0; r < 4; ++r) {
        long row = m0 + wr * 64 + mi * FRAG + l4 * 4 + r;
        long col = n0 + wc * 64 + ni * FRAG + l15;
        if (!EDGE || (row < M && col < N))
          C[row * N + col] = acc[mi][ni][r];
      }
    }
  }
}

}  // namespace

// A [M,K] bf16, B [N,K] bf16 -> C [M,N] (out_dtype bf16 or fp32)
at::Tensor gemm_nt_impl(const at::Tensor& A, const at::Tensor& B,
                        bool out_bf16) {
  TORCH_CHECK(A.dim() == 2 && B.dim() == 2 && A.size(1) == B.size(1),
              "gemm_nt: bad shapes");
  TORCH_CHECK(A.scalar_type() == at::kBFloat16 &&
              B.scalar_type() == at::kBFloat16, "gemm_nt: bf16 only");
  TORCH_CHECK(A.size(1) % 8 == 0, "gemm_nt: K must be a multiple of 8");
  auto Ac = A.contiguous();
  auto Bc = B.contiguous();
  const long M = Ac.size(0), K = Ac.size(1), N = Bc.size(0);
  auto C = at::empty({M, N}, Ac.options().dtype(at::kFloat));
  const int grid_m = (int)((M + BM - 1) / BM);
  const int grid_n = (int)((N + BN - 1) / BN);
  auto s = at::cuda::getCurrentCUDAStream();
  const bool edge = (M % BM) || (N % BN) || (K % BK);
  const uint16_t* a = reinterpret_cast<const uint16_t*>(Ac.data_ptr());
  const uint16_t* b = reinterpret_cast<const uint16_t*>(Bc.data_ptr());
  if (edge)
    gemm_nt_kernel<true><<<grid_m * grid_n, THREADS, 0, s>>>(
        a, b, C.data_ptr<float>(), M, N, K, grid_n);
  else
    gemm_nt_kernel<false><<<grid_m * grid_n, THREADS, 0, s>>>(
        a, b, C.data_ptr<float>(), M, N, K, grid_n);
  return out_bf16 ? C.to(at::kBFloat16) : C;
}

}  // namespace dlx
