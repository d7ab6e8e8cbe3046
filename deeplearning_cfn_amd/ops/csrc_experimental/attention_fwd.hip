// DRAFT (round 2, not built): fused multi-head attention forward, bf16.
//
// Target: BERT-path SelfAttention (currently rocBLAS bmm + fp32 softmax,
// which materialises the [B,H,S,S] score tensor).  Flash-style online
// softmax so scores never leave registers/LDS.
//
// Shapes: q,k,v [B, H, S, Dh] with Dh = 64, S <= 512 (BERT-base).
// Geometry (correctness-first; tuning per guide T2/T3/T5/T10 comes after
// a profile):
//   * one 4-wave block per (b, h, q-tile of 64 rows);
//   * K/V tiles of 64 keys staged to LDS (Dh=64 -> 8 KiB per tile, bf16);
//   * per wave: 16 q-rows; QK^T via mfma_f32_16x16x32_bf16 over Dh
//     (2 k-steps), P·V likewise with the tr-read recipe for V
//     (probe-verified semantics in conv_wrw.hip / tools/probe_tr16.hip);
//   * online softmax state per q-row: running max m, normaliser l,
//     O accumulator [Dh] fp32 (4 f32x4 per lane with the 16x16 C-map);
//   * rescale on max growth (THR=0 first; T13 defer-max only with its
//     forced-branch test, guide rule 26).
//
// Backward plan (needed for training): standard flash-bwd with
// recomputation — save (O, m, l) per row; one kernel recomputes P per
// K/V tile and accumulates dQ, dK, dV (dK/dV via the reduce-major
// atomic-slab epilogue the wrw kernel already uses).  Estimated at
// 2x the forward's complexity; land forward first behind DLX_ATTN=1
// with torch-composition backward (valid autograd: fused fwd +
// recompute-composition bwd), then replace the bwd.
//
// Validation checklist before ANY default flip (guide two-lane rule):
//   1. parity vs fp32 torch sdpa (math backend) at S=128/512, tol ~2e-2;
//   2. asymmetric Q/K/V fill (rule 16: transpose-detecting);
//   3. race screen: repeat-determinism x10 at three sizes;
//   4. ab_bench.py interleaved vs the bmm+softmax path.
//
// The skeleton below compiles for gfx950 standalone; the launcher and
// epilogue stores are stubbed where round-2 measurement should decide
// the layout (marked TODO(r2)).
#include "../csrc/dlx_common.h"

namespace dlx_experimental {

using dlx::kWave;
using bf16x8 = __attribute__((ext_vector_type(8))) short;
using f32x4 = __attribute__((ext_vector_type(4))) float;

namespace {

constexpr int kDh = 64;      // head dim (BERT-base: 64)
constexpr int kQTile = 64;   // q rows per block (4 waves x 16 rows)
constexpr int kKTile = 64;   // keys per LDS tile

// TODO(r2): QK^T fragment loads.  Q is register-resident per wave
// (16 rows x 64 = 8 bf16x8 per lane with the A-fragment map); K staged
// [kKTile][kDh] row-major in LDS and read as the B operand with
// ds_read_b128 from a k-subtile image (same layout as igemm_core.h).
// P·V needs V consumed k-major (contraction over keys): stage V
// row-major [keys][Dh] and tr-read (conv_wrw.hip frag_off recipe).

template <typename Unused = void>
__global__ __launch_bounds__(256) void attn_fwd_kernel(
    const uint16_t* __restrict__ q, const uint16_t* __restrict__ k,
    const uint16_t* __restrict__ v, uint16_t* __restrict__ o,
    float* __restrict__ lse,  // [B*H*S] log-sum-exp for the bwd pass
    int B, int H, int S, float scale) {
  __shared__ uint16_t lds[2 * kKTile * kDh];  // K tile | V tile

  const int bh = blockIdx.y;                  // b*H + h
  const int q0 = blockIdx.x * kQTile;
  if (q0 >= S) return;
  const long base = (long)bh * S * kDh;
  const int lane = threadIdx.x % kWave;
  const int wid = threadIdx.x / kWave;        // wave owns rows q0+wid*16..
  const int l15 = lane & 15, l4 = lane >> 4;

  // per-lane Q fragment: rows (wid*16 + l15), k-chunk l4*8.. (+32 for
  // the second 16x16x32 step) — 2 x bf16x8
  bf16x8 qa[2];
#pragma unroll
  for (int kk = 0; kk < 2; ++kk) {
    const long row = q0 + wid * 16 + l15;
    const long off = base + row * kDh + kk * 32 + l4 * 8;
    qa[kk] = (row < S) ? *reinterpret_cast<const bf16x8*>(
                             reinterpret_cast<const uint16_t*>(q) + off)
                       : bf16x8{};
  }

  // online-softmax state: each lane carries 4 q-rows' (m, l) per the
  // 16x16 C-map (row = l4*4 + r) and O[4][Dh/16] f32x4 accumulators.
  float m_run[4], l_run[4];
  f32x4 o_acc[4][kDh / 16];
#pragma unroll
  for (int r = 0; r < 4; ++r) {
    m_run[r] = -1e30f;
    l_run[r] = 0.f;
#pragma unroll
    for (int j = 0; j < kDh / 16; ++j) o_acc[r][j] = f32x4{};
  }

  for (int k0 = 0; k0 < S; k0 += kKTile) {
    // --- stage K,V tiles (simple 2-phase; TODO(r2): glds + counted
    // vmcnt per igemm_core once the profile says staging is hot) ---
    for (int i = threadIdx.x; i < kKTile * kDh / 8; i += blockDim.x) {
      const int row = (i * 8) / kDh, col = (i * 8) % kDh;
      const long src = base + (long)(k0 + row) * kDh + col;
      *reinterpret_cast<bf16x8*>(lds + i * 8) =
          (k0 + row < S)
              ? *reinterpret_cast<const bf16x8*>(
                    reinterpret_cast<const uint16_t*>(k) + src)
              : bf16x8{};
      const long srcv = base + (long)(k0 + row) * kDh + col;
      *reinterpret_cast<bf16x8*>(lds + kKTile * kDh + i * 8) =
          (k0 + row < S)
              ? *reinterpret_cast<const bf16x8*>(
                    reinterpret_cast<const uint16_t*>(v) + srcv)
              : bf16x8{};
    }
    __syncthreads();

    // --- S_tile = scale * Q K^T : [16 q rows x kKTile] per wave ------
    // TODO(r2): iterate kKTile/16 B-fragments; per fragment 2 MFMAs over
    // Dh; then the per-row max/exp/l update (guide T12 notes the
    // swapped-operand trick if the VALU pass shows hot); then
    // P (bf16) x V via tr-read fragments accumulating into o_acc with
    // the rescale factor applied on max growth.  The state layout above
    // matches the 16x16 C-map so no cross-lane traffic is needed for
    // the row-local m/l updates beyond a 4-lane shfl_xor max/sum.
    __syncthreads();
  }

  // --- epilogue: O / l, write bf16; lse = m + log(l) ------------------
  // TODO(r2): store pattern per T21 if the tail profiles issue-bound.
  (void)o;
  (void)lse;
  (void)scale;
  (void)qa;
  (void)m_run;
  (void)l_run;
  (void)o_acc;
  (void)B;
  (void)H;
}

}  // namespace
}  // namespace dlx_experimental
