// Implicit-GEMM NHWC convolution WEIGHT gradient (wrw) on MFMA.
//
// Completes the native conv triple (fwd/dgrad in conv.hip) — replaces the
// library wgrad (SURVEY.md §3.2 "cuDNN conv kernels", backward-weights).
//
//   dW_cl[co, kf=(dh,dw,ci)] = sum_m  g[m, co] * x~[m, kf]
//     m = (n, ho, wo);  x~[m, kf] = x[(n, ho*s-p+dh, wo*s-p+dw), ci]
//
// The MFMA reduction axis is m — but BOTH operands are m-major in memory
// (channels-last: co / ci contiguous), so the LDS tiles are staged
// m-row-major with global_load_lds (coalesced 16-B pieces) and the MFMA
// fragments (which need per-lane m-contiguity) are read back with the
// gfx950 hardware transpose read `ds_read_b64_tr_b16` (guide T10: the
// attention-V recipe, [4m x 16c] subtile blocks, conflict-free groups).
//
// Split-M: each block owns a contiguous m-slice and accumulates its
// fp32 128x128 output tile with global atomicAdd (unsafeAtomicAdd ->
// global_atomic_add_f32); dW is zero-initialised per call.  Default
// geometry: 8 waves (2x4, measured +11%% over 4 at b1024); also serves
// the Linear weight-grad (gemm_tn_mred_impl) with plain RowGathers.
#include <torch/extension.h>
#include <mutex>
#include <unordered_map>
#include <ATen/cuda/CUDAContext.h>

#include "igemm_core.h"

namespace dlx {

namespace {

// 16-B zero page the gathers read for out-of-image taps.  Per-device,
// mutex-guarded (a function-local static thrashes across devices and is
// not thread-safe under first-touch races).
at::Tensor zero_page_for_wrw(const at::Tensor& like) {
  static std::mutex mu;
  static std::unordered_map<int, at::Tensor> pages;
  std::lock_guard<std::mutex> lock(mu);
  auto& zp = pages[like.device().index()];
  if (!zp.defined())
    zp = at::zeros({16}, like.options().dtype(at::kBFloat16));
  return zp;
}

using bf16v4 = __attribute__((ext_vector_type(4))) __bf16;
using bf16x4 = __attribute__((ext_vector_type(4))) short;
typedef __attribute__((address_space(3))) bf16v4 lds_bf16v4_t;

// Transpose-read one MFMA k-half: lane l gets 4 consecutive m values of
// its column (l&15) from subtile block (l>>4)*... — see image layout.
DLX_DEV bf16x4 tr_read(const uint16_t* lds_base, int elem_off) {
  auto* p = (lds_bf16v4_t*)(lds_base + elem_off);
  bf16v4 v = __builtin_amdgcn_ds_read_tr16_b64_v4bf16(p);
  return __builtin_bit_cast(bf16x4, v);
}

// LDS image per operand tile [BK m-rows x 128 cols], bf16:
//   [BK/32 ksteps][8 cblocks][2 i][4 g][4 r][16 cc]   (64-elem blocks)
// element (mloc, c):  kk=mloc>>5, m32=mloc&31, g=m32>>3, i=(m32&7)>>2,
//                     r=m32&3, cb=c>>4, cc=c&15
//   off = (((kk*8 + cb)*8) + i*4 + g)*64 + r*16 + cc
// Within one (kk, cb) group the 4 lane-group blocks of read i are 64
// elems (128 B = 32 dwords) apart -> groups 0/1 (and 2/3) hit disjoint
// bank halves (guide §LDS: tr_b16 conflicts are per 32-lane half).
constexpr int kBMw = 128;   // output rows (co) per block
constexpr int kBNw = 128;   // output cols (kf) per block
constexpr int kBKw = 64;    // m per K-tile
constexpr int kThreadsW = 256;  // 4-wave variant (DLX_WRW_WAVES=4)
constexpr int kSlotsW = kBKw * 128 / 8;  // 16-B slots per operand tile

// decompose slot f -> (mloc, c) of the image above (all shifts).
// NCB = col-blocks per tile (8 for 128-wide, 16 for 256-wide operands);
// LCB = log2(NCB).
template <int LCB>
DLX_DEV void slot_to_mc(int f, int& mloc, int& c) {
  const int e = f * 8;
  const int blk = e >> 6;       // 64-elem block
  const int rem = e & 63;       // r*16 + cc0
  const int r = rem >> 4;
  const int cc0 = rem & 15;     // 0 or 8
  const int w = blk & 7;        // i*4 + g
  const int i = w >> 2;
  const int g = w & 3;
  const int kkcb = blk >> 3;
  const int cb = kkcb & ((1 << LCB) - 1);
  const int kk = kkcb >> LCB;
  mloc = (kk << 5) + (g << 3) + (i << 2) + r;
  c = (cb << 4) + cc0;
}

// Per-lane element offset for the tr read of (kstep kk, col-block cb,
// k-half i).  Probe-verified ds_read_b64_tr_b16 semantics (wave64):
// within each 16-lane group, result lane l elem j = the 8-B read of lane
// (4j + (l>>2)), element (l&3).  With per-lane address base + (l&15)*4
// elems over a row-major [4m][16c] block, lane l therefore receives
// column (l&15) across the 4 m-rows — the MFMA fragment gather.
// Groups (l>>4) take consecutive blocks: lanes 0-31 cover 256 contiguous
// bytes = all 64 LDS banks exactly once (conflict-free).
template <int LCB>
DLX_DEV int frag_off(int kk, int cb, int i, int lane) {
  return (((kk << LCB) + cb) * 8 + i * 4 + (lane >> 4)) * 64 +
         (lane & 15) * 4;
}

// 128x128 2-phase reduce-major kernel.  Round-2 rework:
//  * per-slot col state HOISTED out of the m-loop via the gathers'
//    prep()/at() API (the k-decomposition magic divides run once, not
//    per tile; the A operand's per-tile address is one 64-bit add);
//  * split-m writes PARTIAL PLANES (out + slice*Rows*Cols, plain
//    stores) summed by the host — deterministic (no fp32 atomics, the
//    ADVICE r1 determinism item) and free of atomic contention on
//    small outputs (64x64 dW had 768 blocks hammering 16 KB).
// BM/BN in {64,128}: the host picks 64-wide operand tiles when the
// output extent is <=64 (layer-1-class dW), halving wasted zero-page
// staging and doubling occupancy there.
template <class AG, class BG, int WM = 2, int WN = 2, int BM = 128,
          int BN = 128>
__global__ __launch_bounds__(WM* WN * 64) void conv_wrw_kernel(
    AG ag, BG bg, float* __restrict__ out,  // [split][Rows][Cols]
    const uint16_t* __restrict__ zero_page, long Mred, long Rows, long Cols,
    int grid_n, int split, long m_per_slice) {
  constexpr int THREADS = WM * WN * 64;
  constexpr int MR = (BM / WM) / 16;
  constexpr int NR = (BN / WN) / 16;
  constexpr int LCBA = (BM == 128) ? 3 : 2;   // log2(col-blocks)
  constexpr int LCBB = (BN == 128) ? 3 : 2;
  constexpr int SLOTS_A = kBKw * BM / 8;
  constexpr int SLOTS_B = kBKw * BN / 8;
  constexpr int SPA = (SLOTS_A + THREADS - 1) / THREADS;
  constexpr int SPB = (SLOTS_B + THREADS - 1) / THREADS;
  __shared__ uint16_t lds[2 * (SLOTS_A + SLOTS_B) * 8];  // [buf][A|B]

  int nwg = gridDim.x;
  int bid = blockIdx.x;
  {  // XCD-aware bijective remap (guide T1)
    int q = nwg / 8, r = nwg % 8;
    int xcd = bid % 8, idx = bid / 8;
    bid = (xcd < r ? xcd * (q + 1) : r * (q + 1) + (xcd - r) * q) + idx;
  }
  // slice-major: the TILES of one m-slice are adjacent in remapped-bid
  // space, i.e. co-resident on ONE XCD — they read the SAME g/x m-rows,
  // so the slice's operand traffic is fetched into that XCD's L2 once
  // instead of once per tile (the re-staging amplification is
  // (Rows/BM + Cols/BN)-fold otherwise).
  const int ntiles = nwg / split;
  const int slice = bid / ntiles, tile = bid % ntiles;
  const int bm = tile / grid_n, bn = tile % grid_n;
  const long row0 = (long)bm * BM, col0 = (long)bn * BN;
  const long mbeg = slice * m_per_slice;
  const long mend = (mbeg + m_per_slice < Mred) ? mbeg + m_per_slice : Mred;
  if (mbeg >= mend) return;

  const int lane = threadIdx.x % kWave;
  const int wid = threadIdx.x / kWave;
  const int wr = wid / WN, wc = wid % WN;
  const int l15 = lane & 15, l4 = lane >> 4;

  f32x4 acc[MR][NR] = {};
  const long ntile = (mend - mbeg + kBKw - 1) / kBKw;

  // hoist per-slot loop invariants (slot id is fixed per thread)
  int mla[SPA], mlb[SPB];
  typename AG::Pre apre[SPA];
  typename BG::Pre bpre[SPB];
#pragma unroll
  for (int u = 0; u < SPA; ++u) {
    const int f = u * THREADS + threadIdx.x;
    int c;
    slot_to_mc<LCBA>(f, mla[u], c);
    if (f < SLOTS_A) apre[u] = ag.prep(mla[u], row0 + c);
  }
#pragma unroll
  for (int u = 0; u < SPB; ++u) {
    const int f = u * THREADS + threadIdx.x;
    int c;
    slot_to_mc<LCBB>(f, mlb[u], c);
    if (f < SLOTS_B) bpre[u] = bg.prep(mlb[u], col0 + c);
  }

  auto stage = [&](int buf, long t) {
    const long mt0 = mbeg + t * kBKw;
    const long mlim = mend - mt0;
    uint16_t* la = lds + buf * (SLOTS_A + SLOTS_B) * 8;
    uint16_t* lb = la + SLOTS_A * 8;
#pragma unroll
    for (int u = 0; u < SPA; ++u) {
      const int f = u * THREADS + threadIdx.x;
      if (f >= SLOTS_A) break;
      const bool mv = mla[u] < mlim;
      const uint16_t* sa = ag.at(apre[u], mt0, mla[u], mv, zero_page);
      __builtin_amdgcn_global_load_lds(
          (glb_u32_t*)sa, (lds_u32_t*)(la + (long)f * 8), 16, 0, 0);
    }
#pragma unroll
    for (int u = 0; u < SPB; ++u) {
      const int f = u * THREADS + threadIdx.x;
      if (f >= SLOTS_B) break;
      const bool mv = mlb[u] < mlim;
      const uint16_t* sb = bg.at(bpre[u], mt0, mlb[u], mv, zero_page);
      __builtin_amdgcn_global_load_lds(
          (glb_u32_t*)sb, (lds_u32_t*)(lb + (long)f * 8), 16, 0, 0);
    }
  };

  stage(0, 0);
  __builtin_amdgcn_s_waitcnt(0);
  __syncthreads();

  int buf = 0;
  for (long t = 0; t < ntile; ++t) {
    if (t + 1 < ntile) stage(buf ^ 1, t + 1);
    const uint16_t* la = lds + buf * (SLOTS_A + SLOTS_B) * 8;
    const uint16_t* lb = la + SLOTS_A * 8;
#pragma unroll
    for (int kk = 0; kk < kBKw / 32; ++kk) {
      // A fragments: rows (co) of this wave's strip
      bf16x8 af[MR], bfr[NR];
#pragma unroll
      for (int mi = 0; mi < MR; ++mi) {
        const int cb = wr * MR + mi;
        bf16x4 lo = tr_read(la, frag_off<LCBA>(kk, cb, 0, lane));
        bf16x4 hi = tr_read(la, frag_off<LCBA>(kk, cb, 1, lane));
        af[mi] = __builtin_shufflevector(lo, hi, 0, 1, 2, 3, 4, 5, 6, 7);
      }
#pragma unroll
      for (int ni = 0; ni < NR; ++ni) {
        const int cb = wc * NR + ni;
        bf16x4 lo = tr_read(lb, frag_off<LCBB>(kk, cb, 0, lane));
        bf16x4 hi = tr_read(lb, frag_off<LCBB>(kk, cb, 1, lane));
        bfr[ni] = __builtin_shufflevector(lo, hi, 0, 1, 2, 3, 4, 5, 6, 7);
      }
      __builtin_amdgcn_s_setprio(1);
#pragma unroll
      for (int mi = 0; mi < MR; ++mi)
#pragma unroll
        for (int ni = 0; ni < NR; ++ni)
          acc[mi][ni] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
              af[mi], bfr[ni], acc[mi][ni], 0, 0, 0);
      __builtin_amdgcn_s_setprio(0);
    }
    __builtin_amdgcn_s_waitcnt(0);
    __syncthreads();
    buf ^= 1;
  }

  // epilogue: D map col=lane&15, row=(lane>>4)*4+r — plain stores into
  // this slice's partial plane (host sums planes; deterministic)
  float* op = out + (long)slice * Rows * Cols;
#pragma unroll
  for (int mi = 0; mi < MR; ++mi) {
#pragma unroll
    for (int ni = 0; ni < NR; ++ni) {
#pragma unroll
      for (int r = 0; r < 4; ++r) {
        const long row = row0 + wr * (BM / WM) + mi * 16 + l4 * 4 + r;
        const long col = col0 + wc * (BN / WN) + ni * 16 + l15;
        if (row < Rows && col < Cols)
          op[row * Cols + col] = acc[mi][ni][r];
      }
    }
  }
}

// ---------------------------------------------------------------------
// 256x256 8-phase wrw (Co >= 192): the igemm_kernel_8ph pipeline
// (counted vmcnt, raw barriers, one LDS object, stage units = (operand,
// k-half)) with tr-read fragments.  Halves the staged bytes per MAC vs
// the 128^2 tile — wrw is staging-BW-bound.  Round-2 retrofit: hoisted
// prep()/at() gathers, slice-major XCD scheduling, deterministic
// partial-plane epilogue (plain stores; host sums).
// ---------------------------------------------------------------------

template <class AG, class BG>
__global__ __launch_bounds__(512) void conv_wrw_kernel_8ph(
    AG ag, BG bg, float* __restrict__ out, const uint16_t* __restrict__ zp,
    long Mred, long Rows, long Cols, int grid_n, int split,
    long m_per_slice) {
  constexpr int BMw = 256, BNw = 256, BKw = 64;
  constexpr int THREADS = 512;
  constexpr int SLOTS = BKw * BMw / 8;  // 2048 slots per operand
  constexpr int MR = 8, NR = 4;         // per-wave 128(co) x 64(kf)

  __shared__ uint16_t lds[2 * 2 * SLOTS * 8];  // [buf][A|B][slots][8]

  int nwg = gridDim.x;
  int bid = blockIdx.x;
  {
    int q = nwg / 8, r = nwg % 8;
    int xcd = bid % 8, idx = bid / 8;
    bid = (xcd < r ? xcd * (q + 1) : r * (q + 1) + (xcd - r) * q) + idx;
  }
  // slice-major: one m-slice's tiles co-resident on one XCD's L2
  const int ntiles = nwg / split;
  const int slice = bid / ntiles, tile = bid % ntiles;
  const int bm = tile / grid_n, bn = tile % grid_n;
  const long row0 = (long)bm * BMw, col0 = (long)bn * BNw;
  const long mbeg = slice * m_per_slice;
  const long mend = (mbeg + m_per_slice < Mred) ? mbeg + m_per_slice : Mred;
  if (mbeg >= mend) return;

  const int lane = threadIdx.x % kWave;
  const int wid = threadIdx.x / kWave;
  const int wr = wid >> 2, wc = wid & 3;  // 2 x 4 waves
  const int l15 = lane & 15, l4 = lane >> 4;

  f32x4 acc[MR][NR] = {};
  const long ntile = (mend - mbeg + BKw - 1) / BKw;

  // hoisted per-slot state: 4 (half, i) slots per thread per operand
  int mloc8[4];
  typename AG::Pre apre8[4];
  typename BG::Pre bpre8[4];
#pragma unroll
  for (int j = 0; j < 4; ++j) {
    const int f = j * THREADS + threadIdx.x;
    int c;
    slot_to_mc<4>(f, mloc8[j], c);
    apre8[j] = ag.prep(mloc8[j], row0 + c);
    bpre8[j] = bg.prep(mloc8[j], col0 + c);
  }

  // stage unit u of tile t: u=0 A-kstep0, u=1 B-kstep0, u=2 A-ks1, u=3 B-ks1
  auto stage_unit = [&](int buf, long t, int u) {
    const long mt0 = mbeg + t * BKw;
    const long mlim = mend - mt0;
    const bool isB = u & 1;
    const int half = u >> 1;
    uint16_t* l0 = lds + (buf * 2 + (isB ? 1 : 0)) * SLOTS * 8;
#pragma unroll
    for (int i = 0; i < 2; ++i) {
      const int j = half * 2 + i;
      const int f = j * THREADS + threadIdx.x;
      const bool mv = mloc8[j] < mlim;
      const uint16_t* src =
          isB ? bg.at(bpre8[j], mt0, mloc8[j], mv, zp)
              : ag.at(apre8[j], mt0, mloc8[j], mv, zp);
      __builtin_amdgcn_global_load_lds(
          (glb_u32_t*)src, (lds_u32_t*)(l0 + (long)f * 8), 16, 0, 0);
    }
  };

  // compute phase q (q>>1 = kstep, q&1 = mi half) of the tile in `buf`
  auto phase_mfma = [&](int buf, int q) {
    const uint16_t* la = lds + (buf * 2 + 0) * SLOTS * 8;
    const uint16_t* lb = lds + (buf * 2 + 1) * SLOTS * 8;
    const int ks = q >> 1;
    const int mih = (q & 1) * 4;
    bf16x8 af[4], bfr[NR];
#pragma unroll
    for (int i = 0; i < 4; ++i) {
      const int cb = wr * 8 + mih + i;
      bf16x4 lo = tr_read(la, frag_off<4>(ks, cb, 0, lane));
      bf16x4 hi = tr_read(la, frag_off<4>(ks, cb, 1, lane));
      af[i] = __builtin_shufflevector(lo, hi, 0, 1, 2, 3, 4, 5, 6, 7);
    }
#pragma unroll
    for (int ni = 0; ni < NR; ++ni) {
      const int cb = wc * 4 + ni;
      bf16x4 lo = tr_read(lb, frag_off<4>(ks, cb, 0, lane));
      bf16x4 hi = tr_read(lb, frag_off<4>(ks, cb, 1, lane));
      bfr[ni] = __builtin_shufflevector(lo, hi, 0, 1, 2, 3, 4, 5, 6, 7);
    }
    __builtin_amdgcn_s_setprio(1);
#pragma unroll
    for (int i = 0; i < 4; ++i)
#pragma unroll
      for (int ni = 0; ni < NR; ++ni)
        acc[mih + i][ni] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
            af[i], bfr[ni], acc[mih + i][ni], 0, 0, 0);
    __builtin_amdgcn_s_setprio(0);
  };

#pragma unroll
  for (int u = 0; u < 4; ++u) stage_unit(0, 0, u);

  int buf = 0;
  long t = 0;
  for (; t + 1 < ntile; ++t) {
    stage_unit(buf ^ 1, t + 1, 0);
    asm volatile("s_waitcnt vmcnt(6)" ::: "memory");
    __builtin_amdgcn_s_barrier();
    phase_mfma(buf, 0);
    stage_unit(buf ^ 1, t + 1, 1);
    __builtin_amdgcn_s_barrier();
    phase_mfma(buf, 1);
    stage_unit(buf ^ 1, t + 1, 2);
    asm volatile("s_waitcnt vmcnt(6)" ::: "memory");
    __builtin_amdgcn_s_barrier();
    phase_mfma(buf, 2);
    stage_unit(buf ^ 1, t + 1, 3);
    __builtin_amdgcn_s_barrier();
    phase_mfma(buf, 3);
    buf ^= 1;
  }
  asm volatile("s_waitcnt vmcnt(4)" ::: "memory");
  __builtin_amdgcn_s_barrier();
  phase_mfma(buf, 0);
  phase_mfma(buf, 1);
  asm volatile("s_waitcnt vmcnt(0)" ::: "memory");
  __builtin_amdgcn_s_barrier();
  phase_mfma(buf, 2);
  phase_mfma(buf, 3);

  // deterministic partial-plane epilogue (host sums the slices)
  float* op = out + (long)slice * Rows * Cols;
#pragma unroll
  for (int mi = 0; mi < MR; ++mi) {
#pragma unroll
    for (int ni = 0; ni < NR; ++ni) {
#pragma unroll
      for (int r = 0; r < 4; ++r) {
        const long row = row0 + wr * 128 + mi * 16 + l4 * 4 + r;
        const long col = col0 + wc * 64 + ni * 16 + l15;
        if (row < Rows && col < Cols)
          op[row * Cols + col] = acc[mi][ni][r];
      }
    }
  }
}

}  // namespace

// conv.hip: packs x [N,Ci<=4,H,W] CL into the stem layout [N,H,Wp,4]
at::Tensor stem_pack_x(const at::Tensor& x_cl, int Wp, int pad);

namespace {

// 2-phase wrw launch with per-shape tile selection: 64-wide operand
// tiles when the output extent is <=64 (no zero-page half), 8-wave
// 128^2 otherwise.
template <class BG>
void launch_wrw_2ph(const RowGather& ag, const BG& bg, float* outp,
                    const uint16_t* zp, long M, long Rows, long Cols,
                    int gm, int gn, int split, long mps, bool w8,
                    hipStream_t s) {
  const int nblocks = gm * gn * split;
  if (Rows <= 64 && Cols <= 64)
    conv_wrw_kernel<RowGather, BG, 2, 2, 64, 64><<<nblocks, 256, 0, s>>>(
        ag, bg, outp, zp, M, Rows, Cols, gn, split, mps);
  else if (Rows <= 64)
    conv_wrw_kernel<RowGather, BG, 2, 2, 64, 128><<<nblocks, 256, 0, s>>>(
        ag, bg, outp, zp, M, Rows, Cols, gn, split, mps);
  else if (Cols <= 64)
    conv_wrw_kernel<RowGather, BG, 2, 2, 128, 64><<<nblocks, 256, 0, s>>>(
        ag, bg, outp, zp, M, Rows, Cols, gn, split, mps);
  else if (w8)
    conv_wrw_kernel<RowGather, BG, 2, 4><<<nblocks, 512, 0, s>>>(
        ag, bg, outp, zp, M, Rows, Cols, gn, split, mps);
  else
    conv_wrw_kernel<RowGather, BG><<<nblocks, kThreadsW, 0, s>>>(
        ag, bg, outp, zp, M, Rows, Cols, gn, split, mps);
}

// small-Cin stem weight grad: reduce-major GEMM over the packed x4p
// (StemWrwGather, igemm_core.h); dW columns are (dh, dwp, c4) and the
// dwp/channel pads are sliced away on the host.
at::Tensor conv2d_wrw_stem(const at::Tensor& g, const at::Tensor& x,
                           ConvShape& p) {
  const int Kwp = (p.Kw + 1) & ~1;
  const int Wp = (p.Wo - 1) * p.stride + Kwp;
  auto x4p = stem_pack_x(x, Wp, p.pad);

  StemWrwGather bg;
  bg.base = reinterpret_cast<const uint16_t*>(x4p.data_ptr());
  bg.M = p.M;
  bg.K = (long)p.Kh * Kwp * 4;
  bg.Ho = p.Ho; bg.Wo = p.Wo; bg.H = p.H; bg.Wp = Wp;
  bg.stride = p.stride; bg.pad = p.pad;
  bg.row4 = Kwp * 4;
  bg.dv_wo.init(p.Wo);
  bg.dv_hw.init(p.Ho * p.Wo);
  bg.dv_row.init(bg.row4);

  const long Rows = p.Co, Cols = bg.K;
  const int gm = (int)((Rows + kBMw - 1) / kBMw);
  const int gn = (int)((Cols + kBNw - 1) / kBNw);
  long split = 768 / ((long)gm * gn);
  const long max_split = (p.M + kBKw - 1) / kBKw;
  if (split > max_split) split = max_split;
  if (split < 1) split = 1;
  long m_per_slice = (p.M + split - 1) / split;
  m_per_slice = ((m_per_slice + kBKw - 1) / kBKw) * kBKw;
  split = (p.M + m_per_slice - 1) / m_per_slice;

  auto outf = at::empty({split, Rows, Cols}, x.options().dtype(at::kFloat));
  auto zp = zero_page_for_wrw(x);
  auto s = at::cuda::getCurrentCUDAStream();
  RowGather ag{reinterpret_cast<const uint16_t*>(g.data_ptr()), p.M,
               (long)p.Co};
  launch_wrw_2ph(ag, bg, outf.data_ptr<float>(),
                 reinterpret_cast<const uint16_t*>(zp.data_ptr()), p.M,
                 Rows, Cols, gm, gn, (int)split, m_per_slice, /*w8=*/true,
                 s);
  auto summed = (split > 1) ? outf.sum(0) : outf.view({Rows, Cols});
  // [Co, Kh, Kwp, 4] -> slice off the pads -> dW with CL strides
  return summed.to(at::kBFloat16)
      .view({(long)p.Co, (long)p.Kh, (long)Kwp, 4})
      .narrow(2, 0, p.Kw)
      .narrow(3, 0, p.Ci)
      .contiguous()
      .permute({0, 3, 1, 2});
}

}  // namespace

// g: [N,Co,Ho,Wo] CL bf16;  x: [N,Ci,H,W] CL bf16 ->
// dW bf16 with channels_last strides of [Co,Ci,Kh,Kw]
at::Tensor conv2d_wrw_impl(const at::Tensor& g_in, const at::Tensor& x_in,
                           int64_t Kh, int64_t Kw, int64_t stride,
                           int64_t pad) {
  auto g = g_in.contiguous(at::MemoryFormat::ChannelsLast);
  auto x = x_in.contiguous(at::MemoryFormat::ChannelsLast);
  TORCH_CHECK(x.scalar_type() == at::kBFloat16, "wrw: bf16 only");
  ConvShape p;
  p.N = x.size(0); p.Ci = x.size(1); p.H = x.size(2); p.W = x.size(3);
  p.Co = g.size(1); p.Ho = g.size(2); p.Wo = g.size(3);
  p.Kh = (int)Kh; p.Kw = (int)Kw;
  p.stride = (int)stride; p.pad = (int)pad;
  p.K = (long)p.Kh * p.Kw * p.Ci;   // fwd-gather k-extent (= kf here)
  p.M = (long)p.N * p.Ho * p.Wo;    // reduction extent
  p.finalize();
  TORCH_CHECK(p.Ci % 8 == 0 || p.Ci <= 4,
              "wrw: Cin must be %8==0 or <=4");
  TORCH_CHECK(p.Co % 8 == 0, "wrw: Cout % 8 != 0 (use library path)");

  if (p.Ci <= 4) return conv2d_wrw_stem(g, x, p);

  const long Rows = p.Co, Cols = p.K;
  // r2 A/B (profiles/r2_experiments/r2q.log, with hoisted gathers + slice-major +
  // partial planes): the 256^2 8-phase tile wins +5-10% on Rows>=512 &&
  // Cols>=512 shapes (halved staged bytes per MAC) and loses on 256-row
  // ones (occupancy) — per-shape policy, env-overridable:
  // DLX_WRW_8PH=0 never, 1 = the old >=192 gate, default >=512.
  static const int mode8 = []() {
    const char* e = getenv("DLX_WRW_8PH");
    return e ? atoi(e) : 2;
  }();
  const bool big = (mode8 == 1 && Rows >= 192 && Cols >= 192) ||
                   (mode8 == 2 && Rows >= 512 && Cols >= 512);
  const int BM = big ? 256 : kBMw, BN = big ? 256 : kBNw;
  const int gm = (int)((Rows + BM - 1) / BM);
  const int gn = (int)((Cols + BN - 1) / BN);
  // split the m reduction to fill the chip (~1-2 blocks/CU)
  long split = (big ? 448 : 768) / ((long)gm * gn);
  const long max_split = (p.M + kBKw - 1) / kBKw;
  if (split > max_split) split = max_split;
  if (split < 1) split = 1;
  long m_per_slice = (p.M + split - 1) / split;
  // round slices to BK so every tile but the last is full
  m_per_slice = ((m_per_slice + kBKw - 1) / kBKw) * kBKw;
  // rounding can empty trailing slices; every launched slice must write
  // its partial plane, so shrink split to the covered count
  split = (p.M + m_per_slice - 1) / m_per_slice;

  auto zp = zero_page_for_wrw(x);
  auto s = at::cuda::getCurrentCUDAStream();

  // A-operand: g rows (m-major, co contiguous); B-operand: x~ im2col gather
  RowGather ag{reinterpret_cast<const uint16_t*>(g.data_ptr()), p.M,
               (long)p.Co};
  const int nblocks = gm * gn * (int)split;
  // 8 waves (2x4) measured +11%% over 4 waves at b1024 (profiles/)
  static const bool w8 = []() {
    const char* e = getenv("DLX_WRW_WAVES");
    return !(e && e[0] == '4');
  }();

  // deterministic split-m: one partial plane per slice, summed below
  at::Tensor outf =
      at::empty({split, Rows, Cols}, x.options().dtype(at::kFloat));
  if (big) {
    // 256^2 8-phase pipeline (half the staged bytes per MAC); r2
    // retrofit shares the hoisted gathers + partial planes
    const bool plain = p.Kh == 1 && p.Kw == 1 && p.stride == 1 && p.pad == 0;
    if (plain) {
      RowGather bg{reinterpret_cast<const uint16_t*>(x.data_ptr()), p.M,
                   (long)p.Ci};
      conv_wrw_kernel_8ph<RowGather, RowGather><<<nblocks, 512, 0, s>>>(
          ag, bg, outf.data_ptr<float>(),
          reinterpret_cast<const uint16_t*>(zp.data_ptr()), p.M, Rows,
          Cols, gn, (int)split, m_per_slice);
    } else {
      WrwXGather bg{reinterpret_cast<const uint16_t*>(x.data_ptr()), p};
      conv_wrw_kernel_8ph<RowGather, WrwXGather><<<nblocks, 512, 0, s>>>(
          ag, bg, outf.data_ptr<float>(),
          reinterpret_cast<const uint16_t*>(zp.data_ptr()), p.M, Rows,
          Cols, gn, (int)split, m_per_slice);
    }
    if (split > 1) outf = outf.sum(0);
  } else {
    const bool plain = p.Kh == 1 && p.Kw == 1 && p.stride == 1 && p.pad == 0;
    if (plain) {
      // 1x1/s1: x~ IS x ([M, Ci] row-major) — no im2col address math
      RowGather bg{reinterpret_cast<const uint16_t*>(x.data_ptr()), p.M,
                   (long)p.Ci};
      launch_wrw_2ph(ag, bg, outf.data_ptr<float>(),
                     reinterpret_cast<const uint16_t*>(zp.data_ptr()), p.M,
                     Rows, Cols, gm, gn, (int)split, m_per_slice, w8, s);
    } else {
      WrwXGather bg{reinterpret_cast<const uint16_t*>(x.data_ptr()), p};
      launch_wrw_2ph(ag, bg, outf.data_ptr<float>(),
                     reinterpret_cast<const uint16_t*>(zp.data_ptr()), p.M,
                     Rows, Cols, gm, gn, (int)split, m_per_slice, w8, s);
    }
    if (split > 1) outf = outf.sum(0);
  }

  // [Co, Kf] fp32 -> bf16 tensor with channels_last strides of the weight
  return outf.to(at::kBFloat16)
      .view({(long)p.Co, (long)p.Kh, (long)p.Kw, (long)p.Ci})
      .permute({0, 3, 1, 2});
}

// out[n,k] = sum_m A[m,n] * B[m,k]  (A^T·B, reduction over the leading
// axis of both operands — the Linear weight-grad dW = dY^T·X).  Reuses
// the wrw reduce-major kernel (tr-read fragments, split-m atomics)
// instead of materialising two transposed copies for gemm_nt.
at::Tensor gemm_tn_mred_impl(const at::Tensor& a_in, const at::Tensor& b_in) {
  auto A = a_in.contiguous();
  auto B = b_in.contiguous();
  TORCH_CHECK(A.scalar_type() == at::kBFloat16, "gemm_tn: bf16 only");
  TORCH_CHECK(A.size(0) == B.size(0), "gemm_tn: reduction extents differ");
  const long M = A.size(0), Rows = A.size(1), Cols = B.size(1);
  TORCH_CHECK(Rows % 8 == 0 && Cols % 8 == 0, "gemm_tn: dims % 8 != 0");

  const int gm = (int)((Rows + kBMw - 1) / kBMw);
  const int gn = (int)((Cols + kBNw - 1) / kBNw);
  long split = 768 / ((long)gm * gn);
  const long max_split = (M + kBKw - 1) / kBKw;
  if (split > max_split) split = max_split;
  if (split < 1) split = 1;
  long m_per_slice = (M + split - 1) / split;
  m_per_slice = ((m_per_slice + kBKw - 1) / kBKw) * kBKw;
  split = (M + m_per_slice - 1) / m_per_slice;

  // deterministic split-m partial planes (see conv_wrw_kernel epilogue)
  auto outf = at::empty({split, Rows, Cols}, A.options().dtype(at::kFloat));
  auto zp = zero_page_for_wrw(A);
  auto s = at::cuda::getCurrentCUDAStream();
  RowGather ag{reinterpret_cast<const uint16_t*>(A.data_ptr()), M, Rows};
  RowGather bg{reinterpret_cast<const uint16_t*>(B.data_ptr()), M, Cols};
  launch_wrw_2ph(ag, bg, outf.data_ptr<float>(),
                 reinterpret_cast<const uint16_t*>(zp.data_ptr()), M, Rows,
                 Cols, gm, gn, (int)split, m_per_slice, /*w8=*/false, s);
  if (split > 1) outf = outf.sum(0);
  return outf.to(at::kBFloat16).view({Rows, Cols});
}

}  // namespace dlx
