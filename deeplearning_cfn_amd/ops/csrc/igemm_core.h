// Unified MFMA implicit-GEMM core (gfx950).
//
// One templated kernel serves plain GEMM (C = A·B^T, both K-contiguous)
// and NHWC convolution fwd/dgrad (the im2col gather fused into staging):
// every operand element is fetched by global_load_lds with a PER-LANE
// source address (guide §5: the glds source is per-lane; only the LDS
// destination must be lane-linear), and any out-of-range tap (padding,
// M/N/K tails) redirects to a 16-B zero page — no separate edge kernels.
//
// LDS layout per operand tile: [BK/8][ROWS][bf16x8] — k-subtile-major, so
//  * glds destination is lane-linear (slot f = kc*ROWS + row), and
//  * fragment ds_read_b128 is conflict-free: each 16-lane group reads 16
//    CONSECUTIVE 16-B slots (one full 256-B bank row).
//
// Geometries: 256x256 8-phase counted-vmcnt pipeline (8 waves, 128 KiB
// LDS) for big problems; 256x64 (narrow N) and 128x128 2-phase tiles
// otherwise.  Output rows go through an OM mapper (identity, or the
// stride-2 dgrad parity-class pixel scatter).  1x1/s1 convs are routed
// with plain RowGathers (no im2col address math) by the conv host code.
#pragma once

#include <cstdlib>

#include "dlx_common.h"

namespace dlx {

using bf16x8 = __attribute__((ext_vector_type(8))) short;
using f32x4 = __attribute__((ext_vector_type(4))) float;

typedef __attribute__((address_space(3))) uint32_t lds_u32_t;
typedef const __attribute__((address_space(1))) uint32_t glb_u32_t;

// ---- gather policies ---------------------------------------------------

// plain row-major [rows, K] operand (GEMM A/B, conv weights)
struct RowGather {
  const uint16_t* base;
  long rows, K;
  DLX_DEV const uint16_t* src(long row, long k,
                              const uint16_t* zp) const {
    return (row < rows && k < K) ? base + row * K + k : zp;
  }
  // Two-phase API for the reduce-major (wrw) kernel: the column index of
  // a staging slot is loop-invariant across m-tiles, so everything
  // col-dependent is hoisted into prep(); at() is one 64-bit add.
  struct Pre { const uint16_t* pbase; };  // nullptr = col out of range
  DLX_DEV Pre prep(int mloc, long col) const {
    Pre p;
    p.pbase = (col < K) ? base + (long)mloc * K + col : nullptr;
    return p;
  }
  DLX_DEV const uint16_t* at(const Pre& p, long mt0, int /*mloc*/,
                             bool mvalid, const uint16_t* zp) const {
    return (p.pbase && mvalid) ? p.pbase + mt0 * K : zp;
  }
  // Output-major twin (igemm_kernel): a staging slot's ROW is fixed
  // across k-tiles — hoist the row base, oat() is one add + bound.
  struct OPre { const uint16_t* rowbase; };
  DLX_DEV OPre oprep(long row) const {
    OPre p;
    p.rowbase = (row < rows) ? base + row * K : nullptr;
    return p;
  }
  DLX_DEV const uint16_t* oat(const OPre& p, long k,
                              const uint16_t* zp) const {
    return (p.rowbase && k < K) ? p.rowbase + k : zp;
  }
};

// Division by a runtime-constant divisor via magic multiply (round-up
// method: m = ceil(2^k/d), k = 31 + ceil_log2(d); exact for 0<=n<2^31).
// A v_mul_hi_u64+shift replaces the ~20-cycle integer-divide expansion in
// the per-slot gather address math (guide: VALU work beside glds staging).
struct U32Div {
  uint64_t mul;
  int sh;
  void init(uint32_t d) {
    int k = 31;
    while ((1ull << (k - 31)) < d) ++k;   // k = 31 + ceil_log2(d)
    mul = ((1ull << k) + d - 1) / d;
    sh = k;
  }
  DLX_DEV uint32_t div(uint32_t n) const {
    return (uint32_t)(((uint64_t)n * mul) >> sh);
  }
};

struct ConvShape {
  int N, H, W, Ci, Ho, Wo, Co, Kh, Kw, stride, pad;
  long K, M;
  U32Div dv_wo, dv_hw, dv_ci, dv_kw;   // fwd/wrw gather divisors
  U32Div dv_w, dv_hw_in, dv_co, dv_s;  // dgrad gather divisors
  void finalize() {
    dv_wo.init(Wo);
    dv_hw.init(Ho * Wo);
    dv_ci.init(Ci);
    dv_kw.init(Kw);
    dv_w.init(W);
    dv_hw_in.init(H * W);
    dv_co.init(Co);
    dv_s.init(stride);
  }
};

// conv fwd activations: row m=(n,ho,wo), k=(dh,dw,ci)
struct ConvFwdGather {
  const uint16_t* base;
  ConvShape p;
  DLX_DEV const uint16_t* src(long m, long k, const uint16_t* zp) const {
    if (m >= p.M || k >= p.K) return zp;
    const int n = (int)p.dv_hw.div((uint32_t)m);
    const int rem = (int)m - n * (p.Ho * p.Wo);
    const int ho = (int)p.dv_wo.div((uint32_t)rem);
    const int wo = rem - ho * p.Wo;
    const int dhw = (int)p.dv_ci.div((uint32_t)k);
    const int ci = (int)k - dhw * p.Ci;
    const int dh = (int)p.dv_kw.div((uint32_t)dhw);
    const int dw = dhw - dh * p.Kw;
    const int hi = ho * p.stride - p.pad + dh;
    const int wi = wo * p.stride - p.pad + dw;
    if (hi < 0 || hi >= p.H || wi < 0 || wi >= p.W) return zp;
    return base + (((long)n * p.H + hi) * p.W + wi) * p.Ci + ci;
  }
  // output-major hoist: m=(n,ho,wo) is fixed per staging slot
  struct OPre {
    long nbase;  // n*H, or -1 when m out of range
    int hi0;     // ho*stride - pad
    int wi0;     // wo*stride - pad
  };
  DLX_DEV OPre oprep(long m) const {
    OPre r;
    if (m >= p.M) {
      r.nbase = -1;
      return r;
    }
    const uint32_t n = p.dv_hw.div((uint32_t)m);
    const uint32_t rem = (uint32_t)m - n * (uint32_t)(p.Ho * p.Wo);
    const uint32_t ho = p.dv_wo.div(rem);
    const uint32_t wo = rem - ho * (uint32_t)p.Wo;
    r.nbase = (long)n * p.H;
    r.hi0 = (int)ho * p.stride - p.pad;
    r.wi0 = (int)wo * p.stride - p.pad;
    return r;
  }
  DLX_DEV const uint16_t* oat(const OPre& r, long k,
                              const uint16_t* zp) const {
    if (r.nbase < 0 || k >= p.K) return zp;
    const uint32_t dhw = p.dv_ci.div((uint32_t)k);
    const int ci = (int)k - (int)dhw * p.Ci;
    const uint32_t dh = p.dv_kw.div(dhw);
    const int dw = (int)(dhw - dh * p.Kw);
    const int hi = r.hi0 + (int)dh;
    const int wi = r.wi0 + dw;
    if ((unsigned)hi >= (unsigned)p.H || (unsigned)wi >= (unsigned)p.W)
      return zp;
    return base + ((r.nbase + hi) * p.W + wi) * p.Ci + ci;
  }
};

// conv wrw B-operand: x~ im2col with the k-decomposition HOISTED.
// prep() resolves col=(dh,dw,ci) once per staging slot (loop-invariant
// across m-tiles); at() only decomposes m=(n,ho,wo) — 2 magic divides
// instead of 4 plus shorter address math in the hot staging loop.
struct WrwXGather {
  const uint16_t* base;
  ConvShape p;
  struct Pre {
    int ci;
    int dh;   // dh - pad (ready to add to ho*stride)
    int dw;   // dw - pad
    int ok;
  };
  DLX_DEV Pre prep(int /*mloc*/, long col) const {
    Pre r;
    r.ok = col < p.K;
    const uint32_t k = (uint32_t)(r.ok ? col : 0);
    const uint32_t dhw = p.dv_ci.div(k);
    r.ci = (int)k - (int)dhw * p.Ci;
    const uint32_t dh = p.dv_kw.div(dhw);
    r.dh = (int)dh - p.pad;
    r.dw = (int)(dhw - dh * p.Kw) - p.pad;
    return r;
  }
  DLX_DEV const uint16_t* at(const Pre& r, long mt0, int mloc, bool mvalid,
                             const uint16_t* zp) const {
    if (!r.ok || !mvalid) return zp;
    const uint32_t m = (uint32_t)(mt0 + mloc);
    const uint32_t n = p.dv_hw.div(m);
    const uint32_t rem = m - n * (uint32_t)(p.Ho * p.Wo);
    const uint32_t ho = p.dv_wo.div(rem);
    const uint32_t wo = rem - ho * (uint32_t)p.Wo;
    const int hi = (int)ho * p.stride + r.dh;
    const int wi = (int)wo * p.stride + r.dw;
    if ((unsigned)hi >= (unsigned)p.H || (unsigned)wi >= (unsigned)p.W)
      return zp;
    return base + (((long)n * p.H + hi) * p.W + wi) * p.Ci + r.ci;
  }
};

// Small-Cin ("stem") conv gathers over a channel+width padded copy of x:
// x4p [N, H, Wp=(Wo-1)*stride+Kwp, 4] bf16, where Kwp = Kw rounded up to
// even and channels padded to 4 with zeros (pack_stem kernel).  The k
// axis is (dh, dwp, c4) with a 16-B slot = 2 adjacent dwp taps x 4
// channels — CONTIGUOUS in x4p (the property plain NHWC c=3 lacks), so
// glds staging works unchanged.  Width padding bakes the wi bounds in;
// only the hi bound remains a zero-page redirect.  (VERDICT r1 #4.)
struct StemGather {
  const uint16_t* base;  // x4p
  long M, K;             // K = Kh * Kwp * 4
  int Ho, Wo, H, Wp, stride, pad, row4;  // row4 = Kwp*4
  U32Div dv_wo, dv_hw, dv_row;
  DLX_DEV const uint16_t* src(long m, long k, const uint16_t* zp) const {
    if (m >= M || k >= K) return zp;
    const uint32_t n = dv_hw.div((uint32_t)m);
    const uint32_t rem = (uint32_t)m - n * (uint32_t)(Ho * Wo);
    const uint32_t ho = dv_wo.div(rem);
    const uint32_t wo = rem - ho * (uint32_t)Wo;
    const uint32_t dh = dv_row.div((uint32_t)k);
    const uint32_t krem = (uint32_t)k - dh * (uint32_t)row4;  // dwp*4 + c
    const int hi = (int)ho * stride - pad + (int)dh;
    if ((unsigned)hi >= (unsigned)H) return zp;
    // col in x4p = wo*stride + dwp (pad folded into the packing)
    return base + (((long)n * H + hi) * Wp + (long)wo * stride) * 4 + krem;
  }
  struct OPre {
    long nbase;    // n*H, or -1 when m out of range
    long colbase;  // wo*stride*4
    int hi0;       // ho*stride - pad
  };
  DLX_DEV OPre oprep(long m) const {
    OPre r;
    if (m >= M) {
      r.nbase = -1;
      return r;
    }
    const uint32_t n = dv_hw.div((uint32_t)m);
    const uint32_t rem = (uint32_t)m - n * (uint32_t)(Ho * Wo);
    const uint32_t ho = dv_wo.div(rem);
    const uint32_t wo = rem - ho * (uint32_t)Wo;
    r.nbase = (long)n * H;
    r.colbase = (long)wo * stride * 4;
    r.hi0 = (int)ho * stride - pad;
    return r;
  }
  DLX_DEV const uint16_t* oat(const OPre& r, long k,
                              const uint16_t* zp) const {
    if (r.nbase < 0 || k >= K) return zp;
    const uint32_t dh = dv_row.div((uint32_t)k);
    const uint32_t krem = (uint32_t)k - dh * (uint32_t)row4;
    const int hi = r.hi0 + (int)dh;
    if ((unsigned)hi >= (unsigned)H) return zp;
    return base + ((r.nbase + hi) * Wp) * 4 + r.colbase + krem;
  }
};

// wrw twin with the hoisted two-phase API (reduce-major kernel B operand)
struct StemWrwGather {
  const uint16_t* base;  // x4p
  long M, K;
  int Ho, Wo, H, Wp, stride, pad, row4;
  U32Div dv_wo, dv_hw, dv_row;
  struct Pre {
    int dh;    // dh - pad (ready to add to ho*stride)
    int krem;  // dwp*4 + c
    int ok;
  };
  DLX_DEV Pre prep(int /*mloc*/, long col) const {
    Pre r;
    r.ok = col < K;
    const uint32_t k = (uint32_t)(r.ok ? col : 0);
    const uint32_t dh = dv_row.div(k);
    r.krem = (int)(k - dh * (uint32_t)row4);
    r.dh = (int)dh - pad;
    return r;
  }
  DLX_DEV const uint16_t* at(const Pre& r, long mt0, int mloc, bool mvalid,
                             const uint16_t* zp) const {
    if (!r.ok || !mvalid) return zp;
    const uint32_t m = (uint32_t)(mt0 + mloc);
    const uint32_t n = dv_hw.div(m);
    const uint32_t rem = m - n * (uint32_t)(Ho * Wo);
    const uint32_t ho = dv_wo.div(rem);
    const uint32_t wo = rem - ho * (uint32_t)Wo;
    const int hi = (int)ho * stride + r.dh;
    if ((unsigned)hi >= (unsigned)H) return zp;
    return base + (((long)n * H + hi) * Wp + (long)wo * stride) * 4 +
           r.krem;
  }
};

// conv dgrad: row m=(n,hi,wi) over INPUT pixels, k=(dh,dw,co) from grad
struct ConvDgradGather {
  const uint16_t* base;
  ConvShape p;
  DLX_DEV const uint16_t* src(long m, long k, const uint16_t* zp) const {
    if (m >= p.M || k >= p.K) return zp;
    const int n = (int)p.dv_hw_in.div((uint32_t)m);
    const int rem = (int)m - n * (p.H * p.W);
    const int hi = (int)p.dv_w.div((uint32_t)rem);
    const int wi = rem - hi * p.W;
    const int dhw = (int)p.dv_co.div((uint32_t)k);
    const int co = (int)k - dhw * p.Co;
    const int dh = (int)p.dv_kw.div((uint32_t)dhw);
    const int dw = dhw - dh * p.Kw;
    const int ht = hi + p.pad - dh;
    const int wt = wi + p.pad - dw;
    if (ht < 0 || wt < 0) return zp;
    const int ho = (int)p.dv_s.div((uint32_t)ht);
    const int wo = (int)p.dv_s.div((uint32_t)wt);
    if (ho * p.stride != ht || wo * p.stride != wt) return zp;
    if (ho >= p.Ho || wo >= p.Wo) return zp;
    return base + (((long)n * p.Ho + ho) * p.Wo + wo) * p.Co + co;
  }
  struct OPre {
    long nbase;  // n*Ho, or -1 when m out of range
    int ht0;     // hi + pad
    int wt0;     // wi + pad
  };
  DLX_DEV OPre oprep(long m) const {
    OPre r;
    if (m >= p.M) {
      r.nbase = -1;
      return r;
    }
    const uint32_t n = p.dv_hw_in.div((uint32_t)m);
    const uint32_t rem = (uint32_t)m - n * (uint32_t)(p.H * p.W);
    const uint32_t hi = p.dv_w.div(rem);
    const uint32_t wi = rem - hi * (uint32_t)p.W;
    r.nbase = (long)n * p.Ho;
    r.ht0 = (int)hi + p.pad;
    r.wt0 = (int)wi + p.pad;
    return r;
  }
  DLX_DEV const uint16_t* oat(const OPre& r, long k,
                              const uint16_t* zp) const {
    if (r.nbase < 0 || k >= p.K) return zp;
    const uint32_t dhw = p.dv_co.div((uint32_t)k);
    const int co = (int)k - (int)dhw * p.Co;
    const uint32_t dh = p.dv_kw.div(dhw);
    const int dw = (int)(dhw - dh * p.Kw);
    const int ht = r.ht0 - (int)dh;
    const int wt = r.wt0 - dw;
    if (ht < 0 || wt < 0) return zp;
    const int ho = (int)p.dv_s.div((uint32_t)ht);
    const int wo = (int)p.dv_s.div((uint32_t)wt);
    if (ho * p.stride != ht || wo * p.stride != wt) return zp;
    if (ho >= p.Ho || wo >= p.Wo) return zp;
    return base + ((r.nbase + ho) * p.Wo + wo) * p.Co + co;
  }
};

// stride-1 dgrad: ho == ht, wo == wt — no /stride divides, no exactness
// checks (NEXT.md r1 dgrad lead)
struct ConvDgradS1Gather {
  const uint16_t* base;
  ConvShape p;
  DLX_DEV const uint16_t* src(long m, long k, const uint16_t* zp) const {
    if (m >= p.M || k >= p.K) return zp;
    const int n = (int)p.dv_hw_in.div((uint32_t)m);
    const int rem = (int)m - n * (p.H * p.W);
    const int hi = (int)p.dv_w.div((uint32_t)rem);
    const int wi = rem - hi * p.W;
    const int dhw = (int)p.dv_co.div((uint32_t)k);
    const int co = (int)k - dhw * p.Co;
    const int dh = (int)p.dv_kw.div((uint32_t)dhw);
    const int dw = dhw - dh * p.Kw;
    const int ho = hi + p.pad - dh;
    const int wo = wi + p.pad - dw;
    if ((unsigned)ho >= (unsigned)p.Ho || (unsigned)wo >= (unsigned)p.Wo)
      return zp;
    return base + (((long)n * p.Ho + ho) * p.Wo + wo) * p.Co + co;
  }
  struct OPre {
    long nbase;
    int ht0, wt0;  // hi + pad, wi + pad
  };
  DLX_DEV OPre oprep(long m) const {
    OPre r;
    if (m >= p.M) {
      r.nbase = -1;
      return r;
    }
    const uint32_t n = p.dv_hw_in.div((uint32_t)m);
    const uint32_t rem = (uint32_t)m - n * (uint32_t)(p.H * p.W);
    const uint32_t hi = p.dv_w.div(rem);
    const uint32_t wi = rem - hi * (uint32_t)p.W;
    r.nbase = (long)n * p.Ho;
    r.ht0 = (int)hi + p.pad;
    r.wt0 = (int)wi + p.pad;
    return r;
  }
  DLX_DEV const uint16_t* oat(const OPre& r, long k,
                              const uint16_t* zp) const {
    if (r.nbase < 0 || k >= p.K) return zp;
    const uint32_t dhw = p.dv_co.div((uint32_t)k);
    const int co = (int)k - (int)dhw * p.Co;
    const uint32_t dh = p.dv_kw.div(dhw);
    const int dw = (int)(dhw - dh * p.Kw);
    const int ho = r.ht0 - (int)dh;
    const int wo = r.wt0 - dw;
    if ((unsigned)ho >= (unsigned)p.Ho || (unsigned)wo >= (unsigned)p.Wo)
      return zp;
    return base + ((r.nbase + ho) * p.Wo + wo) * p.Co + co;
  }
};

// stride-2 dgrad, parity-decomposed: class (a,b) covers input pixels
// (2i+a, 2j+b); only taps with dh = pa+2*di, dw = pb+2*dj (pa=(a+pad)&1)
// can hit them, so k enumerates (di, dj, co) — 4x fewer MACs than the
// generic gather's zero-page-padded taps.  The B operand is the matching
// weight slice wt[Ci, nKh, nKw, Co] (host-sliced, RowGather).
struct ConvDgradS2Gather {
  const uint16_t* base;  // g [N,Ho,Wo,Co] channels-last
  ConvShape p;
  int a, b, pa, pb, nKw, Wb, HaWb;
  U32Div dv_wb, dv_hwb, dv_nkw;
  long M2, K2;
  DLX_DEV const uint16_t* src(long m, long k, const uint16_t* zp) const {
    if (m >= M2 || k >= K2) return zp;
    const int n = (int)dv_hwb.div((uint32_t)m);
    const int rem = (int)m - n * HaWb;
    const int i = (int)dv_wb.div((uint32_t)rem);
    const int j = rem - i * Wb;
    const int dij = (int)p.dv_co.div((uint32_t)k);
    const int co = (int)k - dij * p.Co;
    const int di = (int)dv_nkw.div((uint32_t)dij);
    const int dj = dij - di * nKw;
    const int ht = (2 * i + a) + p.pad - (pa + 2 * di);  // even
    const int wt = (2 * j + b) + p.pad - (pb + 2 * dj);
    if (ht < 0 || wt < 0) return zp;
    const int ho = ht >> 1, wo = wt >> 1;
    if (ho >= p.Ho || wo >= p.Wo) return zp;
    return base + (((long)n * p.Ho + ho) * p.Wo + wo) * p.Co + co;
  }
  struct OPre {
    long nbase;  // n*Ho, or -1 when m out of range
    int hti;     // (2i+a) + pad
    int wtj;     // (2j+b) + pad
  };
  DLX_DEV OPre oprep(long m) const {
    OPre r;
    if (m >= M2) {
      r.nbase = -1;
      return r;
    }
    const uint32_t n = dv_hwb.div((uint32_t)m);
    const uint32_t rem = (uint32_t)m - n * (uint32_t)HaWb;
    const uint32_t i = dv_wb.div(rem);
    const uint32_t j = rem - i * (uint32_t)Wb;
    r.nbase = (long)n * p.Ho;
    r.hti = (int)(2 * i) + a + p.pad;
    r.wtj = (int)(2 * j) + b + p.pad;
    return r;
  }
  DLX_DEV const uint16_t* oat(const OPre& r, long k,
                              const uint16_t* zp) const {
    if (r.nbase < 0 || k >= K2) return zp;
    const uint32_t dij = p.dv_co.div((uint32_t)k);
    const int co = (int)k - (int)dij * p.Co;
    const uint32_t di = dv_nkw.div(dij);
    const int dj = (int)(dij - di * nKw);
    const int ht = r.hti - (pa + 2 * (int)di);
    const int wt = r.wtj - (pb + 2 * dj);
    if (ht < 0 || wt < 0) return zp;
    const int ho = ht >> 1, wo = wt >> 1;
    if (ho >= p.Ho || wo >= p.Wo) return zp;
    return base + ((r.nbase + ho) * p.Wo + wo) * p.Co + co;
  }
};

// ---- output row mappers ------------------------------------------------
// Map an output row index to its element offset in C (row stride Nc for
// the plain case; a scattered pixel address for the stride-2 dgrad
// parity classes, where row enumerates only pixels (2i+a, 2j+b)).

struct IdentityOut {
  long Nc;
  DLX_DEV long off(long row) const { return row * Nc; }
};

struct Stride2Out {
  int H, W, Ci, a, b, Wb;   // Wb = pixels per row in this class
  U32Div dv_wb, dv_hwb;     // /Wb, /(Ha*Wb)
  int HaWb;
  DLX_DEV long off(long row) const {
    const int n = (int)dv_hwb.div((uint32_t)row);
    const int rem = (int)row - n * HaWb;
    const int i = (int)dv_wb.div((uint32_t)rem);
    const int j = rem - i * Wb;
    return (((long)n * H + 2 * i + a) * W + 2 * j + b) * Ci;
  }
};

// ---- the kernel --------------------------------------------------------

// Deterministic per-col (channel) sum / sum-of-squares of this block's
// output tile, written to stats[bm][2][Nc] — lets BatchNorm skip its
// whole-tensor reduce pass (the conv epilogue already holds y in regs).
// sred: LDS scratch (reused staging memory, >= 2*BN*WM floats).
template <int BM, int BN, int WM, int WN, int MR, int NR>
DLX_DEV void igemm_stats_epilogue(const f32x4 (*acc)[NR], float* sred,
                                  float* __restrict__ stats, long bm,
                                  long n0, long Nc, int wr, int wc, int l4,
                                  int l15, int tid) {
#pragma unroll
  for (int ni = 0; ni < NR; ++ni) {
    float s1 = 0.f, s2 = 0.f;
#pragma unroll
    for (int mi = 0; mi < MR; ++mi)
#pragma unroll
      for (int r = 0; r < 4; ++r) {
        const float v = acc[mi][ni][r];
        s1 += v;
        s2 += v * v;
      }
    // sum the 4 row-groups (l4) of this wave: lanes l<16 end up valid
    s1 += __shfl_down(s1, 16, kWave);
    s1 += __shfl_down(s1, 32, kWave);
    s2 += __shfl_down(s2, 16, kWave);
    s2 += __shfl_down(s2, 32, kWave);
    if (l4 == 0) {
      const int cl = wc * (BN / WN) + ni * 16 + l15;  // block-local col
      sred[(0 * BN + cl) * WM + wr] = s1;
      sred[(1 * BN + cl) * WM + wr] = s2;
    }
  }
  __syncthreads();
  // threads 0..BN-1: fixed-order sum over the WM row-waves (deterministic)
  for (int cl = tid; cl < BN; cl += WM * WN * 64) {
    if (n0 + cl >= Nc) continue;
    float t1 = 0.f, t2 = 0.f;
#pragma unroll
    for (int w = 0; w < WM; ++w) {
      t1 += sred[(0 * BN + cl) * WM + w];
      t2 += sred[(1 * BN + cl) * WM + w];
    }
    stats[(bm * 2 + 0) * Nc + n0 + cl] = t1;
    stats[(bm * 2 + 1) * Nc + n0 + cl] = t2;
  }
}

template <int BM, int BN, int BK, int WM, int WN,
          class AG, class BG, bool OUT_BF16, class OM = IdentityOut,
          bool STATS = false>
__global__ __launch_bounds__(WM* WN * 64) void igemm_kernel(
    AG ag, BG bg, void* __restrict__ Cout_,
    const uint16_t* __restrict__ zero_page,
    long M, long Nc, long K, int grid_n, OM om,
    float* __restrict__ stats = nullptr,
    const uint16_t* __restrict__ accum = nullptr) {
  constexpr int THREADS = WM * WN * 64;
  constexpr int SLOTS_A = (BK / 8) * BM;
  constexpr int SLOTS_B = (BK / 8) * BN;
  constexpr int GA = SLOTS_A / THREADS;
  constexpr int GB = SLOTS_B / THREADS;
  constexpr int MR = (BM / WM) / 16;
  constexpr int NR = (BN / WN) / 16;

  __shared__ uint16_t lsA[2 * SLOTS_A * 8];
  __shared__ uint16_t lsB[2 * SLOTS_B * 8];

  int nwg = gridDim.x;
  int bid = blockIdx.x;
  {  // XCD-aware bijective remap (guide T1)
    int q = nwg / 8, r = nwg % 8;
    int xcd = bid % 8, idx = bid / 8;
    bid = (xcd < r ? xcd * (q + 1) : r * (q + 1) + (xcd - r) * q) + idx;
  }
  const int bm = bid / grid_n, bn = bid % grid_n;
  const long m0 = (long)bm * BM, n0 = (long)bn * BN;

  const int lane = threadIdx.x % kWave;
  const int wid = threadIdx.x / kWave;
  const int wr = wid / WN, wc = wid % WN;
  const int l15 = lane & 15, l4 = lane >> 4;

  f32x4 acc[MR][NR] = {};
  const long ntile = (K + BK - 1) / BK;

  // hoist per-slot row state out of the k loop (rows are fixed per slot;
  // only k advances) — the ConvFwd/Dgrad m-decomposition divides run
  // once instead of per tile
  int kca[GA], kcb[GB];
  typename AG::OPre apre[GA];
  typename BG::OPre bpre[GB];
#pragma unroll
  for (int i = 0; i < GA; ++i) {
    const int f = i * THREADS + threadIdx.x;
    kca[i] = f / BM;
    apre[i] = ag.oprep(m0 + (f - kca[i] * BM));
  }
#pragma unroll
  for (int i = 0; i < GB; ++i) {
    const int f = i * THREADS + threadIdx.x;
    kcb[i] = f / BN;
    bpre[i] = bg.oprep(n0 + (f - kcb[i] * BN));
  }

  auto stage = [&](int buf, long t) {
    const long k0 = t * BK;
    uint16_t* la = lsA + buf * SLOTS_A * 8;
    uint16_t* lb = lsB + buf * SLOTS_B * 8;
#pragma unroll
    for (int i = 0; i < GA; ++i) {
      const int f = i * THREADS + threadIdx.x;
      const uint16_t* sa = ag.oat(apre[i], k0 + kca[i] * 8, zero_page);
      __builtin_amdgcn_global_load_lds(
          (glb_u32_t*)sa, (lds_u32_t*)(la + (long)f * 8), 16, 0, 0);
    }
#pragma unroll
    for (int i = 0; i < GB; ++i) {
      const int f = i * THREADS + threadIdx.x;
      const uint16_t* sb = bg.oat(bpre[i], k0 + kcb[i] * 8, zero_page);
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
    const uint16_t* la = lsA + buf * SLOTS_A * 8;
    const uint16_t* lb = lsB + buf * SLOTS_B * 8;
#pragma unroll
    for (int ks = 0; ks < BK / 32; ++ks) {
      bf16x8 af[MR], bfr[NR];
#pragma unroll
      for (int mi = 0; mi < MR; ++mi) {
        int m = wr * (BM / WM) + mi * 16 + l15;
        int kc = ks * 4 + l4;
        af[mi] = *reinterpret_cast<const bf16x8*>(la + ((long)kc * BM + m) * 8);
      }
#pragma unroll
      for (int ni = 0; ni < NR; ++ni) {
        int n = wc * (BN / WN) + ni * 16 + l15;
        int kc = ks * 4 + l4;
        bfr[ni] = *reinterpret_cast<const bf16x8*>(lb + ((long)kc * BN + n) * 8);
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

  // epilogue: C/D map for 16x16x32: col = lane&15, row = (lane>>4)*4 + r
#pragma unroll
  for (int mi = 0; mi < MR; ++mi) {
#pragma unroll
    for (int ni = 0; ni < NR; ++ni) {
#pragma unroll
      for (int r = 0; r < 4; ++r) {
        long row = m0 + wr * (BM / WM) + mi * 16 + l4 * 4 + r;
        long col = n0 + wc * (BN / WN) + ni * 16 + l15;
        if (row < M && col < Nc) {
          const long o = om.off(row) + col;
          float val = acc[mi][ni][r];
          if (accum) val += bf2f(accum[o]);  // fused residual fan-in add
          if (OUT_BF16)
            reinterpret_cast<uint16_t*>(Cout_)[o] = f2bf(val);
          else
            reinterpret_cast<float*>(Cout_)[o] = val;
        }
      }
    }
  }
  if (STATS) {
    // staging LDS is dead after the loop's final barrier — reuse it
    igemm_stats_epilogue<BM, BN, WM, WN, MR, NR>(
        acc, reinterpret_cast<float*>(lsA), stats, bm, n0, Nc, wr, wc, l4,
        l15, threadIdx.x);
  }
}

// ---------------------------------------------------------------------
// 8-phase counted-vmcnt pipeline, 256x256xBK64, 8 waves (guide §5 "256²
// 8-phase template" adapted to the k-subtile LDS layout):
//  * ONE __shared__ array (guide §5 trap 4a: a second LDS object makes
//    hipcc drain vmcnt(0) before every ds_read of a glds pipeline);
//  * stage units = (operand, k-half): 2 glds each; one unit issued per
//    phase, so 3 units stay in flight across barriers -> vmcnt(6);
//  * raw s_barrier (never __syncthreads: its fence drains pending
//    LDS-DMA), each wave waits its OWN glds before the barrier;
//  * waits only at phases 0/2 of each K-tile (k-half boundaries);
//  * s_setprio(1) around each 16-MFMA cluster (T5 — the phase split
//    creates the wave role diversity it needs).
// ---------------------------------------------------------------------

template <class AG, class BG, bool OUT_BF16, class OM = IdentityOut,
          bool STATS = false>
__global__ __launch_bounds__(512) void igemm_kernel_8ph(
    AG ag, BG bg, void* __restrict__ Cout_,
    const uint16_t* __restrict__ zero_page,
    long M, long Nc, long K, int grid_n, OM om,
    float* __restrict__ stats = nullptr,
    const uint16_t* __restrict__ accum = nullptr) {
  constexpr int BM = 256, BN = 256, BK = 64;
  constexpr int THREADS = 512;
  constexpr int SLOTS = (BK / 8) * BM;          // 2048 16-B slots/operand
  constexpr int MR = 8, NR = 4;                 // per-wave 128x64

  __shared__ uint16_t lds[2 * 2 * SLOTS * 8];   // [buf][A|B][SLOTS][8]

  int nwg = gridDim.x;
  int bid = blockIdx.x;
  {
    int q = nwg / 8, r = nwg % 8;
    int xcd = bid % 8, idx = bid / 8;
    bid = (xcd < r ? xcd * (q + 1) : r * (q + 1) + (xcd - r) * q) + idx;
  }
  const int bm = bid / grid_n, bn = bid % grid_n;
  const long m0 = (long)bm * BM, n0 = (long)bn * BN;

  const int lane = threadIdx.x % kWave;
  const int wid = threadIdx.x / kWave;
  const int wr = wid / 4, wc = wid % 4;         // 2 x 4 waves
  const int l15 = lane & 15, l4 = lane >> 4;

  f32x4 acc[MR][NR] = {};
  const long ntile = (K + BK - 1) / BK;

  // hoisted per-slot row state (see igemm_kernel): 4 slots per thread
  // per operand (units x 2 glds)
  int kc8[4];
  typename AG::OPre apre[4];
  typename BG::OPre bpre[4];
#pragma unroll
  for (int j = 0; j < 4; ++j) {
    const int f = j * THREADS + threadIdx.x;
    kc8[j] = f / BM;
    const int r = f - kc8[j] * BM;
    apre[j] = ag.oprep(m0 + r);
    bpre[j] = bg.oprep(n0 + r);  // BN == BM == 256
  }

  // stage unit u of tile t into buffer `buf`:
  //   u=0: A k-half 0 (glds instrs 0,1) ; u=1: B k-half 0
  //   u=2: A k-half 1 (instrs 2,3)      ; u=3: B k-half 1
  auto stage_unit = [&](int buf, long t, int u) {
    const long k0 = t * BK;
    const bool isB = u & 1;
    const int half = u >> 1;
    uint16_t* l0 = lds + (buf * 2 + (isB ? 1 : 0)) * SLOTS * 8;
#pragma unroll
    for (int i = 0; i < 2; ++i) {
      const int j = half * 2 + i;
      const int f = j * THREADS + threadIdx.x;
      const long k = k0 + kc8[j] * 8;
      const uint16_t* src = isB ? bg.oat(bpre[j], k, zero_page)
                                : ag.oat(apre[j], k, zero_page);
      __builtin_amdgcn_global_load_lds(
          (glb_u32_t*)src, (lds_u32_t*)(l0 + (long)f * 8), 16, 0, 0);
    }
  };

  // compute phase q (q>>1 = ks half, q&1 = mi half) of the tile in `buf`
  auto phase_mfma = [&](int buf, int q) {
    const uint16_t* la = lds + (buf * 2 + 0) * SLOTS * 8;
    const uint16_t* lb = lds + (buf * 2 + 1) * SLOTS * 8;
    const int ks = q >> 1;
    const int mih = (q & 1) * 4;
    bf16x8 af[4], bfr[NR];
#pragma unroll
    for (int i = 0; i < 4; ++i) {
      int m = wr * 128 + (mih + i) * 16 + l15;
      int kc = ks * 4 + l4;
      af[i] = *reinterpret_cast<const bf16x8*>(la + ((long)kc * BM + m) * 8);
    }
#pragma unroll
    for (int ni = 0; ni < NR; ++ni) {
      int n = wc * 64 + ni * 16 + l15;
      int kc = ks * 4 + l4;
      bfr[ni] = *reinterpret_cast<const bf16x8*>(lb + ((long)kc * BN + n) * 8);
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

  // prologue: all 4 units of tile 0
#pragma unroll
  for (int u = 0; u < 4; ++u) stage_unit(0, 0, u);

  int buf = 0;
  long t = 0;
  for (; t + 1 < ntile; ++t) {
    // phase 0: needs A/B k-half0 of t; 3 units may remain in flight
    stage_unit(buf ^ 1, t + 1, 0);
    asm volatile("s_waitcnt vmcnt(6)" ::: "memory");
    __builtin_amdgcn_s_barrier();
    phase_mfma(buf, 0);
    // phase 1: same k-half, no new data needed
    stage_unit(buf ^ 1, t + 1, 1);
    __builtin_amdgcn_s_barrier();
    phase_mfma(buf, 1);
    // phase 2: needs k-half1 of t
    stage_unit(buf ^ 1, t + 1, 2);
    asm volatile("s_waitcnt vmcnt(6)" ::: "memory");
    __builtin_amdgcn_s_barrier();
    phase_mfma(buf, 2);
    // phase 3
    stage_unit(buf ^ 1, t + 1, 3);
    __builtin_amdgcn_s_barrier();
    phase_mfma(buf, 3);
    buf ^= 1;
  }
  // epilogue tile: nothing left to stage
  asm volatile("s_waitcnt vmcnt(4)" ::: "memory");
  __builtin_amdgcn_s_barrier();
  phase_mfma(buf, 0);
  phase_mfma(buf, 1);
  asm volatile("s_waitcnt vmcnt(0)" ::: "memory");
  __builtin_amdgcn_s_barrier();
  phase_mfma(buf, 2);
  phase_mfma(buf, 3);

#pragma unroll
  for (int mi = 0; mi < MR; ++mi) {
#pragma unroll
    for (int ni = 0; ni < NR; ++ni) {
#pragma unroll
      for (int r = 0; r < 4; ++r) {
        long row = m0 + wr * 128 + mi * 16 + l4 * 4 + r;
        long col = n0 + wc * 64 + ni * 16 + l15;
        if (row < M && col < Nc) {
          const long o = om.off(row) + col;
          float val = acc[mi][ni][r];
          if (accum) val += bf2f(accum[o]);  // fused residual fan-in add
          if (OUT_BF16)
            reinterpret_cast<uint16_t*>(Cout_)[o] = f2bf(val);
          else
            reinterpret_cast<float*>(Cout_)[o] = val;
        }
      }
    }
  }
  if (STATS) {
    __syncthreads();  // last phase has no trailing barrier; LDS now dead
    igemm_stats_epilogue<256, 256, 2, 4, MR, NR>(
        acc, reinterpret_cast<float*>(lds), stats, bm, n0, Nc, wr, wc, l4,
        l15, threadIdx.x);
  }
}

// launch helper: picks the 8-phase 256^2 pipeline for big problems,
// the simple 128^2 kernel otherwise
template <class AG, class BG, bool OUT_BF16, class OM>
inline void igemm_launch_om(const AG& ag, const BG& bg, void* cptr,
                            const uint16_t* zp, long M, long Nc, long K,
                            hipStream_t s, const OM& om) {
  // 8ph needs a long K-loop to fill its counted-vmcnt pipeline (its one
  // 128-KiB block per CU cannot hide the prologue/drain otherwise)
  static const long mink8 = []() {
    const char* e = getenv("DLX_8PH_MINK");
    return e ? atol(e) : 256;  // 384->256 r2: short-K dgrad -20% (r2d/r2e)
  }();
  if (M >= 256 && Nc >= 192 && K >= mink8) {
    const int gm = (int)((M + 255) / 256), gn = (int)((Nc + 255) / 256);
    igemm_kernel_8ph<AG, BG, OUT_BF16, OM>
        <<<gm * gn, 512, 0, s>>>(ag, bg, cptr, zp, M, Nc, K, gn, om);
  } else if (Nc <= 96 && M >= 1024) {
    // narrow-N (layer1-class: N = Ci/Co = 64): a 128x128 tile wastes
    // half its MFMA work on the zero page — use a 256x64 tile (4 waves
    // of 64x64, 80 KiB LDS, 2 blocks/CU)
    const int gm = (int)((M + 255) / 256), gn = (int)((Nc + 63) / 64);
    igemm_kernel<256, 64, 64, 4, 1, AG, BG, OUT_BF16, OM>
        <<<gm * gn, 256, 0, s>>>(ag, bg, cptr, zp, M, Nc, K, gn, om);
  } else {
    // 8 waves measured +0.7%% whole-step (interleaved A/B, profiles/)
    static const bool w8 = []() {
      const char* e = getenv("DLX_IGEMM_W8");
      return !(e && e[0] == '0');
    }();
    const int gm = (int)((M + 127) / 128), gn = (int)((Nc + 127) / 128);
    if (w8)
      igemm_kernel<128, 128, 64, 2, 4, AG, BG, OUT_BF16, OM>
          <<<gm * gn, 512, 0, s>>>(ag, bg, cptr, zp, M, Nc, K, gn, om);
    else
      igemm_kernel<128, 128, 64, 2, 2, AG, BG, OUT_BF16, OM>
          <<<gm * gn, 256, 0, s>>>(ag, bg, cptr, zp, M, Nc, K, gn, om);
  }
}

template <class AG, class BG, bool OUT_BF16>
inline void igemm_launch(const AG& ag, const BG& bg, void* cptr,
                         const uint16_t* zp, long M, long Nc, long K,
                         hipStream_t s) {
  igemm_launch_om<AG, BG, OUT_BF16, IdentityOut>(ag, bg, cptr, zp, M, Nc, K,
                                                 s, IdentityOut{Nc});
}

// variant with the fused epilogue accumulate: out = GEMM + accum (bf16,
// same [M, Nc] layout) — the autograd residual fan-in add folded into
// the dgrad store
template <class AG, class BG>
inline void igemm_launch_acc(const AG& ag, const BG& bg, void* cptr,
                             const uint16_t* zp, long M, long Nc, long K,
                             hipStream_t s, const uint16_t* accum) {
  static const long mink8 = []() {
    const char* e = getenv("DLX_8PH_MINK");
    return e ? atol(e) : 256;  // 384->256 r2: short-K dgrad -20% (r2d/r2e)
  }();
  const IdentityOut om{Nc};
  if (M >= 256 && Nc >= 192 && K >= mink8) {
    const int gm = (int)((M + 255) / 256), gn = (int)((Nc + 255) / 256);
    igemm_kernel_8ph<AG, BG, true, IdentityOut>
        <<<gm * gn, 512, 0, s>>>(ag, bg, cptr, zp, M, Nc, K, gn, om,
                                 nullptr, accum);
  } else if (Nc <= 96 && M >= 1024) {
    const int gm = (int)((M + 255) / 256), gn = (int)((Nc + 63) / 64);
    igemm_kernel<256, 64, 64, 4, 1, AG, BG, true, IdentityOut>
        <<<gm * gn, 256, 0, s>>>(ag, bg, cptr, zp, M, Nc, K, gn, om,
                                 nullptr, accum);
  } else {
    const int gm = (int)((M + 127) / 128), gn = (int)((Nc + 127) / 128);
    igemm_kernel<128, 128, 64, 2, 4, AG, BG, true, IdentityOut>
        <<<gm * gn, 512, 0, s>>>(ag, bg, cptr, zp, M, Nc, K, gn, om,
                                 nullptr, accum);
  }
}

// geometry the launcher will pick (the stats slab is [gm][2][Nc], so the
// host must know gm before allocating)
inline long igemm_gm(long M, long Nc, long K) {
  static const long mink8 = []() {
    const char* e = getenv("DLX_8PH_MINK");
    return e ? atol(e) : 256;  // 384->256 r2: short-K dgrad -20% (r2d/r2e)
  }();
  if (M >= 256 && Nc >= 192 && K >= mink8) return (M + 255) / 256;
  if (Nc <= 96 && M >= 1024) return (M + 255) / 256;
  return (M + 127) / 128;
}

// conv-fwd launch with the per-block channel-stats epilogue (stats
// slab [igemm_gm][2][Nc] fp32; reduce with the BN finalize kernel)
template <class AG, class BG>
inline void igemm_launch_stats(const AG& ag, const BG& bg, void* cptr,
                               const uint16_t* zp, long M, long Nc, long K,
                               hipStream_t s, float* stats) {
  static const long mink8 = []() {
    const char* e = getenv("DLX_8PH_MINK");
    return e ? atol(e) : 256;  // 384->256 r2: short-K dgrad -20% (r2d/r2e)
  }();
  const IdentityOut om{Nc};
  if (M >= 256 && Nc >= 192 && K >= mink8) {
    const int gm = (int)((M + 255) / 256), gn = (int)((Nc + 255) / 256);
    igemm_kernel_8ph<AG, BG, true, IdentityOut, true>
        <<<gm * gn, 512, 0, s>>>(ag, bg, cptr, zp, M, Nc, K, gn, om, stats);
  } else if (Nc <= 96 && M >= 1024) {
    const int gm = (int)((M + 255) / 256), gn = (int)((Nc + 63) / 64);
    igemm_kernel<256, 64, 64, 4, 1, AG, BG, true, IdentityOut, true>
        <<<gm * gn, 256, 0, s>>>(ag, bg, cptr, zp, M, Nc, K, gn, om, stats);
  } else {
    static const bool w8 = []() {
      const char* e = getenv("DLX_IGEMM_W8");
      return !(e && e[0] == '0');
    }();
    const int gm = (int)((M + 127) / 128), gn = (int)((Nc + 127) / 128);
    if (w8)
      igemm_kernel<128, 128, 64, 2, 4, AG, BG, true, IdentityOut, true>
          <<<gm * gn, 512, 0, s>>>(ag, bg, cptr, zp, M, Nc, K, gn, om,
                                   stats);
    else
      igemm_kernel<128, 128, 64, 2, 2, AG, BG, true, IdentityOut, true>
          <<<gm * gn, 256, 0, s>>>(ag, bg, cptr, zp, M, Nc, K, gn, om,
                                   stats);
  }
}

}  // namespace dlx
