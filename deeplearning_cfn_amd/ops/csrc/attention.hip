// Fused multi-head attention forward (flash-style online softmax), bf16.
//
// Replaces the BERT path's rocBLAS bmm + fp32 softmax (which materialises
// the [B,H,S,S] score tensor) — SURVEY.md §3.2 "cuBLAS GEMM / batched
// attention" row; VERDICT r1 #5.  Scores never leave registers/LDS.
//
// Shapes: q,k,v [B,H,S,Dh], Dh == 64, S % 64 == 0 (BERT-base: S=128/512).
// Geometry: one 4-wave block per (b*h, 64-row q tile); K/V staged in LDS
// per 64-key tile; per wave 16 q rows.
//   * QK^T: Q register-resident A-fragments (lane l: row l&15, k-chunk
//     (l>>4)*8), K staged k-subtile-major ([Dh/8][64 keys][8] slots) and
//     read as conflict-free b128 B-fragments — the igemm_core LDS image.
//   * online softmax per q row: running max m, normaliser l, fp32 O
//     accumulator in the 16x16 C-map (row = (l>>4)*4+r, col = l&15), so
//     the row-local rescale is lane-local; row reductions are 4
//     shfl_xors within a 16-lane group.
//   * P·V: P (bf16) staged per-wave into an A-fragment LDS image, V
//     staged in the wrw transpose image and read with
//     ds_read_b64_tr_b16 (probe-verified semantics: tools/probe_tr16.hip
//     / conv_wrw.hip) giving key-contiguous B-fragments.
//   * epilogue: O/l stores + lse = m + log(l) saved for the backward
//     (recompute-composition backward in ops/functional.py).
#include <torch/extension.h>
#include <ATen/cuda/CUDAContext.h>

#include "igemm_core.h"

namespace dlx {

namespace {

constexpr int kDh = 64;      // head dim
constexpr int kQT = 64;      // q rows per block (4 waves x 16)
constexpr int kKT = 64;      // keys per LDS tile

using bf16v4 = __attribute__((ext_vector_type(4))) __bf16;
using bf16x4a = __attribute__((ext_vector_type(4))) short;
typedef __attribute__((address_space(3))) bf16v4 lds_bf16v4_t;

DLX_DEV bf16x4a attn_tr_read(const uint16_t* lds_base, int elem_off) {
  auto* p = (lds_bf16v4_t*)(lds_base + elem_off);
  bf16v4 v = __builtin_amdgcn_ds_read_tr16_b64_v4bf16(p);
  return __builtin_bit_cast(bf16x4a, v);
}

// wrw transpose-image helpers for a 64-col operand (LCB=2); see
// conv_wrw.hip for the layout derivation.
DLX_DEV void attn_slot_to_mc(int f, int& mloc, int& c) {
  const int e = f * 8;
  const int blk = e >> 6;
  const int rem = e & 63;
  const int r = rem >> 4;
  const int cc0 = rem & 15;
  const int w = blk & 7;
  const int i = w >> 2;
  const int g = w & 3;
  const int kkcb = blk >> 3;
  const int cb = kkcb & 3;
  const int kk = kkcb >> 2;
  mloc = (kk << 5) + (g << 3) + (i << 2) + r;
  c = (cb << 4) + cc0;
}

DLX_DEV int attn_frag_off(int kk, int cb, int i, int lane) {
  return (((kk << 2) + cb) * 8 + i * 4 + (lane >> 4)) * 64 +
         (lane & 15) * 4;
}

DLX_DEV float rowmax16(float v) {
#pragma unroll
  for (int m = 1; m < 16; m <<= 1)
    v = fmaxf(v, __shfl_xor(v, m, kWave));
  return v;
}

DLX_DEV float rowsum16(float v) {
#pragma unroll
  for (int m = 1; m < 16; m <<= 1) v += __shfl_xor(v, m, kWave);
  return v;
}

__global__ __launch_bounds__(256) void attn_fwd_kernel(
    const uint16_t* __restrict__ q, const uint16_t* __restrict__ k,
    const uint16_t* __restrict__ v, uint16_t* __restrict__ o,
    float* __restrict__ lse, int S, float scale) {
  // LDS: K image (8 KiB) | V image (8 KiB) | per-wave P images (8 KiB)
  constexpr int kSlots = kKT * kDh / 8;            // 512 per operand
  __shared__ uint16_t lk[kSlots * 8];
  __shared__ uint16_t lv[kSlots * 8];
  __shared__ uint16_t lp[4][16 * kKT];             // [wave][qrow 16][key 64]

  const long bh = blockIdx.y;
  const int q0 = blockIdx.x * kQT;
  const long base = bh * (long)S * kDh;
  const int lane = threadIdx.x % kWave;
  const int wid = threadIdx.x / kWave;
  const int l15 = lane & 15, l4 = lane >> 4;

  // Q fragments (register-resident): lane l holds row l&15 of this
  // wave's 16 rows, k-chunk (l>>4)*8 (+32 for the second k-step)
  const int qrow_g = q0 + wid * 16 + l15;
  bf16x8 qa[2];
#pragma unroll
  for (int kk = 0; kk < 2; ++kk)
    qa[kk] = *reinterpret_cast<const bf16x8*>(
        q + base + (long)qrow_g * kDh + kk * 32 + l4 * 8);

  float m_run[4], l_run[4];
  f32x4 o_acc[4];  // [dh col-block]: each f32x4 spans 4 q rows (C-map)
#pragma unroll
  for (int r = 0; r < 4; ++r) {
    m_run[r] = -1e30f;
    l_run[r] = 0.f;
  }
#pragma unroll
  for (int cb = 0; cb < 4; ++cb) o_acc[cb] = f32x4{0.f, 0.f, 0.f, 0.f};

  for (int k0 = 0; k0 < S; k0 += kKT) {
    // ---- stage K (k-subtile image) + V (transpose image) via glds ----
#pragma unroll
    for (int u = 0; u < kSlots / 256; ++u) {
      const int f = u * 256 + threadIdx.x;
      {  // K: slot f = kc*64 + key
        const int kc = f >> 6, key = f & 63;
        const uint16_t* src = k + base + (long)(k0 + key) * kDh + kc * 8;
        __builtin_amdgcn_global_load_lds(
            (glb_u32_t*)src, (lds_u32_t*)(lk + (long)f * 8), 16, 0, 0);
      }
      {  // V: wrw image slot
        int key, c;
        attn_slot_to_mc(f, key, c);
        const uint16_t* src = v + base + (long)(k0 + key) * kDh + c;
        __builtin_amdgcn_global_load_lds(
            (glb_u32_t*)src, (lds_u32_t*)(lv + (long)f * 8), 16, 0, 0);
      }
    }
    __builtin_amdgcn_s_waitcnt(0);
    __syncthreads();

    // ---- S tile: 4 C-tiles of [16 qrow x 16 key] per wave ------------
    f32x4 sc[4];
#pragma unroll
    for (int cb = 0; cb < 4; ++cb) {
      sc[cb] = f32x4{0.f, 0.f, 0.f, 0.f};
#pragma unroll
      for (int kk = 0; kk < 2; ++kk) {
        // K B-fragment: lane l = key col cb*16+(l&15), dh chunk
        // (kk*4 + l>>4)*8 — 16 consecutive slots per lane group
        bf16x8 kf = *reinterpret_cast<const bf16x8*>(
            lk + (((kk << 2) + l4) * 64 + (cb << 4) + l15) * 8);
        sc[cb] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(qa[kk], kf,
                                                         sc[cb], 0, 0, 0);
      }
    }

    // ---- online softmax update (rows = l4*4 + r, lane-local) ---------
    float p_val[4][4];  // [cb][r] bf16-bound P values
#pragma unroll
    for (int r = 0; r < 4; ++r) {
      float mx = -1e30f;
#pragma unroll
      for (int cb = 0; cb < 4; ++cb) mx = fmaxf(mx, sc[cb][r] * scale);
      mx = rowmax16(mx);
      const float m_new = fmaxf(m_run[r], mx);
      const float alpha = __expf(m_run[r] - m_new);
      float rs = 0.f;
#pragma unroll
      for (int cb = 0; cb < 4; ++cb) {
        const float p = __expf(sc[cb][r] * scale - m_new);
        p_val[cb][r] = p;
        rs += p;
      }
      rs = rowsum16(rs);
      l_run[r] = alpha * l_run[r] + rs;
      m_run[r] = m_new;
#pragma unroll
      for (int cb = 0; cb < 4; ++cb) o_acc[cb][r] *= alpha;
    }

    // ---- stage P into this wave's A-fragment image -------------------
    // element (qrow, key) at slot (key/8)*16 + qrow, byte key%8
    uint16_t* lpw = lp[wid];
#pragma unroll
    for (int cb = 0; cb < 4; ++cb) {
      const int key = (cb << 4) + l15;
      const int so = (key >> 3) * 128 + (key & 7);
#pragma unroll
      for (int r = 0; r < 4; ++r)
        lpw[so + ((l4 << 2) + r) * 8] = f2bf(p_val[cb][r]);
    }
    __syncthreads();  // P visible across lanes; K/V reads done

    // ---- O += P · V ---------------------------------------------------
#pragma unroll
    for (int kk = 0; kk < 2; ++kk) {
      // P A-fragment: lane l = qrow l&15, keys (kk*32 + (l>>4)*8 ..)
      bf16x8 pf = *reinterpret_cast<const bf16x8*>(
          lpw + (((kk << 2) + l4) * 16 + l15) * 8);
#pragma unroll
      for (int cb = 0; cb < 4; ++cb) {
        bf16x4a lo = attn_tr_read(lv, attn_frag_off(kk, cb, 0, lane));
        bf16x4a hi = attn_tr_read(lv, attn_frag_off(kk, cb, 1, lane));
        bf16x8 vf =
            __builtin_shufflevector(lo, hi, 0, 1, 2, 3, 4, 5, 6, 7);
        o_acc[cb] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(pf, vf,
                                                            o_acc[cb], 0,
                                                            0, 0);
      }
    }
    __syncthreads();  // before the next tile overwrites LDS
  }

  // ---- epilogue: O / l, bf16 stores; lse = m + log(l) -----------------
#pragma unroll
  for (int r = 0; r < 4; ++r) {
    const long row = q0 + wid * 16 + (l4 << 2) + r;
    const float inv_l = 1.f / l_run[r];
#pragma unroll
    for (int cb = 0; cb < 4; ++cb)
      o[base + row * kDh + (cb << 4) + l15] = f2bf(o_acc[cb][r] * inv_l);
    if (l15 == 0) lse[bh * S + row] = m_run[r] + __logf(l_run[r]);
  }
}

// ---------------------------------------------------------------------
// Fused attention BACKWARD (flash-bwd, recompute P^T in-kernel).
//
// One block per (b*h, 64-key tile): K/V rows live in registers per wave
// (16 keys each), Q/dO tiles stream through LDS in BOTH the k-subtile
// image (dh-contiguous fragments) and the transpose image
// (row-contiguous fragments via ds_read_b64_tr_b16).  Computing S^T
// (= K Q^T) instead of S puts P^T and dP^T in the SAME C-map, so
// dS^T = P^T*(dP^T - delta)*scale is lane-local.  dK/dV accumulate in
// registers and store plainly (each block owns its keys); dQ^T partials
// go to a fp32 buffer with atomics (S/64 contributors per element —
// NOT bitwise deterministic; DLX_ATTN=0 gives the deterministic
// composition).  5 MFMA stages per (i,j) tile pair:
//   S^T = K Q^T;  dV += P^T dO;  dP^T = V dO^T;  dK += dS^T Q;
//   dQ^T += K^T dS^T  (block-level dS^T transpose image).
// ---------------------------------------------------------------------

__global__ __launch_bounds__(256) void attn_bwd_kernel(
    const uint16_t* __restrict__ q, const uint16_t* __restrict__ k,
    const uint16_t* __restrict__ v, const uint16_t* __restrict__ dout,
    const float* __restrict__ lse, const float* __restrict__ delta,
    uint16_t* __restrict__ dk_out, uint16_t* __restrict__ dv_out,
    float* __restrict__ dq_f32, int S, float scale) {
  constexpr int kSlots = kKT * kDh / 8;  // 512 per image
  __shared__ uint16_t lq[kSlots * 8];    // Q k-subtile image
  __shared__ uint16_t lqt[kSlots * 8];   // Q transpose image
  __shared__ uint16_t lo[kSlots * 8];    // dO k-subtile image
  __shared__ uint16_t lot[kSlots * 8];   // dO transpose image
  __shared__ uint16_t lkt[kSlots * 8];   // K transpose image (fixed)
  __shared__ uint16_t lst[kSlots * 8];   // dS^T transpose image (block)
  __shared__ uint16_t lp[4][16 * kKT];   // per-wave P^T/dS^T A-image

  const long bh = blockIdx.y;
  const int j0 = blockIdx.x * kKT;       // key-tile origin
  const long base = bh * (long)S * kDh;
  const int lane = threadIdx.x % kWave;
  const int wid = threadIdx.x / kWave;   // wave owns keys j0+wid*16..
  const int l15 = lane & 15, l4 = lane >> 4;

  // K,V register fragments (A-operand map: lane = key row l&15,
  // dh chunk (l>>4)*8 + kk*32) for the wave's 16 keys
  const int krow_g = j0 + wid * 16 + l15;
  bf16x8 ka[2], va[2];
#pragma unroll
  for (int kk = 0; kk < 2; ++kk) {
    const long off = base + (long)krow_g * kDh + kk * 32 + l4 * 8;
    ka[kk] = *reinterpret_cast<const bf16x8*>(k + off);
    va[kk] = *reinterpret_cast<const bf16x8*>(v + off);
  }

  // K transpose image (A-fragments of K^T for the dQ^T GEMM)
#pragma unroll
  for (int u = 0; u < kSlots / 256; ++u) {
    const int f = u * 256 + threadIdx.x;
    int m, c;
    attn_slot_to_mc(f, m, c);
    const uint16_t* src = k + base + (long)(j0 + m) * kDh + c;
    __builtin_amdgcn_global_load_lds(
        (glb_u32_t*)src, (lds_u32_t*)(lkt + (long)f * 8), 16, 0, 0);
  }

  f32x4 dk_acc[4], dv_acc[4];  // [dh col-block], rows = wave's 16 keys
#pragma unroll
  for (int db = 0; db < 4; ++db) {
    dk_acc[db] = f32x4{0.f, 0.f, 0.f, 0.f};
    dv_acc[db] = f32x4{0.f, 0.f, 0.f, 0.f};
  }
  __builtin_amdgcn_s_waitcnt(0);
  __syncthreads();

  uint16_t* lpw = lp[wid];
  for (int i0 = 0; i0 < S; i0 += kKT) {
    // ---- stage Q and dO in both images -----------------------------
#pragma unroll
    for (int u = 0; u < kSlots / 256; ++u) {
      const int f = u * 256 + threadIdx.x;
      const int kc = f >> 6, qr = f & 63;
      int m, c;
      attn_slot_to_mc(f, m, c);
      const uint16_t* sq = q + base + (long)(i0 + qr) * kDh + kc * 8;
      const uint16_t* sqt = q + base + (long)(i0 + m) * kDh + c;
      const uint16_t* so = dout + base + (long)(i0 + qr) * kDh + kc * 8;
      const uint16_t* sot = dout + base + (long)(i0 + m) * kDh + c;
      __builtin_amdgcn_global_load_lds(
          (glb_u32_t*)sq, (lds_u32_t*)(lq + (long)f * 8), 16, 0, 0);
      __builtin_amdgcn_global_load_lds(
          (glb_u32_t*)sqt, (lds_u32_t*)(lqt + (long)f * 8), 16, 0, 0);
      __builtin_amdgcn_global_load_lds(
          (glb_u32_t*)so, (lds_u32_t*)(lo + (long)f * 8), 16, 0, 0);
      __builtin_amdgcn_global_load_lds(
          (glb_u32_t*)sot, (lds_u32_t*)(lot + (long)f * 8), 16, 0, 0);
    }
    __builtin_amdgcn_s_waitcnt(0);
    __syncthreads();

    // ---- S^T = K Q^T (C-map: row = key (l4*4+r), col = qrow l15) ---
    f32x4 st[4];
#pragma unroll
    for (int qc = 0; qc < 4; ++qc) {
      st[qc] = f32x4{0.f, 0.f, 0.f, 0.f};
#pragma unroll
      for (int kk = 0; kk < 2; ++kk) {
        bf16x8 qf = *reinterpret_cast<const bf16x8*>(
            lq + (((kk << 2) + l4) * 64 + (qc << 4) + l15) * 8);
        st[qc] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(ka[kk], qf,
                                                         st[qc], 0, 0, 0);
      }
    }

    // ---- P^T = exp(S^T*scale - lse[qrow]); dP^T = V dO^T -----------
    float p_val[4][4], lse_c[4], del_c[4];
    f32x4 dp[4];
#pragma unroll
    for (int qc = 0; qc < 4; ++qc) {
      lse_c[qc] = lse[bh * S + i0 + (qc << 4) + l15];
      del_c[qc] = delta[bh * S + i0 + (qc << 4) + l15];
      dp[qc] = f32x4{0.f, 0.f, 0.f, 0.f};
#pragma unroll
      for (int kk = 0; kk < 2; ++kk) {
        bf16x8 of = *reinterpret_cast<const bf16x8*>(
            lo + (((kk << 2) + l4) * 64 + (qc << 4) + l15) * 8);
        dp[qc] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(va[kk], of,
                                                         dp[qc], 0, 0, 0);
      }
#pragma unroll
      for (int r = 0; r < 4; ++r)
        p_val[qc][r] = __expf(st[qc][r] * scale - lse_c[qc]);
    }

    // ---- dV += P^T dO (P^T via the per-wave A-image) ---------------
#pragma unroll
    for (int qc = 0; qc < 4; ++qc) {
      const int so = ((qc << 1) + (l15 >> 3)) * 128 + (l15 & 7);
#pragma unroll
      for (int r = 0; r < 4; ++r)
        lpw[so + (((l4 << 2) + r) << 3)] = f2bf(p_val[qc][r]);
    }
    asm volatile("s_waitcnt lgkmcnt(0)" ::: "memory");
    __builtin_amdgcn_wave_barrier();
#pragma unroll
    for (int kk = 0; kk < 2; ++kk) {
      bf16x8 pa = *reinterpret_cast<const bf16x8*>(
          lpw + (((kk << 2) + l4) * 16 + l15) * 8);
#pragma unroll
      for (int db = 0; db < 4; ++db) {
        bf16x4a lo_ = attn_tr_read(lot, attn_frag_off(kk, db, 0, lane));
        bf16x4a hi_ = attn_tr_read(lot, attn_frag_off(kk, db, 1, lane));
        bf16x8 of =
            __builtin_shufflevector(lo_, hi_, 0, 1, 2, 3, 4, 5, 6, 7);
        dv_acc[db] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
            pa, of, dv_acc[db], 0, 0, 0);
      }
    }

    // ---- dS^T = P^T * (dP^T - delta[qrow]) * scale ------------------
    // write into the per-wave A-image (for dK) AND the block transpose
    // image (for dQ^T)
#pragma unroll
    for (int qc = 0; qc < 4; ++qc) {
      const int so = ((qc << 1) + (l15 >> 3)) * 128 + (l15 & 7);
#pragma unroll
      for (int r = 0; r < 4; ++r) {
        const float ds =
            p_val[qc][r] * (dp[qc][r] - del_c[qc]) * scale;
        const uint16_t dsb = f2bf(ds);
        lpw[so + (((l4 << 2) + r) << 3)] = dsb;
        // transpose image element (m = key local, c = qrow)
        const int m = (wid << 4) + (l4 << 2) + r;
        const int c = (qc << 4) + l15;
        const int blk = (((m >> 5) << 2) | (c >> 4)) * 8 +
                        (((m >> 2) & 1) << 2) + ((m >> 3) & 3);
        lst[blk * 64 + (m & 3) * 16 + (c & 15)] = dsb;
      }
    }
    asm volatile("s_waitcnt lgkmcnt(0)" ::: "memory");
    __builtin_amdgcn_wave_barrier();

    // ---- dK += dS^T Q (Q via transpose image) ----------------------
#pragma unroll
    for (int kk = 0; kk < 2; ++kk) {
      bf16x8 dsa = *reinterpret_cast<const bf16x8*>(
          lpw + (((kk << 2) + l4) * 16 + l15) * 8);
#pragma unroll
      for (int db = 0; db < 4; ++db) {
        bf16x4a lo_ = attn_tr_read(lqt, attn_frag_off(kk, db, 0, lane));
        bf16x4a hi_ = attn_tr_read(lqt, attn_frag_off(kk, db, 1, lane));
        bf16x8 qf =
            __builtin_shufflevector(lo_, hi_, 0, 1, 2, 3, 4, 5, 6, 7);
        dk_acc[db] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
            dsa, qf, dk_acc[db], 0, 0, 0);
      }
    }
    __syncthreads();  // all waves' dS^T visible in lst

    // ---- dQ^T += K^T dS^T : wave w computes dh rows [w*16, w*16+16)
    f32x4 dqt[4];
#pragma unroll
    for (int qc = 0; qc < 4; ++qc) {
      dqt[qc] = f32x4{0.f, 0.f, 0.f, 0.f};
#pragma unroll
      for (int kk = 0; kk < 2; ++kk) {
        bf16x4a klo = attn_tr_read(lkt, attn_frag_off(kk, wid, 0, lane));
        bf16x4a khi = attn_tr_read(lkt, attn_frag_off(kk, wid, 1, lane));
        bf16x8 ktf =
            __builtin_shufflevector(klo, khi, 0, 1, 2, 3, 4, 5, 6, 7);
        bf16x4a slo = attn_tr_read(lst, attn_frag_off(kk, qc, 0, lane));
        bf16x4a shi = attn_tr_read(lst, attn_frag_off(kk, qc, 1, lane));
        bf16x8 stf =
            __builtin_shufflevector(slo, shi, 0, 1, 2, 3, 4, 5, 6, 7);
        dqt[qc] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(ktf, stf,
                                                          dqt[qc], 0, 0,
                                                          0);
      }
      // accumulate transposed into dq[qrow][dh] (fp32 atomics)
#pragma unroll
      for (int r = 0; r < 4; ++r) {
        const long qrow = i0 + (qc << 4) + l15;
        const int dh = (wid << 4) + (l4 << 2) + r;
        unsafeAtomicAdd(&dq_f32[(base + qrow * kDh) + dh], dqt[qc][r]);
      }
    }
    __syncthreads();  // before the next i-tile overwrites LDS
  }

  // ---- store dK, dV (plain: this block owns keys j0+wid*16..) -------
#pragma unroll
  for (int r = 0; r < 4; ++r) {
    const long row = j0 + wid * 16 + (l4 << 2) + r;
#pragma unroll
    for (int db = 0; db < 4; ++db) {
      dk_out[base + row * kDh + (db << 4) + l15] = f2bf(dk_acc[db][r]);
      dv_out[base + row * kDh + (db << 4) + l15] = f2bf(dv_acc[db][r]);
    }
  }
}

}  // namespace

// q,k,v: [B,H,S,64] bf16 contiguous -> (o [B,H,S,64] bf16, lse [B,H,S] f32)
std::tuple<at::Tensor, at::Tensor> attn_fwd_impl(const at::Tensor& q_in,
                                                 const at::Tensor& k_in,
                                                 const at::Tensor& v_in,
                                                 double scale) {
  auto q = q_in.contiguous(), k = k_in.contiguous(), v = v_in.contiguous();
  TORCH_CHECK(q.scalar_type() == at::kBFloat16, "attn: bf16 only");
  TORCH_CHECK(q.dim() == 4 && q.size(3) == kDh, "attn: [B,H,S,64] expected");
  TORCH_CHECK(q.size(2) % 64 == 0, "attn: S % 64 != 0 (use library path)");
  TORCH_CHECK(q.sizes() == k.sizes() && q.sizes() == v.sizes(),
              "attn: q/k/v shape mismatch");
  const int B = q.size(0), H = q.size(1), S = q.size(2);
  auto o = at::empty_like(q);
  auto lse = at::empty({B, H, S}, q.options().dtype(at::kFloat));
  auto s = at::cuda::getCurrentCUDAStream();
  dim3 grid((unsigned)(S / kQT), (unsigned)(B * H));
  attn_fwd_kernel<<<grid, 256, 0, s>>>(
      reinterpret_cast<const uint16_t*>(q.data_ptr()),
      reinterpret_cast<const uint16_t*>(k.data_ptr()),
      reinterpret_cast<const uint16_t*>(v.data_ptr()),
      reinterpret_cast<uint16_t*>(o.data_ptr()), lse.data_ptr<float>(), S,
      (float)scale);
  return {o, lse};
}

// grad,q,k,v,o: [B,H,S,64] bf16; lse: [B,H,S] f32 -> (dq, dk, dv) bf16
std::tuple<at::Tensor, at::Tensor, at::Tensor> attn_bwd_impl(
    const at::Tensor& grad_in, const at::Tensor& q_in,
    const at::Tensor& k_in, const at::Tensor& v_in, const at::Tensor& o_in,
    const at::Tensor& lse, double scale) {
  auto g = grad_in.contiguous(), q = q_in.contiguous(),
       k = k_in.contiguous(), v = v_in.contiguous(), o = o_in.contiguous();
  TORCH_CHECK(q.scalar_type() == at::kBFloat16, "attn bwd: bf16 only");
  TORCH_CHECK(q.dim() == 4 && q.size(3) == kDh && q.size(2) % 64 == 0,
              "attn bwd: [B,H,S%64,64] expected");
  const int B = q.size(0), H = q.size(1), S = q.size(2);
  // delta_i = rowsum(dO_i * O_i) — one memory-bound pass
  auto delta = (g.to(at::kFloat) * o.to(at::kFloat)).sum(-1);
  auto dk = at::empty_like(k);
  auto dv = at::empty_like(v);
  auto dq_f32 = at::zeros({(long)B * H * S * kDh},
                          q.options().dtype(at::kFloat));
  auto s = at::cuda::getCurrentCUDAStream();
  dim3 grid((unsigned)(S / kKT), (unsigned)(B * H));
  attn_bwd_kernel<<<grid, 256, 0, s>>>(
      reinterpret_cast<const uint16_t*>(q.data_ptr()),
      reinterpret_cast<const uint16_t*>(k.data_ptr()),
      reinterpret_cast<const uint16_t*>(v.data_ptr()),
      reinterpret_cast<const uint16_t*>(g.data_ptr()),
      lse.data_ptr<float>(), delta.data_ptr<float>(),
      reinterpret_cast<uint16_t*>(dk.data_ptr()),
      reinterpret_cast<uint16_t*>(dv.data_ptr()), dq_f32.data_ptr<float>(),
      S, (float)scale);
  auto dq = dq_f32.to(at::kBFloat16).view({B, H, S, (long)kDh});
  return {dq, dk, dv};
}

}  // namespace dlx
