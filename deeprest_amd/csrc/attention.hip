// Fused multi-head attention forward for gfx950 (flash-style, MFMA).
//
// One workgroup (4 waves) per 64-row Q tile of one (batch, head): QK^T on
// v_mfma_f32_16x16x32_bf16 with the K tile staged in LDS (rows padded +16 B
// so the 16-lane ds_read_b128 groups land on distinct banks), online softmax
// entirely in registers (running row max/denominator, rescale-on-grow), PV
// on MFMA with V staged transposed.  No S x S score tensor ever exists in
// memory; the per-row logsumexp is saved for the recompute backward
// (deeprest_amd/ops/attention.py).
//
// Internal compute is bf16 MFMA with fp32 softmax statistics and fp32 O
// accumulation; fp32 inputs are rounded to bf16 at the MFMA operands only.
// Head dims supported: 8 <= D <= 64, D % 8 == 0 (the traffic encoder uses
// D = d_model / n_heads = 32).
#include "common.h"

namespace dr {

constexpr int QT = 64;     // Q rows per block
constexpr int KT_KEYS = 32;  // keys per tile
constexpr int A_WAVES = 4;
constexpr int A_THREADS = A_WAVES * DR_WAVE;
constexpr int DMAX = 64;
// padded LDS row strides (bytes): +16 B breaks the power-of-2 bank pattern
constexpr int K_STRIDE = DMAX * 2 + 16;      // K tile: [key][d]
constexpr int V_STRIDE = KT_KEYS * 2 + 16;   // V^T tile: [d][key]
constexpr int P_STRIDE = KT_KEYS * 2 + 16;   // P tile:  [qrow][key]

template <typename T>
__global__ __launch_bounds__(A_THREADS) void mha_fwd_kernel(
    const T* __restrict__ q,   // (BH, T, D)
    const T* __restrict__ k,
    const T* __restrict__ v,
    T* __restrict__ o,         // (BH, T, D)
    float* __restrict__ lse,   // (BH, T)
    int T_len, int D, float scale) {
  __shared__ __attribute__((aligned(16))) char k_lds[KT_KEYS * K_STRIDE];
  __shared__ __attribute__((aligned(16))) char vt_lds[DMAX * V_STRIDE];
  __shared__ __attribute__((aligned(16))) char p_lds[QT * P_STRIDE];

  const int tid = threadIdx.x;
  const int wv = tid / DR_WAVE;
  const int lane = tid % DR_WAVE;
  const int64_t bh = blockIdx.y;
  const int q0 = blockIdx.x * QT;

  const T* qb = q + bh * T_len * D;
  const T* kb = k + bh * T_len * D;
  const T* vb = v + bh * T_len * D;

  const int c_col = lane & 15;
  const int rgrp = lane >> 4;

  // ---- Q fragments: this wave's 16 rows, zero-padded past D and T ----
  const int n_kt_qk = (D + 31) / 32;  // K-tiles in QK^T (K dim = D)
  bf16x8 qfrag[2];
  {
    const int qrow = q0 + wv * 16 + (lane & 15);
#pragma unroll
    for (int kt = 0; kt < 2; ++kt) {
      uint16_t tmp[8];
#pragma unroll
      for (int j = 0; j < 8; ++j) {
        int d = kt * 32 + rgrp * 8 + j;
        float val = (qrow < T_len && d < D) ? ldf(qb + (int64_t)qrow * D + d) : 0.f;
        tmp[j] = f2bf(val);
      }
      qfrag[kt] = *reinterpret_cast<bf16x8*>(tmp);
    }
  }

  // ---- per-row online softmax state (4 rows per lane, C-layout) ----
  // o_acc[nt] is the MFMA C fragment of output d-tile nt: element i = row
  // sub-index (row = (lane>>4)*4 + i), col = nt*16 + (lane&15).
  float m_run[4], l_run[4];
  f32x4 o_acc[4];
#pragma unroll
  for (int i = 0; i < 4; ++i) { m_run[i] = -1e30f; l_run[i] = 0.f; }
  const int n_d_tiles = (D + 15) / 16;
#pragma unroll
  for (int nt = 0; nt < 4; ++nt)
#pragma unroll
    for (int e = 0; e < 4; ++e) o_acc[nt][e] = 0.f;

  const int n_key_tiles = (T_len + KT_KEYS - 1) / KT_KEYS;
  for (int ktile = 0; ktile < n_key_tiles; ++ktile) {
    const int key0 = ktile * KT_KEYS;

    // ---- cooperative stage: K tile [key][d], V^T tile [d][key] ----
    __syncthreads();  // previous iteration's reads done
    for (int id = tid; id < KT_KEYS * (DMAX / 8); id += A_THREADS) {
      int key = id / (DMAX / 8);
      int blk = id % (DMAX / 8);
      uint16_t* dst = reinterpret_cast<uint16_t*>(k_lds + key * K_STRIDE + blk * 16);
      int krow = key0 + key;
#pragma unroll
      for (int e = 0; e < 8; ++e) {
        int d = blk * 8 + e;
        dst[e] = f2bf((krow < T_len && d < D) ? ldf(kb + (int64_t)krow * D + d) : 0.f);
      }
    }
    for (int id = tid; id < DMAX * KT_KEYS / 8; id += A_THREADS) {
      int d = id / (KT_KEYS / 8);
      int kb8 = id % (KT_KEYS / 8);
      uint16_t* dst = reinterpret_cast<uint16_t*>(vt_lds + d * V_STRIDE + kb8 * 16);
#pragma unroll
      for (int e = 0; e < 8; ++e) {
        int key = key0 + kb8 * 8 + e;
        dst[e] = f2bf((key < T_len && d < D) ? ldf(vb + (int64_t)key * D + d) : 0.f);
      }
    }
    __syncthreads();

    // ---- S = Q K^T * scale for this wave's 16 rows x 32 keys ----
    f32x4 s_acc[2];
#pragma unroll
    for (int nt = 0; nt < 2; ++nt) {
      f32x4 a = {0.f, 0.f, 0.f, 0.f};
#pragma unroll
      for (int kt = 0; kt < 2; ++kt) {  // compile-time index (rule 20)
        if (kt >= n_kt_qk) continue;
        // B operand: K^T -> B[dk][key]: lane col = key (16-wide), k = d
        int key = nt * 16 + c_col;
        int d0 = kt * 32 + rgrp * 8;
        bf16x8 bfrag = *reinterpret_cast<const bf16x8*>(
            k_lds + key * K_STRIDE + d0 * 2);
        a = __builtin_amdgcn_mfma_f32_16x16x32_bf16(qfrag[kt], bfrag, a, 0, 0, 0);
      }
      s_acc[nt] = a;
    }

    // ---- online softmax (per lane: 4 rows x 2 key-cols) ----
    float p[2][4];
    float pmax[4];
#pragma unroll
    for (int i = 0; i < 4; ++i) pmax[i] = -1e30f;
#pragma unroll
    for (int nt = 0; nt < 2; ++nt) {
      int key = key0 + nt * 16 + c_col;
#pragma unroll
      for (int i = 0; i < 4; ++i) {
        float s = (key < T_len) ? s_acc[nt][i] * scale : -1e30f;
        p[nt][i] = s;
        pmax[i] = fmaxf(pmax[i], s);
      }
    }
#pragma unroll
    for (int i = 0; i < 4; ++i) {
      pmax[i] = group16_max(pmax[i]);
      float m_new = fmaxf(m_run[i], pmax[i]);
      float alpha = __expf(m_run[i] - m_new);  // m_run starts -1e30 -> alpha 0
      m_run[i] = m_new;
      l_run[i] *= alpha;
#pragma unroll
      for (int nt = 0; nt < 4; ++nt) o_acc[nt][i] *= alpha;
    }
    // exponentiate + row sums (per-wave 16-lane col groups)
#pragma unroll
    for (int nt = 0; nt < 2; ++nt)
#pragma unroll
      for (int i = 0; i < 4; ++i) p[nt][i] = __expf(p[nt][i] - m_run[i]);
#pragma unroll
    for (int i = 0; i < 4; ++i)
      l_run[i] += group16_sum(p[0][i] + p[1][i]);
    // P -> bf16 -> LDS (this wave's own rows; wave-coherent, no barrier)
#pragma unroll
    for (int nt = 0; nt < 2; ++nt) {
      int keyc = nt * 16 + c_col;
#pragma unroll
      for (int i = 0; i < 4; ++i) {
        int row = wv * 16 + rgrp * 4 + i;
        *reinterpret_cast<uint16_t*>(p_lds + row * P_STRIDE + keyc * 2) =
            f2bf(p[nt][i]);
      }
    }

    // ---- O += P V  (A = P from LDS, B = V^T from LDS) ----
    {
      bf16x8 pfrag;
      int prow = wv * 16 + (lane & 15);
      pfrag = *reinterpret_cast<const bf16x8*>(
          p_lds + prow * P_STRIDE + (rgrp * 8) * 2);
#pragma unroll
      for (int nt = 0; nt < 4; ++nt) {  // compile-time index (rule 20)
        if (nt >= n_d_tiles) continue;
        int d = nt * 16 + c_col;
        bf16x8 vfrag = *reinterpret_cast<const bf16x8*>(
            vt_lds + d * V_STRIDE + (rgrp * 8) * 2);
        o_acc[nt] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
            pfrag, vfrag, o_acc[nt], 0, 0, 0);
      }
    }
  }

  // ---- epilogue: O / l, store O and lse ----
#pragma unroll
  for (int i = 0; i < 4; ++i) {
    int row = q0 + wv * 16 + rgrp * 4 + i;
    if (row >= T_len) continue;
    float inv_l = (l_run[i] > 0.f) ? 1.f / l_run[i] : 0.f;
#pragma unroll
    for (int nt = 0; nt < 4; ++nt) {  // compile-time index (rule 20)
      if (nt >= n_d_tiles) continue;
      int d = nt * 16 + c_col;
      if (d < D)
        stf(o + (bh * T_len + row) * D + d, o_acc[nt][i] * inv_l);
    }
    if (c_col == 0)
      lse[bh * T_len + row] = m_run[i] + __logf(fmaxf(l_run[i], 1e-30f));
  }
}

// ---------------------------------------------------------------- backward
// Flash-style: one block per (bh, 64-key tile), 4 waves x 16 keys each.
// Everything key-indexed (S^T, dP^T, dS^T) accumulates in the SAME MFMA
// C-layout — key rows per lane, q columns per 16-lane group — so the
// softmax-gradient elementwise math is entirely lane-local with per-column
// lse / D_row reads.  dK/dV accumulate in registers (exclusive key rows);
// dQ contributions scatter with fp32 atomics (few per block at these T).
template <typename T>
__global__ __launch_bounds__(A_THREADS) void mha_bwd_kernel(
    const T* __restrict__ q,    // (BH, T, D)
    const T* __restrict__ k,
    const T* __restrict__ v,
    const T* __restrict__ o,
    const T* __restrict__ dout,
    const float* __restrict__ lse,   // (BH, T)
    float* __restrict__ dq,          // (BH, T, D) fp32 (atomic accumulate)
    T* __restrict__ dk,              // (BH, T, D)
    T* __restrict__ dv,
    int T_len, int D, float scale) {
  // LDS: q / do tiles (32 x D bf16, padded rows) + per-wave P/ds scratch +
  // per-q-tile lse/Drow
  __shared__ __attribute__((aligned(16))) char q_lds[32 * K_STRIDE];
  __shared__ __attribute__((aligned(16))) char do_lds[32 * K_STRIDE];
  __shared__ __attribute__((aligned(16))) char p_sc[A_WAVES * 16 * (32 * 2 + 16)];
  __shared__ float lse_s[32];
  __shared__ float drow_s[32];

  const int tid = threadIdx.x;
  const int wv = tid / DR_WAVE;
  const int lane = tid % DR_WAVE;
  const int64_t bh = blockIdx.y;
  const int key0 = blockIdx.x * 64 + wv * 16;  // this wave's 16 keys

  const T* qb = q + bh * T_len * D;
  const T* kb = k + bh * T_len * D;
  const T* vb = v + bh * T_len * D;
  const T* ob = o + bh * T_len * D;
  const T* dob = dout + bh * T_len * D;

  const int c_col = lane & 15;
  const int rgrp = lane >> 4;
  const int n_kt_d = (D + 31) / 32;   // K-tiles over the D axis
  const int n_d_tiles = (D + 15) / 16;

  // ---- K and V fragments for this wave's 16 keys (A operands, row = key) ----
  bf16x8 kfrag[2], vfrag[2];
#pragma unroll
  for (int kt = 0; kt < 2; ++kt) {
    uint16_t tk[8], tv[8];
    int krow = key0 + (lane & 15);
#pragma unroll
    for (int j = 0; j < 8; ++j) {
      int d = kt * 32 + rgrp * 8 + j;
      bool ok = krow < T_len && d < D;
      tk[j] = f2bf(ok ? ldf(kb + (int64_t)krow * D + d) : 0.f);
      tv[j] = f2bf(ok ? ldf(vb + (int64_t)krow * D + d) : 0.f);
    }
    kfrag[kt] = *reinterpret_cast<bf16x8*>(tk);
    vfrag[kt] = *reinterpret_cast<bf16x8*>(tv);
  }

  // dK/dV accumulators: this wave's 16 key rows x D
  f32x4 dk_acc[4], dv_acc[4];
#pragma unroll
  for (int nt = 0; nt < 4; ++nt)
#pragma unroll
    for (int e = 0; e < 4; ++e) { dk_acc[nt][e] = 0.f; dv_acc[nt][e] = 0.f; }

  const int PW_STRIDE = 32 * 2 + 16;           // bytes per scratch row
  char* my_p = p_sc + wv * 16 * PW_STRIDE;     // 16 key rows x 32 q cols

  const int n_qt = (T_len + 31) / 32;
  for (int qt = 0; qt < n_qt; ++qt) {
    const int q0 = qt * 32;
    // ---- cooperative stage: q / do tiles + lse + Drow ----
    __syncthreads();
    for (int id = tid; id < 32 * (DMAX / 8); id += A_THREADS) {
      int row = id / (DMAX / 8);
      int blk = id % (DMAX / 8);
      uint16_t* dstq = reinterpret_cast<uint16_t*>(q_lds + row * K_STRIDE + blk * 16);
      uint16_t* dstd = reinterpret_cast<uint16_t*>(do_lds + row * K_STRIDE + blk * 16);
      int qrow = q0 + row;
#pragma unroll
      for (int e = 0; e < 8; ++e) {
        int d = blk * 8 + e;
        bool ok = qrow < T_len && d < D;
        dstq[e] = f2bf(ok ? ldf(qb + (int64_t)qrow * D + d) : 0.f);
        dstd[e] = f2bf(ok ? ldf(dob + (int64_t)qrow * D + d) : 0.f);
      }
    }
    // Drow[r] = sum_d do[r,d] * o[r,d]; lse per row (one wave's worth of rows)
    if (tid < 32) {
      int qrow = q0 + tid;
      float s = 0.f;
      if (qrow < T_len) {
        for (int d = 0; d < D; ++d)
          s += ldf(dob + (int64_t)qrow * D + d) * ldf(ob + (int64_t)qrow * D + d);
        lse_s[tid] = lse[bh * T_len + qrow];
      } else {
        lse_s[tid] = 0.f;
      }
      drow_s[tid] = s;
    }
    __syncthreads();

    // ---- S^T = K Q^T * scale; P^T = exp(S^T - lse[col]) ----
    f32x4 st[2];
#pragma unroll
    for (int nt = 0; nt < 2; ++nt) {
      f32x4 a = {0.f, 0.f, 0.f, 0.f};
#pragma unroll
      for (int kt = 0; kt < 2; ++kt) {
        if (kt >= n_kt_d) continue;
        // B operand: Q^T -> B[d][qcol]: lane col = q (16-wide), k = d
        int qc = nt * 16 + c_col;
        bf16x8 bfrag = *reinterpret_cast<const bf16x8*>(
            q_lds + qc * K_STRIDE + (kt * 32 + rgrp * 8) * 2);
        a = __builtin_amdgcn_mfma_f32_16x16x32_bf16(kfrag[kt], bfrag, a, 0, 0, 0);
      }
      st[nt] = a;
    }
    // dP^T = V dO^T (same structure)
    f32x4 dpt[2];
#pragma unroll
    for (int nt = 0; nt < 2; ++nt) {
      f32x4 a = {0.f, 0.f, 0.f, 0.f};
#pragma unroll
      for (int kt = 0; kt < 2; ++kt) {
        if (kt >= n_kt_d) continue;
        int qc = nt * 16 + c_col;
        bf16x8 bfrag = *reinterpret_cast<const bf16x8*>(
            do_lds + qc * K_STRIDE + (kt * 32 + rgrp * 8) * 2);
        a = __builtin_amdgcn_mfma_f32_16x16x32_bf16(vfrag[kt], bfrag, a, 0, 0, 0);
      }
      dpt[nt] = a;
    }

    // ---- elementwise: P^T and dS^T (lane-local; cols give lse/Drow) ----
    // rows = keys (reg i), cols = q.  Write both into per-wave scratch:
    // P^T rows for the dV GEMM, then dS^T for the dK/dQ GEMMs.
#pragma unroll
    for (int nt = 0; nt < 2; ++nt) {
      int qc = nt * 16 + c_col;
      bool qok = (q0 + qc) < T_len;
      float l = lse_s[qc];
      float dr = drow_s[qc];
#pragma unroll
      for (int i = 0; i < 4; ++i) {
        int key = key0 + rgrp * 4 + i;
        float p = 0.f, ds = 0.f;
        if (qok && key < T_len) {
          p = __expf(st[nt][i] * scale - l);
          ds = p * (dpt[nt][i] - dr) * scale;
        }
        int row = rgrp * 4 + i;
        *reinterpret_cast<uint16_t*>(my_p + row * PW_STRIDE + (nt * 16 + c_col) * 2) =
            f2bf(p);
        st[nt][i] = ds;  // reuse register: stash dS^T
      }
    }
    // dV += P^T dO   (A = P^T from scratch, B = dO tile read k-major)
    {
      bf16x8 pfrag;
      int prow = lane & 15;
      pfrag = *reinterpret_cast<const bf16x8*>(
          my_p + prow * PW_STRIDE + (rgrp * 8) * 2);
#pragma unroll
      for (int nt = 0; nt < 4; ++nt) {
        if (nt >= n_d_tiles) continue;
        // B[qrow][d]: lane col = d, k = qrow -> read dO transposed: stride!
        // dO tile is row-major [q][d]; B operand needs contiguous q for a
        // fixed d — strided 8 reads (D*2 apart), packed here.
        uint16_t bq[8];
        int d = nt * 16 + c_col;
#pragma unroll
        for (int j = 0; j < 8; ++j) {
          int qrow = rgrp * 8 + j;
          bq[j] = *reinterpret_cast<const uint16_t*>(
              do_lds + qrow * K_STRIDE + d * 2);
        }
        bf16x8 bfrag = *reinterpret_cast<bf16x8*>(bq);
        dv_acc[nt] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
            pfrag, bfrag, dv_acc[nt], 0, 0, 0);
      }
    }
    // write dS^T into scratch, then dK += dS^T Q (same pattern as dV)
#pragma unroll
    for (int nt = 0; nt < 2; ++nt)
#pragma unroll
      for (int i = 0; i < 4; ++i) {
        int row = rgrp * 4 + i;
        *reinterpret_cast<uint16_t*>(my_p + row * PW_STRIDE + (nt * 16 + c_col) * 2) =
            f2bf(st[nt][i]);
      }
    {
      bf16x8 dsfrag;
      int prow = lane & 15;
      dsfrag = *reinterpret_cast<const bf16x8*>(
          my_p + prow * PW_STRIDE + (rgrp * 8) * 2);
#pragma unroll
      for (int nt = 0; nt < 4; ++nt) {
        if (nt >= n_d_tiles) continue;
        uint16_t bq[8];
        int d = nt * 16 + c_col;
#pragma unroll
        for (int j = 0; j < 8; ++j) {
          int qrow = rgrp * 8 + j;
          bq[j] = *reinterpret_cast<const uint16_t*>(
              q_lds + qrow * K_STRIDE + d * 2);
        }
        bf16x8 bfrag = *reinterpret_cast<bf16x8*>(bq);
        dk_acc[nt] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
            dsfrag, bfrag, dk_acc[nt], 0, 0, 0);
      }
    }
    // dQ[q0+qc][d] += sum over this wave's 16 keys of dS[qc,key] K[key,d]:
    // MFMA with A = dS (q rows x key cols) = transpose of scratch -> read
    // strided; B = K fragments ALREADY row=key (reuse via LDS? K is in regs
    // as A-layout) — do it with A = dS rows from scratch (strided pack) and
    // B[key][d] read from global k (L2).
#pragma unroll
    for (int qh = 0; qh < 2; ++qh) {  // two 16-row halves of the 32-q tile
      uint16_t aq[8];
      int qrow_l = qh * 16 + (lane & 15);  // q row within the tile
#pragma unroll
      for (int j = 0; j < 8; ++j) {
        int keyr = rgrp * 8 + j;
        aq[j] = keyr < 16 ? *reinterpret_cast<const uint16_t*>(
                                my_p + keyr * PW_STRIDE + qrow_l * 2)
                          : (uint16_t)0;
      }
      // wait: scratch holds only 32 q cols; qrow_l in 0..31 OK.
      // K dim = 32 but only this wave's 16 keys are real (upper half zeroed).
      bf16x8 afrag = *reinterpret_cast<bf16x8*>(aq);
#pragma unroll
      for (int nt = 0; nt < 4; ++nt) {
        if (nt >= n_d_tiles) continue;
        uint16_t bk[8];
        int d = nt * 16 + c_col;
#pragma unroll
        for (int j = 0; j < 8; ++j) {
          int keyr = rgrp * 8 + j;
          int krow = key0 + keyr;
          bool ok = keyr < 16 && krow < T_len && d < D;
          bk[j] = f2bf(ok ? ldf(kb + (int64_t)krow * D + d) : 0.f);
        }
        bf16x8 bfrag = *reinterpret_cast<bf16x8*>(bk);
        f32x4 a = {0.f, 0.f, 0.f, 0.f};
        a = __builtin_amdgcn_mfma_f32_16x16x32_bf16(afrag, bfrag, a, 0, 0, 0);
        // scatter-add into dq (fp32)
#pragma unroll
        for (int i = 0; i < 4; ++i) {
          int qrow = q0 + qh * 16 + rgrp * 4 + i;
          int d2 = nt * 16 + c_col;
          if (qrow < T_len && d2 < D)
            atomicAdd(dq + (bh * T_len + qrow) * D + d2, a[i]);
        }
      }
    }
  }

  // ---- store dK / dV (exclusive key rows) ----
#pragma unroll
  for (int i = 0; i < 4; ++i) {
    int krow = key0 + rgrp * 4 + i;
    if (krow >= T_len) continue;
#pragma unroll
    for (int nt = 0; nt < 4; ++nt) {
      if (nt >= n_d_tiles) continue;
      int d = nt * 16 + c_col;
      if (d < D) {
        stf(dk + (bh * T_len + krow) * D + d, dk_acc[nt][i]);
        stf(dv + (bh * T_len + krow) * D + d, dv_acc[nt][i]);
      }
    }
  }
}

template <typename T>
static void mha_fwd_launch_t(const void* q, const void* k, const void* v, void* o,
                             float* lse, int64_t BH, int T_len, int D, float scale,
                             hipStream_t stream) {
  dim3 grid((T_len + QT - 1) / QT, (unsigned)BH);
  hipLaunchKernelGGL((mha_fwd_kernel<T>), grid, dim3(A_THREADS), 0, stream,
                     (const T*)q, (const T*)k, (const T*)v, (T*)o, lse, T_len, D,
                     scale);
}

template <typename T>
static void mha_bwd_launch_t(const void* q, const void* k, const void* v,
                             const void* o, const void* dout, const float* lse,
                             float* dq, void* dk, void* dv, int64_t BH, int T_len,
                             int D, float scale, hipStream_t stream) {
  dim3 grid((T_len + 63) / 64, (unsigned)BH);
  hipLaunchKernelGGL((mha_bwd_kernel<T>), grid, dim3(A_THREADS), 0, stream,
                     (const T*)q, (const T*)k, (const T*)v, (const T*)o,
                     (const T*)dout, lse, dq, (T*)dk, (T*)dv, T_len, D, scale);
}

}  // namespace dr

extern "C" {

void dr_mha_bwd(const void* q, const void* k, const void* v, const void* o,
                const void* dout, const float* lse, float* dq, void* dk, void* dv,
                int64_t BH, int T_len, int D, float scale, int is_bf16,
                hipStream_t stream) {
  if (is_bf16)
    dr::mha_bwd_launch_t<uint16_t>(q, k, v, o, dout, lse, dq, dk, dv, BH, T_len,
                                   D, scale, stream);
  else
    dr::mha_bwd_launch_t<float>(q, k, v, o, dout, lse, dq, dk, dv, BH, T_len,
                                D, scale, stream);
}

void dr_mha_fwd(const void* q, const void* k, const void* v, void* o, float* lse,
                int64_t BH, int T_len, int D, float scale, int is_bf16,
                hipStream_t stream) {
  if (is_bf16)
    dr::mha_fwd_launch_t<uint16_t>(q, k, v, o, lse, BH, T_len, D, scale, stream);
  else
    dr::mha_fwd_launch_t<float>(q, k, v, o, lse, BH, T_len, D, scale, stream);
}

}  // extern "C"
