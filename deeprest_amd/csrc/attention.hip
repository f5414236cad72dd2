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

template <typename T>
static void mha_fwd_launch_t(const void* q, const void* k, const void* v, void* o,
                             float* lse, int64_t BH, int T_len, int D, float scale,
                             hipStream_t stream) {
  dim3 grid((T_len + QT - 1) / QT, (unsigned)BH);
  hipLaunchKernelGGL((mha_fwd_kernel<T>), grid, dim3(A_THREADS), 0, stream,
                     (const T*)q, (const T*)k, (const T*)v, (T*)o, lse, T_len, D,
                     scale);
}

}  // namespace dr

extern "C" {

void dr_mha_fwd(const void* q, const void* k, const void* v, void* o, float* lse,
                int64_t BH, int T_len, int D, float scale, int is_bf16,
                hipStream_t stream) {
  if (is_bf16)
    dr::mha_fwd_launch_t<uint16_t>(q, k, v, o, lse, BH, T_len, D, scale, stream);
  else
    dr::mha_fwd_launch_t<float>(q, k, v, o, lse, BH, T_len, D, scale, stream);
}

}  // extern "C"
