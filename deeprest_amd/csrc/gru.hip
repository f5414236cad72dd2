// Fused GRU sequence kernel for gfx950 — the framework's hot op.
//
// Replaces the reference's cuDNN GRU (reference: resource-estimation/
// qrnn.py:24,41) with an MI355X-native design: the WHOLE sequence runs in one
// kernel launch.  Rows (= batch x component pairs) evolve independently, so
// each 4-wave workgroup owns a 64-row tile, stages the recurrent weight
// matrix W_hh (3H x H bf16 = 96 KB) in XOR-swizzled LDS once, keeps its
// rows' fp32 hidden state entirely in registers, and per time step runs a
// 64x384x128 MFMA GEMM (v_mfma_f32_16x16x32_bf16) fused with the
// sigmoid/tanh gate epilogue and the per-component FiLM conditioning
// (g = xg * gamma + beta).  State stays fp32; GEMM operands are bf16
// (SURVEY.md "keep state in fp32, activations in low precision").
//
// Fixed geometry: H = 128, 3H = 384, 64 rows/block, 4 waves, 256 threads.
// LDS: W 96 KB + h-tile (bf16, swizzled) 16 KB + xg slots 24 KB = 136 KB
// -> one block per CU (the MFMA loop runs from LDS/registers; ILP across 24
// independent accumulator tiles covers the 1-wave/SIMD occupancy).
//
// Backward: the sequential part only — per reversed step compute the gate
// pre-activation gradients (dpre) and the recurrent chain
// dh_prev = dh*z + dpre @ W_hh (second MFMA phase, W^T image in LDS), write
// dpre to global.  The large batched reductions (dW_hh, db_hh, dgamma,
// dbeta, dx_gates) are plain GEMMs done by rocBLAS on the dpre tensor in the
// autograd wrapper (deeprest_amd/ops/gru.py).
#include "common.h"

namespace dr {

constexpr int H = 128;        // hidden size (fixed)
constexpr int G3H = 384;      // 3*H
constexpr int ROWS = 64;      // rows per block
constexpr int WAVES = 4;
constexpr int THREADS = WAVES * DR_WAVE;
constexpr int KT = H / 32;    // K-tiles of 32 in the fwd GEMM (4)
constexpr int NT = G3H / 16;  // N-tiles of 16 (24)
constexpr int XG_SLOTS = 32;  // distinct batch indices a 64-row tile may span

// LDS byte offsets (single dynamic region, all 16B aligned)
constexpr int LDS_W = 0;                       // 384 x 128 bf16 swizzled (98304 B)
constexpr int LDS_H = LDS_W + G3H * H * 2;     // 64 x 128 bf16 swizzled (16384 B)
constexpr int LDS_XG = LDS_H + ROWS * H * 2;   // 32 x 384 bf16 (24576 B)
constexpr int LDS_FWD_TOTAL = LDS_XG + XG_SLOTS * G3H * 2;  // 139264 B

// bwd LDS: W^T image (128 x 384 bf16) + dpre tile (64 x 384 bf16)
constexpr int LDS_WT = 0;                       // 98304 B
constexpr int LDS_DPRE = LDS_WT + H * G3H * 2;  // 49152 B
constexpr int LDS_BWD_TOTAL = LDS_DPRE + ROWS * G3H * 2;  // 147456 B

// swizzled byte address inside a row-major [rows][128] bf16 tile (256 B rows):
// 16B block index ^= (row & 15) — spreads a 16-lane ds_read_b128 group over
// 16 distinct banks (guide T2 / Guideline 4).
__device__ __forceinline__ int swz(int row, int k_elem) {
  int blk = k_elem >> 3;            // 8 bf16 = 16 B per block
  int within = (k_elem & 7) * 2;
  return row * 256 + ((blk ^ (row & 15)) << 4) + within;
}

// same swizzle for a row-major [rows][384] bf16 tile (768 B rows)
__device__ __forceinline__ int swz768(int row, int k_elem) {
  int blk = k_elem >> 3;
  int within = (k_elem & 7) * 2;
  return row * 768 + ((blk ^ (row & 15)) << 4) + within;
}

__device__ __forceinline__ bf16x8 lds_read8(const char* base, int byte_off) {
  return *reinterpret_cast<const bf16x8*>(base + byte_off);
}

// ---------------------------------------------------------------- forward
template <typename T, bool SAVE>
__global__ __launch_bounds__(THREADS) void gru_fwd_kernel(
    const T* __restrict__ xg,      // (B, TT, 3H)
    const T* __restrict__ gamma,   // (C, 3H)
    const T* __restrict__ beta,    // (C, 3H)
    const T* __restrict__ w_hh,    // (3H, H)
    const float* __restrict__ b_hh,  // (3H,)
    const T* __restrict__ h0,      // (B, C, H)
    T* __restrict__ h_all,         // (B, TT, C, H)
    T* __restrict__ saves,         // (B, TT, C, 4H): r|z|n|hh_n  (SAVE only)
    int B, int TT, int C, int reverse) {
  extern __shared__ __attribute__((aligned(16))) char smem[];
  char* Wl = smem + LDS_W;
  char* Hl = smem + LDS_H;
  char* XGl = smem + LDS_XG;

  const int tid = threadIdx.x;
  const int wv = tid / DR_WAVE;
  const int lane = tid % DR_WAVE;
  const int64_t R = (int64_t)B * C;
  const int64_t r0 = (int64_t)blockIdx.x * ROWS;
  const int b_lo = (int)(r0 / C);

  // ---- prologue: stage W into swizzled LDS ----
  for (int id = tid; id < G3H * (H / 8); id += THREADS) {
    int j = id / (H / 8);
    int blk = id % (H / 8);
    float v[8];
#pragma unroll
    for (int e = 0; e < 8; ++e) v[e] = ldf(w_hh + (int64_t)j * H + blk * 8 + e);
    uint16_t* dst = reinterpret_cast<uint16_t*>(Wl + j * 256 + ((blk ^ (j & 15)) << 4));
#pragma unroll
    for (int e = 0; e < 8; ++e) dst[e] = f2bf(v[e]);
  }

  // ---- per-lane static geometry (C-layout of the 16x16 MFMA tile) ----
  const int c_col = lane & 15;           // col within a 16-wide N-tile
  const int rgrp = lane >> 4;            // row group (0..3)
  int row_of[4];                         // absolute tile row per acc reg
  int64_t r_abs[4];
  int b_of[4], comp_of[4];
  bool live[4];
#pragma unroll
  for (int i = 0; i < 4; ++i) {
    row_of[i] = wv * 16 + rgrp * 4 + i;
    r_abs[i] = r0 + row_of[i];
    live[i] = r_abs[i] < R;
    int64_t rr = live[i] ? r_abs[i] : (R - 1);
    b_of[i] = (int)(rr / C);
    comp_of[i] = (int)(rr % C);
  }

  // ---- preload T-invariant per-lane values: gamma/beta and b_hh ----
  // gamma/beta for (row i, gate g, h-col tile nt) at col c_col.
  float gm[4][3][8], bt[4][3][8];
#pragma unroll
  for (int i = 0; i < 4; ++i) {
    const T* grow = gamma + (int64_t)comp_of[i] * G3H;
    const T* brow = beta + (int64_t)comp_of[i] * G3H;
#pragma unroll
    for (int g = 0; g < 3; ++g)
#pragma unroll
      for (int nt = 0; nt < 8; ++nt) {
        int col = g * H + nt * 16 + c_col;
        gm[i][g][nt] = live[i] ? ldf(grow + col) : 0.f;
        bt[i][g][nt] = live[i] ? ldf(brow + col) : 0.f;
      }
  }
  float bh[3][8];
#pragma unroll
  for (int g = 0; g < 3; ++g)
#pragma unroll
    for (int nt = 0; nt < 8; ++nt) bh[g][nt] = b_hh[g * H + nt * 16 + c_col];

  // ---- fp32 hidden state in registers + bf16 tile in LDS ----
  float h[4][8];
#pragma unroll
  for (int i = 0; i < 4; ++i)
#pragma unroll
    for (int nt = 0; nt < 8; ++nt) {
      int col = nt * 16 + c_col;
      float v = live[i] ? ldf(h0 + ((int64_t)b_of[i] * C + comp_of[i]) * H + col) : 0.f;
      h[i][nt] = v;
      *reinterpret_cast<uint16_t*>(Hl + swz(row_of[i], col)) = f2bf(v);
    }
  __syncthreads();

  // ---- time loop ----
  for (int step = 0; step < TT; ++step) {
    const int t = reverse ? (TT - 1 - step) : step;

    // A-fragments: this wave's 16 rows of the h tile, all 4 K-tiles
    bf16x8 afrag[KT];
    {
      int arow = wv * 16 + (lane & 15);
      int k0 = (lane >> 4) * 8;
#pragma unroll
      for (int kt = 0; kt < KT; ++kt)
        afrag[kt] = lds_read8(Hl, swz(arow, kt * 32 + k0));
    }

    // stage xg[b_lo.., t, :] into LDS (vectorized, cooperative)
    {
      int n_b = (int)(std::min<int64_t>(r0 + ROWS - 1, R - 1) / C) - b_lo + 1;
      for (int id = tid; id < n_b * (G3H / 8); id += THREADS) {
        int slot = id / (G3H / 8);
        int blk = id % (G3H / 8);
        const T* src = xg + (((int64_t)(b_lo + slot) * TT) + t) * G3H + blk * 8;
        uint16_t* dst = reinterpret_cast<uint16_t*>(XGl + slot * (G3H * 2) + blk * 16);
#pragma unroll
        for (int e = 0; e < 8; ++e) dst[e] = f2bf(ldf(src + e));
      }
    }
    __syncthreads();

    // ---- MFMA: hh = h_tile @ W^T -> (64, 384), this wave's 16 rows ----
    f32x4 acc[NT];
#pragma unroll
    for (int nt = 0; nt < NT; ++nt) {
      f32x4 a = {0.f, 0.f, 0.f, 0.f};
#pragma unroll
      for (int kt = 0; kt < KT; ++kt) {
        int j = nt * 16 + c_col;                       // W row (gate col)
        int k0 = kt * 32 + (lane >> 4) * 8;
        bf16x8 bfrag = lds_read8(Wl, swz(j, k0));
        a = __builtin_amdgcn_mfma_f32_16x16x32_bf16(afrag[kt], bfrag, a, 0, 0, 0);
      }
      acc[nt] = a;
    }

    // ---- fused gate epilogue ----
    const uint16_t* xg_rows[4];
#pragma unroll
    for (int i = 0; i < 4; ++i)
      xg_rows[i] = reinterpret_cast<const uint16_t*>(
          XGl + (b_of[i] - b_lo) * (G3H * 2));

#pragma unroll
    for (int nt = 0; nt < 8; ++nt) {
      int col = nt * 16 + c_col;
#pragma unroll
      for (int i = 0; i < 4; ++i) {
        float xr = bf2f(xg_rows[i][col]);
        float xz = bf2f(xg_rows[i][H + col]);
        float xn = bf2f(xg_rows[i][2 * H + col]);
        float g_r = xr * gm[i][0][nt] + bt[i][0][nt];
        float g_z = xz * gm[i][1][nt] + bt[i][1][nt];
        float g_n = xn * gm[i][2][nt] + bt[i][2][nt];
        float rp = sigmoidf_(acc[nt][i] + bh[0][nt] + g_r);
        float zp = sigmoidf_(acc[nt + 8][i] + bh[1][nt] + g_z);
        float hn = acc[nt + 16][i] + bh[2][nt];
        float nn = tanhf(g_n + rp * hn);
        float hnew = (1.f - zp) * nn + zp * h[i][nt];
        h[i][nt] = hnew;
        *reinterpret_cast<uint16_t*>(Hl + swz(row_of[i], col)) = f2bf(hnew);
        if (SAVE && live[i]) {
          T* sv = saves + (((int64_t)b_of[i] * TT + t) * C + comp_of[i]) * (4 * H);
          stf(sv + col, rp);
          stf(sv + H + col, zp);
          stf(sv + 2 * H + col, nn);
          stf(sv + 3 * H + col, hn);
        }
      }
    }
    __syncthreads();

    // ---- cooperative vectorized h_all store (reads the bf16 LDS tile) ----
    for (int id = tid; id < ROWS * (H / 8); id += THREADS) {
      int row = id / (H / 8);
      int blk = id % (H / 8);
      int64_t r = r0 + row;
      if (r >= R) continue;
      int b = (int)(r / C), c = (int)(r % C);
      bf16x8 v = lds_read8(Hl, row * 256 + ((blk ^ (row & 15)) << 4));
      T* dst = h_all + (((int64_t)b * TT + t) * C + c) * H + blk * 8;
      const uint16_t* vu = reinterpret_cast<const uint16_t*>(&v);
#pragma unroll
      for (int e = 0; e < 8; ++e) stf(dst + e, bf2f(vu[e]));
    }
    // barrier covers both: h-store reads done AND next step's a-frag reads
    // see the same consistent tile until the next epilogue writes it.
    __syncthreads();
  }
}

// ---------------------------------------------------------------- backward
// Computes dpre_x (B,TT,C,3H: dr_pre|dz_pre|dn_pre) and dh0; the caller does
// the batched reductions with rocBLAS.
template <typename T>
__global__ __launch_bounds__(THREADS) void gru_bwd_kernel(
    const T* __restrict__ grad_h,   // (B, TT, C, H)
    const T* __restrict__ w_hh,     // (3H, H)
    const T* __restrict__ h0,       // (B, C, H)
    const T* __restrict__ h_all,    // (B, TT, C, H)
    const T* __restrict__ saves,    // (B, TT, C, 4H)
    T* __restrict__ dpre_x,         // (B, TT, C, 4H): dr_pre|dz_pre|dn_pre|d_hh_n
    float* __restrict__ dh0,        // (B, C, H)
    int B, int TT, int C, int reverse) {
  extern __shared__ __attribute__((aligned(16))) char smem[];
  char* WTl = smem + LDS_WT;
  char* DPl = smem + LDS_DPRE;

  const int tid = threadIdx.x;
  const int wv = tid / DR_WAVE;
  const int lane = tid % DR_WAVE;
  const int64_t R = (int64_t)B * C;
  const int64_t r0 = (int64_t)blockIdx.x * ROWS;

  // ---- stage W^T (128 x 384) into swizzled LDS ----
  // WT[k][j] = W[j][k]; rows are 384 bf16 = 768 B -> use 16B-block swizzle
  // with blk ^ (row & 15) on a 48-block row (blk 0..47; XOR over low 4 bits).
  for (int id = tid; id < G3H * (H / 8); id += THREADS) {
    int j = id / (H / 8);          // 0..383
    int kblk = id % (H / 8);       // 0..15 (k block of 8)
#pragma unroll
    for (int e = 0; e < 8; ++e) {
      int k = kblk * 8 + e;
      float v = ldf(w_hh + (int64_t)j * H + k);
      int blk = j >> 3;            // 16B block within WT row k (j/8)
      int within = (j & 7) * 2;
      int byte_off = k * 768 + ((blk ^ (k & 15)) << 4) + within;
      *reinterpret_cast<uint16_t*>(WTl + byte_off) = f2bf(v);
    }
  }

  const int c_col = lane & 15;
  const int rgrp = lane >> 4;
  int row_of[4];
  int64_t r_abs[4];
  int b_of[4], comp_of[4];
  bool live[4];
#pragma unroll
  for (int i = 0; i < 4; ++i) {
    row_of[i] = wv * 16 + rgrp * 4 + i;
    r_abs[i] = r0 + row_of[i];
    live[i] = r_abs[i] < R;
    int64_t rr = live[i] ? r_abs[i] : (R - 1);
    b_of[i] = (int)(rr / C);
    comp_of[i] = (int)(rr % C);
  }

  float dh_carry[4][8];
#pragma unroll
  for (int i = 0; i < 4; ++i)
#pragma unroll
    for (int nt = 0; nt < 8; ++nt) dh_carry[i][nt] = 0.f;

  // zero the dpre tile once: pad rows (r >= R) never write it afterwards, so
  // their MFMA inputs stay finite
  for (int id = tid; id < ROWS * G3H; id += THREADS)
    reinterpret_cast<uint16_t*>(DPl)[id] = 0;
  __syncthreads();

  for (int step = 0; step < TT; ++step) {
    // reversed traversal of the forward processing order
    const int t = reverse ? step : (TT - 1 - step);
    const int tprev = reverse ? (t + 1) : (t - 1);
    const bool use_h0 = reverse ? (t == TT - 1) : (t == 0);

    float zs[4][8];  // saved z, needed again after the MFMA phase
#pragma unroll
    for (int nt = 0; nt < 8; ++nt) {
      int col = nt * 16 + c_col;
#pragma unroll
      for (int i = 0; i < 4; ++i) {
        if (!live[i]) { zs[i][nt] = 0.f; continue; }
        const int64_t bc = (int64_t)b_of[i] * TT;
        const T* sv = saves + ((bc + t) * C + comp_of[i]) * (4 * H);
        float rp = ldf(sv + col);
        float zp = ldf(sv + H + col);
        float nn = ldf(sv + 2 * H + col);
        float hn = ldf(sv + 3 * H + col);
        float hp = use_h0
            ? ldf(h0 + ((int64_t)b_of[i] * C + comp_of[i]) * H + col)
            : ldf(h_all + ((bc + tprev) * C + comp_of[i]) * H + col);
        float g = ldf(grad_h + ((bc + t) * C + comp_of[i]) * H + col);
        float dht = g + dh_carry[i][nt];
        float dz = dht * (hp - nn);
        float dn = dht * (1.f - zp);
        float dnp = dn * (1.f - nn * nn);
        float dhhn = dnp * rp;
        float dr = dnp * hn;
        float drp = dr * rp * (1.f - rp);
        float dzp = dz * zp * (1.f - zp);
        dh_carry[i][nt] = dht * zp;    // partial; MFMA adds dpre @ W
        zs[i][nt] = zp;

        T* dx = dpre_x + ((bc + t) * C + comp_of[i]) * (4 * H);
        stf(dx + col, drp);
        stf(dx + H + col, dzp);
        stf(dx + 2 * H + col, dnp);
        stf(dx + 3 * H + col, dhhn);
        // LDS dpre_W image (768 B rows): dr_pre | dz_pre | d_hh_n
        *reinterpret_cast<uint16_t*>(DPl + swz768(row_of[i], col)) = f2bf(drp);
        *reinterpret_cast<uint16_t*>(DPl + swz768(row_of[i], H + col)) = f2bf(dzp);
        *reinterpret_cast<uint16_t*>(DPl + swz768(row_of[i], 2 * H + col)) = f2bf(dhhn);
      }
    }
    __syncthreads();

    // ---- MFMA: delta = dpre_W (64x384) @ W (384x128), wave's 16 rows ----
    // A from DPl rows (768 B rows, swizzle over 48 blocks), B from WT image.
    {
      bf16x8 afrag[12];
      int arow = wv * 16 + (lane & 15);
      int k0 = (lane >> 4) * 8;
#pragma unroll
      for (int kt = 0; kt < 12; ++kt) {
        int k = kt * 32 + k0;                         // 0..383 (j index)
        int blk = k >> 3;
        afrag[kt] = lds_read8(
            DPl, arow * 768 + ((blk ^ (arow & 15)) << 4) + ((k & 7) * 2));
      }
#pragma unroll
      for (int nt = 0; nt < 8; ++nt) {
        f32x4 a = {0.f, 0.f, 0.f, 0.f};
#pragma unroll
        for (int kt = 0; kt < 12; ++kt) {
          int n = nt * 16 + c_col;                    // output h-col
          int j0 = kt * 32 + (lane >> 4) * 8;         // K (j) index
          int blk = j0 >> 3;
          bf16x8 bfrag = lds_read8(
              WTl, n * 768 + ((blk ^ (n & 15)) << 4) + ((j0 & 7) * 2));
          a = __builtin_amdgcn_mfma_f32_16x16x32_bf16(afrag[kt], bfrag, a, 0, 0, 0);
        }
#pragma unroll
        for (int i = 0; i < 4; ++i) dh_carry[i][nt] += a[i];
      }
    }
    __syncthreads();  // DPl consumed; next step may overwrite
  }

  // ---- dh0 = final carry ----
#pragma unroll
  for (int nt = 0; nt < 8; ++nt) {
    int col = nt * 16 + c_col;
#pragma unroll
    for (int i = 0; i < 4; ++i) {
      if (!live[i]) continue;
      dh0[((int64_t)b_of[i] * C + comp_of[i]) * H + col] = dh_carry[i][nt];
    }
  }
}

// ------------------------------------------------- backward reductions
// One pass over dpre per kernel (replaces einsum broadcasts that dominated
// the profiled step: 44% elementwise time before this existed).

// dxg[bt, j] = sum_c dpre[bt, c, j] * gamma[c, j]   (j in [0, 3H))
template <typename T>
__global__ void gru_dxg_kernel(const T* __restrict__ dpre,   // (BT, C, 4H)
                               const T* __restrict__ gamma,  // (C, 3H)
                               T* __restrict__ dxg,          // (BT, 3H)
                               int64_t BT, int C) {
  const int64_t n_chunks = BT * (G3H / 8);
  for (int64_t idx = (int64_t)blockIdx.x * blockDim.x + threadIdx.x; idx < n_chunks;
       idx += (int64_t)gridDim.x * blockDim.x) {
    const int64_t bt = idx / (G3H / 8);
    const int j0 = (int)(idx % (G3H / 8)) * 8;
    float acc[8];
#pragma unroll
    for (int e = 0; e < 8; ++e) acc[e] = 0.f;
    const T* base = dpre + bt * C * (4 * H) + j0;
    for (int c = 0; c < C; ++c) {
      const T* dp = base + (int64_t)c * (4 * H);
      const T* gm = gamma + (int64_t)c * G3H + j0;
#pragma unroll
      for (int e = 0; e < 8; ++e) acc[e] += ldf(dp + e) * ldf(gm + e);
    }
    T* out = dxg + bt * G3H + j0;
#pragma unroll
    for (int e = 0; e < 8; ++e) stf(out + e, acc[e]);
  }
}

// dgamma[c, j] = sum_bt dpre[bt, c, j] * xg[bt, j];  dbeta[c, j] = sum_bt dpre
// grid.y slices the BT axis; f32 atomics finalize (few M adds total).
template <typename T>
__global__ void gru_dgamma_kernel(const T* __restrict__ dpre,  // (BT, C, 4H)
                                  const T* __restrict__ xg,    // (BT, 3H)
                                  float* __restrict__ dgamma,  // (C, 3H) f32, zeroed
                                  float* __restrict__ dbeta,   // (C, 3H) f32, zeroed
                                  int64_t BT, int C) {
  const int n_threads_needed = C * (G3H / 8);
  const int tid_g = blockIdx.x * blockDim.x + threadIdx.x;
  if (tid_g >= n_threads_needed) return;
  const int c = tid_g / (G3H / 8);
  const int j0 = (tid_g % (G3H / 8)) * 8;
  const int64_t bt_lo = BT * blockIdx.y / gridDim.y;
  const int64_t bt_hi = BT * (blockIdx.y + 1) / gridDim.y;
  float accg[8], accb[8];
#pragma unroll
  for (int e = 0; e < 8; ++e) { accg[e] = 0.f; accb[e] = 0.f; }
  for (int64_t bt = bt_lo; bt < bt_hi; ++bt) {
    const T* dp = dpre + (bt * C + c) * (4 * H) + j0;
    const T* x = xg + bt * G3H + j0;
#pragma unroll
    for (int e = 0; e < 8; ++e) {
      float d = ldf(dp + e);
      accg[e] += d * ldf(x + e);
      accb[e] += d;
    }
  }
  float* g_out = dgamma + (int64_t)c * G3H + j0;
  float* b_out = dbeta + (int64_t)c * G3H + j0;
#pragma unroll
  for (int e = 0; e < 8; ++e) {
    atomicAdd(g_out + e, accg[e]);
    atomicAdd(b_out + e, accb[e]);
  }
}

template <typename T>
static void gru_fwd_launch_t(const void* xg, const void* gamma, const void* beta,
                             const void* w_hh, const float* b_hh, const void* h0,
                             void* h_all, void* saves, int B, int TT, int C,
                             int reverse, int save, hipStream_t stream) {
  int64_t R = (int64_t)B * C;
  int grid = (int)((R + ROWS - 1) / ROWS);
  static bool attr_set = false;
  if (!attr_set) {
    DR_HIP_CHECK(hipFuncSetAttribute(
        reinterpret_cast<const void*>(&gru_fwd_kernel<T, true>),
        hipFuncAttributeMaxDynamicSharedMemorySize, LDS_FWD_TOTAL));
    DR_HIP_CHECK(hipFuncSetAttribute(
        reinterpret_cast<const void*>(&gru_fwd_kernel<T, false>),
        hipFuncAttributeMaxDynamicSharedMemorySize, LDS_FWD_TOTAL));
    attr_set = true;
  }
  if (save)
    hipLaunchKernelGGL((gru_fwd_kernel<T, true>), dim3(grid), dim3(THREADS),
                       LDS_FWD_TOTAL, stream, (const T*)xg, (const T*)gamma,
                       (const T*)beta, (const T*)w_hh, b_hh, (const T*)h0,
                       (T*)h_all, (T*)saves, B, TT, C, reverse);
  else
    hipLaunchKernelGGL((gru_fwd_kernel<T, false>), dim3(grid), dim3(THREADS),
                       LDS_FWD_TOTAL, stream, (const T*)xg, (const T*)gamma,
                       (const T*)beta, (const T*)w_hh, b_hh, (const T*)h0,
                       (T*)h_all, (T*)saves, B, TT, C, reverse);
}

template <typename T>
static void gru_bwd_launch_t(const void* grad_h, const void* w_hh, const void* h0,
                             const void* h_all, const void* saves, void* dpre_x,
                             float* dh0, int B, int TT, int C, int reverse,
                             hipStream_t stream) {
  int64_t R = (int64_t)B * C;
  int grid = (int)((R + ROWS - 1) / ROWS);
  static bool attr_set = false;
  if (!attr_set) {
    DR_HIP_CHECK(hipFuncSetAttribute(
        reinterpret_cast<const void*>(&gru_bwd_kernel<T>),
        hipFuncAttributeMaxDynamicSharedMemorySize, LDS_BWD_TOTAL));
    attr_set = true;
  }
  hipLaunchKernelGGL((gru_bwd_kernel<T>), dim3(grid), dim3(THREADS),
                     LDS_BWD_TOTAL, stream, (const T*)grad_h, (const T*)w_hh,
                     (const T*)h0, (const T*)h_all, (const T*)saves, (T*)dpre_x,
                     dh0, B, TT, C, reverse);
}

template <typename T>
static void gru_reduce_launch_t(const void* dpre, const void* gamma, const void* xg,
                                void* dxg, float* dgamma, float* dbeta, int64_t BT,
                                int C, hipStream_t stream) {
  {
    int64_t n = BT * (G3H / 8);
    int grid = (int)std::min<int64_t>((n + 255) / 256, 4096);
    hipLaunchKernelGGL((gru_dxg_kernel<T>), dim3(grid), dim3(256), 0, stream,
                       (const T*)dpre, (const T*)gamma, (T*)dxg, BT, C);
  }
  {
    int n_threads = C * (G3H / 8);
    int gx = (n_threads + 255) / 256;
    int gy = 32;  // BT slices
    hipLaunchKernelGGL((gru_dgamma_kernel<T>), dim3(gx, gy), dim3(256), 0, stream,
                       (const T*)dpre, (const T*)xg, dgamma, dbeta, BT, C);
  }
}

}  // namespace dr

extern "C" {

void dr_gru_bwd_reduce(const void* dpre, const void* gamma, const void* xg,
                       void* dxg, float* dgamma, float* dbeta, int64_t BT, int C,
                       int is_bf16, hipStream_t stream) {
  if (is_bf16)
    dr::gru_reduce_launch_t<uint16_t>(dpre, gamma, xg, dxg, dgamma, dbeta, BT, C,
                                      stream);
  else
    dr::gru_reduce_launch_t<float>(dpre, gamma, xg, dxg, dgamma, dbeta, BT, C,
                                   stream);
}

void dr_gru_fwd(const void* xg, const void* gamma, const void* beta,
                const void* w_hh, const float* b_hh, const void* h0, void* h_all,
                void* saves, int B, int TT, int C, int reverse, int save,
                int is_bf16, hipStream_t stream) {
  if (is_bf16)
    dr::gru_fwd_launch_t<uint16_t>(xg, gamma, beta, w_hh, b_hh, h0, h_all, saves,
                                   B, TT, C, reverse, save, stream);
  else
    dr::gru_fwd_launch_t<float>(xg, gamma, beta, w_hh, b_hh, h0, h_all, saves,
                                B, TT, C, reverse, save, stream);
}

void dr_gru_bwd(const void* grad_h, const void* w_hh, const void* h0,
                const void* h_all, const void* saves, void* dpre_x, float* dh0,
                int B, int TT, int C, int reverse, int is_bf16, hipStream_t stream) {
  if (is_bf16)
    dr::gru_bwd_launch_t<uint16_t>(grad_h, w_hh, h0, h_all, saves, dpre_x, dh0,
                                   B, TT, C, reverse, stream);
  else
    dr::gru_bwd_launch_t<float>(grad_h, w_hh, h0, h_all, saves, dpre_x, dh0,
                                B, TT, C, reverse, stream);
}

}  // extern "C"
