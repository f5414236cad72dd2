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
// IO layout note: the training-only side tensors (forward saves r|z|n|hh_n,
// backward dpre dr|dz|dn|d_hhn) use a within-row permutation "pi":
//   pi(gate, col) = gate*H + (col & 15)*8 + (col >> 4)
// so each lane's 8 values per gate (its MFMA C-fragment columns, stride 16)
// are CONTIGUOUS 16 bytes -> one vector load/store instead of 8 scalars.
// Only this file and ops/gru.py (the dW row unpermute) know about pi.
//
// Backward: the sequential chain only — per reversed step compute the gate
// pre-activation grads (phase A, vectorized pi IO) and the recurrent term
// dh_prev = dh*z + dpre @ W_hh.  The dpre tile round-trips through L2 (same
// CU writes then reads it after __syncthreads; per-CU L1 is updated by its
// own stores), against a pi-row-permuted W image in LDS, so no LDS dpre
// image and one barrier per step.  Batched reductions (dW_hh, db_hh, dgamma,
// dbeta, dx_gates) happen outside: rocBLAS GEMMs + the fused reduce kernels
// below (driven by ops/gru.py).
//
// Fixed geometry: H = 128, 3H = 384, 64 rows/block, 4 waves, 256 threads.
#include "common.h"

namespace dr {

constexpr int H = 128;        // hidden size (fixed)
constexpr int G3H = 384;      // 3*H
constexpr int G4H = 512;      // 4*H
constexpr int ROWS = 64;      // rows per block
constexpr int WAVES = 4;
constexpr int THREADS = WAVES * DR_WAVE;
constexpr int KT = H / 32;    // K-tiles of 32 in the fwd GEMM (4)
constexpr int NT = G3H / 16;  // N-tiles of 16 (24)
constexpr int XG_SLOTS = 32;  // distinct batch indices a 64-row tile may span

// forward LDS offsets (single dynamic region, all 16B aligned).
// Forward stages W in LDS (an L2-streamed W was measured SLOWER for the
// forward: 96 KB/wave/step of L2 B-fragment traffic outweighed the
// occupancy gain); the backward streams W from L2 instead (its register
// budget allows 2 waves/SIMD there, and it has no LDS at all).
constexpr int LDS_W = 0;                       // 384 x 128 bf16 swizzled (98304 B)
constexpr int LDS_H = LDS_W + G3H * H * 2;     // 64 x 128 bf16 swizzled (16384 B)
constexpr int LDS_XG = LDS_H + ROWS * H * 2;   // 32 x 384 bf16 (24576 B)
constexpr int LDS_FWD_TOTAL = LDS_XG + XG_SLOTS * G3H * 2;  // 139264 B

// backward uses NO LDS: the pi-permuted W image stays in L2 (96 KB, shared
// by every block) so occupancy is VGPR-bound (2-3 waves/SIMD) instead of
// LDS-bound (1 wave/SIMD) — the kernel is latency-bound, occupancy hides it.

// swizzled byte address inside a row-major [rows][128] bf16 tile (256 B rows)
__device__ __forceinline__ int swz(int row, int k_elem) {
  int blk = k_elem >> 3;            // 8 bf16 = 16 B per block
  int within = (k_elem & 7) * 2;
  return row * 256 + ((blk ^ (row & 15)) << 4) + within;
}

// same for a row-major [rows][128] fp8 tile (128 B rows, 8 B blocks)
__device__ __forceinline__ int swz8(int row, int k_elem) {
  int blk = k_elem >> 3;
  return row * 128 + ((blk ^ (row & 15)) << 3) + (k_elem & 7);
}

__device__ __forceinline__ uint8_t f2fp8(float v) {
  __hip_fp8_e4m3 x(v);              // OCP e4m3fn on gfx950 (hardware cvt)
  return x.__x;
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

// ---- pi-layout vector IO: 8 lane-owned values <-> 16/32 contiguous bytes ----
template <typename T>
__device__ __forceinline__ void st8(T* dst, const float* v);
template <>
__device__ __forceinline__ void st8<uint16_t>(uint16_t* dst, const float* v) {
  uint4 p;
  p.x = f2bf2(v[0], v[1]);
  p.y = f2bf2(v[2], v[3]);
  p.z = f2bf2(v[4], v[5]);
  p.w = f2bf2(v[6], v[7]);
  *reinterpret_cast<uint4*>(dst) = p;
}
template <>
__device__ __forceinline__ void st8<float>(float* dst, const float* v) {
  *reinterpret_cast<float4*>(dst) = *reinterpret_cast<const float4*>(v);
  *reinterpret_cast<float4*>(dst + 4) = *reinterpret_cast<const float4*>(v + 4);
}

template <typename T>
__device__ __forceinline__ void ld8(const T* src, float* v);
template <>
__device__ __forceinline__ void ld8<uint16_t>(const uint16_t* src, float* v) {
  uint4 raw = *reinterpret_cast<const uint4*>(src);
  const uint16_t* p = reinterpret_cast<const uint16_t*>(&raw);
#pragma unroll
  for (int e = 0; e < 8; ++e) v[e] = bf2f(p[e]);
}
template <>
__device__ __forceinline__ void ld8<float>(const float* src, float* v) {
  *reinterpret_cast<float4*>(v) = *reinterpret_cast<const float4*>(src);
  *reinterpret_cast<float4*>(v + 4) = *reinterpret_cast<const float4*>(src + 4);
}

// store an 8-wide bf16 fragment to a T-typed tensor (bf16: one 16 B store)
template <typename T>
__device__ __forceinline__ void st_frag(T* dst, bf16x8 v);
template <>
__device__ __forceinline__ void st_frag<uint16_t>(uint16_t* dst, bf16x8 v) {
  *reinterpret_cast<uint4*>(dst) = *reinterpret_cast<const uint4*>(&v);
}
template <>
__device__ __forceinline__ void st_frag<float>(float* dst, bf16x8 v) {
  const uint16_t* p = reinterpret_cast<const uint16_t*>(&v);
#pragma unroll
  for (int e = 0; e < 8; ++e) dst[e] = bf2f(p[e]);
}

// load 8 T values as raw bf16 bit patterns (bf16: one 16 B load)
template <typename T>
__device__ __forceinline__ void ld_raw8(const T* src, uint16_t* out);
template <>
__device__ __forceinline__ void ld_raw8<uint16_t>(const uint16_t* src, uint16_t* out) {
  uint4 raw = *reinterpret_cast<const uint4*>(src);
  *reinterpret_cast<uint4*>(out) = raw;
}
template <>
__device__ __forceinline__ void ld_raw8<float>(const float* src, uint16_t* out) {
#pragma unroll
  for (int e = 0; e < 8; ++e) out[e] = f2bf(src[e]);
}

// load 8 values directly as an MFMA bf16 fragment (no f32 round trip)
template <typename T>
__device__ __forceinline__ bf16x8 ld_frag(const T* src);
template <>
__device__ __forceinline__ bf16x8 ld_frag<uint16_t>(const uint16_t* src) {
  return *reinterpret_cast<const bf16x8*>(src);
}
template <>
__device__ __forceinline__ bf16x8 ld_frag<float>(const float* src) {
  uint16_t p[8];
#pragma unroll
  for (int e = 0; e < 8; ++e) p[e] = f2bf(src[e]);
  return *reinterpret_cast<bf16x8*>(p);
}

// ---------------------------------------------------------------- forward
// FP8 = true: W and the h mirror live in LDS as OCP e4m3 and the recurrent
// GEMM runs on v_mfma_f32_16x16x32_fp8_fp8 (state stays fp32 in registers;
// activations/outputs keep dtype T).  Inference-only (BASELINE config 5:
// fp8 MFMA for the long-horizon path); training uses bf16.
// NSUB = 2: 8 waves / 128-row tile — 2 waves/SIMD (the kernel is
// latency-stalled at 1 wave/SIMD); gamma/beta/b_hh then stream from L1/L2
// instead of registers so each wave fits the 256-reg budget.
template <typename T, bool SAVE, bool FP8 = false, int NSUB = 1>
__global__ __launch_bounds__(THREADS * NSUB) void gru_fwd_kernel(
    const T* __restrict__ xg,      // (B, TT, 3H)
    const T* __restrict__ gamma,   // (C, 3H)
    const T* __restrict__ beta,    // (C, 3H)
    const uint16_t* __restrict__ w_gemm,  // (3H, H) bf16 GEMM weight image
    const float* __restrict__ b_hh,  // (3H,)
    const T* __restrict__ h0,      // (B, C, H)
    T* __restrict__ h_all,         // (B, TT, C, H)
    T* __restrict__ saves,         // (B, TT, C, 4H) pi layout (SAVE only)
    int B, int TT, int C, int reverse) {
  extern __shared__ __attribute__((aligned(16))) char smem[];
  constexpr int ELT = FP8 ? 1 : 2;             // GEMM-tile element bytes
  constexpr int BROWS = ROWS * NSUB;           // rows per block
  constexpr int BTHREADS = THREADS * NSUB;
  char* Wl = smem;
  char* Hl = smem + G3H * H * ELT;
  char* XGl = Hl + BROWS * H * ELT;

  const int tid = threadIdx.x;
  const int wv = tid / DR_WAVE;
  const int lane = tid % DR_WAVE;
  const int64_t R = (int64_t)B * C;
  const int64_t r0 = (int64_t)blockIdx.x * BROWS;
  const int b_lo = (int)(r0 / C);

  // ---- prologue: stage the W image into swizzled LDS ----
  for (int id = tid; id < G3H * (H / 8); id += BTHREADS) {
    int j = id / (H / 8);
    int blk = id % (H / 8);
    bf16x8 v = *reinterpret_cast<const bf16x8*>(w_gemm + (int64_t)j * H + blk * 8);
    if constexpr (FP8) {
      const uint16_t* bv = reinterpret_cast<const uint16_t*>(&v);
      uint8_t q[8];
#pragma unroll
      for (int e = 0; e < 8; ++e) q[e] = f2fp8(bf2f(bv[e]));
      *reinterpret_cast<uint64_t*>(Wl + j * 128 + ((blk ^ (j & 15)) << 3)) =
          *reinterpret_cast<const uint64_t*>(q);
    } else {
      *reinterpret_cast<bf16x8*>(Wl + j * 256 + ((blk ^ (j & 15)) << 4)) = v;
    }
  }

  // ---- per-lane static geometry (C-layout of the 16x16 MFMA tile) ----
  const int c_col = lane & 15;           // col within a 16-wide N-tile
  const int rgrp = lane >> 4;            // row group (0..3)
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

  // ---- preload T-invariant per-lane values ----
  // NSUB==1: gamma/beta packed as bf16 pairs in one u32 per (row, gate,
  // tile) — LDS caps this variant at 1 block/CU so the registers are free.
  // NSUB==2: two waves/SIMD need the registers back; gamma/beta/b_hh are
  // L1/L2-resident and reload per step (opaque pointers in the epilogue).
  constexpr int GBN = (NSUB == 1) ? 8 : 1;
  uint32_t gb[4][3][GBN];
  float bh[3][GBN];
  int gb_off[4];
#pragma unroll
  for (int i = 0; i < 4; ++i) gb_off[i] = comp_of[i] * G3H;
  if constexpr (NSUB == 1) {
#pragma unroll
    for (int i = 0; i < 4; ++i) {
      const T* grow = gamma + gb_off[i];
      const T* brow = beta + gb_off[i];
#pragma unroll
      for (int g = 0; g < 3; ++g)
#pragma unroll
        for (int nt = 0; nt < 8; ++nt) {
          int col = g * H + nt * 16 + c_col;
          uint16_t gv = live[i] ? f2bf(ldf(grow + col)) : 0;
          uint16_t bv = live[i] ? f2bf(ldf(brow + col)) : 0;
          gb[i][g][nt] = ((uint32_t)bv << 16) | gv;
        }
    }
#pragma unroll
    for (int g = 0; g < 3; ++g)
#pragma unroll
      for (int nt = 0; nt < 8; ++nt) bh[g][nt] = b_hh[g * H + nt * 16 + c_col];
  }

  // ---- fp32 hidden state in registers + bf16 tile in LDS ----
  float h[4][8];
#pragma unroll
  for (int i = 0; i < 4; ++i)
#pragma unroll
    for (int nt = 0; nt < 8; ++nt) {
      int col = nt * 16 + c_col;
      float v = live[i] ? ldf(h0 + ((int64_t)b_of[i] * C + comp_of[i]) * H + col) : 0.f;
      h[i][nt] = v;
      if constexpr (FP8)
        *reinterpret_cast<uint8_t*>(Hl + swz8(row_of[i], col)) = f2fp8(v);
      else
        *reinterpret_cast<uint16_t*>(Hl + swz(row_of[i], col)) = f2bf(v);
    }

  // ---- t-invariant descriptors ----
  // The expensive per-step address math was the runtime divisions (r / C)
  // and 64-bit muls: precompute ROW indices once, share ONE per-step time
  // offset, and derive addresses with shifts/adds.
  const int t_first = reverse ? (TT - 1) : 0;
  const int dirC = (reverse ? -1 : 1) * C;           // row-index per-t stride

  // xg staging chunks owned by this thread (slot, blk fixed over t);
  // offsets are 32-bit (binding checks xg.numel() < 2^31)
  constexpr int MAXCH = 6;  // covers n_b <= 32 slots at 256 threads
  int stage_soff[MAXCH];    // xg element offset at t_first
  int stage_pi0[MAXCH];     // pi-layout LDS element offset of the chunk
  int n_stage = 0;
  {
    int n_b = (int)(std::min<int64_t>(r0 + BROWS - 1, R - 1) / C) - b_lo + 1;
    for (int id = tid; id < n_b * (G3H / 8) && n_stage < MAXCH; id += BTHREADS) {
      int slot = id / (G3H / 8);
      int blk = id % (G3H / 8);
      int j0 = blk * 8;
      int g = j0 >> 7;
      int col0 = j0 & 127;
      stage_soff[n_stage] = (int)((((int64_t)(b_lo + slot) * TT) + t_first) * G3H) + j0;
      stage_pi0[n_stage] =
          slot * G3H + g * H + ((col0 & 15) << 3) + (col0 >> 4);
      ++n_stage;
    }
  }
  // h_all cooperative store chunks (4 per thread; row index fixed over t)
  int64_t hout_bc[4];       // (b*TT + t_first)*C + c
  int hout_blk[4];
  int hout_lds[4];
  bool hout_live[4];
#pragma unroll
  for (int u = 0; u < 4; ++u) {
    int id = tid + u * BTHREADS;            // BROWS*(H/8) = 4*BTHREADS
    int row = id / (H / 8);
    int blk = id % (H / 8);
    int64_t r = r0 + row;
    hout_live[u] = r < R;
    int64_t rr = hout_live[u] ? r : (R - 1);
    int b = (int)(rr / C), c = (int)(rr % C);
    hout_bc[u] = ((int64_t)b * TT + t_first) * C + c;
    hout_blk[u] = blk * 8;
    hout_lds[u] = FP8 ? (row * 128 + ((blk ^ (row & 15)) << 3))
                      : (row * 256 + ((blk ^ (row & 15)) << 4));
  }
  // per-lane-row indices for the saves stores (pi layout)
  int64_t sv_bc[4];
#pragma unroll
  for (int i = 0; i < 4; ++i)
    sv_bc[i] = ((int64_t)b_of[i] * TT + t_first) * C + comp_of[i];
  // per-wave A-fragment LDS offsets
  int afrag_off[KT];
  {
    int arow = wv * 16 + (lane & 15);
    int k0 = (lane >> 4) * 8;
#pragma unroll
    for (int kt = 0; kt < KT; ++kt)
      afrag_off[kt] = FP8 ? swz8(arow, kt * 32 + k0) : swz(arow, kt * 32 + k0);
  }
  // xg LDS row bases (pi layout, element-indexed)
  const uint16_t* xg_rows[4];
#pragma unroll
  for (int i = 0; i < 4; ++i)
    xg_rows[i] = reinterpret_cast<const uint16_t*>(XGl) +
                 (b_of[i] - b_lo) * G3H + c_col * 8;

  __syncthreads();

  // ---- time loop ----
  int stage_toff = 0;                        // += dir*G3H per step (32-bit)
  int64_t bc_toff = 0;                       // += dirC per step (row-index units)
  const int stage_tstep = (reverse ? -1 : 1) * G3H;
  for (int step = 0; step < TT; ++step) {
    // A-fragments: this wave's 16 rows of the h tile, all 4 K-tiles
    bf16x8 afrag[KT];
    int64_t afrag8[KT];
#pragma unroll
    for (int kt = 0; kt < KT; ++kt) {
      if constexpr (FP8)
        afrag8[kt] = *reinterpret_cast<const int64_t*>(Hl + afrag_off[kt]);
      else
        afrag[kt] = lds_read8(Hl, afrag_off[kt]);
    }

    // stage xg[., t, :] into LDS in pi layout: the 8 loaded naturals land at
    // positions pi_0 + 8e (16 B stride), so the epilogue reads each (row,
    // gate) slice as ONE ds_read_b128 instead of 8 scalar reads.
#pragma unroll
    for (int u = 0; u < MAXCH; ++u) {        // compile-time index (rule 20)
      if (u < n_stage) {
        const T* src = xg + stage_soff[u] + stage_toff;
        uint16_t* dst = reinterpret_cast<uint16_t*>(XGl) + stage_pi0[u];
        uint16_t vals[8];
        ld_raw8(src, vals);
#pragma unroll
        for (int e = 0; e < 8; ++e) dst[e * 8] = vals[e];
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
        if constexpr (FP8) {
          int64_t bfrag = *reinterpret_cast<const int64_t*>(Wl + swz8(j, k0));
          a = __builtin_amdgcn_mfma_f32_16x16x32_fp8_fp8(afrag8[kt], bfrag, a, 0, 0, 0);
        } else {
          bf16x8 bfrag = lds_read8(Wl, swz(j, k0));
          a = __builtin_amdgcn_mfma_f32_16x16x32_bf16(afrag[kt], bfrag, a, 0, 0, 0);
        }
      }
      acc[nt] = a;
    }

    // ---- fused gate epilogue (vectorized pi-layout saves) ----
    const T* gm_v = gamma;
    const T* bt_v = beta;
    const float* bh_v = b_hh;
    if constexpr (NSUB == 2) asm volatile("" : "+v"(gm_v), "+v"(bt_v), "+v"(bh_v));
#pragma unroll
    for (int i = 0; i < 4; ++i) {
      const T* grow = gm_v + gb_off[i];
      const T* brow = bt_v + gb_off[i];
      float fr[8], fz[8], fn[8];
#pragma unroll
      for (int nt = 0; nt < 8; ++nt) {
        // pi layout: per-gate values sit at CONSECUTIVE LDS addresses over
        // nt, so these scalar reads coalesce/merge cheaply
        float xr = bf2f(xg_rows[i][nt]);
        float xz = bf2f(xg_rows[i][H + nt]);
        float xn = bf2f(xg_rows[i][2 * H + nt]);
        int col = nt * 16 + c_col;
        float gmr, gmz, gmn, btr, btz, btn, bhr, bhz, bhn;
        if constexpr (NSUB == 1) {
          gmr = bf2f((uint16_t)gb[i][0][nt]); btr = bf2f((uint16_t)(gb[i][0][nt] >> 16));
          gmz = bf2f((uint16_t)gb[i][1][nt]); btz = bf2f((uint16_t)(gb[i][1][nt] >> 16));
          gmn = bf2f((uint16_t)gb[i][2][nt]); btn = bf2f((uint16_t)(gb[i][2][nt] >> 16));
          bhr = bh[0][nt]; bhz = bh[1][nt]; bhn = bh[2][nt];
        } else {
          gmr = ldf(grow + col); btr = ldf(brow + col);
          gmz = ldf(grow + H + col); btz = ldf(brow + H + col);
          gmn = ldf(grow + 2 * H + col); btn = ldf(brow + 2 * H + col);
          bhr = bh_v[col]; bhz = bh_v[H + col]; bhn = bh_v[2 * H + col];
        }
        float g_r = xr * gmr + btr;
        float g_z = xz * gmz + btz;
        float g_n = xn * gmn + btn;
        float rp = sigmoidf_(acc[nt][i] + bhr + g_r);
        float zp = sigmoidf_(acc[nt + 8][i] + bhz + g_z);
        float hn = acc[nt + 16][i] + bhn;
        float nn = tanhf_(g_n + rp * hn);
        float hnew = (1.f - zp) * nn + zp * h[i][nt];
        h[i][nt] = hnew;
        if constexpr (FP8)
          *reinterpret_cast<uint8_t*>(Hl + swz8(row_of[i], col)) = f2fp8(hnew);
        else
          *reinterpret_cast<uint16_t*>(Hl + swz(row_of[i], col)) = f2bf(hnew);
        fr[nt] = rp; fz[nt] = zp; fn[nt] = nn; (void)fn;
      }
      if (SAVE && live[i]) {
        // only r and z are saved (2H); the backward recomputes n and hh_n
        // (an extra H x H GEMM there is cheaper than 2H of HBM both ways)
        T* sv = saves + (sv_bc[i] + bc_toff) * (2 * H) + c_col * 8;
        st8(sv, fr);
        st8(sv + H, fz);
      }
      // fence: stop the scheduler interleaving all 4 rows' live ranges
      __builtin_amdgcn_sched_barrier(0);
    }
    __syncthreads();

    // ---- cooperative vectorized h_all store (reads the bf16 LDS tile) ----
#pragma unroll
    for (int u = 0; u < 4; ++u) {
      if (hout_live[u]) {
        T* dst = h_all + (hout_bc[u] + bc_toff) * H + hout_blk[u];
        if constexpr (FP8) {
          uint64_t raw = *reinterpret_cast<const uint64_t*>(Hl + hout_lds[u]);
          const uint8_t* q = reinterpret_cast<const uint8_t*>(&raw);
#pragma unroll
          for (int e = 0; e < 8; ++e) {
            __hip_fp8_e4m3 x;
            x.__x = q[e];
            stf(dst + e, float(x));
          }
        } else {
          st_frag(dst, lds_read8(Hl, hout_lds[u]));
        }
      }
    }
    stage_toff += stage_tstep;
    bc_toff += dirC;
    // barrier covers both: h-store reads done AND next step's a-frag reads
    // see the same consistent tile until the next epilogue writes it.
    __syncthreads();
  }
}

// ---------------------------------------------------------------- backward
// The MFMA K axis runs over pi positions m in [0, 384): regions {dr, dz,
// d_hhn} of the dpre tensor (skipping the dn region, which only feeds the
// x-side).  A-fragments read the just-written GLOBAL dpre rows (same-CU L2
// after __syncthreads); the B image in LDS is W^T with rows permuted to the
// same m ordering, so both sides agree on any K permutation.
// NSUB = 1: 4 waves / 64 rows, no LDS, W streamed from L2 (2 blocks/CU can
//           co-reside when the grid exceeds the CU count).
// NSUB = 2: 8 waves / 128 rows, the pi-permuted W image staged ONCE in LDS
//           (96 KB) — 2 waves/SIMD from one block with low-latency
//           B-fragments; used when the grid still fills the chip.
template <typename T, int NSUB>
__global__ __launch_bounds__(NSUB * THREADS) void gru_bwd_kernel(
    const T* __restrict__ grad_h,   // (B, TT, C, H)
    const uint16_t* __restrict__ w_img,  // (H, 384) bf16: w_img[k][m] = W[natJ(m)][k]
    const uint16_t* __restrict__ w_fwd,  // (3H, H) bf16 natural layout
    const T* __restrict__ xg,       // (B, TT, 3H)
    const T* __restrict__ gamma,    // (C, 3H)
    const T* __restrict__ beta,     // (C, 3H)
    const float* __restrict__ b_hh, // (3H,)
    const T* __restrict__ h0,       // (B, C, H)
    const T* __restrict__ h_all,    // (B, TT, C, H)
    const T* __restrict__ saves,    // (B, TT, C, 2H) pi layout: r|z
    T* __restrict__ dpre,           // (B, TT, C, 4H) pi: dr|dz|dn|d_hhn
    float* __restrict__ dh0,        // (B, C, H)
    int B, int TT, int C, int reverse) {
  extern __shared__ __attribute__((aligned(16))) char smem[];
  char* wn_lds = smem + H * G3H * 2;   // (NSUB==2) 128 x 128 bf16 swizzled
  const int tid = threadIdx.x;
  const int wv = tid / DR_WAVE;
  const int lane = tid % DR_WAVE;
  const int64_t R = (int64_t)B * C;
  const int64_t r0 = (int64_t)blockIdx.x * (ROWS * NSUB);

  if constexpr (NSUB == 2) {
    // stage the pi W image (dh GEMM), swz768-swizzled rows [k_h][m]
    for (int id = tid; id < H * (G3H / 8); id += NSUB * THREADS) {
      int k = id / (G3H / 8);
      int mblk = id % (G3H / 8);
      bf16x8 v = *reinterpret_cast<const bf16x8*>(w_img + (int64_t)k * G3H + mblk * 8);
      *reinterpret_cast<bf16x8*>(smem + k * 768 + ((mblk ^ (k & 15)) << 4)) = v;
    }
    // stage W_hn rows (hh_n recompute GEMM), fwd-style [j][k] swizzled
    for (int id = tid; id < H * (H / 8); id += NSUB * THREADS) {
      int jl = id / (H / 8);
      int blk = id % (H / 8);
      bf16x8 v = *reinterpret_cast<const bf16x8*>(
          w_fwd + (int64_t)(2 * H + jl) * H + blk * 8);
      *reinterpret_cast<bf16x8*>(wn_lds + jl * 256 + ((blk ^ (jl & 15)) << 4)) = v;
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
  if constexpr (NSUB == 2) __syncthreads();  // W staging visible to all waves

  // ---- compact addressing: per-row indices + ONE shared per-step offset ----
  // (pointer arrays spilled; full recompute burned VALU on divisions — this
  // derives every address from bc[i] + bc_toff with shifts/adds)
  const int t_first = reverse ? 0 : (TT - 1);
  const int dirC = (reverse ? 1 : -1) * C;
  int64_t bc0[4];           // (b*TT + t_first)*C + c per owned row
  int64_t h0_off[4];        // h0 row offset (t-invariant)
#pragma unroll
  for (int i = 0; i < 4; ++i) {
    bc0[i] = ((int64_t)b_of[i] * TT + t_first) * C + comp_of[i];
    h0_off[i] = ((int64_t)b_of[i] * C + comp_of[i]) * H;
  }
  const int arow = wv * 16 + (lane & 15);
  const int64_t ra = (r0 + arow < R) ? (r0 + arow) : (R - 1);
  const int ab = (int)(ra / C), ac = (int)(ra % C);
  const int64_t abc0 = ((int64_t)ab * TT + t_first) * C + ac;
  const int64_t h0_off_a = ((int64_t)ab * C + ac) * H;
  const int k0 = (lane >> 4) * 8;
  // per-row xg bases + t-invariant b_hh n-slice
  int64_t xgb0[4];
#pragma unroll
  for (int i = 0; i < 4; ++i)
    xgb0[i] = ((int64_t)b_of[i] * TT + t_first) * G3H;
  float bhn[8];
#pragma unroll
  for (int nt = 0; nt < 8; ++nt) bhn[nt] = b_hh[2 * H + nt * 16 + c_col];

  int64_t bc_toff = 0;      // += dirC per step
  int64_t xg_toff = 0;      // += dir*G3H per step
  const int64_t xg_step = (reverse ? 1 : -1) * (int64_t)G3H;
  for (int step = 0; step < TT; ++step) {
    // the FIRST processed step is the forward pass's LAST, whose h_prev is
    // h_all of the step before; h0 is the prev only at the LAST processed
    // step (= forward t boundary)
    const bool use_h0 = (step == TT - 1);

    // ---- recompute hh_n = h_prev @ W_hn^T for this wave's 16 rows ----
    // (saving it cost 2H of HBM each way; one extra HxH MFMA pass is cheaper)
    f32x4 hhn_acc[8];
    {
      const uint16_t* wfv = w_fwd;
      if constexpr (NSUB == 1) asm volatile("" : "+v"(wfv));
      const T* hsrc_a = use_h0 ? (h0 + h0_off_a)
                               : (h_all + (abc0 + bc_toff + dirC) * H);
      bf16x8 hfrag[KT];
#pragma unroll
      for (int kt = 0; kt < KT; ++kt)
        hfrag[kt] = ld_frag(hsrc_a + kt * 32 + k0);
#pragma unroll
      for (int nt = 0; nt < 8; ++nt) {
        f32x4 a = {0.f, 0.f, 0.f, 0.f};
        const int jl = nt * 16 + c_col;
#pragma unroll
        for (int kt = 0; kt < KT; ++kt) {
          bf16x8 bfrag;
          if constexpr (NSUB == 2)
            bfrag = lds_read8(wn_lds, swz(jl, kt * 32 + k0));
          else
            bfrag = *reinterpret_cast<const bf16x8*>(
                wfv + (int64_t)(2 * H + jl) * H + kt * 32 + k0);
          a = __builtin_amdgcn_mfma_f32_16x16x32_bf16(hfrag[kt], bfrag, a, 0, 0, 0);
        }
        hhn_acc[nt] = a;
      }
    }

    const T* gm_v = gamma;
    const T* bt_v = beta;
    asm volatile("" : "+v"(gm_v), "+v"(bt_v));
#pragma unroll
    for (int i = 0; i < 4; ++i) {
      if (!live[i]) continue;
      const int64_t bc = bc0[i] + bc_toff;
      const T* sv_p = saves + bc * (2 * H) + c_col * 8;
      float rp[8], zp[8];
      ld8(sv_p, rp);
      ld8(sv_p + H, zp);
      float fdr[8], fdz[8], fdn[8], fdh[8];
      // h_prev sits at bc + dirC rows (tprev = t+dir); h0 at the boundary.
      // pointer select -> ONE load (the OOB address is never dereferenced)
      const T* hsrc = use_h0 ? (h0 + h0_off[i])
                             : (h_all + (bc + dirC) * H);
      const T* gr_p = grad_h + bc * H;
      const T* xg_n = xg + xgb0[i] + xg_toff + 2 * H;
      const T* gmn = gm_v + comp_of[i] * G3H + 2 * H;
      const T* btn = bt_v + comp_of[i] * G3H + 2 * H;
#pragma unroll
      for (int nt = 0; nt < 8; ++nt) {
        int col = nt * 16 + c_col;
        float hp = ldf(hsrc + col);
        float g = ldf(gr_p + col);
        // recompute the n gate: hh_n from the GEMM above, g_n from xg/FiLM
        float hn = hhn_acc[nt][i] + bhn[nt];
        float g_n = ldf(xg_n + col) * ldf(gmn + col) + ldf(btn + col);
        float nn = tanhf_(g_n + rp[nt] * hn);
        float dht = g + dh_carry[i][nt];
        float dz = dht * (hp - nn);
        float dn = dht * (1.f - zp[nt]);
        float dnp = dn * (1.f - nn * nn);
        float dhhn = dnp * rp[nt];
        float dr = dnp * hn;
        fdr[nt] = dr * rp[nt] * (1.f - rp[nt]);
        fdz[nt] = dz * zp[nt] * (1.f - zp[nt]);
        fdn[nt] = dnp;
        fdh[nt] = dhhn;
        dh_carry[i][nt] = dht * zp[nt];    // partial; MFMA adds dpre @ W
      }
      T* dx_p = dpre + bc * G4H + c_col * 8;
      st8(dx_p, fdr);
      st8(dx_p + H, fdz);
      st8(dx_p + 2 * H, fdn);
      st8(dx_p + 3 * H, fdh);
      // fence the scheduler: without this it interleaves all 4 row
      // iterations and the combined live ranges spill to scratch
      __builtin_amdgcn_sched_barrier(0);
    }
    // Every A-fragment row this wave reads was written by THIS wave's own
    // lanes (rows wv*16..wv*16+15), so no cross-wave barrier is needed —
    // only completion of our own stores (same-CU L1/L2 then serves the
    // cross-lane reads).  Waves free-run, which is worth real latency
    // hiding on this memory-bound kernel.
    asm volatile("s_waitcnt vmcnt(0)" ::: "memory");

    // ---- MFMA: delta = dpre_pi (64 x 384 over m) @ W_pi, wave's 16 rows ----
    {
      // NSUB==1: opaque copy of the W pointer stops LLVM from hoisting all
      // 96 loop-invariant B-fragment loads out of the t loop (384 registers
      // -> scratch); W streams from L1/L2 each step by design.
      const uint16_t* w_imgv = w_img;
      if constexpr (NSUB == 1) asm volatile("" : "+v"(w_imgv));
      const T* arow_p = dpre + (abc0 + bc_toff) * G4H;
      bf16x8 afrag[12];
#pragma unroll
      for (int kt = 0; kt < 12; ++kt) {
        int m = kt * 32 + k0;
        int off = (m < 256) ? m : (m + 128);   // skip the dn region (g == 2)
        afrag[kt] = ld_frag(arow_p + off);
      }
#pragma unroll
      for (int nt = 0; nt < 8; ++nt) {
        f32x4 a = {0.f, 0.f, 0.f, 0.f};
        const int n = nt * 16 + c_col;
        const uint16_t* wrow = w_imgv + (int64_t)n * G3H;
#pragma unroll
        for (int kt = 0; kt < 12; ++kt) {
          bf16x8 bfrag;
          if constexpr (NSUB == 2)
            bfrag = lds_read8(smem, swz768(n, kt * 32 + k0));
          else
            bfrag = *reinterpret_cast<const bf16x8*>(wrow + kt * 32 + k0);
          a = __builtin_amdgcn_mfma_f32_16x16x32_bf16(afrag[kt], bfrag, a, 0, 0, 0);
        }
#pragma unroll
        for (int i = 0; i < 4; ++i) dh_carry[i][nt] += a[i];
      }
    }
    bc_toff += dirC;
    xg_toff += xg_step;
    // no end-of-step barrier: the next step's stores go to DIFFERENT dpre
    // addresses (per-t storage); the W images are read-only.
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
// One pass over dpre per kernel (replaces the einsum broadcasts that
// dominated the profiled step).  Both kernels understand the pi layout and
// emit NATURAL-order outputs.

// natural (r, 3H) -> pi-layout copy: position g*H + cc*8 + e holds natural
// column g*H + cc + 16e.  Run ONCE per tensor so the reduction kernels'
// inner loops replace 8 scalar loads with one 16-byte vector load (the
// scalar-load issue rate, not HBM bandwidth, bounded the old versions).
template <typename T>
__global__ void pi_permute_rows_kernel(const T* __restrict__ in,
                                       T* __restrict__ out, int64_t N) {
  const int64_t n_chunks = N * 48;
  for (int64_t idx = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
       idx < n_chunks; idx += (int64_t)gridDim.x * blockDim.x) {
    const int64_t r = idx / 48;
    const int rem = (int)(idx % 48);
    const int g = rem / 16;
    const int cc = rem % 16;
    const T* src = in + r * G3H + g * H + cc;
    T* dst = out + r * G3H + g * H + cc * 8;
#pragma unroll
    for (int e = 0; e < 8; ++e) dst[e] = src[e * 16];
  }
}

// dxg[bt, j] = sum_c dpre[bt, c, pi(j)] * gamma[c, j]   (j in [0, 3H))
// thread chunk = (bt, gate, c_col): 8 pi-contiguous dpre values per c.
// gamma_pi is the pi-permuted gamma image: one ld8 per c instead of 8
// scalar loads (tiny tensor, L1/L2-resident across all bt chunks).
template <typename T>
__global__ void gru_dxg_kernel(const T* __restrict__ dpre,     // (BT, C, 4H) pi
                               const T* __restrict__ gamma_pi, // (C, 3H) pi
                               T* __restrict__ dxg,            // (BT, 3H)
                               int64_t BT, int C) {
  const int64_t n_chunks = BT * 48;      // 3 gates x 16 c_col groups
  for (int64_t idx = (int64_t)blockIdx.x * blockDim.x + threadIdx.x; idx < n_chunks;
       idx += (int64_t)gridDim.x * blockDim.x) {
    const int64_t bt = idx / 48;
    const int rem = (int)(idx % 48);
    const int g = rem / 16;
    const int cc = rem % 16;
    const int pi0 = g * H + cc * 8;
    float acc[8];
#pragma unroll
    for (int e = 0; e < 8; ++e) acc[e] = 0.f;
    const T* base = dpre + bt * C * G4H + pi0;
    // 4x unroll: the c loop has a runtime bound, so without it the
    // compiler keeps ONE 1-KB wave read in flight and the kernel is
    // latency-bound at ~2.7 TB/s (measured; 4 concurrent reads ~ 4x MLP)
    int c = 0;
    for (; c + 4 <= C; c += 4) {
      float d[4][8], gm[4][8];
#pragma unroll
      for (int u = 0; u < 4; ++u)
        ld8(base + (int64_t)(c + u) * G4H, d[u]);
#pragma unroll
      for (int u = 0; u < 4; ++u)
        ld8(gamma_pi + (int64_t)(c + u) * G3H + pi0, gm[u]);
#pragma unroll
      for (int u = 0; u < 4; ++u)
#pragma unroll
        for (int e = 0; e < 8; ++e) acc[e] += d[u][e] * gm[u][e];
    }
    for (; c < C; ++c) {
      float d[8], gm[8];
      ld8(base + (int64_t)c * G4H, d);
      ld8(gamma_pi + (int64_t)c * G3H + pi0, gm);
#pragma unroll
      for (int e = 0; e < 8; ++e) acc[e] += d[e] * gm[e];
    }
    T* out = dxg + bt * G3H + g * H + cc;
#pragma unroll
    for (int e = 0; e < 8; ++e) stf(out + e * 16, acc[e]);
  }
}

// dgamma[c, j<3H] = sum_bt dpre[bt, c, pi(j)] * xg[bt, j]
// dbeta4[c, j<4H] = sum_bt dpre[bt, c, pi(j)]   (4th slice -> db_hh_n)
// grid.y slices BT; f32 atomics finalize.  Natural-order outputs.
// xg_pi is the pi-permuted xg image: one ld8 per bt instead of 8 scalars.
template <typename T>
__global__ void gru_dgamma_kernel(const T* __restrict__ dpre,   // (BT, C, 4H) pi
                                  const T* __restrict__ xg_pi,  // (BT, 3H) pi
                                  float* __restrict__ dgamma,   // (C, 3H) zeroed
                                  float* __restrict__ dbeta4,   // (C, 4H) zeroed
                                  int64_t BT, int C) {
  const int n_threads_needed = C * 64;   // 4 gates x 16 c_col groups
  const int tid_g = blockIdx.x * blockDim.x + threadIdx.x;
  if (tid_g >= n_threads_needed) return;
  const int c = tid_g / 64;
  const int rem = tid_g % 64;
  const int g = rem / 16;
  const int cc = rem % 16;
  const bool has_x = g < 3;
  const int pi0 = g * H + cc * 8;
  const int64_t bt_lo = BT * blockIdx.y / gridDim.y;
  const int64_t bt_hi = BT * (blockIdx.y + 1) / gridDim.y;
  float accg[8], accb[8];
#pragma unroll
  for (int e = 0; e < 8; ++e) { accg[e] = 0.f; accb[e] = 0.f; }
  // 4x unroll over bt for memory-level parallelism (see gru_dxg_kernel)
  int64_t bt = bt_lo;
  for (; bt + 4 <= bt_hi; bt += 4) {
    float d[4][8];
#pragma unroll
    for (int u = 0; u < 4; ++u)
      ld8(dpre + ((bt + u) * C + c) * G4H + pi0, d[u]);
    if (has_x) {
      float xv[4][8];
#pragma unroll
      for (int u = 0; u < 4; ++u)
        ld8(xg_pi + (bt + u) * G3H + pi0, xv[u]);
#pragma unroll
      for (int u = 0; u < 4; ++u)
#pragma unroll
        for (int e = 0; e < 8; ++e) accg[e] += d[u][e] * xv[u][e];
    }
#pragma unroll
    for (int u = 0; u < 4; ++u)
#pragma unroll
      for (int e = 0; e < 8; ++e) accb[e] += d[u][e];
  }
  for (; bt < bt_hi; ++bt) {
    float d[8];
    ld8(dpre + (bt * C + c) * G4H + pi0, d);
    if (has_x) {
      float xv[8];
      ld8(xg_pi + bt * G3H + pi0, xv);
#pragma unroll
      for (int e = 0; e < 8; ++e) accg[e] += d[e] * xv[e];
    }
#pragma unroll
    for (int e = 0; e < 8; ++e) accb[e] += d[e];
  }
  float* b_out = dbeta4 + (int64_t)c * G4H + g * H + cc;
#pragma unroll
  for (int e = 0; e < 8; ++e) atomicAdd(b_out + e * 16, accb[e]);
  if (has_x) {
    float* g_out = dgamma + (int64_t)c * G3H + g * H + cc;
#pragma unroll
    for (int e = 0; e < 8; ++e) atomicAdd(g_out + e * 16, accg[e]);
  }
}

// ------------------------------------------------------------- launchers
// fp8 tiles halve the GEMM-side LDS: 48K W + 8K h + 24K xg
constexpr int LDS_FWD_FP8 = G3H * H + ROWS * H + XG_SLOTS * G3H * 2;  // 81920 B
// 8-wave variant: 96K W + 32K h(128 rows) + 24K xg
constexpr int LDS_FWD_8W = G3H * H * 2 + 2 * ROWS * H * 2 + XG_SLOTS * G3H * 2;

template <typename T>
static void gru_fwd_launch_t(const void* xg, const void* gamma, const void* beta,
                             const void* w_gemm, const float* b_hh, const void* h0,
                             void* h_all, void* saves, int B, int TT, int C,
                             int reverse, int save, int fp8, hipStream_t stream) {
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
    DR_HIP_CHECK(hipFuncSetAttribute(
        reinterpret_cast<const void*>(&gru_fwd_kernel<T, false, true>),
        hipFuncAttributeMaxDynamicSharedMemorySize, LDS_FWD_FP8));
    DR_HIP_CHECK(hipFuncSetAttribute(
        reinterpret_cast<const void*>(&gru_fwd_kernel<T, true, false, 2>),
        hipFuncAttributeMaxDynamicSharedMemorySize, LDS_FWD_8W));
    DR_HIP_CHECK(hipFuncSetAttribute(
        reinterpret_cast<const void*>(&gru_fwd_kernel<T, false, false, 2>),
        hipFuncAttributeMaxDynamicSharedMemorySize, LDS_FWD_8W));
    attr_set = true;
  }
  int tiles128 = (int)((R + 2 * ROWS - 1) / (2 * ROWS));
  if (!fp8 && tiles128 >= 192 && C >= 5) {
    // 8-wave 128-row variant: 2 waves/SIMD (needs enough tiles + xg slots)
    if (save)
      hipLaunchKernelGGL((gru_fwd_kernel<T, true, false, 2>), dim3(tiles128),
                         dim3(2 * THREADS), LDS_FWD_8W, stream, (const T*)xg,
                         (const T*)gamma, (const T*)beta, (const uint16_t*)w_gemm,
                         b_hh, (const T*)h0, (T*)h_all, (T*)saves, B, TT, C,
                         reverse);
    else
      hipLaunchKernelGGL((gru_fwd_kernel<T, false, false, 2>), dim3(tiles128),
                         dim3(2 * THREADS), LDS_FWD_8W, stream, (const T*)xg,
                         (const T*)gamma, (const T*)beta, (const uint16_t*)w_gemm,
                         b_hh, (const T*)h0, (T*)h_all, (T*)saves, B, TT, C,
                         reverse);
    return;
  }
  if (fp8)
    hipLaunchKernelGGL((gru_fwd_kernel<T, false, true>), dim3(grid), dim3(THREADS),
                       LDS_FWD_FP8, stream, (const T*)xg, (const T*)gamma,
                       (const T*)beta, (const uint16_t*)w_gemm, b_hh, (const T*)h0,
                       (T*)h_all, (T*)saves, B, TT, C, reverse);
  else if (save)
    hipLaunchKernelGGL((gru_fwd_kernel<T, true>), dim3(grid), dim3(THREADS),
                       LDS_FWD_TOTAL, stream, (const T*)xg, (const T*)gamma,
                       (const T*)beta, (const uint16_t*)w_gemm, b_hh, (const T*)h0,
                       (T*)h_all, (T*)saves, B, TT, C, reverse);
  else
    hipLaunchKernelGGL((gru_fwd_kernel<T, false>), dim3(grid), dim3(THREADS),
                       LDS_FWD_TOTAL, stream, (const T*)xg, (const T*)gamma,
                       (const T*)beta, (const uint16_t*)w_gemm, b_hh, (const T*)h0,
                       (T*)h_all, (T*)saves, B, TT, C, reverse);
}

template <typename T>
static void gru_bwd_launch_t(const void* grad_h, const void* w_img,
                             const void* w_fwd, const void* xg, const void* gamma,
                             const void* beta, const float* b_hh, const void* h0,
                             const void* h_all, const void* saves, void* dpre,
                             float* dh0, int B, int TT, int C, int reverse,
                             hipStream_t stream) {
  int64_t R = (int64_t)B * C;
  int tiles128 = (int)((R + 2 * ROWS - 1) / (2 * ROWS));
  if (tiles128 >= 192) {
    // enough 128-row tiles to fill the chip at 1 block/CU: LDS-W variant
    constexpr int LDS_WB = H * G3H * 2 + H * H * 2;  // 96 + 32 KB
    static bool attr_set = false;
    if (!attr_set) {
      DR_HIP_CHECK(hipFuncSetAttribute(
          reinterpret_cast<const void*>(&gru_bwd_kernel<T, 2>),
          hipFuncAttributeMaxDynamicSharedMemorySize, LDS_WB));
      attr_set = true;
    }
    hipLaunchKernelGGL((gru_bwd_kernel<T, 2>), dim3(tiles128), dim3(2 * THREADS),
                       LDS_WB, stream, (const T*)grad_h, (const uint16_t*)w_img,
                       (const uint16_t*)w_fwd, (const T*)xg, (const T*)gamma,
                       (const T*)beta, b_hh, (const T*)h0, (const T*)h_all,
                       (const T*)saves, (T*)dpre, dh0, B, TT, C, reverse);
    return;
  }
  int grid = (int)((R + ROWS - 1) / ROWS);
  hipLaunchKernelGGL((gru_bwd_kernel<T, 1>), dim3(grid), dim3(THREADS), 0, stream,
                     (const T*)grad_h, (const uint16_t*)w_img,
                     (const uint16_t*)w_fwd, (const T*)xg, (const T*)gamma,
                     (const T*)beta, b_hh, (const T*)h0, (const T*)h_all,
                     (const T*)saves, (T*)dpre, dh0, B, TT, C, reverse);
}

template <typename T>
static void gru_reduce_launch_t(const void* dpre, const void* gamma, const void* xg,
                                void* dxg, float* dgamma, float* dbeta,
                                void* gamma_pi, void* xg_pi, int64_t BT,
                                int C, hipStream_t stream) {
  // stage pi-layout images of gamma and xg (once per backward, ~47 MB at
  // bench scale) so the reduction inner loops are pure 16-byte loads
  {
    int64_t n = (int64_t)C * 48;
    int grid = (int)std::min<int64_t>((n + 255) / 256, 4096);
    hipLaunchKernelGGL((pi_permute_rows_kernel<T>), dim3(grid), dim3(256), 0,
                       stream, (const T*)gamma, (T*)gamma_pi, (int64_t)C);
  }
  {
    int64_t n = BT * 48;
    int grid = (int)std::min<int64_t>((n + 255) / 256, 4096);
    hipLaunchKernelGGL((pi_permute_rows_kernel<T>), dim3(grid), dim3(256), 0,
                       stream, (const T*)xg, (T*)xg_pi, BT);
  }
  {
    int64_t n = BT * 48;
    int grid = (int)std::min<int64_t>((n + 255) / 256, 4096);
    hipLaunchKernelGGL((gru_dxg_kernel<T>), dim3(grid), dim3(256), 0, stream,
                       (const T*)dpre, (const T*)gamma_pi, (T*)dxg, BT, C);
  }
  {
    int n_threads = C * 64;
    int gx = (n_threads + 255) / 256;
    int gy = 128;  // BT slices (parallelism for small C; atomics stay cheap)
    hipLaunchKernelGGL((gru_dgamma_kernel<T>), dim3(gx, gy), dim3(256), 0, stream,
                       (const T*)dpre, (const T*)xg_pi, dgamma, dbeta, BT, C);
  }
}

}  // namespace dr

extern "C" {

void dr_gru_fwd(const void* xg, const void* gamma, const void* beta,
                const void* w_hh, const float* b_hh, const void* h0, void* h_all,
                void* saves, int B, int TT, int C, int reverse, int save,
                int fp8, int is_bf16, hipStream_t stream) {
  if (is_bf16)
    dr::gru_fwd_launch_t<uint16_t>(xg, gamma, beta, w_hh, b_hh, h0, h_all, saves,
                                   B, TT, C, reverse, save, fp8, stream);
  else
    dr::gru_fwd_launch_t<float>(xg, gamma, beta, w_hh, b_hh, h0, h_all, saves,
                                B, TT, C, reverse, save, fp8, stream);
}

void dr_gru_bwd(const void* grad_h, const void* w_img, const void* w_fwd,
                const void* xg, const void* gamma, const void* beta,
                const float* b_hh, const void* h0, const void* h_all,
                const void* saves, void* dpre, float* dh0, int B, int TT, int C,
                int reverse, int is_bf16, hipStream_t stream) {
  if (is_bf16)
    dr::gru_bwd_launch_t<uint16_t>(grad_h, w_img, w_fwd, xg, gamma, beta, b_hh,
                                   h0, h_all, saves, dpre, dh0, B, TT, C,
                                   reverse, stream);
  else
    dr::gru_bwd_launch_t<float>(grad_h, w_img, w_fwd, xg, gamma, beta, b_hh,
                                h0, h_all, saves, dpre, dh0, B, TT, C,
                                reverse, stream);
}

void dr_gru_bwd_reduce(const void* dpre, const void* gamma, const void* xg,
                       void* dxg, float* dgamma, float* dbeta, void* gamma_pi,
                       void* xg_pi, int64_t BT, int C, int is_bf16,
                       hipStream_t stream) {
  if (is_bf16)
    dr::gru_reduce_launch_t<uint16_t>(dpre, gamma, xg, dxg, dgamma, dbeta,
                                      gamma_pi, xg_pi, BT, C, stream);
  else
    dr::gru_reduce_launch_t<float>(dpre, gamma, xg, dxg, dgamma, dbeta,
                                   gamma_pi, xg_pi, BT, C, stream);
}

}  // extern "C"
