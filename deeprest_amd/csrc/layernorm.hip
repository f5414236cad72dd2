// LayerNorm forward/backward for gfx950.
//
// Memory-bound op (guide Appendix B): one wave per row, vectorized loads,
// in-register two-statistic reduction via 64-wide shuffles, fused affine.
// Backward avoids global atomics for dweight/dbias by accumulating per-block
// partials in LDS and letting the host finalize with one column-sum GEMM-free
// reduction (torch .sum(0)).
#include "common.h"

namespace dr {

// ---- packed 4-column IO: one 8 B (bf16) / 16 B (f32) op per row per lane ----
__device__ __forceinline__ void ld4(const uint16_t* p, float v[4]) {
  uint2 u = *reinterpret_cast<const uint2*>(p);
  v[0] = bf2f((uint16_t)(u.x & 0xffff));
  v[1] = bf2f((uint16_t)(u.x >> 16));
  v[2] = bf2f((uint16_t)(u.y & 0xffff));
  v[3] = bf2f((uint16_t)(u.y >> 16));
}
__device__ __forceinline__ void ld4(const float* p, float v[4]) {
  float4 u = *reinterpret_cast<const float4*>(p);
  v[0] = u.x; v[1] = u.y; v[2] = u.z; v[3] = u.w;
}
__device__ __forceinline__ void st4(uint16_t* p, const float v[4]) {
  uint2 u;
  u.x = f2bf2(v[0], v[1]);
  u.y = f2bf2(v[2], v[3]);
  *reinterpret_cast<uint2*>(p) = u;
}
__device__ __forceinline__ void st4(float* p, const float v[4]) {
  *reinterpret_cast<float4*>(p) = make_float4(v[0], v[1], v[2], v[3]);
}

// Fast path for D % 4 == 0 && D <= 256 (lane owns 4 contiguous columns, the
// whole row is one wave-wide vector op).  The column set per lane is
// row-invariant, so w/b live in registers across the row loop.
template <typename T>
__global__ void ln_fwd_vec4(const T* __restrict__ x, const float* __restrict__ w,
                            const float* __restrict__ b, T* __restrict__ y,
                            float* __restrict__ mean_out, float* __restrict__ rstd_out,
                            int64_t n_rows, int D, float eps) {
  const int wave = threadIdx.x / DR_WAVE;
  const int lane = threadIdx.x % DR_WAVE;
  const int waves_per_block = blockDim.x / DR_WAVE;
  const int64_t row0 = (int64_t)blockIdx.x * waves_per_block + wave;
  const int64_t stride_rows = (int64_t)gridDim.x * waves_per_block;
  const int j0 = lane * 4;
  const bool active = j0 < D;
  const float invD = 1.f / D;
  float wr[4] = {0.f, 0.f, 0.f, 0.f}, br[4] = {0.f, 0.f, 0.f, 0.f};
  if (active) { ld4(w + j0, wr); ld4(b + j0, br); }

  for (int64_t row = row0; row < n_rows; row += stride_rows) {
    float v[4] = {0.f, 0.f, 0.f, 0.f};
    if (active) ld4(x + row * D + j0, v);
    float mu = wave_sum(v[0] + v[1] + v[2] + v[3]) * invD;
    float var = 0.f;
    if (active) {
#pragma unroll
      for (int i = 0; i < 4; ++i) { float d = v[i] - mu; var += d * d; }
    }
    const float rstd = rsqrtf(wave_sum(var) * invD + eps);
    if (lane == 0) {
      mean_out[row] = mu;
      rstd_out[row] = rstd;
    }
    if (active) {
      float o[4];
#pragma unroll
      for (int i = 0; i < 4; ++i) o[i] = (v[i] - mu) * rstd * wr[i] + br[i];
      st4(y + row * D + j0, o);
    }
  }
}

// vec4 backward: dw/db accumulate in per-wave REGISTERS across the row loop
// (same lane -> same columns every row); the LDS combine + partial-slab write
// happen once per block, so the row loop is pure streaming with no atomics.
template <typename T>
__global__ void ln_bwd_vec4(const T* __restrict__ dy, const T* __restrict__ x,
                            const float* __restrict__ w,
                            const float* __restrict__ mean,
                            const float* __restrict__ rstd, T* __restrict__ dx,
                            float* __restrict__ dwdb_part,  // (gridDim.x, 2, D)
                            int64_t n_rows, int D) {
  extern __shared__ float smem[];  // 2 * D floats: [dw | db]
  float* dw_s = smem;
  float* db_s = smem + D;
  for (int j = threadIdx.x; j < 2 * D; j += blockDim.x) smem[j] = 0.f;
  __syncthreads();

  const int wave = threadIdx.x / DR_WAVE;
  const int lane = threadIdx.x % DR_WAVE;
  const int waves_per_block = blockDim.x / DR_WAVE;
  const int64_t row0 = (int64_t)blockIdx.x * waves_per_block + wave;
  const int64_t stride_rows = (int64_t)gridDim.x * waves_per_block;
  const int j0 = lane * 4;
  const bool active = j0 < D;
  const float invD = 1.f / D;
  float wr[4] = {0.f, 0.f, 0.f, 0.f};
  if (active) ld4(w + j0, wr);
  float dwacc[4] = {0.f, 0.f, 0.f, 0.f}, dbacc[4] = {0.f, 0.f, 0.f, 0.f};

  for (int64_t row = row0; row < n_rows; row += stride_rows) {
    float g[4] = {0.f, 0.f, 0.f, 0.f}, xv[4] = {0.f, 0.f, 0.f, 0.f};
    if (active) {
      ld4(dy + row * D + j0, g);
      ld4(x + row * D + j0, xv);
    }
    const float mu = mean[row];
    const float rs = rstd[row];
    float xh[4], gw[4];
    float a = 0.f, bsum = 0.f;
#pragma unroll
    for (int i = 0; i < 4; ++i) {
      xh[i] = (xv[i] - mu) * rs;
      gw[i] = g[i] * wr[i];
      a += gw[i] * xh[i];
      bsum += gw[i];
    }
    a = wave_sum(a) * invD;
    bsum = wave_sum(bsum) * invD;
    if (active) {
      float o[4];
#pragma unroll
      for (int i = 0; i < 4; ++i) {
        o[i] = rs * (gw[i] - bsum - xh[i] * a);
        dwacc[i] += g[i] * xh[i];
        dbacc[i] += g[i];
      }
      st4(dx + row * D + j0, o);
    }
  }
  if (active) {
#pragma unroll
    for (int i = 0; i < 4; ++i) {
      atomicAdd(&dw_s[j0 + i], dwacc[i]);
      atomicAdd(&db_s[j0 + i], dbacc[i]);
    }
  }
  __syncthreads();
  float* out = dwdb_part + (int64_t)blockIdx.x * 2 * D;
  for (int j = threadIdx.x; j < 2 * D; j += blockDim.x) out[j] = smem[j];
}

// rows handled per block = (blockDim.x / 64); each wave owns one row.
template <typename T, int MAX_PER_LANE>
__global__ void ln_fwd_kernel(const T* __restrict__ x, const float* __restrict__ w,
                              const float* __restrict__ b, T* __restrict__ y,
                              float* __restrict__ mean_out, float* __restrict__ rstd_out,
                              int64_t n_rows, int D, float eps) {
  const int wave = threadIdx.x / DR_WAVE;
  const int lane = threadIdx.x % DR_WAVE;
  const int waves_per_block = blockDim.x / DR_WAVE;
  const int64_t row0 = (int64_t)blockIdx.x * waves_per_block + wave;
  const int64_t stride_rows = (int64_t)gridDim.x * waves_per_block;

  for (int64_t row = row0; row < n_rows; row += stride_rows) {
    const T* xr = x + row * D;
    float v[MAX_PER_LANE];
    float s = 0.f;
    int nper = 0;
    for (int j = lane; j < D; j += DR_WAVE) {
      v[nper] = ldf(xr + j);
      s += v[nper];
      ++nper;
    }
    s = wave_sum(s);
    const float mean = s / D;
    float var = 0.f;
    for (int i = 0; i < nper; ++i) {
      float d = v[i] - mean;
      var += d * d;
    }
    var = wave_sum(var) / D;
    const float rstd = rsqrtf(var + eps);
    if (lane == 0) {
      mean_out[row] = mean;
      rstd_out[row] = rstd;
    }
    T* yr = y + row * D;
    int i = 0;
    for (int j = lane; j < D; j += DR_WAVE, ++i) {
      stf(yr + j, (v[i] - mean) * rstd * w[j] + b[j]);
    }
  }
}

// dweight/dbias partials: one (2, D) f32 slab per block, summed on host.
template <typename T, int MAX_PER_LANE>
__global__ void ln_bwd_kernel(const T* __restrict__ dy, const T* __restrict__ x,
                              const float* __restrict__ w,
                              const float* __restrict__ mean,
                              const float* __restrict__ rstd, T* __restrict__ dx,
                              float* __restrict__ dwdb_part,  // (gridDim.x, 2, D)
                              int64_t n_rows, int D) {
  extern __shared__ float smem[];  // 2 * D floats: [dw | db] partials
  float* dw_s = smem;
  float* db_s = smem + D;
  for (int j = threadIdx.x; j < 2 * D; j += blockDim.x) smem[j] = 0.f;
  __syncthreads();

  const int wave = threadIdx.x / DR_WAVE;
  const int lane = threadIdx.x % DR_WAVE;
  const int waves_per_block = blockDim.x / DR_WAVE;
  const int64_t row0 = (int64_t)blockIdx.x * waves_per_block + wave;
  const int64_t stride_rows = (int64_t)gridDim.x * waves_per_block;

  for (int64_t row = row0; row < n_rows; row += stride_rows) {
    const T* dyr = dy + row * D;
    const T* xr = x + row * D;
    const float mu = mean[row];
    const float rs = rstd[row];
    float gy[MAX_PER_LANE], xh[MAX_PER_LANE];
    float a = 0.f, bsum = 0.f;
    int nper = 0;
    for (int j = lane; j < D; j += DR_WAVE) {
      float g = ldf(dyr + j);
      float h = (ldf(xr + j) - mu) * rs;
      float gw = g * w[j];
      gy[nper] = gw;
      xh[nper] = h;
      a += gw * h;
      bsum += gw;
      ++nper;
    }
    a = wave_sum(a) / D;
    bsum = wave_sum(bsum) / D;
    T* dxr = dx + row * D;
    int i = 0;
    for (int j = lane; j < D; j += DR_WAVE, ++i) {
      stf(dxr + j, rs * (gy[i] - bsum - xh[i] * a));
      float g = ldf(dyr + j);
      atomicAdd(&dw_s[j], g * xh[i]);
      atomicAdd(&db_s[j], g);
    }
  }
  __syncthreads();
  float* out = dwdb_part + (int64_t)blockIdx.x * 2 * D;
  for (int j = threadIdx.x; j < 2 * D; j += blockDim.x) out[j] = smem[j];
}

template <typename T>
static void ln_fwd_launch_t(const void* x, const float* w, const float* b, void* y,
                            float* mean, float* rstd, int64_t n_rows, int D, float eps,
                            hipStream_t stream) {
  const int block = 256;
  const int waves = block / DR_WAVE;
  int grid = (int)std::min<int64_t>((n_rows + waves - 1) / waves, 2048);
  if (grid == 0) grid = 1;
  if ((D % 4) == 0 && D <= 4 * DR_WAVE) {
    hipLaunchKernelGGL((ln_fwd_vec4<T>), dim3(grid), dim3(block), 0, stream,
                       (const T*)x, w, b, (T*)y, mean, rstd, n_rows, D, eps);
  } else if (D <= 256) {
    hipLaunchKernelGGL((ln_fwd_kernel<T, 4>), dim3(grid), dim3(block), 0, stream,
                       (const T*)x, w, b, (T*)y, mean, rstd, n_rows, D, eps);
  } else if (D <= 1024) {
    hipLaunchKernelGGL((ln_fwd_kernel<T, 16>), dim3(grid), dim3(block), 0, stream,
                       (const T*)x, w, b, (T*)y, mean, rstd, n_rows, D, eps);
  } else {
    hipLaunchKernelGGL((ln_fwd_kernel<T, 64>), dim3(grid), dim3(block), 0, stream,
                       (const T*)x, w, b, (T*)y, mean, rstd, n_rows, D, eps);
  }
}

template <typename T>
static void ln_bwd_launch_t(const void* dy, const void* x, const float* w,
                            const float* mean, const float* rstd, void* dx,
                            float* dwdb_part, int n_blocks, int64_t n_rows, int D,
                            hipStream_t stream) {
  const int block = 256;
  size_t smem = 2 * D * sizeof(float);
  if ((D % 4) == 0 && D <= 4 * DR_WAVE) {
    hipLaunchKernelGGL((ln_bwd_vec4<T>), dim3(n_blocks), dim3(block), smem, stream,
                       (const T*)dy, (const T*)x, w, mean, rstd, (T*)dx, dwdb_part,
                       n_rows, D);
  } else if (D <= 256) {
    hipLaunchKernelGGL((ln_bwd_kernel<T, 4>), dim3(n_blocks), dim3(block), smem, stream,
                       (const T*)dy, (const T*)x, w, mean, rstd, (T*)dx, dwdb_part,
                       n_rows, D);
  } else if (D <= 1024) {
    hipLaunchKernelGGL((ln_bwd_kernel<T, 16>), dim3(n_blocks), dim3(block), smem, stream,
                       (const T*)dy, (const T*)x, w, mean, rstd, (T*)dx, dwdb_part,
                       n_rows, D);
  } else {
    hipLaunchKernelGGL((ln_bwd_kernel<T, 64>), dim3(n_blocks), dim3(block), smem, stream,
                       (const T*)dy, (const T*)x, w, mean, rstd, (T*)dx, dwdb_part,
                       n_rows, D);
  }
}

}  // namespace dr

// ---- C ABI launchers (called from bindings.cpp) ----
extern "C" {

void dr_layernorm_fwd(const void* x, const float* w, const float* b, void* y,
                      float* mean, float* rstd, int64_t n_rows, int D, float eps,
                      int is_bf16, hipStream_t stream) {
  if (is_bf16)
    dr::ln_fwd_launch_t<uint16_t>(x, w, b, y, mean, rstd, n_rows, D, eps, stream);
  else
    dr::ln_fwd_launch_t<float>(x, w, b, y, mean, rstd, n_rows, D, eps, stream);
}

void dr_layernorm_bwd(const void* dy, const void* x, const float* w, const float* mean,
                      const float* rstd, void* dx, float* dwdb_part, int n_blocks,
                      int64_t n_rows, int D, int is_bf16, hipStream_t stream) {
  if (is_bf16)
    dr::ln_bwd_launch_t<uint16_t>(dy, x, w, mean, rstd, dx, dwdb_part, n_blocks,
                                  n_rows, D, stream);
  else
    dr::ln_bwd_launch_t<float>(dy, x, w, mean, rstd, dx, dwdb_part, n_blocks,
                               n_rows, D, stream);
}

}  // extern "C"
