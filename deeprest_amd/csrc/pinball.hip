// Fused quantile (pinball) loss, forward + backward.
//
// Semantics match reference resource-estimation/qrnn.py:58-67 for equal-sized
// metrics: loss = (1/(B*T*M)) * sum_{b,t,m,q} max((q-1)e, q*e), e = y - o_q.
// Forward is one grid-stride pass with wave+block reduction and a single
// atomicAdd per block; backward is pure elementwise.
//
// Templated on the output dtype: under bf16 autocast the model's (B,T,M,Q)
// predictions feed the loss DIRECTLY (math still fp32 in-register), instead
// of paying a ~280 MB f32 cast of the output tensor each step (the gradient
// is sign(e)-based quantile constants, so bf16 inputs lose nothing).
#include "common.h"

namespace dr {

template <typename T, int QMAX>
__global__ void pinball_fwd_kernel(const T* __restrict__ out,      // (N, Q)
                                   const float* __restrict__ labels,  // (N,)
                                   const float* __restrict__ quantiles, int Q,
                                   int64_t N, float inv_count,
                                   float* __restrict__ loss) {
  float q[QMAX];
  for (int i = 0; i < Q; ++i) q[i] = quantiles[i];
  float acc = 0.f;
  for (int64_t n = (int64_t)blockIdx.x * blockDim.x + threadIdx.x; n < N;
       n += (int64_t)gridDim.x * blockDim.x) {
    const float y = labels[n];
    const T* o = out + n * Q;
    for (int i = 0; i < Q; ++i) {
      float e = y - ldf(o + i);
      acc += fmaxf((q[i] - 1.f) * e, q[i] * e);
    }
  }
  acc = wave_sum(acc);
  __shared__ float partials[16];
  const int wave = threadIdx.x / DR_WAVE;
  const int lane = threadIdx.x % DR_WAVE;
  if (lane == 0) partials[wave] = acc;
  __syncthreads();
  if (threadIdx.x == 0) {
    float s = 0.f;
    for (int wv = 0; wv < blockDim.x / DR_WAVE; ++wv) s += partials[wv];
    atomicAdd(loss, s * inv_count);
  }
}

template <typename T, int QMAX>
__global__ void pinball_bwd_kernel(const T* __restrict__ out,
                                   const float* __restrict__ labels,
                                   const float* __restrict__ quantiles, int Q,
                                   int64_t N, const float* __restrict__ grad_loss,
                                   float inv_n, T* __restrict__ dout) {
  // upstream grad read on-device (no host .item() sync -> hipGraph-capturable)
  const float gscale = grad_loss[0] * inv_n;
  float q[QMAX];
  for (int i = 0; i < Q; ++i) q[i] = quantiles[i];
  for (int64_t n = (int64_t)blockIdx.x * blockDim.x + threadIdx.x; n < N;
       n += (int64_t)gridDim.x * blockDim.x) {
    const float y = labels[n];
    const T* o = out + n * Q;
    T* d = dout + n * Q;
    for (int i = 0; i < Q; ++i) {
      float e = y - ldf(o + i);
      // d/do max((q-1)e, qe): e>0 -> -q ; e<0 -> (1-q) ; e==0 -> 0
      float g = (e > 0.f) ? -q[i] : ((e < 0.f) ? (1.f - q[i]) : 0.f);
      stf(d + i, gscale * g);
    }
  }
}

}  // namespace dr

extern "C" {

void dr_pinball_fwd(const void* out, const float* labels, const float* quantiles,
                    int Q, int64_t N, float inv_count, float* loss, int is_bf16,
                    hipStream_t stream) {
  const int block = 256;
  int grid = (int)std::min<int64_t>((N + block - 1) / block, 2048);
  if (grid == 0) grid = 1;
  if (is_bf16)
    hipLaunchKernelGGL((dr::pinball_fwd_kernel<uint16_t, 8>), dim3(grid),
                       dim3(block), 0, stream, (const uint16_t*)out, labels,
                       quantiles, Q, N, inv_count, loss);
  else
    hipLaunchKernelGGL((dr::pinball_fwd_kernel<float, 8>), dim3(grid),
                       dim3(block), 0, stream, (const float*)out, labels,
                       quantiles, Q, N, inv_count, loss);
}

void dr_pinball_bwd(const void* out, const float* labels, const float* quantiles,
                    int Q, int64_t N, const float* grad_loss, float inv_n,
                    void* dout, int is_bf16, hipStream_t stream) {
  const int block = 256;
  int grid = (int)std::min<int64_t>((N + block - 1) / block, 2048);
  if (grid == 0) grid = 1;
  if (is_bf16)
    hipLaunchKernelGGL((dr::pinball_bwd_kernel<uint16_t, 8>), dim3(grid),
                       dim3(block), 0, stream, (const uint16_t*)out, labels,
                       quantiles, Q, N, grad_loss, inv_n, (uint16_t*)dout);
  else
    hipLaunchKernelGGL((dr::pinball_bwd_kernel<float, 8>), dim3(grid),
                       dim3(block), 0, stream, (const float*)out, labels,
                       quantiles, Q, N, grad_loss, inv_n, (float*)dout);
}

}  // extern "C"
