// Fused multi-tensor Adam: ONE kernel launch updates every parameter tensor.
//
// The host packs [p_ptr[i], g_ptr[i], m_ptr[i], v_ptr[i]] x nt plus the
// cumulative-numel table into a single int64 device buffer; the kernel
// grid-strides over the total element count and binary-searches the table
// (log2(nt) steps) to locate each element's tensor. All math fp32.
#include "common.h"

namespace dr {

// STEP_DEV = true: the step counter lives in device memory (incremented by a
// captured device op between replays), so bias correction stays exact inside
// a hipGraph-captured training step; host-arg bias factors would be frozen
// at their capture-time values.
template <bool STEP_DEV>
__global__ void fused_adam_kernel(const int64_t* __restrict__ meta, int nt,
                                  int64_t total, float lr, float beta1, float beta2,
                                  float eps, float weight_decay, float bias_c1,
                                  float bias_c2, const int* __restrict__ step_ptr,
                                  const float* __restrict__ lr_ptr) {
  if (STEP_DEV) {
    // step counter AND learning rate live in device memory so a captured
    // replay keeps exact bias correction and follows LR schedules (the
    // host updates the scalar between replays; a by-value lr would be
    // frozen at its capture-time value)
    const float s = (float)*step_ptr;
    bias_c1 = 1.f - __powf(beta1, s);
    bias_c2 = 1.f - __powf(beta2, s);
    lr = *lr_ptr;
  }
  const int64_t* p_ptrs = meta;
  const int64_t* g_ptrs = meta + nt;
  const int64_t* m_ptrs = meta + 2 * nt;
  const int64_t* v_ptrs = meta + 3 * nt;
  const int64_t* cum = meta + 4 * nt;  // cum[i] = end offset of tensor i (nt entries)

  for (int64_t idx = (int64_t)blockIdx.x * blockDim.x + threadIdx.x; idx < total;
       idx += (int64_t)gridDim.x * blockDim.x) {
    // binary search: first i with cum[i] > idx
    int lo = 0, hi = nt - 1;
    while (lo < hi) {
      int mid = (lo + hi) >> 1;
      if (cum[mid] > idx) hi = mid; else lo = mid + 1;
    }
    const int i = lo;
    const int64_t base = (i == 0) ? 0 : cum[i - 1];
    const int64_t off = idx - base;

    float* p = reinterpret_cast<float*>(p_ptrs[i]) + off;
    const float* g = reinterpret_cast<const float*>(g_ptrs[i]) + off;
    float* m = reinterpret_cast<float*>(m_ptrs[i]) + off;
    float* v = reinterpret_cast<float*>(v_ptrs[i]) + off;

    float grad = *g;
    float pv = *p;
    if (weight_decay != 0.f) grad += weight_decay * pv;
    float mv = beta1 * (*m) + (1.f - beta1) * grad;
    float vv = beta2 * (*v) + (1.f - beta2) * grad * grad;
    *m = mv;
    *v = vv;
    float denom = sqrtf(vv / bias_c2) + eps;
    *p = pv - (lr / bias_c1) * mv / denom;
  }
}

}  // namespace dr

extern "C" {

void dr_fused_adam(const int64_t* meta, int nt, int64_t total, float lr, float beta1,
                   float beta2, float eps, float weight_decay, int step,
                   hipStream_t stream) {
  const int block = 256;
  int grid = (int)std::min<int64_t>((total + block - 1) / block, 4096);
  if (grid == 0) grid = 1;
  float bias_c1 = 1.f - powf(beta1, (float)step);
  float bias_c2 = 1.f - powf(beta2, (float)step);
  hipLaunchKernelGGL((dr::fused_adam_kernel<false>), dim3(grid), dim3(block), 0,
                     stream, meta, nt, total, lr, beta1, beta2, eps, weight_decay,
                     bias_c1, bias_c2, nullptr, nullptr);
}

void dr_fused_adam_dev(const int64_t* meta, int nt, int64_t total,
                       const float* lr_ptr, float beta1, float beta2, float eps,
                       float weight_decay, const int* step_ptr,
                       hipStream_t stream) {
  const int block = 256;
  int grid = (int)std::min<int64_t>((total + block - 1) / block, 4096);
  if (grid == 0) grid = 1;
  hipLaunchKernelGGL((dr::fused_adam_kernel<true>), dim3(grid), dim3(block), 0,
                     stream, meta, nt, total, 0.f, beta1, beta2, eps, weight_decay,
                     0.f, 0.f, step_ptr, lr_ptr);
}

}  // extern "C"
