// Empirical probe of the v_mfma_f32_16x16x32_fp8_fp8 A-operand byte->k map.
//
// Feeds A bytes = 1.0 for byte index e < 4, = 2.0 for e >= 4 (all lanes),
// and B = identity over k < 16 (one-hot columns).  D[i][j] = A[i][k = j].
//   contiguous map (k = 8*(l>>4) + e):      D row = 1 1 1 1 2 2 2 2 ...
//   split map (k = 4*(l>>4) + e%4 + 16*(e/4)): D row = all 1s
// Also prints the same for B to confirm symmetry.
//
//   hipcc --offload-arch=gfx950 tools/fp8_probe.hip -o /tmp/fp8probe && /tmp/fp8probe
#include <hip/hip_runtime.h>
#include <hip/hip_fp8.h>
#include <stdio.h>

typedef __attribute__((__vector_size__(4 * sizeof(float)))) float f32x4;

__device__ unsigned char q8(float v) {
  __hip_fp8_e4m3 x(v);
  return x.__x;
}

__global__ void probeA(float* out /* 16x16 */) {
  int lane = threadIdx.x;
  unsigned char a[8], b[8];
  // A: byte e -> 1.0 (e<4) or 2.0 (e>=4), every lane
  for (int e = 0; e < 8; ++e) a[e] = q8(e < 4 ? 1.0f : 2.0f);
  // B one-hot: we need B[k][j] = (k == j && k < 16). Build per the CONTIGUOUS
  // hypothesis for B (k = 8*(l>>4)+e, col j = l&15): byte e of lane l is
  // B[8*(l>>4)+e][l&15] -> set to 1 iff 8*(l>>4)+e == (l&15).
  int col = lane & 15, g = lane >> 4;
  for (int e = 0; e < 8; ++e) b[e] = q8((8 * g + e) == col ? 1.0f : 0.0f);
  f32x4 acc = {0, 0, 0, 0};
  long av = *reinterpret_cast<long*>(a);
  long bv = *reinterpret_cast<long*>(b);
  acc = __builtin_amdgcn_mfma_f32_16x16x32_fp8_fp8(av, bv, acc, 0, 0, 0);
  // C layout: col = l&15, row = (l>>4)*4 + r
  for (int r = 0; r < 4; ++r) out[((lane >> 4) * 4 + r) * 16 + (lane & 15)] = acc[r];
}

// second probe: B built per the SPLIT hypothesis (k = 4*(l>>4) + e%4 + 16*(e/4))
__global__ void probeA_splitB(float* out) {
  int lane = threadIdx.x;
  unsigned char a[8], b[8];
  for (int e = 0; e < 8; ++e) a[e] = q8(e < 4 ? 1.0f : 2.0f);
  int col = lane & 15, g = lane >> 4;
  for (int e = 0; e < 8; ++e) {
    int k = 4 * g + (e & 3) + 16 * (e >> 2);
    b[e] = q8(k == col ? 1.0f : 0.0f);
  }
  f32x4 acc = {0, 0, 0, 0};
  long av = *reinterpret_cast<long*>(a);
  long bv = *reinterpret_cast<long*>(b);
  acc = __builtin_amdgcn_mfma_f32_16x16x32_fp8_fp8(av, bv, acc, 0, 0, 0);
  for (int r = 0; r < 4; ++r) out[((lane >> 4) * 4 + r) * 16 + (lane & 15)] = acc[r];
}

int main() {
  float* d;
  hipMalloc(&d, 16 * 16 * sizeof(float));
  float h[256];

  hipLaunchKernelGGL(probeA, dim3(1), dim3(64), 0, 0, d);
  hipMemcpy(h, d, sizeof(h), hipMemcpyDeviceToHost);
  printf("B-contiguous hypothesis, D row 0: ");
  for (int j = 0; j < 16; ++j) printf("%.0f ", h[j]);
  printf("\n");

  hipLaunchKernelGGL(probeA_splitB, dim3(1), dim3(64), 0, 0, d);
  hipMemcpy(h, d, sizeof(h), hipMemcpyDeviceToHost);
  printf("B-split hypothesis,      D row 0: ");
  for (int j = 0; j < 16; ++j) printf("%.0f ", h[j]);
  printf("\n");
  printf("interpretation: with the CORRECT B map, D[0][j] = A[0][k=j];\n"
         "A contiguous -> quads 1 1 1 1 2 2 2 2 1 1 1 1 2 2 2 2 over j=k<16\n"
         "A split      -> first 8 of row are 1s at k<16? see bytes: k<16 uses e<4 -> all 1\n");
  return 0;
}
