// MFMA fragment-layout probe (gfx950).
//
// Verifies on hardware the lane→element mappings the attention kernels
// assume for v_mfma_f32_32x32x16_bf16:
//   A[m][k]: lane l holds m = l&31, k = (l>>5)*8 + e   (e = 0..7)
//   B[k][n]: lane l holds n = l&31, k = (l>>5)*8 + e
//   D[m][n]: lane l holds n = l&31, m = (reg&3) + 8*(reg>>2) + 4*(l>>5)
// (D comes from the CDNA4 guide §3; A/B are the canonical CDNA pattern.)
// The GPU test builds fragments from matrices with these maps, runs one
// MFMA, and compares against a host matmul — asymmetric inputs so a
// transposed mapping cannot pass.
#include "common.h"

typedef __bf16 bf16x8_v __attribute__((ext_vector_type(8)));
typedef float f32x16_v __attribute__((ext_vector_type(16)));

__global__ void mfma_probe_32x32x16(
    float* __restrict__ d_out,          // [32, 32] row-major
    const bf16_t* __restrict__ a_in,    // [32, 16] row-major (M x K)
    const bf16_t* __restrict__ b_in) {  // [16, 32] row-major (K x N)
  const int lane = threadIdx.x & 63;
  bf16x8_v a, b;
#pragma unroll
  for (int e = 0; e < 8; ++e) {
    const int k = ((lane >> 5) << 3) + e;
    a[e] = (__bf16)(float)bf2f(a_in[(lane & 31) * 16 + k]);
    b[e] = (__bf16)(float)bf2f(b_in[k * 32 + (lane & 31)]);
  }
  f32x16_v c = {};
  c = __builtin_amdgcn_mfma_f32_32x32x16_bf16(a, b, c, 0, 0, 0);
#pragma unroll
  for (int r = 0; r < 16; ++r) {
    const int m = (r & 3) + 8 * (r >> 2) + 4 * (lane >> 5);
    d_out[m * 32 + (lane & 31)] = c[r];
  }
}

extern "C" void launch_mfma_probe(float* d, const void* a, const void* b,
                                  hipStream_t stream) {
  hipLaunchKernelGGL(mfma_probe_32x32x16, dim3(1), dim3(64), 0, stream, d,
                     (const bf16_t*)a, (const bf16_t*)b);
}
