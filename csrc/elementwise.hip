// Memory-bound elementwise / normalization kernels (gfx950).
//
// RMSNorm + fused residual-add RMSNorm + SwiGLU — the between-GEMM glue of
// the Llama/Mixtral layer, fused so each activation makes one HBM round
// trip.  All bf16 I/O vectorized as 16-B lanes (guide G13: scalar bf16 is
// 2-2.5x slower), fp32 math.
#include "common.h"

// One block per row.  hidden sizes of interest: 4096 (8B), 8192 (70B) —
// bf16x8 lanes: 512 / 1024 vectors per row.
template <bool FUSED_ADD>
__global__ void rmsnorm_kernel(
    bf16_t* __restrict__ out,          // [n, h] normed
    bf16_t* __restrict__ residual,     // [n, h] in: residual, out: x+residual (FUSED_ADD)
    const bf16_t* __restrict__ x,      // [n, h]
    const bf16_t* __restrict__ weight, // [h]
    const float eps, const int h) {
  const int row = blockIdx.x;
  const bf16_t* xr = x + (int64_t)row * h;
  bf16_t* rr = FUSED_ADD ? residual + (int64_t)row * h : nullptr;
  bf16_t* orow = out + (int64_t)row * h;
  const int nvec = h >> 3;

  float ss = 0.f;
  // pass 1: (x [+ residual]) sum of squares; fused path writes the new
  // residual so pass 2 can re-read it from L2/LDS-free path
  for (int i = threadIdx.x; i < nvec; i += blockDim.x) {
    BF16x8 vx = *(const BF16x8*)(xr + (i << 3));
    if (FUSED_ADD) {
      BF16x8 vr = *(const BF16x8*)(rr + (i << 3));
#pragma unroll
      for (int j = 0; j < 8; ++j) {
        float s = bf2f(vx.h[j]) + bf2f(vr.h[j]);
        vx.h[j] = f2bf(s);
      }
      *(BF16x8*)(rr + (i << 3)) = vx;  // new residual
    }
#pragma unroll
    for (int j = 0; j < 8; ++j) {
      float f = bf2f(vx.h[j]);
      ss += f * f;
    }
  }
  __shared__ float scratch[16];
  ss = block_sum(ss, scratch);
  const float inv = rsqrtf(ss / h + eps);

  const bf16_t* src = FUSED_ADD ? rr : xr;
  for (int i = threadIdx.x; i < nvec; i += blockDim.x) {
    BF16x8 v = *(const BF16x8*)(src + (i << 3));
    BF16x8 w = *(const BF16x8*)(weight + (i << 3));
#pragma unroll
    for (int j = 0; j < 8; ++j) v.h[j] = f2bf(bf2f(v.h[j]) * inv * bf2f(w.h[j]));
    *(BF16x8*)(orow + (i << 3)) = v;
  }
}

// SwiGLU: out[n, i] = silu(gu[n, i]) * gu[n, I + i]
__global__ void swiglu_kernel(
    bf16_t* __restrict__ out, const bf16_t* __restrict__ gate_up,
    const int64_t n_rows, const int inter) {
  const int64_t nvec = n_rows * (inter >> 3);
  const int64_t stride = (int64_t)gridDim.x * blockDim.x;
  for (int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x; i < nvec; i += stride) {
    const int64_t row = i / (inter >> 3);
    const int64_t col = i % (inter >> 3);
    const bf16_t* base = gate_up + row * (2 * (int64_t)inter);
    BF16x8 g = *(const BF16x8*)(base + (col << 3));
    BF16x8 u = *(const BF16x8*)(base + inter + (col << 3));
    BF16x8 o;
#pragma unroll
    for (int j = 0; j < 8; ++j) {
      float gf = bf2f(g.h[j]);
      float s = gf / (1.f + __expf(-gf));
      o.h[j] = f2bf(s * bf2f(u.h[j]));
    }
    *(BF16x8*)(out + row * inter + (col << 3)) = o;
  }
}

extern "C" {

void launch_rmsnorm(void* out, const void* x, const void* weight,
                    float eps, int n, int h, hipStream_t stream) {
  dim3 grid(n), block(256);
  hipLaunchKernelGGL((rmsnorm_kernel<false>), grid, block, 0, stream,
                     (bf16_t*)out, nullptr, (const bf16_t*)x,
                     (const bf16_t*)weight, eps, h);
}

void launch_fused_add_rmsnorm(void* out, void* residual, const void* x,
                              const void* weight, float eps, int n, int h,
                              hipStream_t stream) {
  dim3 grid(n), block(256);
  hipLaunchKernelGGL((rmsnorm_kernel<true>), grid, block, 0, stream,
                     (bf16_t*)out, (bf16_t*)residual, (const bf16_t*)x,
                     (const bf16_t*)weight, eps, h);
}

void launch_swiglu(void* out, const void* gate_up, int64_t n, int inter,
                   hipStream_t stream) {
  int64_t nvec = n * (inter >> 3);
  int64_t want = (nvec + 255) / 256;
  int blocks = (int)(want < 2048 ? want : 2048);
  hipLaunchKernelGGL(swiglu_kernel, dim3(blocks), dim3(256), 0, stream,
                     (bf16_t*)out, (const bf16_t*)gate_up, n, inter);
}

}  // extern "C"
