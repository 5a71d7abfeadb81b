// Fused rotary embedding + paged-KV scatter (gfx950).
//
// One kernel applies rotate-half RoPE to Q and K in place and scatters the
// post-rope K and raw V of every token into the paged cache at its flat
// slot — one pass over the QKV projection output instead of the three the
// unfused form costs (rope q, rope k, cache write).
//
// Layouts: q [N, Hq, D], k/v [N, Hkv, D] bf16 contiguous; cache
// [num_blocks, block_size, Hkv, D] viewed flat as [slots, Hkv, D];
// cos_sin [2, P, D/2] fp32.  D = head_dim (128 for the model family).
#include "common.h"

__global__ void rope_cache_kernel(
    bf16_t* __restrict__ q,            // [N, Hq, D], token stride q_stride
    bf16_t* __restrict__ k,            // [N, Hkv, D], token stride kv_stride
    const bf16_t* __restrict__ v,      // [N, Hkv, D], token stride kv_stride
    bf16_t* __restrict__ k_cache,      // [slots, Hkv, D]
    bf16_t* __restrict__ v_cache,
    const int64_t* __restrict__ positions,     // [N]
    const int64_t* __restrict__ slot_mapping,  // [N]
    const float* __restrict__ cos_sin,         // [2, P, D/2]
    const int Hq, const int Hkv, const int D, const int64_t P,
    const int64_t q_stride, const int64_t kv_stride) {
  const int token = blockIdx.x;
  const int64_t pos = positions[token];
  const int64_t slot = slot_mapping[token];
  const int d2 = D >> 1;
  const float* cos_t = cos_sin + pos * d2;
  const float* sin_t = cos_sin + (P + pos) * d2;

  const int wid = threadIdx.x >> 6;      // wave id
  const int lane = threadIdx.x & 63;
  const int nw = blockDim.x >> 6;
  const int per_lane = d2 / WAVE;        // rotate pairs per lane (D=128 → 1)

  // Q heads: rope in place
  for (int hq = wid; hq < Hq; hq += nw) {
    bf16_t* qh = q + (int64_t)token * q_stride + (int64_t)hq * D;
#pragma unroll
    for (int r = 0; r < 4; ++r) {        // supports D up to 512
      if (r >= per_lane) break;
      const int i = lane + (r << 6);
      const float c = cos_t[i], s = sin_t[i];
      const float x1 = bf2f(qh[i]), x2 = bf2f(qh[i + d2]);
      qh[i] = f2bf(x1 * c - x2 * s);
      qh[i + d2] = f2bf(x2 * c + x1 * s);
    }
  }
  // K heads: rope in place + cache scatter; V heads: cache scatter
  for (int hk = wid; hk < Hkv; hk += nw) {
    bf16_t* kh = k + (int64_t)token * kv_stride + (int64_t)hk * D;
    const bf16_t* vh = v + (int64_t)token * kv_stride + (int64_t)hk * D;
    bf16_t* kc = k_cache + (slot * Hkv + hk) * D;
    bf16_t* vc = v_cache + (slot * Hkv + hk) * D;
#pragma unroll
    for (int r = 0; r < 4; ++r) {
      if (r >= per_lane) break;
      const int i = lane + (r << 6);
      const float c = cos_t[i], s = sin_t[i];
      const float x1 = bf2f(kh[i]), x2 = bf2f(kh[i + d2]);
      const bf16_t k1 = f2bf(x1 * c - x2 * s), k2 = f2bf(x2 * c + x1 * s);
      kh[i] = k1;
      kh[i + d2] = k2;
      kc[i] = k1;
      kc[i + d2] = k2;
      vc[i] = vh[i];
      vc[i + d2] = vh[i + d2];
    }
  }
}

extern "C" void launch_rope_cache(void* q, void* k, const void* v,
                                  void* k_cache, void* v_cache,
                                  const int64_t* positions,
                                  const int64_t* slot_mapping,
                                  const float* cos_sin, int N, int Hq,
                                  int Hkv, int D, int64_t P,
                                  int64_t q_stride, int64_t kv_stride,
                                  hipStream_t stream) {
  dim3 grid(N), block(256);
  hipLaunchKernelGGL(rope_cache_kernel, grid, block, 0, stream, (bf16_t*)q,
                     (bf16_t*)k, (const bf16_t*)v, (bf16_t*)k_cache,
                     (bf16_t*)v_cache, positions, slot_mapping, cos_sin, Hq,
                     Hkv, D, P, q_stride, kv_stride);
}
