// Shared device helpers for the CDNA4 (gfx950) kernel set.
//
// Conventions (per /opt/skills/guides/cdna_hip_programming.md):
//  - wave width is 64; block sizes are multiples of 64
//  - bf16 tensors are loaded vectorized (short4/short8 reinterpret, G13)
//  - fp32 accumulation everywhere; bf16 only at the memory boundary
#pragma once

#include <hip/hip_runtime.h>
#include <hip/hip_bf16.h>
#include <stdint.h>

#define WAVE 64
#define DEV __device__ __forceinline__

typedef __hip_bfloat16 bf16_t;

union BF16x8 {
  ulonglong2 u128;   // 16 B — one global_load_dwordx4
  short s[8];
  bf16_t h[8];
};

union BF16x4 {
  uint2 u64;
  short s[4];
  bf16_t h[4];
};

DEV float bf2f(bf16_t v) { return __bfloat162float(v); }
DEV bf16_t f2bf(float v) { return __float2bfloat16(v); }

// wave-wide reductions (64 lanes)
DEV float wave_sum(float v) {
#pragma unroll
  for (int off = 32; off > 0; off >>= 1) v += __shfl_xor(v, off, 64);
  return v;
}

DEV float wave_max(float v) {
#pragma unroll
  for (int off = 32; off > 0; off >>= 1) v = fmaxf(v, __shfl_xor(v, off, 64));
  return v;
}

// reduction within a 16-lane group (lanes l, l^1, ..., l^8)
DEV float group16_sum(float v) {
#pragma unroll
  for (int off = 8; off > 0; off >>= 1) v += __shfl_xor(v, off, 64);
  return v;
}

// block-wide sum across waves through LDS; `scratch` must hold
// blockDim.x/64 floats.  Every thread returns the total.
DEV float block_sum(float v, float* scratch) {
  const int lane = threadIdx.x & 63;
  const int wid = threadIdx.x >> 6;
  const int nw = blockDim.x >> 6;
  v = wave_sum(v);
  if (lane == 0) scratch[wid] = v;
  __syncthreads();
  float total = 0.f;
#pragma unroll 4
  for (int i = 0; i < nw; ++i) total += scratch[i];
  return total;
}
