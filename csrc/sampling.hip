// Batch token sampling (gfx950): temperature → optional grammar mask →
// top-k → top-p → inverse-CDF draw, one workgroup per row.
//
// The engine samples over the tokenizer's live vocabulary (bytes +
// specials, ≤512 ids) after the full-vocab LM-head GEMM; a full bitonic
// sort of the padded 512-entry row in LDS gives exact top-k and top-p in a
// few microseconds.  Uniform draws come in from the host-side torch
// generator so sampling stays reproducible against the fp32 reference.
#include "common.h"

#define VPAD 512  // padded live-vocab size (power of two for bitonic)

__global__ __launch_bounds__(256) void sample_kernel(
    int64_t* __restrict__ out,            // [B]
    const float* __restrict__ logits,     // [B, V] (live slice)
    const float* __restrict__ temps,      // [B]
    const int64_t* __restrict__ top_ks,   // [B] (0 = off)
    const float* __restrict__ top_ps,     // [B]
    const float* __restrict__ uniforms,   // [B] in [0,1)
    const uint8_t* __restrict__ mask,     // [B, V] bool, may be null
    const int V) {
  const int row = blockIdx.x;
  __shared__ float val[VPAD];
  __shared__ short idx[VPAD];
  __shared__ float red[VPAD / 64];

  const float T = temps[row];
  const bool greedy = T <= 0.f;
  const float invT = greedy ? 1.f : 1.f / T;
  const float* lrow = logits + (int64_t)row * V;
  const uint8_t* mrow = mask ? mask + (int64_t)row * V : nullptr;

  for (int i = threadIdx.x; i < VPAD; i += blockDim.x) {
    float v = -1e30f;
    if (i < V && (!mrow || mrow[i])) v = lrow[i] * invT;
    val[i] = v;
    idx[i] = (short)i;
  }
  __syncthreads();

  // bitonic sort descending on val (stable enough; ties broken by index move)
  for (int ksz = 2; ksz <= VPAD; ksz <<= 1) {
    for (int j = ksz >> 1; j > 0; j >>= 1) {
      for (int t = threadIdx.x; t < VPAD / 2; t += blockDim.x) {
        const int i = 2 * t - (t & (j - 1));
        const int ixj = i + j;
        const bool up = ((i & ksz) == 0);
        const float a = val[i], b = val[ixj];
        if ((a < b) == up) {
          val[i] = b;
          val[ixj] = a;
          const short tmp = idx[i];
          idx[i] = idx[ixj];
          idx[ixj] = tmp;
        }
      }
      __syncthreads();
    }
  }

  if (greedy) {
    if (threadIdx.x == 0) out[row] = (int64_t)idx[0];
    return;
  }

  // keep = min(top_k or V, V); probabilities over the kept prefix
  int keep = V;
  const int tk = (int)top_ks[row];
  if (tk > 0 && tk < keep) keep = tk;

  // exps relative to the max (val[0]); masked/padded entries are ~exp(-1e30)=0
  const float vmax = val[0];
  float part = 0.f;
  for (int i = threadIdx.x; i < VPAD; i += blockDim.x) {
    const float e = (i < keep) ? __expf(val[i] - vmax) : 0.f;
    val[i] = e;
    if (i < keep) part += e;
  }
  __syncthreads();
  const float total = block_sum(part, red);

  // top-p on the kept prefix: cumulative (exclusive) ≤ p * total
  const float p = top_ps[row];
  const float cut = (p < 1.f) ? p * total : 3.4e38f;

  // serial-ish scan is fine at 512 entries: one wave scans with shfl
  // (exclusive prefix), writing cumsum back to val
  if (threadIdx.x < 64) {
    float carry = 0.f;
    for (int base = 0; base < VPAD; base += 64) {
      float x = val[base + threadIdx.x];
      // inclusive scan within the wave
      float sc = x;
#pragma unroll
      for (int off = 1; off < 64; off <<= 1) {
        const float n = __shfl_up(sc, off, 64);
        if ((threadIdx.x & 63) >= off) sc += n;
      }
      val[base + threadIdx.x] = carry + sc - x;  // exclusive prefix
      carry += __shfl(sc, 63, 64);
    }
  }
  __syncthreads();

  // kept mass after top-p: largest i with excl_prefix(i) <= cut gets kept
  // through; find kept_total = prefix of first dropped element
  __shared__ float kept_total_s;
  __shared__ int kept_n_s;
  if (threadIdx.x == 0) {
    int n = 1;  // always keep the head token
    while (n < keep) {
      if (val[n] > cut) break;  // exclusive prefix beyond the budget
      ++n;
    }
    kept_n_s = n;
    kept_total_s = (n < VPAD) ? val[n] : total;
    if (n == keep) {
      // prefix of element `keep` = sum of kept — val[keep] holds it unless
      // keep == VPAD
      kept_total_s = (keep < VPAD) ? val[keep] : total;
    }
  }
  __syncthreads();
  const int kept_n = kept_n_s;
  const float target = uniforms[row] * kept_total_s;

  // inverse CDF: first element whose inclusive prefix exceeds target
  __shared__ int pick_s;
  if (threadIdx.x == 0) pick_s = kept_n - 1;
  __syncthreads();
  for (int i = threadIdx.x; i < kept_n; i += blockDim.x) {
    const float lo = val[i];
    const float hi = (i + 1 < VPAD) ? val[i + 1] : kept_total_s;
    if (lo <= target && target < hi) pick_s = i;
  }
  __syncthreads();
  if (threadIdx.x == 0) out[row] = (int64_t)idx[pick_s];
}

extern "C" void launch_sample(int64_t* out, const float* logits,
                              const float* temps, const int64_t* top_ks,
                              const float* top_ps, const float* uniforms,
                              const uint8_t* mask, int B, int V,
                              hipStream_t stream) {
  hipLaunchKernelGGL(sample_kernel, dim3(B), dim3(256), 0, stream, out,
                     logits, temps, top_ks, top_ps, uniforms, mask, V);
}
