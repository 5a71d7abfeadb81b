// Full-vocabulary token sampling (gfx950): temperature → optional grammar
// mask → exact top-k → top-p (nucleus) → inverse-CDF draw in token-id
// order; one workgroup per row, vocab up to 256k (Llama-3: 128256).
//
// The ≤512-entry LDS bitonic sampler (sampling.hip) covers the byte
// tokenizer; a real BPE vocabulary needs per-row selection over 128k
// entries.  Sorting is out at B≈1k rows; instead thresholds come from LDS
// radix-select: logits map to order-preserving uint32 keys and a
// 3-level histogram walk (10+11+11 bits; ≤2048 buckets × 8 B = 16 KB LDS,
// well inside the 160 KB CU budget) pins the exact k-th-largest key; the
// nucleus (top-p) boundary key falls out of the same walk with per-bucket
// exp-mass instead of counts.  Every pass streams the row coalesced
// (thread t reads i = t, t+WG, …); only the final draw walks blocked
// stripes so the inverse CDF runs in token-id order and draws match the
// fp32 oracle (ops/reference.py softmax_sample) for a given uniform.
//
// Semantics match the oracle:
//   greedy (T<=0): argmax, lowest index on ties
//   top-k:         keep {x >= kth-largest}   (ties at the k-th all kept)
//   top-p:         keep the minimal high-prob set whose exclusive
//                  cumulative <= p            (boundary-key ties all kept)
#include "common.h"

#define WG 256
#define NBUCKET 2048

// order-preserving float -> uint32 key
__device__ __forceinline__ uint32_t f2key(float x) {
  uint32_t u = __float_as_uint(x);
  return (u & 0x80000000u) ? ~u : (u | 0x80000000u);
}

template <typename T>
__device__ __forceinline__ float ldf(const T* p);
template <>
__device__ __forceinline__ float ldf<float>(const float* p) { return *p; }
template <>
__device__ __forceinline__ float ldf<bf16_t>(const bf16_t* p) {
  return bf2f(*p);
}

__device__ __forceinline__ float blk_reduce_max(float v, float* red) {
#pragma unroll
  for (int off = 32; off > 0; off >>= 1) v = fmaxf(v, __shfl_down(v, off, 64));
  if ((threadIdx.x & 63) == 0) red[threadIdx.x >> 6] = v;
  __syncthreads();
  if (threadIdx.x == 0) {
    float r = red[0];
    for (int w = 1; w < WG / 64; ++w) r = fmaxf(r, red[w]);
    red[0] = r;
  }
  __syncthreads();
  float r = red[0];
  __syncthreads();
  return r;
}

__device__ __forceinline__ float blk_reduce_sum(float v, float* red) {
#pragma unroll
  for (int off = 32; off > 0; off >>= 1) v += __shfl_down(v, off, 64);
  if ((threadIdx.x & 63) == 0) red[threadIdx.x >> 6] = v;
  __syncthreads();
  if (threadIdx.x == 0) {
    float r = red[0];
    for (int w = 1; w < WG / 64; ++w) r += red[w];
    red[0] = r;
  }
  __syncthreads();
  float r = red[0];
  __syncthreads();
  return r;
}

// radix levels: bits [22,32) [11,22) [0,11) — widths 10/11/11
__constant__ int LVL_SHIFT[3] = {22, 11, 0};
__constant__ int LVL_BITS[3] = {10, 11, 11};

// Find the boundary key where the descending cumulative weight crosses
// `target`.  Weight per element = 1 (BY_MASS=false, top-k) or
// exp((x-m)/T) (BY_MASS=true, top-p).  Only elements with key >= kmin and
// allowed by the mask participate.  Returns the fully-resolved 32-bit
// boundary key: kept = {key >= boundary}.
template <typename T, bool BY_MASS>
__device__ uint32_t radix_walk(const T* lrow, const uint8_t* mrow, int V,
                               float m, float invT, uint32_t kmin,
                               double target, uint32_t* hist_c,
                               float* hist_m) {
  uint32_t prefix = 0;
  double above = 0.0;  // weight strictly above the current prefix range
  __shared__ int sel_s;
  __shared__ double above_s;
  for (int level = 0; level < 3; ++level) {
    const int shift = LVL_SHIFT[level];
    const int nb = 1 << LVL_BITS[level];
    const uint32_t pmask =
        (level == 0) ? 0u : (0xFFFFFFFFu << (LVL_SHIFT[level - 1]));
    for (int i = threadIdx.x; i < nb; i += WG) {
      hist_c[i] = 0;
      hist_m[i] = 0.f;
    }
    __syncthreads();
    for (int i = threadIdx.x; i < V; i += WG) {
      if (mrow && !mrow[i]) continue;
      const float x = ldf<T>(lrow + i);
      const uint32_t key = f2key(x);
      if (key < kmin) continue;
      if ((key & pmask) != (prefix & pmask)) continue;
      const uint32_t b = (key >> shift) & (nb - 1);
      if (BY_MASS)
        atomicAdd(&hist_m[b], __expf((x - m) * invT));
      else
        atomicAdd(&hist_c[b], 1u);
    }
    __syncthreads();
    if (threadIdx.x == 0) {
      // serial walk from the top bucket: ≤2048 iterations, ~µs
      double cum = above;
      int sel = 0;
      for (int b = nb - 1; b >= 0; --b) {
        const double w = BY_MASS ? (double)hist_m[b] : (double)hist_c[b];
        if (cum + w >= target || b == 0) {
          sel = b;
          break;
        }
        cum += w;
      }
      sel_s = sel;
      above_s = cum;
    }
    __syncthreads();
    prefix |= ((uint32_t)sel_s) << shift;
    above = above_s;
    __syncthreads();
  }
  return prefix;
}

template <typename T>
__device__ __forceinline__ float kept_mass(const T* lrow, const uint8_t* mrow,
                                           int V, float m, float invT,
                                           uint32_t kmin, float* red) {
  float part = 0.f;
  for (int i = threadIdx.x; i < V; i += WG) {
    if (mrow && !mrow[i]) continue;
    const float x = ldf<T>(lrow + i);
    if (f2key(x) < kmin) continue;
    part += __expf((x - m) * invT);
  }
  return blk_reduce_sum(part, red);
}

template <typename T>
__global__ __launch_bounds__(WG) void sample_fullvocab_kernel(
    int64_t* __restrict__ out,           // [B]
    const T* __restrict__ logits,        // [B, V]
    const float* __restrict__ temps,     // [B]
    const int64_t* __restrict__ top_ks,  // [B] (0 = off)
    const float* __restrict__ top_ps,    // [B]
    const float* __restrict__ uniforms,  // [B] in [0,1)
    const uint8_t* __restrict__ mask,    // [M, V] bool, may be null
    const int32_t* __restrict__ mask_map,// [B] row -> mask row (-1 = none)
    const int V) {
  const int row = blockIdx.x;
  const T* lrow = logits + (int64_t)row * V;
  // compact masks: only constrained rows carry one (a dense [B,V] mask at
  // B=1k, V=128k would be 128 MB of pinned staging per step)
  const uint8_t* mrow = nullptr;
  if (mask && mask_map && mask_map[row] >= 0)
    mrow = mask + (int64_t)mask_map[row] * V;
  const float Tmp = temps[row];
  const bool greedy = Tmp <= 0.f;
  const float invT = greedy ? 1.f : 1.f / Tmp;

  __shared__ uint32_t hist_c[NBUCKET];
  __shared__ float hist_m[NBUCKET];
  __shared__ float red[WG / 64];

  // ---- pass 1: masked row max
  float mx = -3.4e38f;
  for (int i = threadIdx.x; i < V; i += WG) {
    if (mrow && !mrow[i]) continue;
    mx = fmaxf(mx, ldf<T>(lrow + i));
  }
  const float m = blk_reduce_max(mx, red);

  if (greedy) {
    // lowest index attaining the max (oracle argmax tie rule)
    int best = V;
    for (int i = threadIdx.x; i < V; i += WG) {
      if (mrow && !mrow[i]) continue;
      if (ldf<T>(lrow + i) == m) {
        best = i;
        break;  // strided loop: the first hit is this thread's lowest
      }
    }
#pragma unroll
    for (int off = 32; off > 0; off >>= 1)
      best = min(best, __shfl_down(best, off, 64));
    __shared__ int ired[WG / 64];
    if ((threadIdx.x & 63) == 0) ired[threadIdx.x >> 6] = best;
    __syncthreads();
    if (threadIdx.x == 0) {
      int b = ired[0];
      for (int w = 1; w < WG / 64; ++w) b = min(b, ired[w]);
      out[row] = (int64_t)(b < V ? b : 0);
    }
    return;
  }

  // ---- top-k: kmin = key of the k-th largest; kept = {key >= kmin}
  uint32_t kmin = 0;
  const int tk = (int)top_ks[row];
  if (tk > 0 && tk < V) {
    kmin = radix_walk<T, false>(lrow, mrow, V, m, invT, 0u, (double)tk,
                                hist_c, hist_m);
  }

  // ---- top-p: raise kmin to the nucleus boundary key
  const float p = top_ps[row];
  if (p < 1.f) {
    const float M = kept_mass<T>(lrow, mrow, V, m, invT, kmin, red);
    const uint32_t pmin = radix_walk<T, true>(
        lrow, mrow, V, m, invT, kmin, (double)p * (double)M, hist_c, hist_m);
    if (pmin > kmin) kmin = pmin;
  }

  // ---- inverse-CDF draw in token-id order over the kept set.
  // Thread t owns the blocked stripe [t*stride, (t+1)*stride).
  const int stride = (V + WG - 1) / WG;
  const int lo = min(threadIdx.x * stride, V);
  const int hi = min(lo + stride, V);
  float mine = 0.f;
  for (int i = lo; i < hi; ++i) {
    if (mrow && !mrow[i]) continue;
    const float x = ldf<T>(lrow + i);
    if (f2key(x) < kmin) continue;
    mine += __expf((x - m) * invT);
  }
  __shared__ float stripe[WG + 1];
  stripe[threadIdx.x + 1] = mine;
  if (threadIdx.x == 0) stripe[0] = 0.f;
  __syncthreads();
  if (threadIdx.x == 0) {
    for (int i = 1; i <= WG; ++i) stripe[i] += stripe[i - 1];
  }
  __syncthreads();
  const float total = stripe[WG];
  const float target = uniforms[row] * total;
  __shared__ int out_s;
  if (threadIdx.x == 0) out_s = -1;
  __syncthreads();
  // exactly one stripe satisfies prefix <= target < prefix + mine
  // (mine > 0 guaranteed for the owner since the CDF strictly increases
  // across it); fp edge cases fall through to the fallback below
  if (mine > 0.f && stripe[threadIdx.x] <= target &&
      target < stripe[threadIdx.x + 1]) {
    float cum = stripe[threadIdx.x];
    int pick = -1;
    for (int i = lo; i < hi; ++i) {
      if (mrow && !mrow[i]) continue;
      const float x = ldf<T>(lrow + i);
      if (f2key(x) < kmin) continue;
      cum += __expf((x - m) * invT);
      pick = i;
      if (cum > target) break;
    }
    if (pick >= 0) out_s = pick;
  }
  __syncthreads();
  if (threadIdx.x == 0 && out_s < 0) {
    // fallback (target == total to fp precision, or an empty kept set):
    // the last kept token, matching searchsorted's clamp in the oracle
    for (int i = V - 1; i >= 0; --i) {
      if (mrow && !mrow[i]) continue;
      if (f2key(ldf<T>(lrow + i)) < kmin) continue;
      out_s = i;
      break;
    }
    if (out_s < 0) out_s = 0;
  }
  __syncthreads();
  if (threadIdx.x == 0) out[row] = (int64_t)out_s;
}

extern "C" void launch_sample_fullvocab(int64_t* out, const void* logits, bool bf16,
                             const float* temps, const int64_t* top_ks,
                             const float* top_ps, const float* uniforms,
                             const uint8_t* mask, const int32_t* mask_map,
                             int B, int V, hipStream_t stream) {
  dim3 grid(B), block(WG);
  if (bf16) {
    hipLaunchKernelGGL((sample_fullvocab_kernel<bf16_t>), grid, block, 0,
                       stream, out, (const bf16_t*)logits, temps,
                       top_ks, top_ps, uniforms, mask, mask_map, V);
  } else {
    hipLaunchKernelGGL((sample_fullvocab_kernel<float>), grid, block, 0,
                       stream, out, (const float*)logits, temps, top_ks,
                       top_ps, uniforms, mask, mask_map, V);
  }
}
