// Prefill (chunked) causal attention over the paged KV cache (gfx950) — v0.
//
// Q-tiled flash attention: each workgroup owns a 64-row Q tile of one
// (sequence, q-head); K/V tiles of 64 keys are staged through LDS once and
// reused by all 64 q-rows — the Q-tiling that makes prefill
// arithmetic-bound instead of re-streaming KV per row.
//
//   workgroup: 256 threads = 4 waves; wave owns 16 q-rows
//   lane layout: 16 rows × 4 lanes, each lane covers 32 of the 128 dims
//   KV staging: 64 keys × 128 dims bf16 → 16 KiB K + 16 KiB V in LDS
//   online softmax per row (4-lane shfl reduction for scores)
//
// v0 is VALU-based (correctness + the LDS/tiling structure); the MFMA
// version (32x32x16 bf16 tiles per §5/§B of the CDNA4 guide) replaces the
// inner product once its fragment layouts are probe-verified on hardware —
// at agent-serving shapes prefill attention is ≪ the layer GEMM cost, so
// v0 does not gate the end-to-end path.
#include "common.h"

#define D_HEAD 128
#define TILE_Q 64
#define TILE_K 64
#define LANES_PER_ROW 4
#define DIMS_PER_LANE 32  // D_HEAD / LANES_PER_ROW

__global__ __launch_bounds__(256) void prefill_attn_kernel(
    bf16_t* __restrict__ out,            // [T, Hq, D]
    const bf16_t* __restrict__ q,        // [T, Hq, D]
    const bf16_t* __restrict__ k_cache,  // [slots, Hkv, D]
    const bf16_t* __restrict__ v_cache,
    const int* __restrict__ block_tables,  // [num_seqs, max_blocks]
    const int* __restrict__ seq_lens,      // [num_seqs] (after this chunk)
    const int* __restrict__ ctx_lens,      // [num_seqs] (before this chunk)
    const int* __restrict__ row_starts,    // [num_seqs] row offset into q/out
    const int* __restrict__ tile_seq,      // [num_tiles] seq index per tile
    const int* __restrict__ tile_q0,       // [num_tiles] first row in chunk
    const float scale, const int Hq, const int Hkv, const int max_blocks,
    const int kv_block, const int64_t q_stride) {
  const int tile = blockIdx.x;
  const int head = blockIdx.y;
  const int kvh = head / (Hq / Hkv);
  const int seq = tile_seq[tile];
  const int q0 = tile_q0[tile];           // row offset within the chunk
  const int S = seq_lens[seq];
  const int ctx = ctx_lens[seq];
  const int row_base = row_starts[seq];
  const int q_len = S - ctx;              // chunk length
  const int rows_here = min(TILE_Q, q_len - q0);

  const int lane = threadIdx.x & 63;
  const int wid = threadIdx.x >> 6;
  const int row_in_wave = lane >> 2;            // 16 rows per wave
  const int row = (wid << 4) + row_in_wave;     // 0..63 within tile
  const int sub = lane & 3;                     // dim quarter
  const int d0 = sub * DIMS_PER_LANE;

  __shared__ bf16_t kt[TILE_K][D_HEAD];
  __shared__ bf16_t vt[TILE_K][D_HEAD];
  __shared__ int bt[512];
  const int nblk = (S + kv_block - 1) / kv_block;
  for (int i = threadIdx.x; i < nblk; i += blockDim.x)
    bt[i] = block_tables[(int64_t)seq * max_blocks + i];
  __syncthreads();

  // Q row → registers (32 dims per lane), pre-scaled
  float qreg[DIMS_PER_LANE];
  const bool live = row < rows_here;
  const int qpos = ctx + q0 + row;        // global position of this q-row
  if (live) {
    const bf16_t* qrow =
        q + (int64_t)(row_base + q0 + row) * q_stride + (int64_t)head * D_HEAD + d0;
#pragma unroll
    for (int e = 0; e < DIMS_PER_LANE; e += 8) {
      BF16x8 v8 = *(const BF16x8*)(qrow + e);
#pragma unroll
      for (int j = 0; j < 8; ++j) qreg[e + j] = bf2f(v8.h[j]) * scale;
    }
  }

  float m = -1e30f, l = 0.f, acc[DIMS_PER_LANE];
#pragma unroll
  for (int e = 0; e < DIMS_PER_LANE; ++e) acc[e] = 0.f;

  // causal bound: the last live row of this tile sees keys < ctx+q0+rows
  const int k_end = min(S, ctx + q0 + rows_here);
  for (int kbase = 0; kbase < k_end; kbase += TILE_K) {
    const int kn = min(TILE_K, k_end - kbase);
    // stage K/V tile: 256 threads × 8 bf16 lanes; 64*128/8 = 1024 vectors
    __syncthreads();
    for (int i = threadIdx.x; i < (kn * D_HEAD) / 8; i += blockDim.x) {
      const int kk = (i << 3) / D_HEAD;
      const int dd = (i << 3) % D_HEAD;
      const int j = kbase + kk;
      const int64_t slot = (int64_t)bt[j / kv_block] * kv_block + j % kv_block;
      *(BF16x8*)(&kt[kk][dd]) =
          *(const BF16x8*)(k_cache + (slot * Hkv + kvh) * D_HEAD + dd);
      *(BF16x8*)(&vt[kk][dd]) =
          *(const BF16x8*)(v_cache + (slot * Hkv + kvh) * D_HEAD + dd);
    }
    __syncthreads();

    if (live) {
      for (int kk = 0; kk < kn; ++kk) {
        const int kpos = kbase + kk;
        if (kpos > qpos) break;  // keys are in position order: done for row
        float dot = 0.f;
#pragma unroll
        for (int e = 0; e < DIMS_PER_LANE; e += 2) {
          // ds_read_b64-width reads
          dot += qreg[e] * bf2f(kt[kk][d0 + e]);
          dot += qreg[e + 1] * bf2f(kt[kk][d0 + e + 1]);
        }
        // 4-lane reduce within the row group
        dot += __shfl_xor(dot, 1, 64);
        dot += __shfl_xor(dot, 2, 64);
        const float mn = fmaxf(m, dot);
        const float corr = __expf(m - mn);
        const float w = __expf(dot - mn);
        l = l * corr + w;
        m = mn;
#pragma unroll
        for (int e = 0; e < DIMS_PER_LANE; ++e)
          acc[e] = acc[e] * corr + w * bf2f(vt[kk][d0 + e]);
      }
    }
  }

  if (live) {
    const float inv = (l > 0.f) ? 1.f / l : 0.f;
    bf16_t* orow = out + ((int64_t)(row_base + q0 + row) * Hq + head) * D_HEAD + d0;
#pragma unroll
    for (int e = 0; e < DIMS_PER_LANE; e += 8) {
      BF16x8 v8;
#pragma unroll
      for (int j = 0; j < 8; ++j) v8.h[j] = f2bf(acc[e + j] * inv);
      *(BF16x8*)(orow + e) = v8;
    }
  }
}

extern "C" void launch_prefill_attn(
    void* out, const void* q, const void* k_cache, const void* v_cache,
    const int* block_tables, const int* seq_lens, const int* ctx_lens,
    const int* row_starts, const int* tile_seq, const int* tile_q0,
    float scale, int num_tiles, int Hq, int Hkv, int D, int max_blocks,
    int kv_block, int64_t q_stride, hipStream_t stream) {
  if (D != D_HEAD) return;
  dim3 grid(num_tiles, Hq), block(256);
  hipLaunchKernelGGL(prefill_attn_kernel, grid, block, 0, stream,
                     (bf16_t*)out, (const bf16_t*)q, (const bf16_t*)k_cache,
                     (const bf16_t*)v_cache, block_tables, seq_lens,
                     ctx_lens, row_starts, tile_seq, tile_q0, scale, Hq,
                     Hkv, max_blocks, kv_block, q_stride);
}
