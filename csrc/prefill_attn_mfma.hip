// MFMA prefill attention (gfx950) — v1.
//
// Flash-style causal attention over the paged KV cache on the matrix
// cores, replacing the VALU v0 kernel (161 ms / 26% of engine GPU time in
// profiles/r01: ~20 TF; 32x32x16 bf16 MFMA lifts the QK^T/PV inner
// products onto the 2.5 PF pipe).
//
// Geometry (CDNA4 guide §B, simplified to a verified-layout subset):
//   workgroup = 256 threads = 4 waves, one (seq-tile, head)
//   wave owns QBLK=32 q-rows; workgroup covers 128 rows
//   KV tile = 32 keys; K staged row-major in LDS, V transposed in LDS
//   QK^T swapped (mfma(A=K, B=Q)) so each lane owns ONE q-column: the
//     online-softmax row reduction is 16 regs + one shfl_xor(32) (T12)
//   P→bf16 A-fragments via v_cvt_pk_bf16_f32 + permlane32_swap (T12/T21)
//   O accumulates in 4 D-tile f32x16 fragments; per-q alpha/normalizer
//     cross the S→O layout change through a per-wave LDS broadcast
//
// Fragment layouts are the probe-verified maps (csrc/mfma_probe.hip).
#include "common.h"

typedef __bf16 bf16x8_v __attribute__((ext_vector_type(8)));
typedef float f32x16_v __attribute__((ext_vector_type(16)));
typedef __attribute__((ext_vector_type(2))) int int2_v;

#define D_HEAD 128
#define QWG 128     // q rows per workgroup
#define QBLK 32     // q rows per wave
#define KVBLK 32    // keys per tile

__global__ __launch_bounds__(256) void prefill_attn_mfma_kernel(
    bf16_t* __restrict__ out,            // [T, Hq, D]
    const bf16_t* __restrict__ q,        // [T, Hq, D]
    const bf16_t* __restrict__ k_cache,  // [slots, Hkv, D]
    const bf16_t* __restrict__ v_cache,
    const int* __restrict__ block_tables,
    const int* __restrict__ seq_lens,
    const int* __restrict__ ctx_lens,
    const int* __restrict__ row_starts,
    const int* __restrict__ tile_seq,
    const int* __restrict__ tile_q0,
    const float scale, const int Hq, const int Hkv, const int max_blocks,
    const int kv_block, const int64_t q_stride) {
  const int tile = blockIdx.x;
  const int head = blockIdx.y;
  const int kvh = head / (Hq / Hkv);
  const int seq = tile_seq[tile];
  const int q0 = tile_q0[tile];
  const int S = seq_lens[seq];
  const int ctx = ctx_lens[seq];
  const int row_base = row_starts[seq];
  const int q_len = S - ctx;
  const int rows_here = min(QWG, q_len - q0);

  const int lane = threadIdx.x & 63;
  const int wid = threadIdx.x >> 6;
  const int qcol = lane & 31;           // this lane's q within the wave (S layout)
  const int khalf = lane >> 5;

  // LDS: K row-major [32][128] with XOR swizzle on 16-B units to break the
  // ds_read_b128 16-way conflict (guide G4: byte ^= (row&7)<<4); V
  // transposed [128][32+2] — the 2-element row pad makes the row stride
  // 17 words (coprime with the 32 banks), so the B-fragment b128 reads
  // (one V row per lane) start on all-distinct banks instead of the
  // 2-bank/16-way pileup a 64-B stride gives, and the 2-B transpose
  // scatter drops from 16-way to ~4-way (r02 PMC: this kernel's LDS
  // conflict cycles matched its MFMA-busy cycles before the pad).
  // Double buffered: tile t+1 stages while tile t computes.
  __shared__ bf16_t Kt[2][KVBLK][D_HEAD];
  __shared__ bf16_t Vt[2][D_HEAD][KVBLK + 2];
  __shared__ float bcast[4][QBLK];
  __shared__ int bt[512];
  const int nblk = (S + kv_block - 1) / kv_block;
  for (int i = threadIdx.x; i < nblk; i += blockDim.x)
    bt[i] = block_tables[(int64_t)seq * max_blocks + i];

  // Q fragments: lane's q-row, 8 d-steps x 8 bf16 (B-operand of the swapped
  // QK^T: B[k=d][n=q] → lane holds its q column's 8 contiguous d per step)
  const int my_qrow = q0 + wid * QBLK + qcol;   // row within the chunk
  const bool q_live = (wid * QBLK + qcol) < rows_here && my_qrow < q_len;
  const int my_qpos = q_live ? (ctx + my_qrow) : 0x3fffffff;  // OOB: never masked
  bf16x8_v qfrag[8];
  if (q_live) {
    const bf16_t* qrow =
        q + (int64_t)(row_base + my_qrow) * q_stride + (int64_t)head * D_HEAD;
#pragma unroll
    for (int kk = 0; kk < 8; ++kk)
      qfrag[kk] = *(const bf16x8_v*)(qrow + kk * 16 + khalf * 8);
  } else {
#pragma unroll
    for (int kk = 0; kk < 8; ++kk) qfrag[kk] = bf16x8_v{};
  }

  float m_run = -1e30f, l_run = 0.f;
  f32x16_v o0 = {}, o1 = {}, o2 = {}, o3 = {};

  const int k_end = min(S, ctx + q0 + rows_here);
  const int n_tiles = (k_end + KVBLK - 1) / KVBLK;

  auto stage_tile = [&](int tile_idx, int buf) {
    const int kv0 = tile_idx * KVBLK;
    const int kn0 = min(KVBLK, k_end - kv0);
    // 32x128 bf16 = 512 16-B pieces; 256 threads x 2
    for (int i = threadIdx.x; i < (KVBLK * D_HEAD) / 8; i += blockDim.x) {
      const int kk = i >> 4;           // key row (128 d / 8 = 16 pieces per row)
      const int dd = (i & 15) << 3;    // d offset
      const int j = kv0 + kk;
      BF16x8 v8, vv;
      if (kk < kn0) {
        const int64_t slot = (int64_t)bt[j / kv_block] * kv_block + j % kv_block;
        v8 = *(const BF16x8*)(k_cache + (slot * Hkv + kvh) * D_HEAD + dd);
        vv = *(const BF16x8*)(v_cache + (slot * Hkv + kvh) * D_HEAD + dd);
      } else {
        v8.u128 = ulonglong2{0, 0};
        vv.u128 = ulonglong2{0, 0};
      }
      // swizzled K store: 16-B unit index dd/8 XORed with row&7
      const int sw = (dd >> 3) ^ (kk & 7);
      *(BF16x8*)(&Kt[buf][kk][sw << 3]) = v8;
      // V^T scatter (8 x 2-B stores)
#pragma unroll
      for (int e = 0; e < 8; ++e) Vt[buf][dd + e][kk] = vv.h[e];
    }
  };

  __syncthreads();
  if (n_tiles > 0) stage_tile(0, 0);
  __syncthreads();
  int cur = 0;
  for (int t = 0; t < n_tiles; ++t) {
    const int kv = t * KVBLK;
    const int kn = min(KVBLK, k_end - kv);
    // stage the NEXT tile into the other buffer while computing this one
    if (t + 1 < n_tiles) stage_tile(t + 1, cur ^ 1);

    // QK^T: S_tile[k][q] = sum_d K[k][d] * Q[q][d]
    f32x16_v s = {};
#pragma unroll
    for (int kk = 0; kk < 8; ++kk) {
      // A-frag: lane holds K[row = lane&31][d = kk*16 + khalf*8 + e]
      const int unit = ((kk * 16 + khalf * 8) >> 3) ^ ((lane & 31) & 7);
      bf16x8_v a = *(const bf16x8_v*)(&Kt[cur][lane & 31][unit << 3]);
      s = __builtin_amdgcn_mfma_f32_32x32x16_bf16(a, qfrag[kk], s, 0, 0, 0);
    }

    // scale + causal mask; lane's 16 regs are k rows of its q column
    float p[16];
    float mt = -1e30f;
#pragma unroll
    for (int r = 0; r < 16; ++r) {
      const int krow = (r & 3) + 8 * (r >> 2) + 4 * khalf;
      float v = s[r] * scale;
      if (kv + krow > my_qpos || krow >= kn) v = -1e30f;
      p[r] = v;
      mt = fmaxf(mt, v);
    }
    mt = fmaxf(mt, __shfl_xor(mt, 32, 64));
    const float m_new = fmaxf(m_run, mt);
    const float alpha = __expf(m_run - m_new);
    m_run = m_new;
    float psum = 0.f;
#pragma unroll
    for (int r = 0; r < 16; ++r) {
      p[r] = __expf(p[r] - m_new);
      psum += p[r];
    }
    psum += __shfl_xor(psum, 32, 64);
    l_run = l_run * alpha + psum;

    // broadcast alpha (per q) to the O layout via LDS
    if (lane < 32) bcast[wid][lane] = alpha;
    // P → bf16 A-fragments: cvt_pk pairs + permlane32_swap (T12)
    // lo half (k 0..15): regs p0..p7; hi half (k 16..31): p8..p15
    int c01, c23, c45, c67, c89, cab, ccd, cef;
    asm("v_cvt_pk_bf16_f32 %0, %1, %2" : "=v"(c01) : "v"(p[0]), "v"(p[1]));
    asm("v_cvt_pk_bf16_f32 %0, %1, %2" : "=v"(c23) : "v"(p[2]), "v"(p[3]));
    asm("v_cvt_pk_bf16_f32 %0, %1, %2" : "=v"(c45) : "v"(p[4]), "v"(p[5]));
    asm("v_cvt_pk_bf16_f32 %0, %1, %2" : "=v"(c67) : "v"(p[6]), "v"(p[7]));
    asm("v_cvt_pk_bf16_f32 %0, %1, %2" : "=v"(c89) : "v"(p[8]), "v"(p[9]));
    asm("v_cvt_pk_bf16_f32 %0, %1, %2" : "=v"(cab) : "v"(p[10]), "v"(p[11]));
    asm("v_cvt_pk_bf16_f32 %0, %1, %2" : "=v"(ccd) : "v"(p[12]), "v"(p[13]));
    asm("v_cvt_pk_bf16_f32 %0, %1, %2" : "=v"(cef) : "v"(p[14]), "v"(p[15]));
    int2_v s1 = __builtin_amdgcn_permlane32_swap(c01, c45, false, false);
    int2_v s2 = __builtin_amdgcn_permlane32_swap(c23, c67, false, false);
    int2_v s3 = __builtin_amdgcn_permlane32_swap(c89, ccd, false, false);
    int2_v s4 = __builtin_amdgcn_permlane32_swap(cab, cef, false, false);
    int pa_lo_i[4] = {s1[0], s2[0], s1[1], s2[1]};   // A-frag k 0..15
    int pa_hi_i[4] = {s3[0], s4[0], s3[1], s4[1]};   // A-frag k 16..31
    bf16x8_v pa_lo = *(bf16x8_v*)pa_lo_i;
    bf16x8_v pa_hi = *(bf16x8_v*)pa_hi_i;

    // rescale O by alpha of each reg's q-row (LDS broadcast; wave-local)
    float al[16];
#pragma unroll
    for (int r = 0; r < 16; ++r) {
      const int qrow = (r & 3) + 8 * (r >> 2) + 4 * khalf;
      al[r] = bcast[wid][qrow];
    }
#pragma unroll
    for (int r = 0; r < 16; ++r) {
      o0[r] *= al[r];
      o1[r] *= al[r];
      o2[r] *= al[r];
      o3[r] *= al[r];
    }

    // PV: O_tile[q][d] += P[q][k] * V[k][d], d-tiles of 32
    // B-frag: lane holds V[k = kh*16 + khalf*8 + e][d = dtile*32 + (lane&31)]
    //         = Vt[dtile*32 + (lane&31)][k…] — 16 B contiguous
#pragma unroll
    for (int dt = 0; dt < 4; ++dt) {
      const bf16_t* vrow = &Vt[cur][dt * 32 + (lane & 31)][khalf * 8];
      bf16x8_v b_lo = *(const bf16x8_v*)(vrow);
      bf16x8_v b_hi = *(const bf16x8_v*)(vrow + 16);
      f32x16_v* od = dt == 0 ? &o0 : dt == 1 ? &o1 : dt == 2 ? &o2 : &o3;
      *od = __builtin_amdgcn_mfma_f32_32x32x16_bf16(pa_lo, b_lo, *od, 0, 0, 0);
      *od = __builtin_amdgcn_mfma_f32_32x32x16_bf16(pa_hi, b_hi, *od, 0, 0, 0);
    }
    __syncthreads();  // staging of t+1 done AND everyone done reading cur
    cur ^= 1;
  }

  // epilogue: normalize by 1/l_run per q (broadcast through LDS), store
  __syncthreads();
  if (lane < 32) bcast[wid][lane] = (l_run > 0.f) ? 1.f / l_run : 0.f;
  float inv[16];
#pragma unroll
  for (int r = 0; r < 16; ++r) {
    const int qrow = (r & 3) + 8 * (r >> 2) + 4 * khalf;
    inv[r] = bcast[wid][qrow];
  }
  const int dcol = lane & 31;
#pragma unroll
  for (int r = 0; r < 16; ++r) {
    const int qrow_w = (r & 3) + 8 * (r >> 2) + 4 * khalf;  // within wave
    const int qrow_c = q0 + wid * QBLK + qrow_w;            // within chunk
    if (wid * QBLK + qrow_w >= rows_here) continue;
    bf16_t* orow = out + ((int64_t)(row_base + qrow_c) * Hq + head) * D_HEAD;
    orow[0 * 32 + dcol] = f2bf(o0[r] * inv[r]);
    orow[1 * 32 + dcol] = f2bf(o1[r] * inv[r]);
    orow[2 * 32 + dcol] = f2bf(o2[r] * inv[r]);
    orow[3 * 32 + dcol] = f2bf(o3[r] * inv[r]);
  }
}

extern "C" void launch_prefill_attn_mfma(
    void* out, const void* q, const void* k_cache, const void* v_cache,
    const int* block_tables, const int* seq_lens, const int* ctx_lens,
    const int* row_starts, const int* tile_seq, const int* tile_q0,
    float scale, int num_tiles, int Hq, int Hkv, int D, int max_blocks,
    int kv_block, int64_t q_stride, hipStream_t stream) {
  if (D != D_HEAD) return;
  dim3 grid(num_tiles, Hq), block(256);
  hipLaunchKernelGGL(prefill_attn_mfma_kernel, grid, block, 0, stream,
                     (bf16_t*)out, (const bf16_t*)q, (const bf16_t*)k_cache,
                     (const bf16_t*)v_cache, block_tables, seq_lens,
                     ctx_lens, row_starts, tile_seq, tile_q0, scale, Hq,
                     Hkv, max_blocks, kv_block, q_stride);
}
