// Hand-written bf16 GEMM for the prefill projections (gfx950) — the
// 256²-tile 8-phase schedule from the CDNA4 guide (§5 template / T1-T5).
//
// C[M,N] = A[M,K] · B[N,K]^T   (torch Linear layout: B is [out, in]),
// bf16 in, fp32 MFMA accumulation, bf16 out.  Replaces hipBLASLt for the
// large-M prefill projections (53% of engine GPU time in profiles/r01 at
// ~1.0-1.1 PF effective): the 8-phase glds pipeline + LDS swizzle targets
// the guide's measured 1.3+ PF regime on the same shapes.
//
// Geometry
//   tile 256(M) × 256(N), K consumed in 32-wide chunks
//   512 threads = 8 waves as 2(M) × 4(N); wave tile 128×64
//   per wave: 4×2 fragments of v_mfma_f32_32x32x16_bf16 (layouts are the
//     probe-verified maps, csrc/mfma_probe.hip)
//   LDS: ring of 8 panels × 16 KiB (128 KiB dynamic).  A panel is one
//     k-chunk of A or B: [256 rows][32 k] bf16, staged by
//     global_load_lds_dwordx4 (2 per wave), XOR-swizzled on the SOURCE
//     address (16-B slot ^= (row>>2)&3) so ds_read_b128 fragment reads
//     are bank-conflict-free while the glds LDS image stays lane-linear
//   schedule: phase = (k-chunk, m-half); one panel staged per phase at
//     distance 6 ahead of its consumption; counted s_waitcnt vmcnt(8) at
//     even phases only (never 0 in the loop); raw s_barrier (no vmcnt
//     drain) + lgkmcnt(0); s_setprio(1) around the 8-MFMA cluster
//   XCD-aware bijective workgroup remap (8 XCDs, private L2s) so
//     consecutive tiles share the B (weight) panel within one XCD
#include "common.h"

typedef __bf16 bf16x8_v __attribute__((ext_vector_type(8)));
typedef float f32x16_v __attribute__((ext_vector_type(16)));

#define BM 256
#define BN 256
#define PANEL_BYTES 16384  // 256 rows x 32 k x 2 B
#define NSLOTS 8

typedef __attribute__((address_space(3))) char lds_char;
typedef __attribute__((address_space(1))) const char glob_char;

// stage one panel (k-chunk `chunk` of A or B) into LDS slot `slot`.
// Every thread issues 2 glds; per wave that is 2 vmcnt units.
// src rows are clamped to rows-1 (ragged M): clamped rows hold duplicate
// data and are never read by a guarded consumer.
static DEV void stage_panel(const bf16_t* __restrict__ src, int64_t ld,
                            int rows, int row0, int chunk, char* lds,
                            int slot) {
  const int lane = threadIdx.x & 63;
  const int wid = threadIdx.x >> 6;
#pragma unroll
  for (int i = 0; i < 2; ++i) {
    const int off = wid * 2048 + i * 1024;           // wave-uniform LDS byte
    const int row = (off >> 6) + (lane >> 2);        // panel row of this lane
    const int logcol = (lane & 3) ^ ((row >> 2) & 3);  // source pre-swizzle
    const int grow = min(row0 + row, rows - 1);
    const bf16_t* g = src + (int64_t)grow * ld + chunk * 32 + logcol * 8;
    __builtin_amdgcn_global_load_lds((glob_char*)g,
                                     (lds_char*)(lds + slot * PANEL_BYTES + off),
                                     16, 0, 0);
  }
}

#define RAW_BARRIER() __builtin_amdgcn_s_barrier()
#define WAIT_VM(n) asm volatile("s_waitcnt vmcnt(" #n ")" ::: "memory")
#define WAIT_LGKM0() asm volatile("s_waitcnt lgkmcnt(0)" ::: "memory")

__global__ __launch_bounds__(512, 1) void gemm_bf16_kernel(
    bf16_t* __restrict__ C, const bf16_t* __restrict__ A,
    const bf16_t* __restrict__ B, const int M, const int N, const int K,
    const int tiles_m) {
  extern __shared__ char lds[];

  // bijective XCD remap: each XCD gets a contiguous chunk of the grid
  const int nwg = gridDim.x;
  const int q = nwg >> 3, r = nwg & 7;
  const int xcd = blockIdx.x & 7, idx = blockIdx.x >> 3;
  const int wg = (xcd < r ? xcd * (q + 1) : r * (q + 1) + (xcd - r) * q) + idx;
  // 8(m)x4(n) supertile rasterization: concurrent blocks share 8 A-tiles +
  // 4 B-tiles (k-slab fronts stay L2-resident) instead of sweeping all of
  // M per N-column, which re-streams A from HBM tiles_n times (ragged
  // edges handled by clamped band/column widths; bijective)
  const int tiles_n = nwg / tiles_m;
  const int per_sc = tiles_m * 4;          // tiles per 4-wide super-column
  const int sc = wg / per_sc;
  const int rsc = wg % per_sc;
  const int h = min(4, tiles_n - sc * 4);  // n-width of this super-column
  const int band = 8 * h;                  // tiles per 8-high m-band
  const int b = rsc / band, rb = rsc % band;
  const int w = min(8, tiles_m - b * 8);   // m-height of this band
  const int tm0 = (b * 8 + rb % w) * BM;
  const int tn0 = (sc * 4 + rb / w) * BN;

  const int lane = threadIdx.x & 63;
  const int wid = threadIdx.x >> 6;
  const int wm = wid >> 2, wn = wid & 3;  // wave grid 2(M) x 4(N)
  const int laneM = lane & 31;            // fragment row within a 32-row frag
  const int khalf = lane >> 5;            // fragment k-half (8 of 16)

  // per-lane LDS read offsets (swizzled), invariant over the K loop:
  //   A frag (f, s): row = wm*128 + f*32 + laneM, 16-B slot = (2s+khalf)^xor
  const int rowA = wm * 128 + laneM;
  const int rowB = wn * 64 + laneM;
  const int xorA = (rowA >> 2) & 3, xorB = (rowB >> 2) & 3;
  int offA[2], offB[2];  // [s] — f/g add 32 rows = 2048 B
#pragma unroll
  for (int s = 0; s < 2; ++s) {
    offA[s] = rowA * 64 + (((s << 1) | khalf) ^ xorA) * 16;
    offB[s] = rowB * 64 + (((s << 1) | khalf) ^ xorB) * 16;
  }

  const int nchunks = K >> 5;           // 32-wide k-chunks
  const int nphases = nchunks << 1;     // (chunk, m-half) pairs; K%128==0 → %8==0

  // prologue: 6 panels in flight (panels 0..5 = chunks 0,1,2 interleaved A,B)
#pragma unroll
  for (int p = 0; p < 6; ++p) {
    const int chunk = p >> 1;
    if (p & 1)
      stage_panel(B, K, N, tn0, chunk, lds, p);
    else
      stage_panel(A, K, M, tm0, chunk, lds, p);
  }
  // confirm panels 0,1 (leave 4 panels = 8 loads in flight).  vmcnt is
  // per-wave, so the wait must be sealed by a barrier BEFORE any wave
  // reads the panel — each wave only stages its own 2 KiB share.
  WAIT_VM(8);
  RAW_BARRIER();

  f32x16_v acc[4][2] = {};

  for (int ph0 = 0; ph0 < nphases; ph0 += 8) {
#pragma unroll
    for (int j = 0; j < 8; ++j) {
      const int phase = ph0 + j;
      const int chunk = phase >> 1;     // consumed this phase
      const int h = phase & 1;          // m-half: fragments {2h, 2h+1}
      const int sA = (2 * (j >> 1)) & 7, sB = sA + 1;  // consume slots
      // ds_read the phase's register subtile: 4 A + 4 B ds_read_b128
      bf16x8_v af[2][2], bf[2][2];  // [f'][s], [g][s]
#pragma unroll
      for (int s = 0; s < 2; ++s) {
#pragma unroll
        for (int f = 0; f < 2; ++f)
          af[f][s] = *(const bf16x8_v*)(lds + sA * PANEL_BYTES + offA[s] +
                                        (2 * h + f) * 2048);
#pragma unroll
        for (int g = 0; g < 2; ++g)
          bf[g][s] = *(const bf16x8_v*)(lds + sB * PANEL_BYTES + offB[s] +
                                        g * 2048);
      }

      // stage the panel consumed 6 phases from now (chunk clamped at the
      // tail: the extra panels land in dead slots and are never read)
      {
        const int p = phase + 6;  // panel index; slot = index mod 8
        const int pchunk = min(p >> 1, nchunks - 1);
        if (p & 1)
          stage_panel(B, K, N, tn0, pchunk, lds, (j + 6) & 7);
        else
          stage_panel(A, K, M, tm0, pchunk, lds, (j + 6) & 7);
      }

      RAW_BARRIER();
      WAIT_LGKM0();
      __builtin_amdgcn_s_setprio(1);
#pragma unroll
      for (int s = 0; s < 2; ++s)
#pragma unroll
        for (int f = 0; f < 2; ++f)
#pragma unroll
          for (int g = 0; g < 2; ++g)
            acc[2 * h + f][g] = __builtin_amdgcn_mfma_f32_32x32x16_bf16(
                af[f][s], bf[g][s], acc[2 * h + f][g], 0, 0, 0);
      __builtin_amdgcn_s_setprio(0);
      // at odd phases confirm the two panels the NEXT two phases read
      // (staged 6/5 phases ago): counted — leave 4 panels = 8 loads in
      // flight, never drain to 0 in the loop (T4).  The closing barrier
      // turns the per-wave wait into a workgroup-wide guarantee.
      if (j & 1) WAIT_VM(8);
      RAW_BARRIER();
    }
  }

  // epilogue: D layout (probe-verified) row = (reg&3) + 8*(reg>>2) + 4*khalf,
  // col = laneM; adjacent lanes write adjacent columns (coalesced rows)
#pragma unroll
  for (int f = 0; f < 4; ++f)
#pragma unroll
    for (int g = 0; g < 2; ++g) {
      const int col = tn0 + wn * 64 + g * 32 + laneM;
#pragma unroll
      for (int reg = 0; reg < 16; ++reg) {
        const int row =
            tm0 + wm * 128 + f * 32 + (reg & 3) + 8 * (reg >> 2) + 4 * khalf;
        if (row < M) C[(int64_t)row * N + col] = f2bf(acc[f][g][reg]);
      }
    }
}

extern "C" void launch_gemm_bf16(void* C, const void* A, const void* B, int M,
                                 int N, int K, hipStream_t stream) {
  static bool attr_set = false;
  if (!attr_set) {
    (void)hipFuncSetAttribute((const void*)gemm_bf16_kernel,
                              hipFuncAttributeMaxDynamicSharedMemorySize,
                              NSLOTS * PANEL_BYTES);
    attr_set = true;
  }
  const int tiles_m = (M + BM - 1) / BM;
  const int tiles_n = N / BN;
  dim3 grid(tiles_m * tiles_n);
  gemm_bf16_kernel<<<grid, 512, NSLOTS * PANEL_BYTES, stream>>>(
      (bf16_t*)C, (const bf16_t*)A, (const bf16_t*)B, M, N, K, tiles_m);
}
