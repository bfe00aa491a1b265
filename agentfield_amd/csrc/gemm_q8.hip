// Block-quadrant 8-phase 256x256 bf16 MFMA GEMM (the guide's verified
// deep-pipelined structure, §5 "256² 8-phase template"):
//   C[M,N] = A[M,K] @ W[N,K]^T
//
// Geometry: BM=BN=256, BK=64.  Operand halves are the staging granule:
//   A0/A1 = M-halves [128 rows x 64 K] (16 KiB), B0/B1 = N-halves.
//   LDS ring: 8 granule slots x 16 KiB = 128 KiB.
//
// Each K-tile runs FOUR phases, one 128x128 block-QUADRANT each, in
// gray order (0,0) (0,1) (1,1) (1,0) so each phase changes ONE operand
// half; all 8 waves work the same quadrant (64x32 slice each -> 16
// MFMA/phase/wave).  Fragments persist in registers across phases:
// phase reads are [A0+B0, B1, A1, none] = 24 x ds_read_b128 per tile.
//
// Stage schedule (one granule per phase, two tiles of lookahead):
//   phase 0 of tile t: stage A1(t+1);  phase 1: A0(t+2);
//   phase 2: B0(t+2);  phase 3: B1(t+2)
// Every granule is staged >= 6 stage-slots before its first LDS read, so
// a per-phase `s_waitcnt vmcnt(8)` (4 granules allowed in flight, never
// drained to 0 — T3+T4) plus the phase barrier guarantees residency; and
// every slot's last LDS read is >= 1 phase before the granule that
// overwrites it is issued, so DMA writes can never race a pending read.
//
// Per phase: { ds_read frags | stage 1 granule | vmcnt(8) | barrier |
//              setprio(1) 16 MFMA setprio(0) | barrier }.
#include "common.h"
#include <cstdlib>

#define GQ_BM 256
#define GQ_BK 64
#define GQ_GRAN (128 * 64)  // u16 elems = 16 KiB

typedef __attribute__((ext_vector_type(8))) __bf16 bf16x8q;

__device__ __forceinline__ bf16x8q as_bfq(s16x8 v) {
  union { s16x8 s; bf16x8q b; } u;
  u.s = v;
  return u.b;
}

// Stage one granule: 128 rows x 64 K bf16 from global (row-major, ld
// elements) into a linear LDS slot; 2 x 16 B per thread (512 threads).
// Read-side XOR swizzle (byte_in_row ^ ((row&7)<<4)) pre-applied to the
// global source (ERRATA #21: both-sides involution, linear LDS dest).
__device__ __forceinline__ void gq_stage(const u16* __restrict__ g,
                                         size_t ld, u16* lds, int row0,
                                         int k0, int max_row) {
  const int tid = threadIdx.x;
#pragma unroll
  for (int it = 0; it < 2; ++it) {
    const int e = (it * 512 + tid) * 8;  // element offset in granule
    const int row = e >> 6;              // /64 elements per row
    const int wb = (e & 63) * 2;         // byte within 128-B row
    const int wsw = wb ^ ((row & 7) << 4);
    const u16* src = g + (size_t)min(row0 + row, max_row - 1) * ld + k0 +
                     (wsw >> 1);
    u16* dst = lds + (size_t)(it * 512 + (tid & ~63)) * 8;
    __builtin_amdgcn_global_load_lds(
        (const __attribute__((address_space(1))) uint32_t*)src,
        (__attribute__((address_space(3))) uint32_t*)dst, 16, 0, 0);
  }
}

// byte offset of fragment [row][k-slab s + kk*32] in a granule
__device__ __forceinline__ int gq_off(int row, int kk, int s) {
  return row * 128 + ((kk * 64 + s * 16) ^ ((row & 7) << 4));
}

__device__ __forceinline__ s16x8 gq_read(const u16* lds, int off) {
  return *reinterpret_cast<const s16x8*>(
      reinterpret_cast<const char*>(lds) + off);
}

// AF_GEMM_WAIT selects the per-phase counted wait depth (A/B lever):
// vmcnt(N) admits N outstanding global_load_lds per thread; 8 = 4 granules.
#if !defined(GQ_WAITN)
#define GQ_WAITN 8
#endif
#define GQ_STR2(x) #x
#define GQ_STR(x) GQ_STR2(x)
#define GQ_WAIT8   asm volatile("s_waitcnt vmcnt(" GQ_STR(GQ_WAITN) ")" ::: "memory")

__global__ void __launch_bounds__(512, 1) gemm_bf16_q8_kernel(
    u16* __restrict__ C, const u16* __restrict__ A, const u16* __restrict__ W,
    int M, int N, int K, int tiles_m, int tiles_n, int order) {
  __shared__ u16 ring[8][GQ_GRAN];

  int tm, tn;
  xcd_tile_map(blockIdx.x, tiles_m, tiles_n, order, &tm, &tn);
  const int m0 = tm * GQ_BM, n0 = tn * GQ_BM;
  const int wid = threadIdx.x >> 6, lane = threadIdx.x & 63;
  // 8 waves tile each 128x128 quadrant as 2x4 of 64x32 slices
  const int wqm = (wid >> 2) * 64, wqn = (wid & 3) * 32;
  const int fr = lane & 15, s = lane >> 4;

  // frag byte offsets inside a granule (row-in-half, kk, slab)
  int offA[4][2], offB[2][2];
#pragma unroll
  for (int i = 0; i < 4; ++i)
#pragma unroll
    for (int kk = 0; kk < 2; ++kk)
      offA[i][kk] = gq_off(wqm + i * 16 + fr, kk, s);
#pragma unroll
  for (int j = 0; j < 2; ++j)
#pragma unroll
    for (int kk = 0; kk < 2; ++kk)
      offB[j][kk] = gq_off(wqn + j * 16 + fr, kk, s);

  // acc[quadrant][i][j]: quadrants in gray order (0,0)(0,1)(1,1)(1,0)
  f32x4 acc[4][4][2];
#pragma unroll
  for (int q = 0; q < 4; ++q)
#pragma unroll
    for (int i = 0; i < 4; ++i)
#pragma unroll
      for (int j = 0; j < 2; ++j) acc[q][i][j] = f32x4{0, 0, 0, 0};

  const int ntiles = K / GQ_BK;
  // granule sequence: g = 4t + {0:A0, 1:B0, 2:B1, 3:A1}; slot = g & 7
  // prologue: stage A0,B0,B1,A1 of tile 0 and A0,B0,B1 of tile 1
  gq_stage(A, K, ring[0], m0, 0, M);
  gq_stage(W, K, ring[1], n0, 0, N);
  gq_stage(W, K, ring[2], n0 + 128, 0, N);
  gq_stage(A, K, ring[3], m0 + 128, 0, M);
  gq_stage(A, K, ring[4], m0, GQ_BK, M);
  gq_stage(W, K, ring[5], n0, GQ_BK, N);
  gq_stage(W, K, ring[6], n0 + 128, GQ_BK, N);

  s16x8 af[4][2], b0[2][2], b1[2][2];

  // phase 0 of tile 0 reads A0(0)/B0(0) BEFORE its own wait point: the
  // prologue's 7 staged granules must have the two oldest retired first
  asm volatile("s_waitcnt vmcnt(10)" ::: "memory");
  __builtin_amdgcn_s_barrier();

  for (int t = 0; t < ntiles; ++t) {
    const int k2 = (t + 2) * GQ_BK;  // staged K offset (tiles t+1/t+2)
    const bool deep = t + 2 < ntiles;    // full pipeline
    const bool mid = t + 1 < ntiles;     // A1(t+1) still needed
    const u16* sA0 = ring[(4 * t + 0) & 7];
    const u16* sB0 = ring[(4 * t + 1) & 7];
    const u16* sB1 = ring[(4 * t + 2) & 7];
    const u16* sA1 = ring[(4 * t + 3) & 7];

    // -- phase 0: quadrant (A0,B0); stage A1(t+1) --------------------
#pragma unroll
    for (int i = 0; i < 4; ++i)
#pragma unroll
      for (int kk = 0; kk < 2; ++kk) af[i][kk] = gq_read(sA0, offA[i][kk]);
#pragma unroll
    for (int j = 0; j < 2; ++j)
#pragma unroll
      for (int kk = 0; kk < 2; ++kk) b0[j][kk] = gq_read(sB0, offB[j][kk]);
    if (mid) gq_stage(A, K, ring[(4 * t + 7) & 7], m0 + 128, k2 - GQ_BK, M);
    if (deep) { GQ_WAIT8; } else {
      asm volatile("s_waitcnt vmcnt(0)" ::: "memory");
    }
    __builtin_amdgcn_s_barrier();
    __builtin_amdgcn_s_setprio(1);
#pragma unroll
    for (int i = 0; i < 4; ++i)
#pragma unroll
      for (int j = 0; j < 2; ++j)
#pragma unroll
        for (int kk = 0; kk < 2; ++kk)
          acc[0][i][j] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
              as_bfq(af[i][kk]), as_bfq(b0[j][kk]), acc[0][i][j], 0, 0, 0);
    __builtin_amdgcn_s_setprio(0);
    __builtin_amdgcn_s_barrier();

    // -- phase 1: quadrant (A0,B1); stage A0(t+2) --------------------
#pragma unroll
    for (int j = 0; j < 2; ++j)
#pragma unroll
      for (int kk = 0; kk < 2; ++kk) b1[j][kk] = gq_read(sB1, offB[j][kk]);
    if (deep) {
      gq_stage(A, K, ring[(4 * t + 8) & 7], m0, k2, M);
      GQ_WAIT8;
    }
    __builtin_amdgcn_s_barrier();
    __builtin_amdgcn_s_setprio(1);
#pragma unroll
    for (int i = 0; i < 4; ++i)
#pragma unroll
      for (int j = 0; j < 2; ++j)
#pragma unroll
        for (int kk = 0; kk < 2; ++kk)
          acc[1][i][j] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
              as_bfq(af[i][kk]), as_bfq(b1[j][kk]), acc[1][i][j], 0, 0, 0);
    __builtin_amdgcn_s_setprio(0);
    __builtin_amdgcn_s_barrier();

    // -- phase 2: quadrant (A1,B1); stage B0(t+2) --------------------
#pragma unroll
    for (int i = 0; i < 4; ++i)
#pragma unroll
      for (int kk = 0; kk < 2; ++kk) af[i][kk] = gq_read(sA1, offA[i][kk]);
    if (deep) {
      gq_stage(W, K, ring[(4 * t + 9) & 7], n0, k2, N);
      GQ_WAIT8;
    }
    __builtin_amdgcn_s_barrier();
    __builtin_amdgcn_s_setprio(1);
#pragma unroll
    for (int i = 0; i < 4; ++i)
#pragma unroll
      for (int j = 0; j < 2; ++j)
#pragma unroll
        for (int kk = 0; kk < 2; ++kk)
          acc[2][i][j] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
              as_bfq(af[i][kk]), as_bfq(b1[j][kk]), acc[2][i][j], 0, 0, 0);
    __builtin_amdgcn_s_setprio(0);
    __builtin_amdgcn_s_barrier();

    // -- phase 3: quadrant (A1,B0); stage B1(t+2) --------------------
    if (deep) {
      gq_stage(W, K, ring[(4 * t + 10) & 7], n0 + 128, k2, N);
      GQ_WAIT8;
    }
    __builtin_amdgcn_s_barrier();
    __builtin_amdgcn_s_setprio(1);
#pragma unroll
    for (int i = 0; i < 4; ++i)
#pragma unroll
      for (int j = 0; j < 2; ++j)
#pragma unroll
        for (int kk = 0; kk < 2; ++kk)
          acc[3][i][j] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
              as_bfq(af[i][kk]), as_bfq(b0[j][kk]), acc[3][i][j], 0, 0, 0);
    __builtin_amdgcn_s_setprio(0);
    __builtin_amdgcn_s_barrier();
  }

  // epilogue: quadrant q -> (ha, hb) = gray[(0,0),(0,1),(1,1),(1,0)]
  const int HA[4] = {0, 0, 1, 1}, HB[4] = {0, 1, 1, 0};
#pragma unroll
  for (int q = 0; q < 4; ++q) {
#pragma unroll
    for (int i = 0; i < 4; ++i) {
      const int mrow = m0 + HA[q] * 128 + wqm + i * 16 + (lane >> 4) * 4;
#pragma unroll
      for (int j = 0; j < 2; ++j) {
        const int ncol = n0 + HB[q] * 128 + wqn + j * 16 + (lane & 15);
        if (ncol >= N) continue;
#pragma unroll
        for (int r = 0; r < 4; ++r) {
          if (mrow + r >= M) continue;
          C[(size_t)(mrow + r) * N + ncol] = f2bf(acc[q][i][j][r]);
        }
      }
    }
  }
}

AF_EXPORT int af_gemm_bf16_q8(void* C, const void* A, const void* W,
                              int M, int N, int K, void* stream) {
  if (K % GQ_BK || K < 2 * GQ_BK) return 9004;
  if (M == 0) return 0;
  const int tiles_m = (M + GQ_BM - 1) / GQ_BM;
  const int tiles_n = (N + GQ_BM - 1) / GQ_BM;
  static int order = -1;  // AF_GEMM_ORDER: 1 = row-chunk, 2 = 8x4 supertile
  if (order < 0) {
    const char* e = getenv("AF_GEMM_ORDER");
    order = e ? atoi(e) : 2;
  }
  gemm_bf16_q8_kernel<<<tiles_m * tiles_n, 512, 0, (hipStream_t)stream>>>(
      (u16*)C, (const u16*)A, (const u16*)W, M, N, K, tiles_m, tiles_n,
      order);
  return af_last_err();
}
