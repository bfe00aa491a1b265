// Block-quadrant 8-phase 256x256 MX-fp8 GEMM on the CDNA4 scaled MFMA
// (mfma_scale_f32_16x16x128_f8f6f4 — hardware-fused dequant+matmul at
// 2x the bf16 MFMA rate):
//   C[M,N] (bf16) = dequant(A8,SA)[M,K] @ dequant(W8,SW)[N,K]^T
//
// Quantization: OCP e4m3 bytes + E8M0 scale per 32-element K-block
// (agentfield_amd/quant).  Operand layouts were mapped EMPIRICALLY on
// gfx950 (tools/mx_probe*.py, csrc/probe.hip):
//   * scale lane (row, s = lane>>4) byte0 (op_sel 0) covers hw K-block s
//   * data lane (row, g) byte j -> hw k = 32*(2*(j>>4)+(g>>1))
//       + 16*(g&1) + (j&15)
//   so a lane's 32 data bytes are two b128 reads at row-byte offsets
//   16g and 64+16g (the same fragment offsets as the bf16 q8 kernel),
//   and memory-k == hw-k so per-32-byte-block scales line up.
//
// Geometry mirrors gemm_q8.hip (512 threads, granule = operand half
// [128 rows x 128 B] = 16 KiB, 8-slot LDS ring, 4 gray-order quadrant
// phases per K-tile with REGISTER-PERSISTENT fragments, counted vmcnt,
// setprio MFMA clusters) with BK = 128 bytes per K-tile (one MFMA
// pass) and 8 MFMA/phase/wave.  A 4-wave/512-VGPR restructure (64x64
// slices at 1 wave/SIMD) was measured at 373-463 TF — half this
// kernel's rate: single-wave SIMDs cannot hide the ds_read/barrier
// latency that co-resident wave pairs cover.
//
// Scales ride in each slot's tail: 512 bytes per granule staged as ONE
// size-1 global_load_lds per thread.  Sub-dword loads land each lane's
// byte in its own LDS DWORD slot (measured, tools/lds_stride.py), so
// the region is dword-pitched: byte for (row, blk) at
// GX_GRAN + (row*4+blk)*4.  3 counted loads per granule per thread:
// in-loop wait vmcnt(12) (4 granules in flight), prologue drains to
// 15 = (7-2)*3.
#include "common.h"
#include <cstdlib>

#define GX_BM 256
#define GX_BK 128
#define GX_GRAN (128 * 128)        // data bytes
#define GX_SLOT (GX_GRAN + 2048)   // + dword-pitched scale region

typedef __attribute__((ext_vector_type(8))) int i32x8x;

union gx_frag {
  struct { s16x8 lo, hi; } h;
  i32x8x v;
};

// Stage one granule: 128 rows x 128 K-bytes e4m3 + scales.  Data:
// 2 x 16 B per thread with the read-side XOR swizzle pre-applied to
// the source (both-sides involution, linear LDS dest).
__device__ __forceinline__ void gx_stage(const unsigned char* __restrict__ g,
                                         const unsigned char* __restrict__ sc,
                                         size_t ld, unsigned char* lds,
                                         int row0, int kt, int max_row) {
  const int tid = threadIdx.x;
#pragma unroll
  for (int it = 0; it < 2; ++it) {
    const int byte = (it * 512 + tid) * 16;  // byte offset in granule
    const int row = byte >> 7;
    const int wb = byte & 127;
    const int wsw = wb ^ ((row & 7) << 4);
    const unsigned char* src =
        g + (size_t)min(row0 + row, max_row - 1) * ld + kt * GX_BK + wsw;
    unsigned char* dst = lds + (size_t)(it * 512 + (tid & ~63)) * 16;
    __builtin_amdgcn_global_load_lds(
        (const __attribute__((address_space(1))) uint32_t*)src,
        (__attribute__((address_space(3))) uint32_t*)dst, 16, 0, 0);
  }
  {
    const int row = tid >> 2, c = tid & 3;  // scale byte tid = row*4+blk
    const unsigned char* src =
        sc + (size_t)min(row0 + row, max_row - 1) * (ld >> 5) + kt * 4 + c;
    unsigned char* dst = lds + GX_GRAN + (size_t)(tid & ~63) * 4;
    __builtin_amdgcn_global_load_lds(
        (const __attribute__((address_space(1))) uint32_t*)src,
        (__attribute__((address_space(3))) uint32_t*)dst, 1, 0, 0);
  }
}

// byte offset of data fragment [row][half h, slab s] in a granule
__device__ __forceinline__ int gx_off(int row, int h, int s) {
  return row * 128 + ((h * 64 + s * 16) ^ ((row & 7) << 4));
}

// scale byte for (row, k-block s): dword-pitched tail region
__device__ __forceinline__ int gx_soff(int row, int s) {
  return GX_GRAN + (row * 4 + s) * 4;
}

__device__ __forceinline__ s16x8 gx_read(const unsigned char* lds, int off) {
  return *reinterpret_cast<const s16x8*>(lds + off);
}

#define GX_WAIT   asm volatile("s_waitcnt vmcnt(12)" ::: "memory")

#define GX_MFMA(q, BF, SB)                                                    \
  do {                                                                        \
    _Pragma("unroll")                                                         \
    for (int i = 0; i < 4; ++i) {                                             \
      _Pragma("unroll")                                                       \
      for (int j = 0; j < 2; ++j)                                             \
        acc[q][i][j] = __builtin_amdgcn_mfma_scale_f32_16x16x128_f8f6f4(      \
            af[i].v, BF[j].v, acc[q][i][j], 0, 0, 0, sa[i], 0, SB[j]);        \
    }                                                                         \
  } while (0)

__global__ void __launch_bounds__(512, 1) gemm_mxfp8_kernel(
    u16* __restrict__ C, const unsigned char* __restrict__ A,
    const unsigned char* __restrict__ SA, const unsigned char* __restrict__ W,
    const unsigned char* __restrict__ SW, int M, int N, int K, int tiles_m,
    int tiles_n, int order) {
  __shared__ unsigned char ring[8][GX_SLOT];

  int tm, tn;
  xcd_tile_map(blockIdx.x, tiles_m, tiles_n, order, &tm, &tn);
  const int m0 = tm * GX_BM, n0 = tn * GX_BM;
  const int wid = threadIdx.x >> 6, lane = threadIdx.x & 63;
  // 8 waves tile each 128x128 quadrant as 2x4 of 64x32 slices
  const int fr = lane & 15, s = lane >> 4;
  const int rowA = (wid >> 2) * 64 + fr, rowB = (wid & 3) * 32 + fr;
  // cached fragment/scale offsets (A/B: caching beat inline recompute,
  // 28 vs 43 spilled VGPRs)
  int offA[4][2], offB[2][2], scA[4], scB[2];
#pragma unroll
  for (int i = 0; i < 4; ++i) {
#pragma unroll
    for (int h = 0; h < 2; ++h) offA[i][h] = gx_off(rowA + i * 16, h, s);
    scA[i] = gx_soff(rowA + i * 16, s);
  }
#pragma unroll
  for (int j = 0; j < 2; ++j) {
#pragma unroll
    for (int h = 0; h < 2; ++h) offB[j][h] = gx_off(rowB + j * 16, h, s);
    scB[j] = gx_soff(rowB + j * 16, s);
  }

  // acc[quadrant][i][j]: quadrants in gray order (0,0)(0,1)(1,1)(1,0)
  f32x4 acc[4][4][2];
#pragma unroll
  for (int q = 0; q < 4; ++q)
#pragma unroll
    for (int i = 0; i < 4; ++i)
#pragma unroll
      for (int j = 0; j < 2; ++j) acc[q][i][j] = f32x4{0, 0, 0, 0};

  const int ntiles = K / GX_BK;
  // granule sequence: g = 4t + {0:A0, 1:B0, 2:B1, 3:A1}; slot = g & 7
  gx_stage(A, SA, K, ring[0], m0, 0, M);
  gx_stage(W, SW, K, ring[1], n0, 0, N);
  gx_stage(W, SW, K, ring[2], n0 + 128, 0, N);
  gx_stage(A, SA, K, ring[3], m0 + 128, 0, M);
  gx_stage(A, SA, K, ring[4], m0, 1, M);
  gx_stage(W, SW, K, ring[5], n0, 1, N);
  gx_stage(W, SW, K, ring[6], n0 + 128, 1, N);

  gx_frag af[4], b0[2], b1[2];
  int sa[4], sb0[2], sb1[2];

  asm volatile("s_waitcnt vmcnt(15)" ::: "memory");
  __builtin_amdgcn_s_barrier();

  for (int t = 0; t < ntiles; ++t) {
    const bool deep = t + 2 < ntiles;
    const bool mid = t + 1 < ntiles;
    const unsigned char* sA0 = ring[(4 * t + 0) & 7];
    const unsigned char* sB0 = ring[(4 * t + 1) & 7];
    const unsigned char* sB1 = ring[(4 * t + 2) & 7];
    const unsigned char* sA1 = ring[(4 * t + 3) & 7];

    // -- phase 0: quadrant (A0,B0); stage A1(t+1) --------------------
#pragma unroll
    for (int i = 0; i < 4; ++i) {
      af[i].h.lo = gx_read(sA0, offA[i][0]);
      af[i].h.hi = gx_read(sA0, offA[i][1]);
      sa[i] = sA0[scA[i]];
    }
#pragma unroll
    for (int j = 0; j < 2; ++j) {
      b0[j].h.lo = gx_read(sB0, offB[j][0]);
      b0[j].h.hi = gx_read(sB0, offB[j][1]);
      sb0[j] = sB0[scB[j]];
    }
    if (mid) gx_stage(A, SA, K, ring[(4 * t + 7) & 7], m0 + 128, t + 1, M);
    if (deep) { GX_WAIT; } else {
      asm volatile("s_waitcnt vmcnt(0)" ::: "memory");
    }
    __builtin_amdgcn_s_barrier();
    __builtin_amdgcn_s_setprio(1);
    GX_MFMA(0, b0, sb0);
    __builtin_amdgcn_s_setprio(0);
    __builtin_amdgcn_s_barrier();

    // -- phase 1: quadrant (A0,B1); stage A0(t+2) --------------------
#pragma unroll
    for (int j = 0; j < 2; ++j) {
      b1[j].h.lo = gx_read(sB1, offB[j][0]);
      b1[j].h.hi = gx_read(sB1, offB[j][1]);
      sb1[j] = sB1[scB[j]];
    }
    if (deep) {
      gx_stage(A, SA, K, ring[(4 * t + 8) & 7], m0, t + 2, M);
      GX_WAIT;
    }
    __builtin_amdgcn_s_barrier();
    __builtin_amdgcn_s_setprio(1);
    GX_MFMA(1, b1, sb1);
    __builtin_amdgcn_s_setprio(0);
    __builtin_amdgcn_s_barrier();

    // -- phase 2: quadrant (A1,B1); stage B0(t+2) --------------------
#pragma unroll
    for (int i = 0; i < 4; ++i) {
      af[i].h.lo = gx_read(sA1, offA[i][0]);
      af[i].h.hi = gx_read(sA1, offA[i][1]);
      sa[i] = sA1[scA[i]];
    }
    if (deep) {
      gx_stage(W, SW, K, ring[(4 * t + 9) & 7], n0, t + 2, N);
      GX_WAIT;
    }
    __builtin_amdgcn_s_barrier();
    __builtin_amdgcn_s_setprio(1);
    GX_MFMA(2, b1, sb1);
    __builtin_amdgcn_s_setprio(0);
    __builtin_amdgcn_s_barrier();

    // -- phase 3: quadrant (A1,B0); stage B1(t+2) --------------------
    if (deep) {
      gx_stage(W, SW, K, ring[(4 * t + 10) & 7], n0 + 128, t + 2, N);
      GX_WAIT;
    }
    __builtin_amdgcn_s_barrier();
    __builtin_amdgcn_s_setprio(1);
    GX_MFMA(3, b0, sb0);
    __builtin_amdgcn_s_setprio(0);
    __builtin_amdgcn_s_barrier();
  }

  // epilogue: quadrant q -> (ha, hb) = gray[(0,0),(0,1),(1,1),(1,0)]
  const int HA[4] = {0, 0, 1, 1}, HB[4] = {0, 1, 1, 0};
#pragma unroll
  for (int q = 0; q < 4; ++q) {
#pragma unroll
    for (int i = 0; i < 4; ++i) {
      const int mrow = m0 + HA[q] * 128 + (wid >> 2) * 64 + i * 16 +
                       (lane >> 4) * 4;
#pragma unroll
      for (int j = 0; j < 2; ++j) {
        const int ncol = n0 + HB[q] * 128 + (wid & 3) * 32 + j * 16 +
                         (lane & 15);
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

AF_EXPORT int af_gemm_mxfp8(void* C, const void* A, const void* SA,
                            const void* W, const void* SW, int M, int N,
                            int K, void* stream) {
  if (K % GX_BK || K < 2 * GX_BK) return 9004;
  if (M == 0) return 0;
  const int tiles_m = (M + GX_BM - 1) / GX_BM;
  const int tiles_n = (N + GX_BM - 1) / GX_BM;
  static int order = -1;
  if (order < 0) {
    const char* e = getenv("AF_GEMM_ORDER");
    order = e ? atoi(e) : 2;
  }
  gemm_mxfp8_kernel<<<tiles_m * tiles_n, 512, 0, (hipStream_t)stream>>>(
      (u16*)C, (const unsigned char*)A, (const unsigned char*)SA,
      (const unsigned char*)W, (const unsigned char*)SW, M, N, K, tiles_m,
      tiles_n, order);
  return af_last_err();
}
