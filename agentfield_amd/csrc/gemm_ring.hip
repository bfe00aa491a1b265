// Deep-pipelined 256x256 bf16 MFMA GEMM:  C[M,N] = A[M,K] @ W[N,K]^T.
//
// The 128x128 2-barrier kernel (gemm.hip, ~830-860 TF) is capped by the
// structural barrier drain: __syncthreads forces s_waitcnt vmcnt(0) once
// per K-step, so no global_load_lds ever survives a barrier (guide §5:
// the ~20% stall; the m97-structure ceiling is ~900 TF).  This kernel is
// the guide's verified deep-pipelined 256² structure (§5 "8-phase
// template", 1563-1728 TF = 62-69% of the 2.5 PF dense peak) expressed
// as a K-chunk ring with a provably-correct counted-vmcnt schedule:
//
//   tile: BM=BN=256, staged in K-chunks of 32 (A-chunk = 256x32 bf16 =
//   16 KiB, B-chunk same).  LDS = ring of 4 chunk slots per tensor
//   (4 x 2 x 16 KiB = 128 KiB of the 160 KiB/CU).
//
//   8 waves (512 threads) in a 2(M) x 4(N) grid; each wave owns a
//   128x64 output block = 8x4 MFMA 16x16x32 fragments.
//
//   Phase c (one per K-chunk): the ring is 3 chunks deep, so while
//   chunk c is consumed, chunks c+1 and c+2 are in flight and chunk
//   c+3 is being issued:
//
//     s_waitcnt vmcnt(8)    // own-wave: chunks c+1,c+2 (4 loads each)
//                           // may still fly; in-order retirement means
//                           // chunk c's loads have landed
//     s_barrier             // all waves agree: slot[c] readable,
//                           // slot[c+3 mod 4] (chunk c-1) drained
//     issue chunk c+3 -> slot[(c+3) mod 4]   (4 x global_load_lds)
//     ds_read 12 x b128 (8 A-frags + 4 B-frags)
//     s_setprio(1); 32 x mfma_f32_16x16x32_bf16; s_setprio(0)
//
//   vmcnt never drains to 0 in the main loop (T3+T4, guide §5.5); the
//   tail shrinks the count as issues stop (8 -> 4 -> 0).
//
//   LDS chunk layout: [128 rows x 128 B]; LDS-row r packs M-rows 2r and
//   2r+1 (32 K-elems = 64 B each).  16-B units within a row are XOR-
//   swizzled by (r&7) so a fragment read (lanes 0-15 = 16 consecutive
//   M-rows at one K-slab) spreads uniformly over all banks (T2).  The
//   swizzle is applied on the global SOURCE address (global_load_lds
//   writes linearly) and on the ds_read address — both-sides involution
//   (guide ERRATA #21).
//
//   Safety of slot reuse: chunk c+3's stores target the slot chunk c-1
//   lived in.  Every wave's phase-(c-1) fragment reads completed before
//   its MFMAs issued (register dependency), and every wave passed phase
//   c's barrier after those MFMAs were issued — so by the time any wave
//   issues chunk c+3's loads, no lane still has a pending read from
//   that slot.
#include "common.h"

#define GR_BM 256      // C tile M = N = 256
#define GR_KC 32       // K-chunk
#define GR_SLOTS 4     // ring depth (chunks resident)
#define GR_CHUNK_U16 (128 * 64)  // one chunk: 128 LDS-rows x 64 u16

typedef __attribute__((ext_vector_type(8))) __bf16 bf16x8r;

__device__ __forceinline__ bf16x8r as_bfr(s16x8 v) {
  union { s16x8 s; bf16x8r b; } u;
  u.s = v;
  return u.b;
}

// Issue one tensor-chunk stage: 256 rows x 32 K bf16 from global
// (row-major, ld elements) into a linear LDS slot.  2 x 16 B per thread.
// 16-B unit d of the slot holds, after the inverse swizzle, the slab
//   u = (d&7) ^ (r&7),  m = 2r + (u>>2),  k-slab s = u&3
// so the ds_read-side swizzle (same XOR) finds element [m][s*8..+8].
__device__ __forceinline__ void gr_stage(const u16* __restrict__ g,
                                         size_t ld, u16* lds, int row0,
                                         int k0, int max_row) {
  const int tid = threadIdx.x;
#pragma unroll
  for (int it = 0; it < 2; ++it) {
    const int d = it * 512 + tid;
    const int r = d >> 3;
    const int u = (d & 7) ^ (r & 7);
    const int m = 2 * r + (u >> 2);
    const int s = u & 3;
    const u16* src = g + (size_t)min(row0 + m, max_row - 1) * ld + k0 + s * 8;
    u16* dst = lds + (size_t)(it * 512 + (tid & ~63)) * 8;
    __builtin_amdgcn_global_load_lds(
        (const __attribute__((address_space(1))) uint32_t*)src,
        (__attribute__((address_space(3))) uint32_t*)dst, 16, 0, 0);
  }
}

// Byte offset (within a chunk slot) of fragment element [m][s*8] —
// LDS-row r = m>>1, unit u = (m&1)*4 + s, swizzled u^(r&7).
__device__ __forceinline__ int gr_off(int m, int s) {
  const int r = m >> 1;
  const int u = (m & 1) * 4 + s;
  return r * 128 + (u ^ (r & 7)) * 16;
}

__device__ __forceinline__ s16x8 gr_read(const u16* lds, int off) {
  return *reinterpret_cast<const s16x8*>(
      reinterpret_cast<const char*>(lds) + off);
}

__global__ void __launch_bounds__(512, 1) gemm_bf16_ring_kernel(
    u16* __restrict__ C, const u16* __restrict__ A, const u16* __restrict__ W,
    int M, int N, int K, int tiles_m, int tiles_n) {
  __shared__ u16 sA[GR_SLOTS][GR_CHUNK_U16];
  __shared__ u16 sB[GR_SLOTS][GR_CHUNK_U16];

  const int bid = xcd_swizzle(blockIdx.x, tiles_m * tiles_n);
  const int tm = bid / tiles_n, tn = bid % tiles_n;
  const int m0 = tm * GR_BM, n0 = tn * GR_BM;
  const int wid = threadIdx.x >> 6, lane = threadIdx.x & 63;
  const int wm = (wid >> 2) * 128;   // 2 wave rows x 128
  const int wn = (wid & 3) * 64;     // 4 wave cols x 64

  // per-lane fragment byte offsets within a chunk slot (constant)
  const int s = lane >> 4;
  int offA[8], offB[4];
#pragma unroll
  for (int i = 0; i < 8; ++i) offA[i] = gr_off(wm + i * 16 + (lane & 15), s);
#pragma unroll
  for (int j = 0; j < 4; ++j) offB[j] = gr_off(wn + j * 16 + (lane & 15), s);

  f32x4 acc[8][4];
#pragma unroll
  for (int i = 0; i < 8; ++i)
#pragma unroll
    for (int j = 0; j < 4; ++j) acc[i][j] = f32x4{0, 0, 0, 0};

  const int nchunks = K / GR_KC;
  // prologue: fill the ring 3 deep
  for (int c = 0; c < 3 && c < nchunks; ++c) {
    gr_stage(A, K, sA[c], m0, c * GR_KC, M);
    gr_stage(W, K, sB[c], n0, c * GR_KC, N);
  }

  // chunk 0 must be resident before the first sub-phase reads it: allow
  // only the chunks issued BEYOND chunk 0 to stay in flight
  if (nchunks >= 3) {
    asm volatile("s_waitcnt vmcnt(8)" ::: "memory");
  } else if (nchunks == 2) {
    asm volatile("s_waitcnt vmcnt(4)" ::: "memory");
  } else {
    asm volatile("s_waitcnt vmcnt(0)" ::: "memory");
  }
  __builtin_amdgcn_s_barrier();

  for (int c = 0; c < nchunks; ++c) {
    const u16* a = sA[c & (GR_SLOTS - 1)];
    const u16* b = sB[c & (GR_SLOTS - 1)];
    const bool prefetch = c + 3 < nchunks;
    const int slot = (c + 3) & (GR_SLOTS - 1);
    s16x8 af[4], bf[4];
    // ---- sub-phase 1: quadrant i=0..3 (16 MFMA between barriers) ----
#pragma unroll
    for (int i = 0; i < 4; ++i) af[i] = gr_read(a, offA[i]);
#pragma unroll
    for (int j = 0; j < 4; ++j) bf[j] = gr_read(b, offB[j]);
    if (prefetch) gr_stage(A, K, sA[slot], m0, (c + 3) * GR_KC, M);
    __builtin_amdgcn_s_barrier();
    __builtin_amdgcn_s_setprio(1);
#pragma unroll
    for (int i = 0; i < 4; ++i)
#pragma unroll
      for (int j = 0; j < 4; ++j)
        acc[i][j] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
            as_bfr(af[i]), as_bfr(bf[j]), acc[i][j], 0, 0, 0);
    __builtin_amdgcn_s_setprio(0);
    __builtin_amdgcn_s_barrier();
    // ---- sub-phase 2: quadrant i=4..7 (B frags stay in registers) ----
#pragma unroll
    for (int i = 0; i < 4; ++i) af[i] = gr_read(a, offA[4 + i]);
    if (prefetch) gr_stage(W, K, sB[slot], n0, (c + 3) * GR_KC, N);
    // counted wait, once per chunk: chunk c+1 must be resident for the
    // next iteration's reads; chunks c+2, c+3 (4 loads each) stay in
    // flight.  Own-wave count + barrier ⇒ all waves' loads retired.
    {
      const int ahead = min(nchunks - 2 - c, 2);
      if (ahead >= 2) {
        asm volatile("s_waitcnt vmcnt(8)" ::: "memory");
      } else if (ahead == 1) {
        asm volatile("s_waitcnt vmcnt(4)" ::: "memory");
      } else {
        asm volatile("s_waitcnt vmcnt(0)" ::: "memory");
      }
    }
    __builtin_amdgcn_s_barrier();
    __builtin_amdgcn_s_setprio(1);
#pragma unroll
    for (int i = 0; i < 4; ++i)
#pragma unroll
      for (int j = 0; j < 4; ++j)
        acc[4 + i][j] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
            as_bfr(af[i]), as_bfr(bf[j]), acc[4 + i][j], 0, 0, 0);
    __builtin_amdgcn_s_setprio(0);
    __builtin_amdgcn_s_barrier();
  }

  // epilogue: C[m0+wm+i*16+(lane>>4)*4+r][n0+wn+j*16+(lane&15)]
#pragma unroll
  for (int i = 0; i < 8; ++i) {
    const int mrow = m0 + wm + i * 16 + (lane >> 4) * 4;
#pragma unroll
    for (int j = 0; j < 4; ++j) {
      const int ncol = n0 + wn + j * 16 + (lane & 15);
      if (ncol >= N) continue;
#pragma unroll
      for (int r = 0; r < 4; ++r) {
        if (mrow + r >= M) continue;
        C[(size_t)(mrow + r) * N + ncol] = f2bf(acc[i][j][r]);
      }
    }
  }
}

AF_EXPORT int af_gemm_bf16_ring(void* C, const void* A, const void* W,
                                int M, int N, int K, void* stream) {
  if (K % GR_KC || K < GR_KC) return 9004;
  if (M == 0) return 0;
  const int tiles_m = (M + GR_BM - 1) / GR_BM;
  const int tiles_n = (N + GR_BM - 1) / GR_BM;
  gemm_bf16_ring_kernel<<<tiles_m * tiles_n, 512, 0, (hipStream_t)stream>>>(
      (u16*)C, (const u16*)A, (const u16*)W, M, N, K, tiles_m, tiles_n);
  return af_last_err();
}
