// Pipelined 256x256 bf16 MFMA GEMM:  C[M,N] = A[M,K] @ W[N,K]^T.
//
// The 128x128 2-barrier kernel (gemm.hip) tops out near ~830 TF because its
// __syncthreads drains every in-flight global_load_lds once per K-step
// (guide §5: the barrier drain is the ~20% structural stall).  This kernel
// keeps loads in flight ACROSS barriers with counted s_waitcnt vmcnt(N):
//
//   tile geometry: BM=BN=256, BK=64, 8 waves (512 threads) in a 2x4 grid,
//   each wave owns a 128x64 output block (8x4 MFMA 16x16x32 fragments).
//
//   LDS: per tensor, two K-HALF buffers (256 rows x 32 K = 16 KiB) double
//   buffered -> 4 x 16 KiB x 2 tensors = 128 KiB.  Stage granularity is one
//   K-half (2 x global_load_lds x 16 B per thread), source pre-swizzled so
//   fragment ds_read_b128s spread banks (ERRATA #21 pattern).
//
//   Schedule per K-tile t (stage order of tile t+1: Ak0, Bk0, Ak1, Bk1):
//     s_waitcnt vmcnt(4); s_barrier          // Ak0,Bk0(t) resident
//     P1: read kk0 frags, MFMA n-half 0      + issue Ak0(t+1)
//     P2:              MFMA n-half 1         + issue Bk0(t+1)
//     s_waitcnt vmcnt(4); s_barrier          // Ak1,Bk1(t) resident
//     P3: read kk1 frags, MFMA n-half 0      + issue Ak1(t+1)
//     P4:              MFMA n-half 1         + issue Bk1(t+1)
//   vmcnt retires in order, so waiting to <=4 outstanding guarantees the
//   oldest 4 loads (= the two halves needed next) are complete while the
//   newest two halves keep streaming — the barrier never drains the queue.
#include "common.h"

#define GP_BM 256
#define GP_BK 64
#define GP_KH 32  // K-half

typedef __attribute__((ext_vector_type(8))) __bf16 bf16x8p2;

__device__ __forceinline__ bf16x8p2 as_bf(s16x8 v) {
  union { s16x8 s; bf16x8p2 b; } u;
  u.s = v;
  return u.b;
}

// Stage one K-half: 256 rows x 32 K bf16 (16 KiB) into linear LDS with the
// read-side XOR swizzle pre-applied to the global source.  Row = 64 B.
// 512 threads x 2 x 16 B.  Swizzle: byte_in_row ^ ((row&3)<<4).
__device__ __forceinline__ void gp_stage(const u16* __restrict__ g, size_t ld,
                                         u16* lds, int row0, int k0,
                                         int max_row) {
  const int tid = threadIdx.x;
#pragma unroll
  for (int it = 0; it < 2; ++it) {
    const int e = (it * 512 + tid) * 8;   // element index in the half-tile
    const int row = e >> 5;               // /32
    const int wb = (e & 31) * 2;          // byte within 64 B row
    const int wsw = wb ^ ((row & 3) << 4);
    const u16* src = g + (size_t)min(row0 + row, max_row - 1) * ld + k0 + (wsw >> 1);
    u16* dst = lds + (size_t)(it * 512 + (tid & ~63)) * 8;
    __builtin_amdgcn_global_load_lds(
        (const __attribute__((address_space(1))) uint32_t*)src,
        (__attribute__((address_space(3))) uint32_t*)dst, 16, 0, 0);
  }
}

__device__ __forceinline__ s16x8 gp_frag(const u16* lds, int row, int g8) {
  // element [row][g8*8 .. +8] of a [256][32] half-tile, swizzled
  const int byte = row * 64 + ((g8 * 16) ^ ((row & 3) << 4));
  return *reinterpret_cast<const s16x8*>(
      reinterpret_cast<const char*>(lds) + byte);
}

#define GP_WAIT_BAR(N)                                         \
  asm volatile("s_waitcnt vmcnt(" #N ")" ::: "memory");        \
  __builtin_amdgcn_s_barrier()

__global__ void __launch_bounds__(512, 1) gemm_bf16_pipelined_kernel(
    u16* __restrict__ C, const u16* __restrict__ A, const u16* __restrict__ W,
    int M, int N, int K, int tiles_m, int tiles_n) {
  // [dbuf][khalf][256*32]
  __shared__ u16 sA[2][2][GP_BM * GP_KH];
  __shared__ u16 sB[2][2][GP_BM * GP_KH];

  const int bid = xcd_swizzle(blockIdx.x, tiles_m * tiles_n);
  const int tm = bid / tiles_n, tn = bid % tiles_n;
  const int m0 = tm * GP_BM, n0 = tn * GP_BM;
  const int wid = threadIdx.x >> 6, lane = threadIdx.x & 63;
  const int wm = (wid >> 2) * 128;        // 2 wave rows
  const int wn = (wid & 3) * 64;          // 4 wave cols

  f32x4 acc[8][4];
#pragma unroll
  for (int i = 0; i < 8; ++i)
#pragma unroll
    for (int j = 0; j < 4; ++j) acc[i][j] = f32x4{0, 0, 0, 0};

  const int ksteps = K / GP_BK;
  // prologue: stage tile 0 fully (4 halves = 8 loads/thread)
  gp_stage(A, K, sA[0][0], m0, 0, M);
  gp_stage(W, K, sB[0][0], n0, 0, N);
  gp_stage(A, K, sA[0][1], m0, GP_KH, M);
  gp_stage(W, K, sB[0][1], n0, GP_KH, N);

  for (int t = 0; t < ksteps; ++t) {
    const int cur = t & 1, nxt = cur ^ 1;
    const int knext = (t + 1) * GP_BK;
    const bool more = t + 1 < ksteps;
    const int arow = wm + (lane & 15);
    const int brow = wn + (lane & 15);
    const int g8 = lane >> 4;

    // ---- k-half 0: Ak0,Bk0 of tile t are the 2 oldest in-flight halves
    GP_WAIT_BAR(4);
    if (more) gp_stage(A, K, sA[nxt][0], m0, knext, M);
    s16x8 af[8], bf0, bf1;
#pragma unroll
    for (int i = 0; i < 8; ++i) af[i] = gp_frag(sA[cur][0], arow + i * 16, g8);
    bf0 = gp_frag(sB[cur][0], brow, g8);
    bf1 = gp_frag(sB[cur][0], brow + 16, g8);
#pragma unroll
    for (int i = 0; i < 8; ++i)
      acc[i][0] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(as_bf(af[i]),
                                                          as_bf(bf0), acc[i][0], 0, 0, 0);
#pragma unroll
    for (int i = 0; i < 8; ++i)
      acc[i][1] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(as_bf(af[i]),
                                                          as_bf(bf1), acc[i][1], 0, 0, 0);
    if (more) gp_stage(W, K, sB[nxt][0], n0, knext, N);
    bf0 = gp_frag(sB[cur][0], brow + 32, g8);
    bf1 = gp_frag(sB[cur][0], brow + 48, g8);
#pragma unroll
    for (int i = 0; i < 8; ++i)
      acc[i][2] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(as_bf(af[i]),
                                                          as_bf(bf0), acc[i][2], 0, 0, 0);
#pragma unroll
    for (int i = 0; i < 8; ++i)
      acc[i][3] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(as_bf(af[i]),
                                                          as_bf(bf1), acc[i][3], 0, 0, 0);

    // ---- k-half 1.  On the last tile P1/P2 issued nothing, so the 4
    // outstanding loads ARE this k-half — vmcnt(4) would not retire them.
    if (more) { GP_WAIT_BAR(4); } else { GP_WAIT_BAR(0); }
    if (more) gp_stage(A, K, sA[nxt][1], m0, knext + GP_KH, M);
#pragma unroll
    for (int i = 0; i < 8; ++i) af[i] = gp_frag(sA[cur][1], arow + i * 16, g8);
    bf0 = gp_frag(sB[cur][1], brow, g8);
    bf1 = gp_frag(sB[cur][1], brow + 16, g8);
#pragma unroll
    for (int i = 0; i < 8; ++i)
      acc[i][0] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(as_bf(af[i]),
                                                          as_bf(bf0), acc[i][0], 0, 0, 0);
#pragma unroll
    for (int i = 0; i < 8; ++i)
      acc[i][1] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(as_bf(af[i]),
                                                          as_bf(bf1), acc[i][1], 0, 0, 0);
    if (more) gp_stage(W, K, sB[nxt][1], n0, knext + GP_KH, N);
    bf0 = gp_frag(sB[cur][1], brow + 32, g8);
    bf1 = gp_frag(sB[cur][1], brow + 48, g8);
#pragma unroll
    for (int i = 0; i < 8; ++i)
      acc[i][2] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(as_bf(af[i]),
                                                          as_bf(bf0), acc[i][2], 0, 0, 0);
#pragma unroll
    for (int i = 0; i < 8; ++i)
      acc[i][3] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(as_bf(af[i]),
                                                          as_bf(bf1), acc[i][3], 0, 0, 0);
  }

  // ---- epilogue: C[m0+wm+i*16+(l>>4)*4+r][n0+wn+j*16+(l&15)] ----
  asm volatile("s_waitcnt vmcnt(0)" ::: "memory");
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

AF_EXPORT int af_gemm_bf16_pipelined(void* C, const void* A, const void* W,
                                     int M, int N, int K, void* stream) {
  if (K % GP_BK || K < 2 * GP_BK) return 9004;
  if (M == 0) return 0;
  const int tiles_m = (M + GP_BM - 1) / GP_BM, tiles_n = (N + GP_BM - 1) / GP_BM;
  gemm_bf16_pipelined_kernel<<<tiles_m * tiles_n, 512, 0, (hipStream_t)stream>>>(
      (u16*)C, (const u16*)A, (const u16*)W, M, N, K, tiles_m, tiles_n);
  return af_last_err();
}
