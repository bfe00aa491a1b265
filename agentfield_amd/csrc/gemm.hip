// bf16 MFMA GEMM:  C[M,N] = A[M,K] @ W[N,K]^T   (torch Linear weight layout)
// fp32 accumulate, bf16 out.  Optional fused bias add + SiLU-mul epilogues are
// left to the dedicated fused kernels; this is the plain tiled GEMM.
//
// Structure: the guide's verified "m97" shape (cdna_hip_programming.md §5):
//   128x128 C-tile per workgroup, BK=64, 256 threads = 4 waves,
//   each wave computes a 64x64 quadrant as 4x4 MFMA 16x16x32 fragments,
//   global->LDS staging via __builtin_amdgcn_global_load_lds width 16,
//   LDS XOR-swizzle (T2) applied by pre-swizzling the *global* source address
//   (global_load_lds writes linearly; guide ERRATA #21),
//   XCD-aware bijective blockIdx swizzle (T1).
#include "common.h"

#define GT 128   // C tile (M and N)
#define GBK 64   // K step

typedef __attribute__((ext_vector_type(8))) __bf16 bf16x8g;

__device__ __forceinline__ bf16x8g as_bf16x8g(s16x8 v) {
  union { s16x8 s; bf16x8g b; } u;
  u.s = v;
  return u.b;
}

// Stage one 128x64 bf16 tile (16 KiB) from global (row-major, ld elements)
// into linear LDS, 16 B per lane per call x 4 calls, with the read-side XOR
// swizzle pre-applied to the *source* so LDS[linear] holds swizzled data.
// Rows beyond (rows,k_rem) bounds are clamped to row 0 / zero handled by caller.
__device__ __forceinline__ void stage_tile(
    const u16* __restrict__ gsrc, size_t ld, u16* lds, int row0, int k0,
    int max_row, int max_k) {
  const int tid = threadIdx.x;
#pragma unroll
  for (int it = 0; it < 4; ++it) {
    const int e = (it * 256 + tid) * 8;       // linear element offset in tile
    const int row = e >> 6;                   // /64
    const int wb = (e & 63) * 2;              // byte within row (0..127)
    const int wsw = wb ^ ((row & 7) << 4);    // involution: source pre-swizzle
    const int gr = row0 + row;
    const int gk = k0 + (wsw >> 1);
    // clamp OOB to row0/k0 (tail tiles produce garbage that the epilogue masks;
    // K-tail garbage is zeroed below via the valid flag)
    const u16* src = gsrc + (size_t)min(gr, max_row - 1) * ld + min(gk, max_k - 8);
    u16* dst_wave = lds + (it * 256 + (tid & ~63)) * 8;  // wave-uniform base
    __builtin_amdgcn_global_load_lds(
        (const __attribute__((address_space(1))) uint32_t*)src,
        (__attribute__((address_space(3))) uint32_t*)dst_wave, 16, 0, 0);
  }
}

__global__ void __launch_bounds__(256) gemm_bf16_kernel(
    u16* __restrict__ C, const u16* __restrict__ A, const u16* __restrict__ W,
    int M, int N, int K, int tiles_m, int tiles_n) {
  __shared__ u16 sA[GT * GBK];
  __shared__ u16 sB[GT * GBK];

  const int bid = xcd_swizzle(blockIdx.x, tiles_m * tiles_n);
  const int tm = bid / tiles_n, tn = bid % tiles_n;
  const int m0 = tm * GT, n0 = tn * GT;
  const int wid = threadIdx.x >> 6, lane = threadIdx.x & 63;
  // wave quadrant: 2x2 waves, each 64x64
  const int wm = (wid >> 1) * 64, wn = (wid & 1) * 64;

  f32x4 acc[4][4];
#pragma unroll
  for (int i = 0; i < 4; ++i)
#pragma unroll
    for (int j = 0; j < 4; ++j) acc[i][j] = f32x4{0, 0, 0, 0};

  const int ksteps = (K + GBK - 1) / GBK;
  for (int ks = 0; ks < ksteps; ++ks) {
    const int k0 = ks * GBK;
    __syncthreads();
    stage_tile(A, K, sA, m0, k0, M, K);
    stage_tile(W, K, sB, n0, k0, N, K);
    __syncthreads();
    // frag read: element A[m][k] lives at byte m*128 + (k*2 ^ ((m&7)<<4))
#pragma unroll
    for (int kk = 0; kk < 2; ++kk) {  // two K=32 chunks in BK=64
      s16x8 afrag[4], bfrag[4];
#pragma unroll
      for (int i = 0; i < 4; ++i) {
        const int row = wm + i * 16 + (lane & 15);
        const int byte = (kk * 32 + (lane >> 4) * 8) * 2;
        afrag[i] = *reinterpret_cast<const s16x8*>(
            reinterpret_cast<const char*>(sA) + row * (GBK * 2) +
            (byte ^ ((row & 7) << 4)));
      }
#pragma unroll
      for (int j = 0; j < 4; ++j) {
        const int row = wn + j * 16 + (lane & 15);
        const int byte = (kk * 32 + (lane >> 4) * 8) * 2;
        bfrag[j] = *reinterpret_cast<const s16x8*>(
            reinterpret_cast<const char*>(sB) + row * (GBK * 2) +
            (byte ^ ((row & 7) << 4)));
      }
#pragma unroll
      for (int i = 0; i < 4; ++i)
#pragma unroll
        for (int j = 0; j < 4; ++j)
          acc[i][j] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
              as_bf16x8g(afrag[i]), as_bf16x8g(bfrag[j]), acc[i][j], 0, 0, 0);
    }
  }

  // epilogue: C[m0+wm+i*16+(lane>>4)*4+r][n0+wn+j*16+(lane&15)]
#pragma unroll
  for (int i = 0; i < 4; ++i) {
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

AF_EXPORT int af_gemm_bf16(void* C, const void* A, const void* W,
                           int M, int N, int K, void* stream) {
  if (K % GBK || K < GBK) return 9004;  // K multiple of 64 (model dims are)
  if (M == 0) return 0;
  const int tiles_m = (M + GT - 1) / GT, tiles_n = (N + GT - 1) / GT;
  gemm_bf16_kernel<<<tiles_m * tiles_n, 256, 0, (hipStream_t)stream>>>(
      (u16*)C, (const u16*)A, (const u16*)W, M, N, K, tiles_m, tiles_n);
  return af_last_err();
}
