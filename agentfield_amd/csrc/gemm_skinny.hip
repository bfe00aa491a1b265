// Skinny-M GEMM for the decode path:  Y[M,N] = X[M,K] @ W[N,K]^T, M <= 256.
//
// At decode batch sizes this is a pure weight-stream (W is N*K*2 bytes, X is
// L2-resident), so the kernel is built to read W at HBM rate with MFMA doing
// the math for free:
//   * grid (N/64, SK): block = 4 waves staging one W tile [64 rows x 64 K]
//     (8 KiB) plus the X tile [<=64 x 64] per K-step through LDS with
//     __builtin_amdgcn_global_load_lds (16 B lanes) and the T2 XOR swizzle
//     pre-applied on the *source* address (guide ERRATA #21) so fragment
//     ds_read_b128s are bank-conflict-free
//   * K is split SK ways so the launch fills 256 CUs even at N=4096;
//     fp32 partials [SK, M, N] are folded by a combine kernel that also
//     carries the fused epilogue: plain bf16 store, SwiGLU
//     (out = silu(gate)*up) for the gate_up projection, or residual-add
//   * hipBLASLt's MT16 kernels run at ~1.7-2.6 TB/s on the N=4096/6144
//     decode shapes; this path targets the ~6 TB/s achievable ceiling
#include "common.h"

typedef __attribute__((ext_vector_type(8))) __bf16 bf16x8s;

__device__ __forceinline__ bf16x8s as_bf16x8s(s16x8 v) {
  union { s16x8 s; bf16x8s b; } u;
  u.s = v;
  return u.b;
}

// Stage a 64x64 bf16 tile (8 KiB) into linear LDS with read-side swizzle
// pre-applied to the global source.  2 x 16 B per thread.
__device__ __forceinline__ void stage64(const u16* __restrict__ g, size_t ld,
                                        u16* lds, int row0, int k0,
                                        int max_row) {
  const int tid = threadIdx.x;
#pragma unroll
  for (int it = 0; it < 2; ++it) {
    const int e = (it * 256 + tid) * 8;
    const int row = e >> 6;
    const int wb = (e & 63) * 2;
    const int wsw = wb ^ ((row & 7) << 4);
    const u16* src = g + (size_t)min(row0 + row, max_row - 1) * ld + k0 + (wsw >> 1);
    u16* dst = lds + (size_t)(it * 256 + (tid & ~63)) * 8;
    __builtin_amdgcn_global_load_lds(
        (const __attribute__((address_space(1))) uint32_t*)src,
        (__attribute__((address_space(3))) uint32_t*)dst, 16, 0, 0);
  }
}

__device__ __forceinline__ s16x8 frag64(const u16* lds, int row, int byte) {
  return *reinterpret_cast<const s16x8*>(
      reinterpret_cast<const char*>(lds) + row * 128 + (byte ^ ((row & 7) << 4)));
}

// MT = number of 16-row M tiles (ceil(M/16)); grid (N/64, SK), block 256.
template <int MT>
__global__ void __launch_bounds__(256) gemm_skinny_kernel(
    float* __restrict__ partial, const u16* __restrict__ X,
    const u16* __restrict__ W, int M, int N, int K, int Kc) {
  __shared__ u16 sW[2][64 * 64];
  __shared__ u16 sX[2][((MT + 3) / 4) * 64 * 64];  // 64-row stage blocks
  const int wid = threadIdx.x >> 6, lane = threadIdx.x & 63;
  const int nblk = blockIdx.x * 64;
  const int sk = blockIdx.y;
  const int k0 = sk * Kc;
  const int k1 = min(K, k0 + Kc);

  f32x4 acc[MT];
#pragma unroll
  for (int mt = 0; mt < MT; ++mt) acc[mt] = f32x4{0, 0, 0, 0};

  // 2-phase pipeline (guide T3 minimum): stage tile t+1 while computing
  // tile t; the single __syncthreads (vmcnt(0)+barrier) at loop end drains
  // the in-flight loads after compute has covered their latency.
  constexpr int XBLKS = (MT + 3) / 4;  // 64-row stage blocks for X
  int cur = 0;
  stage64(W, K, sW[0], nblk, k0, N);
#pragma unroll
  for (int xb = 0; xb < XBLKS; ++xb)
    stage64(X + (size_t)(xb * 64) * K, K, sX[0] + xb * 64 * 64, 0, k0,
            M - xb * 64);
  __syncthreads();
  for (int k = k0; k < k1; k += 64) {
    if (k + 64 < k1) {
      stage64(W, K, sW[cur ^ 1], nblk, k + 64, N);
#pragma unroll
      for (int xb = 0; xb < XBLKS; ++xb)
        stage64(X + (size_t)(xb * 64) * K, K, sX[cur ^ 1] + xb * 64 * 64, 0,
                k + 64, M - xb * 64);
    }
#pragma unroll
    for (int kk = 0; kk < 2; ++kk) {
      const int byte = (kk * 32 + (lane >> 4) * 8) * 2;
      const s16x8 bf = frag64(sW[cur], wid * 16 + (lane & 15), byte);
#pragma unroll
      for (int mt = 0; mt < MT; ++mt) {
        const s16x8 af = frag64(sX[cur], mt * 16 + (lane & 15), byte);
        acc[mt] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
            as_bf16x8s(af), as_bf16x8s(bf), acc[mt], 0, 0, 0);
      }
    }
    __syncthreads();
    cur ^= 1;
  }

  // C layout: lane l reg r -> C[m=(l>>4)*4+r][n=l&15]
  float* pbase = partial + (size_t)sk * M * N;
  const int ncol = nblk + wid * 16 + (lane & 15);
#pragma unroll
  for (int mt = 0; mt < MT; ++mt) {
    const int mrow = mt * 16 + (lane >> 4) * 4;
#pragma unroll
    for (int r = 0; r < 4; ++r) {
      if (mrow + r < M)
        pbase[(size_t)(mrow + r) * N + ncol] = acc[mt][r];
    }
  }
}

// mode 0: out[M,N] bf16 = sum_sk partial
// mode 1: out[M,N] bf16 = sum + residual read/write (residual updated in place)
// mode 2: N = 2I; out[M,I] bf16 = silu(sum[:, :I]) * sum[:, I:]
__global__ void __launch_bounds__(256) gemm_skinny_combine_kernel(
    u16* __restrict__ out, u16* __restrict__ residual,
    const float* __restrict__ partial, int M, int N, int SK, int mode,
    const float* __restrict__ scale_ss, float inv_k, float eps) {
  const int cols = (mode == 2) ? (N >> 1) : N;
  const i64 total = (i64)M * (cols >> 2);  // 4 outputs per thread
  const i64 stride = (i64)gridDim.x * blockDim.x;
  for (i64 i = (i64)blockIdx.x * blockDim.x + threadIdx.x; i < total;
       i += stride) {
    const int m = (int)(i / (cols >> 2));
    const int c = (int)(i % (cols >> 2)) * 4;
    float v[4] = {0, 0, 0, 0};
    for (int s = 0; s < SK; ++s) {
      const float* p = partial + ((size_t)s * M + m) * N + c;
#pragma unroll
      for (int j = 0; j < 4; ++j) v[j] += p[j];
    }
    float rstd = 1.f;
    if (scale_ss) {
      float st = 0.f;
#pragma unroll
      for (int j = 0; j < 8; ++j) st += scale_ss[m * 8 + j];
      rstd = rsqrtf(st * inv_k + eps);
#pragma unroll
      for (int j = 0; j < 4; ++j) v[j] *= rstd;
    }
    if (mode == 2) {
      float u[4] = {0, 0, 0, 0};
      for (int s = 0; s < SK; ++s) {
        const float* p = partial + ((size_t)s * M + m) * N + (N >> 1) + c;
#pragma unroll
        for (int j = 0; j < 4; ++j) u[j] += p[j];
      }
#pragma unroll
      for (int j = 0; j < 4; ++j) u[j] *= rstd;
      s16x4 o;
#pragma unroll
      for (int j = 0; j < 4; ++j)
        o[j] = (short)f2bf(v[j] / (1.f + __expf(-v[j])) * u[j]);
      *reinterpret_cast<s16x4*>(out + (size_t)m * cols + c) = o;
    } else if (mode == 1) {
      s16x4 rv = *reinterpret_cast<const s16x4*>(residual + (size_t)m * N + c);
      s16x4 o;
#pragma unroll
      for (int j = 0; j < 4; ++j) o[j] = (short)f2bf(v[j] + bf2f((u16)rv[j]));
      *reinterpret_cast<s16x4*>(out + (size_t)m * N + c) = o;
      *reinterpret_cast<s16x4*>(residual + (size_t)m * N + c) = o;
    } else {
      s16x4 o;
#pragma unroll
      for (int j = 0; j < 4; ++j) o[j] = (short)f2bf(v[j]);
      *reinterpret_cast<s16x4*>(out + (size_t)m * N + c) = o;
    }
  }
}

// mode 4: grid (M, 8): block (m, cb) combines column block cb of row m:
// out = sum + residual (residual updated in place) AND
// ss_out[m*8+cb] = partial sum of out^2 over the block — together the 8
// partials feed the NEXT skinny GEMM's fused input RMSNorm, eliminating
// standalone rmsnorm kernels at decode.
__global__ void __launch_bounds__(256) gemm_skinny_combine_row_kernel(
    u16* __restrict__ out, u16* __restrict__ residual,
    const float* __restrict__ partial, float* __restrict__ ss_out,
    int M, int N, int SK) {
  const int m = blockIdx.x;
  const int nb = N / 8;          // N % 64 == 0 => nb % 4 == 0... (N/8 cols)
  const int c0 = blockIdx.y * nb;
  __shared__ float red[4];
  float local = 0.f;
  for (int c = c0 + threadIdx.x * 4; c < c0 + nb; c += 256 * 4) {
    float v[4] = {0, 0, 0, 0};
    for (int s = 0; s < SK; ++s) {
      const float* p = partial + ((size_t)s * M + m) * N + c;
#pragma unroll
      for (int j = 0; j < 4; ++j) v[j] += p[j];
    }
    s16x4 rv = *reinterpret_cast<const s16x4*>(residual + (size_t)m * N + c);
    s16x4 o;
#pragma unroll
    for (int j = 0; j < 4; ++j) {
      const float h = v[j] + bf2f((u16)rv[j]);
      o[j] = (short)f2bf(h);
      const float hb = bf2f((u16)o[j]);  // stats on the bf16-rounded value
      local += hb * hb;
    }
    *reinterpret_cast<s16x4*>(out + (size_t)m * N + c) = o;
    *reinterpret_cast<s16x4*>(residual + (size_t)m * N + c) = o;
  }
  local = wave_sum_f32(local);
  const int wid2 = threadIdx.x >> 6;
  if ((threadIdx.x & 63) == 0) red[wid2] = local;
  __syncthreads();
  if (threadIdx.x == 0)
    ss_out[m * 8 + blockIdx.y] = red[0] + red[1] + red[2] + red[3];
}

AF_EXPORT int af_gemm_skinny(void* out, void* partial, void* residual,
                             const void* X, const void* W, int M, int N,
                             int K, int SK, int mode,
                             const void* scale_ss, float eps, void* ss_out,
                             void* stream) {
  if (M < 1 || M > 256) return 9005;
  if (N % 64 || K % 64) return 9006;
  hipStream_t st = (hipStream_t)stream;
  int Kc = ((K / SK + 63) / 64) * 64;
  while ((SK - 1) * Kc >= K) --SK;  // drop empty splits
  dim3 grid(N / 64, SK), blk(256);
  const int MT = (M + 15) / 16;
#define AF_SK_LAUNCH(MTV)                                                      \
  gemm_skinny_kernel<MTV><<<grid, blk, 0, st>>>(                               \
      (float*)partial, (const u16*)X, (const u16*)W, M, N, K, Kc)
  switch (MT) {
    case 1: AF_SK_LAUNCH(1); break;
    case 2: AF_SK_LAUNCH(2); break;
    case 3: AF_SK_LAUNCH(3); break;
    case 4: AF_SK_LAUNCH(4); break;
    case 5: AF_SK_LAUNCH(5); break;
    case 6: AF_SK_LAUNCH(6); break;
    case 7: AF_SK_LAUNCH(7); break;
    case 8: AF_SK_LAUNCH(8); break;
    case 9: AF_SK_LAUNCH(9); break;
    case 10: AF_SK_LAUNCH(10); break;
    case 11: AF_SK_LAUNCH(11); break;
    case 12: AF_SK_LAUNCH(12); break;
    case 13: AF_SK_LAUNCH(13); break;
    case 14: AF_SK_LAUNCH(14); break;
    case 15: AF_SK_LAUNCH(15); break;
    default: AF_SK_LAUNCH(16); break;
  }
#undef AF_SK_LAUNCH
  if (mode == 4) {
    if (N % 32) return 9007;
    dim3 g4(M, 8);
    gemm_skinny_combine_row_kernel<<<g4, 256, 0, st>>>(
        (u16*)out, (u16*)residual, (const float*)partial, (float*)ss_out,
        M, N, SK);
    return af_last_err();
  }
  const int cols = (mode == 2) ? N / 2 : N;
  i64 total = (i64)M * (cols / 4);
  int blocks = (int)((total + 255) / 256);
  if (blocks > 1024) blocks = 1024;
  gemm_skinny_combine_kernel<<<blocks, 256, 0, st>>>(
      (u16*)out, (u16*)residual, (const float*)partial, M, N, SK, mode,
      (const float*)scale_ss, 1.f / (float)K, eps);
  return af_last_err();
}
