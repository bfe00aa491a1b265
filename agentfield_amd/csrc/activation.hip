// Fused SwiGLU activation: out[t, i] = silu(gate[t, i]) * up[t, i]
// where input is packed [T, 2I] = [gate | up] (the fused gate_up projection
// output), out is [T, I].  bf16 in/out, fp32 math, 8x-vectorized.
#include "common.h"

// scale_ss (optional): [T,8] stats -> rows scaled by rsqrt(sum/K+eps)
// before the activation (the fused-norm chain's output-side scalar).
__global__ void __launch_bounds__(256) silu_mul_kernel(
    u16* __restrict__ out, const u16* __restrict__ gate_up,
    const float* __restrict__ scale_ss, float inv_k, float eps, i64 T, i64 I) {
  const i64 nvec = T * (I >> 3);
  const i64 stride = (i64)gridDim.x * blockDim.x;
  for (i64 i = (i64)blockIdx.x * blockDim.x + threadIdx.x; i < nvec; i += stride) {
    const i64 t = i / (I >> 3);
    const i64 c = (i % (I >> 3)) << 3;
    float rstd = 1.f;
    if (scale_ss) {
      float st = 0.f;
#pragma unroll
      for (int j = 0; j < 8; ++j) st += scale_ss[t * 8 + j];
      rstd = rsqrtf(st * inv_k + eps);
    }
    s16x8 g = *reinterpret_cast<const s16x8*>(gate_up + t * 2 * I + c);
    s16x8 u = *reinterpret_cast<const s16x8*>(gate_up + t * 2 * I + I + c);
    s16x8 o;
#pragma unroll
    for (int j = 0; j < 8; ++j) {
      const float gf = bf2f((u16)g[j]) * rstd;
      const float uf = bf2f((u16)u[j]) * rstd;
      o[j] = (short)f2bf(gf / (1.f + __expf(-gf)) * uf);
    }
    *reinterpret_cast<s16x8*>(out + t * I + c) = o;
  }
}

AF_EXPORT int af_silu_mul(void* out, const void* gate_up, const void* scale_ss,
                          float inv_k, float eps, i64 T, i64 I, void* stream) {
  if (I % 8) return 9001;
  if (T == 0) return 0;
  i64 nvec = T * (I >> 3);
  int blocks = (int)((nvec + 255) / 256);
  if (blocks > 2048) blocks = 2048;
  silu_mul_kernel<<<blocks, 256, 0, (hipStream_t)stream>>>(
      (u16*)out, (const u16*)gate_up, (const float*)scale_ss, inv_k, eps, T, I);
  return af_last_err();
}

// Elementwise add (bf16): out = a + b — used for residual paths not covered
// by the fused rmsnorm.
__global__ void __launch_bounds__(256) add_kernel(
    u16* __restrict__ out, const u16* __restrict__ a, const u16* __restrict__ b, i64 n8) {
  const i64 stride = (i64)gridDim.x * blockDim.x;
  for (i64 i = (i64)blockIdx.x * blockDim.x + threadIdx.x; i < n8; i += stride) {
    s16x8 av = *reinterpret_cast<const s16x8*>(a + i * 8);
    s16x8 bv = *reinterpret_cast<const s16x8*>(b + i * 8);
    s16x8 o;
#pragma unroll
    for (int j = 0; j < 8; ++j) o[j] = (short)f2bf(bf2f((u16)av[j]) + bf2f((u16)bv[j]));
    *reinterpret_cast<s16x8*>(out + i * 8) = o;
  }
}

AF_EXPORT int af_add(void* out, const void* a, const void* b, i64 n, void* stream) {
  if (n % 8) return 9001;
  if (n == 0) return 0;
  i64 n8 = n / 8;
  int blocks = (int)((n8 + 255) / 256);
  if (blocks > 2048) blocks = 2048;
  add_kernel<<<blocks, 256, 0, (hipStream_t)stream>>>(
      (u16*)out, (const u16*)a, (const u16*)b, n8);
  return af_last_err();
}
