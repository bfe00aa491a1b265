// Fused residual-add RMSNorm for bf16 rows, fp32 accumulation.
//
// out[t]     = h[t] * rsqrt(mean(h[t]^2) + eps) * w,  h = x (+ residual)
// res_out[t] = h[t]                        (written only when residual given)
//
// Replaces the torch composition (add + pow + mean + rsqrt + mul + mul) with
// a single HBM-bound pass: one read of x (+res +w), one write of out (+res).
// Vectorized 8x bf16 (16 B/lane) per guide Guideline 13; row kept in
// registers between the sum-of-squares pass and the scale pass.
#include "common.h"

template <bool HAS_RES>
__global__ void __launch_bounds__(256) rmsnorm_kernel(
    u16* __restrict__ out, u16* __restrict__ res_out,
    const u16* __restrict__ x, const u16* __restrict__ res,
    const u16* __restrict__ w, float eps, int T, int H) {
  constexpr int VEC = 8;
  constexpr int NTHR = 256;
  constexpr int MAX_IT = 8;  // supports H <= 8*256*8 = 16384
  const int per_row = H / VEC;
  __shared__ float red[4];

  // statically indexed register cache (a runtime-indexed array spills to
  // scratch — guide rule #20; the spill measured 0.85 TB/s at prefill T)
  float cache[MAX_IT][VEC];

  for (int row = blockIdx.x; row < T; row += gridDim.x) {
    const u16* xrow = x + (size_t)row * H;
    const u16* rrow = HAS_RES ? res + (size_t)row * H : nullptr;
    float ss = 0.f;
#pragma unroll
    for (int it = 0; it < MAX_IT; ++it) {
      const int c = threadIdx.x + it * NTHR;
      if (c >= per_row) break;
      s16x8 xv = *reinterpret_cast<const s16x8*>(xrow + c * VEC);
      float* f = cache[it];
#pragma unroll
      for (int j = 0; j < VEC; ++j) f[j] = bf2f((u16)xv[j]);
      if (HAS_RES) {
        s16x8 rv = *reinterpret_cast<const s16x8*>(rrow + c * VEC);
#pragma unroll
        for (int j = 0; j < VEC; ++j) f[j] += bf2f((u16)rv[j]);
      }
#pragma unroll
      for (int j = 0; j < VEC; ++j) ss += f[j] * f[j];
    }
    ss = wave_sum_f32(ss);
    const int wid = threadIdx.x >> 6, lid = threadIdx.x & 63;
    if (lid == 0) red[wid] = ss;
    __syncthreads();
    const float rstd = rsqrtf((red[0] + red[1] + red[2] + red[3]) / (float)H + eps);

#pragma unroll
    for (int it = 0; it < MAX_IT; ++it) {
      const int c = threadIdx.x + it * NTHR;
      if (c >= per_row) break;
      float* f = cache[it];
      s16x8 wv = *reinterpret_cast<const s16x8*>(w + c * VEC);
      s16x8 ov, hv;
#pragma unroll
      for (int j = 0; j < VEC; ++j) {
        if (HAS_RES) hv[j] = (short)f2bf(f[j]);
        ov[j] = (short)f2bf(f[j] * rstd * bf2f((u16)wv[j]));
      }
      *reinterpret_cast<s16x8*>(out + (size_t)row * H + c * VEC) = ov;
      if (HAS_RES)
        *reinterpret_cast<s16x8*>(res_out + (size_t)row * H + c * VEC) = hv;
    }
    __syncthreads();  // red[] reused next row iteration
  }
}

// Low-latency variant for decode-sized T: one wave per row, no LDS/barriers
// (the 256-thread block version costs ~14 us at T=64 from barrier latency).
template <bool HAS_RES>
__global__ void __launch_bounds__(64) rmsnorm_wave_kernel(
    u16* __restrict__ out, u16* __restrict__ res_out,
    const u16* __restrict__ x, const u16* __restrict__ res,
    const u16* __restrict__ w, float eps, int T, int H) {
  constexpr int VEC = 8;
  const int row = blockIdx.x;
  if (row >= T) return;
  const int per_row = H / VEC;
  const u16* xrow = x + (size_t)row * H;
  const u16* rrow = HAS_RES ? res + (size_t)row * H : nullptr;
  float ss = 0.f;
  for (int c = threadIdx.x; c < per_row; c += 64) {
    s16x8 xv = *reinterpret_cast<const s16x8*>(xrow + c * VEC);
    float f[VEC];
#pragma unroll
    for (int j = 0; j < VEC; ++j) f[j] = bf2f((u16)xv[j]);
    if (HAS_RES) {
      s16x8 rv = *reinterpret_cast<const s16x8*>(rrow + c * VEC);
#pragma unroll
      for (int j = 0; j < VEC; ++j) f[j] += bf2f((u16)rv[j]);
    }
#pragma unroll
    for (int j = 0; j < VEC; ++j) ss += f[j] * f[j];
  }
  const float rstd = rsqrtf(wave_sum_f32(ss) / (float)H + eps);
  // second pass re-reads from L2 (rows are hot at decode sizes); avoids a
  // runtime-indexed register cache which would spill to scratch (rule #20)
  for (int c = threadIdx.x; c < per_row; c += 64) {
    s16x8 xv = *reinterpret_cast<const s16x8*>(xrow + c * VEC);
    float f[VEC];
#pragma unroll
    for (int j = 0; j < VEC; ++j) f[j] = bf2f((u16)xv[j]);
    if (HAS_RES) {
      s16x8 rv = *reinterpret_cast<const s16x8*>(rrow + c * VEC);
#pragma unroll
      for (int j = 0; j < VEC; ++j) f[j] += bf2f((u16)rv[j]);
    }
    s16x8 wv = *reinterpret_cast<const s16x8*>(w + c * VEC);
    s16x8 ov, hv;
#pragma unroll
    for (int j = 0; j < VEC; ++j) {
      if (HAS_RES) hv[j] = (short)f2bf(f[j]);
      ov[j] = (short)f2bf(f[j] * rstd * bf2f((u16)wv[j]));
    }
    *reinterpret_cast<s16x8*>(out + (size_t)row * H + c * VEC) = ov;
    if (HAS_RES)
      *reinterpret_cast<s16x8*>(res_out + (size_t)row * H + c * VEC) = hv;
  }
}

AF_EXPORT int af_rmsnorm(void* out, void* res_out, const void* x, const void* res,
                         const void* w, float eps, int T, int H, void* stream) {
  if (H % 8 || H > 16384) return 9001;
  if (T == 0) return 0;
  hipStream_t st = (hipStream_t)stream;
  if (T <= 256 && H <= 16384) {
    if (res)
      rmsnorm_wave_kernel<true><<<T, 64, 0, st>>>(
          (u16*)out, (u16*)res_out, (const u16*)x, (const u16*)res,
          (const u16*)w, eps, T, H);
    else
      rmsnorm_wave_kernel<false><<<T, 64, 0, st>>>(
          (u16*)out, nullptr, (const u16*)x, nullptr, (const u16*)w, eps, T, H);
    return af_last_err();
  }
  int blocks = T < 16384 ? T : 16384;  // one row per block fills 256 CUs
  if (res)
    rmsnorm_kernel<true><<<blocks, 256, 0, st>>>(
        (u16*)out, (u16*)res_out, (const u16*)x, (const u16*)res, (const u16*)w, eps, T, H);
  else
    rmsnorm_kernel<false><<<blocks, 256, 0, st>>>(
        (u16*)out, nullptr, (const u16*)x, nullptr, (const u16*)w, eps, T, H);
  return af_last_err();
}
