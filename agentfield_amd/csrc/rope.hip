// Rotary position embedding (Llama rotate-half convention), in-place on Q and K.
//
// For pair (d, d+D/2):  x'[d]     = x[d]*cos - x[d+D/2]*sin
//                       x'[d+D/2] = x[d+D/2]*cos + x[d]*sin
//
// cos/sin are a host-precomputed fp32 table [max_pos, D] laid out
// [cos(0..D/2) | sin(0..D/2)] per position — on-device trig turns this
// memory-bound op VALU-bound (guide Appendix B).
#include "common.h"

__global__ void __launch_bounds__(256) rope_kernel(
    u16* __restrict__ q, u16* __restrict__ k,
    const i32* __restrict__ pos, const float* __restrict__ table,
    int T, int Hq, int Hk, int D) {
  const int half = D >> 1;
  const int qh4 = half >> 2;              // 4 pairs per thread
  const int total = (Hq + Hk) * qh4;
  for (int t = blockIdx.x; t < T; t += gridDim.x) {
    const int p = pos[t];
    const float* cosr = table + (size_t)p * D;
    const float* sinr = cosr + half;
    for (int i = threadIdx.x; i < total; i += blockDim.x) {
      const int h = i / qh4;
      const int dp = (i % qh4) * 4;
      u16* base = (h < Hq) ? q + ((size_t)t * Hq + h) * D
                           : k + ((size_t)t * Hk + (h - Hq)) * D;
      s16x4 a = *reinterpret_cast<const s16x4*>(base + dp);
      s16x4 b = *reinterpret_cast<const s16x4*>(base + dp + half);
      f32x4 c = *reinterpret_cast<const f32x4*>(cosr + dp);
      f32x4 s = *reinterpret_cast<const f32x4*>(sinr + dp);
      s16x4 oa, ob;
#pragma unroll
      for (int j = 0; j < 4; ++j) {
        const float x1 = bf2f((u16)a[j]), x2 = bf2f((u16)b[j]);
        oa[j] = (short)f2bf(x1 * c[j] - x2 * s[j]);
        ob[j] = (short)f2bf(x2 * c[j] + x1 * s[j]);
      }
      *reinterpret_cast<s16x4*>(base + dp) = oa;
      *reinterpret_cast<s16x4*>(base + dp + half) = ob;
    }
  }
}

AF_EXPORT int af_rope(void* q, void* k, const void* pos, const void* table,
                      int T, int Hq, int Hk, int D, void* stream) {
  if (D % 8) return 9001;
  if (T == 0) return 0;
  int blocks = T < 2048 ? T : 2048;
  rope_kernel<<<blocks, 256, 0, (hipStream_t)stream>>>(
      (u16*)q, (u16*)k, (const i32*)pos, (const float*)table, T, Hq, Hk, D);
  return af_last_err();
}
