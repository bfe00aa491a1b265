// Fused RoPE + paged-KV append.
//
// One kernel per layer step does all position-dependent work between the QKV
// GEMM and attention:
//   * in-place rotate-half RoPE on q and k (strided rows straight out of the
//     fused QKV projection — no .contiguous() copies)
//   * scatter the rotated k and raw v into the paged KV cache
//
// cos/sin are a host-precomputed fp32 table [max_pos, D] laid out
// [cos(0..D/2) | sin(0..D/2)] per position (on-device trig would turn this
// memory-bound op VALU-bound, guide Appendix B).
//
// q: [T, ...] rows at stride qs, head h at offset h*D
// k, v: rows at stride ks/vs
// caches: [npages, Hk, page_size, D];  slot[t] < 0 skips the cache write
#include "common.h"

// scale_ss (optional): [T,8] column-block sum-of-squares stats; when given,
// rows are multiplied by rsqrt(sum/K+eps) — the RMSNorm scalar left over
// after folding the norm weight into the QKV projection.  RoPE rotation is
// linear, so pre-rotation scaling is exact.
__global__ void __launch_bounds__(256) rope_cache_kernel(
    u16* __restrict__ q, u16* __restrict__ k, const u16* __restrict__ v,
    const i32* __restrict__ pos, const float* __restrict__ table,
    u16* __restrict__ kc, u16* __restrict__ vc, const i64* __restrict__ slot,
    const float* __restrict__ scale_ss, float inv_k, float eps,
    int T, int Hq, int Hk, int D, i64 qs, i64 ks, i64 vs, int page_size) {
  const int half = D >> 1;
  const int qh4 = half >> 2;              // 4 rotation pairs per thread
  const int rope_work = (Hq + Hk) * qh4;
  const int v_work = Hk * (D >> 3);       // v copy, 8 elems per thread
  for (int t = blockIdx.x; t < T; t += gridDim.x) {
    const int p = pos[t];
    float rstd = 1.f;
    if (scale_ss) {
      float st = 0.f;
#pragma unroll
      for (int j = 0; j < 8; ++j) st += scale_ss[t * 8 + j];
      rstd = rsqrtf(st * inv_k + eps);
    }
    const float* cosr = table + (size_t)p * D;
    const float* sinr = cosr + half;
    const i64 s = slot[t];
    const i64 page = (s >= 0) ? s / page_size : 0;
    const i64 off = (s >= 0) ? s % page_size : 0;
    for (int i = threadIdx.x; i < rope_work; i += blockDim.x) {
      const int h = i / qh4;
      const int dp = (i % qh4) * 4;
      const bool is_q = h < Hq;
      u16* base = is_q ? q + (size_t)t * qs + h * D
                       : k + (size_t)t * ks + (h - Hq) * D;
      s16x4 a = *reinterpret_cast<const s16x4*>(base + dp);
      s16x4 b = *reinterpret_cast<const s16x4*>(base + dp + half);
      f32x4 c = *reinterpret_cast<const f32x4*>(cosr + dp);
      f32x4 sn = *reinterpret_cast<const f32x4*>(sinr + dp);
      s16x4 oa, ob;
#pragma unroll
      for (int j = 0; j < 4; ++j) {
        const float x1 = bf2f((u16)a[j]) * rstd, x2 = bf2f((u16)b[j]) * rstd;
        oa[j] = (short)f2bf(x1 * c[j] - x2 * sn[j]);
        ob[j] = (short)f2bf(x2 * c[j] + x1 * sn[j]);
      }
      *reinterpret_cast<s16x4*>(base + dp) = oa;
      *reinterpret_cast<s16x4*>(base + dp + half) = ob;
      if (!is_q && s >= 0) {
        // mirror the rotated k into the cache (4+4 elems, two halves)
        const int kh = h - Hq;
        u16* krow = kc + (((size_t)page * Hk + kh) * page_size + off) * D;
        *reinterpret_cast<s16x4*>(krow + dp) = oa;
        *reinterpret_cast<s16x4*>(krow + dp + half) = ob;
      }
    }
    if (s >= 0) {
      for (int i = threadIdx.x; i < v_work; i += blockDim.x) {
        const int h = (i * 8) / D, d = (i * 8) % D;
        u16* vrow = vc + (((size_t)page * Hk + h) * page_size + off) * D;
        s16x8 vv = *reinterpret_cast<const s16x8*>(v + (size_t)t * vs + h * D + d);
        if (scale_ss) {
#pragma unroll
          for (int j = 0; j < 8; ++j)
            vv[j] = (short)f2bf(bf2f((u16)vv[j]) * rstd);
        }
        *reinterpret_cast<s16x8*>(vrow + d) = vv;
      }
    }
  }
}

AF_EXPORT int af_rope_cache(void* q, void* k, const void* v, const void* pos,
                            const void* table, void* kc, void* vc,
                            const void* slot, const void* scale_ss,
                            float inv_k, float eps, int T, int Hq, int Hk,
                            int D, i64 qs, i64 ks, i64 vs, int page_size,
                            void* stream) {
  if (D % 8) return 9001;
  if (T == 0) return 0;
  int blocks = T < 2048 ? T : 2048;
  rope_cache_kernel<<<blocks, 256, 0, (hipStream_t)stream>>>(
      (u16*)q, (u16*)k, (const u16*)v, (const i32*)pos, (const float*)table,
      (u16*)kc, (u16*)vc, (const i64*)slot, (const float*)scale_ss, inv_k,
      eps, T, Hq, Hk, D, qs, ks, vs, page_size);
  return af_last_err();
}
