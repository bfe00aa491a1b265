// Paged decode attention (one query token per sequence) with GQA and
// split-K over the KV length ("flash-decoding" style).
//
// Shapes: q   [B, Hq, D=128] bf16     (post-RoPE)
//         kc  [npages, Hk, page_size, D] bf16
//         vc  [npages, Hk, page_size, D] bf16
//         bt  [B, max_pages] i32      block table
//         len [B] i32                 total tokens per seq (incl. current)
//         out [B, Hq, D] bf16
// GQA group G = Hq/Hk, 1 <= G <= 8.
//
// Work decomposition (memory-bound op, CDNA4-shaped):
//   grid (B, Hk, NSPLIT), block = 256 threads = 4 waves.
//   Each wave walks tokens of its split range with stride 4; a wave's 64
//   lanes cover the 128-dim head (2 dims/lane) so every K/V row is one
//   coalesced 256 B wave read (u16x2 per lane).  Scores for the G query
//   heads sharing this KV head are wave-reduced; online softmax keeps
//   (m, l, o[G][2]) in registers.  Wave partials merge through LDS; split
//   partials merge in a second kernel (af_attn_decode_combine).
#include "common.h"

#define AD_D 128
#define AD_MAXG 8

// Partial layout: po [B, Hk, NSPLIT, G, D] f32; pml [B, Hk, NSPLIT, G, 2] f32.
template <int G>
__global__ void __launch_bounds__(256) attn_decode_kernel(
    u16* __restrict__ out, float* __restrict__ po, float* __restrict__ pml,
    const u16* __restrict__ q, const u16* __restrict__ kc, const u16* __restrict__ vc,
    const i32* __restrict__ bt, const i32* __restrict__ len,
    float scale, int Hk, int page_size, int max_pages, int nsplit) {
  const int b = blockIdx.x, kvh = blockIdx.y, split = blockIdx.z;
  const int wid = threadIdx.x >> 6, lane = threadIdx.x & 63;
  const int d0 = lane * 2;
  const int Hq = Hk * G;

  const int L = len[b];
  const int chunk = (L + nsplit - 1) / nsplit;
  const int t0 = split * chunk;
  const int t1 = min(L, t0 + chunk);

  // Q for the G heads of this group, pre-scaled.
  float qr[G][2];
#pragma unroll
  for (int g = 0; g < G; ++g) {
    const u16* qp = q + ((size_t)b * Hq + kvh * G + g) * AD_D + d0;
    qr[g][0] = bf2f(qp[0]) * scale;
    qr[g][1] = bf2f(qp[1]) * scale;
  }

  float m[G], l[G], acc[G][2];
#pragma unroll
  for (int g = 0; g < G; ++g) { m[g] = AF_NEG_INF; l[g] = 0.f; acc[g][0] = 0.f; acc[g][1] = 0.f; }

  const i32* btrow = bt + (size_t)b * max_pages;
  for (int t = t0 + wid; t < t1; t += 4) {
    const i64 page = btrow[t / page_size];
    const size_t base = (((size_t)page * Hk + kvh) * page_size + (t % page_size)) * AD_D + d0;
    const u16x2 kv = *reinterpret_cast<const u16x2*>(kc + base);
    const float k0 = bf2f(kv.x), k1 = bf2f(kv.y);
    const u16x2 vv = *reinterpret_cast<const u16x2*>(vc + base);
    const float v0 = bf2f(vv.x), v1 = bf2f(vv.y);
#pragma unroll
    for (int g = 0; g < G; ++g) {
      const float s = wave_sum_f32(qr[g][0] * k0 + qr[g][1] * k1);
      const float mn = fmaxf(m[g], s);
      const float corr = __expf(m[g] - mn);
      const float p = __expf(s - mn);
      l[g] = l[g] * corr + p;
      acc[g][0] = acc[g][0] * corr + p * v0;
      acc[g][1] = acc[g][1] * corr + p * v1;
      m[g] = mn;
    }
  }

  // ---- merge 4 wave partials through LDS ----
  __shared__ float s_ml[4][G][2];
  __shared__ float s_o[4][G][AD_D];
  if (lane == 0) {
#pragma unroll
    for (int g = 0; g < G; ++g) { s_ml[wid][g][0] = m[g]; s_ml[wid][g][1] = l[g]; }
  }
#pragma unroll
  for (int g = 0; g < G; ++g) {
    s_o[wid][g][d0] = acc[g][0];
    s_o[wid][g][d0 + 1] = acc[g][1];
  }
  __syncthreads();
  if (wid != 0) return;

#pragma unroll
  for (int g = 0; g < G; ++g) {
    float M = AF_NEG_INF;
#pragma unroll
    for (int w = 0; w < 4; ++w) M = fmaxf(M, s_ml[w][g][0]);
    float L2 = 0.f, o0 = 0.f, o1 = 0.f;
#pragma unroll
    for (int w = 0; w < 4; ++w) {
      const float c = (s_ml[w][g][0] <= AF_NEG_INF) ? 0.f : __expf(s_ml[w][g][0] - M);
      L2 += s_ml[w][g][1] * c;
      o0 += s_o[w][g][d0] * c;
      o1 += s_o[w][g][d0 + 1] * c;
    }
    if (nsplit == 1) {
      const float inv = (L2 > 0.f) ? 1.f / L2 : 0.f;
      u16* op = out + ((size_t)b * Hq + kvh * G + g) * AD_D + d0;
      op[0] = f2bf(o0 * inv);
      op[1] = f2bf(o1 * inv);
    } else {
      const size_t pbase = ((((size_t)b * Hk + kvh) * nsplit + split) * G + g);
      float* od = po + pbase * AD_D + d0;
      od[0] = o0; od[1] = o1;
      if (lane == 0) { pml[pbase * 2] = M; pml[pbase * 2 + 1] = L2; }
    }
  }
}

// Combine split partials: grid (B, Hq), 64 threads (one wave, 2 dims/lane).
__global__ void __launch_bounds__(64) attn_decode_combine_kernel(
    u16* __restrict__ out, const float* __restrict__ po, const float* __restrict__ pml,
    int Hq, int G, int nsplit) {
  const int b = blockIdx.x, h = blockIdx.y;
  const int kvh = h / G, g = h % G;
  const int Hk = Hq / G;
  const int d0 = threadIdx.x * 2;

  float M = AF_NEG_INF;
  for (int s = 0; s < nsplit; ++s) {
    const size_t pbase = ((((size_t)b * Hk + kvh) * nsplit + s) * G + g);
    M = fmaxf(M, pml[pbase * 2]);
  }
  float L2 = 0.f, o0 = 0.f, o1 = 0.f;
  for (int s = 0; s < nsplit; ++s) {
    const size_t pbase = ((((size_t)b * Hk + kvh) * nsplit + s) * G + g);
    const float pm = pml[pbase * 2];
    const float c = (pm <= AF_NEG_INF) ? 0.f : __expf(pm - M);
    L2 += pml[pbase * 2 + 1] * c;
    o0 += po[pbase * AD_D + d0] * c;
    o1 += po[pbase * AD_D + d0 + 1] * c;
  }
  const float inv = (L2 > 0.f) ? 1.f / L2 : 0.f;
  u16* op = out + ((size_t)b * Hq + h) * AD_D + d0;
  op[0] = f2bf(o0 * inv);
  op[1] = f2bf(o1 * inv);
}

AF_EXPORT int af_attn_decode(void* out, void* po, void* pml, const void* q,
                             const void* kc, const void* vc, const void* bt,
                             const void* len, float scale, int B, int Hq, int Hk,
                             int D, int page_size, int max_pages, int nsplit,
                             void* stream) {
  if (D != AD_D) return 9002;
  const int G = Hq / Hk;
  if (G < 1 || G > AD_MAXG || G * Hk != Hq) return 9003;
  if (B == 0) return 0;
  dim3 grid(B, Hk, nsplit), blk(256);
  hipStream_t st = (hipStream_t)stream;
#define AF_LAUNCH(GG)                                                            \
  attn_decode_kernel<GG><<<grid, blk, 0, st>>>(                                  \
      (u16*)out, (float*)po, (float*)pml, (const u16*)q, (const u16*)kc,         \
      (const u16*)vc, (const i32*)bt, (const i32*)len, scale, Hk, page_size,     \
      max_pages, nsplit)
  switch (G) {
    case 1: AF_LAUNCH(1); break;
    case 2: AF_LAUNCH(2); break;
    case 4: AF_LAUNCH(4); break;
    case 8: AF_LAUNCH(8); break;
    default: return 9003;
  }
#undef AF_LAUNCH
  if (nsplit > 1) {
    dim3 g2(B, Hq), b2(64);
    attn_decode_combine_kernel<<<g2, b2, 0, st>>>(
        (u16*)out, (const float*)po, (const float*)pml, Hq, G, nsplit);
  }
  return af_last_err();
}
