// Paged decode attention (one query token per sequence) with GQA and
// split-K over the KV length ("flash-decoding" style).
//
// Shapes: q   [B, rows at stride qs] bf16, head h at offset h*D (post-RoPE)
//         kc  [npages, Hk, page_size, D] bf16
//         vc  [npages, Hk, page_size, D] bf16
//         bt  [B, max_pages] i32      block table
//         len [B] i32                 total tokens per seq (incl. current)
//         out [B, Hq*D] bf16 (contiguous)
// GQA group G = Hq/Hk in {1,2,4,8}.  D = 128.
//
// Work decomposition (memory-bound; CDNA4-shaped):
//   grid (B, Hk, NSPLIT), block = 256 threads = 4 waves.
//   A wave iteration covers EIGHT tokens (two unrolled groups of four):
//   lane = tg*16 + dl, token subgroup tg = lane>>4, dim-group dl = lane&15
//   owning dims [dl*8, dl*8+8).  Each K/V row is a 16-lane x 16 B coalesced
//   read; within-page consecutive tokens make a 4-token group's reads 1 KiB
//   contiguous.  The score reduction is 4 xor-shuffles within the 16-lane
//   group + 2 across groups — ~2 shuffles per token vs 6 for a
//   whole-wave-per-token layout, with 8x the load ILP.
//   Online softmax state (m, l) is tracked wave-wide over the 4-token tile;
//   o accumulates per-lane (8 dims) and is tg-reduced once at the end.
//   Wave partials merge through LDS; split partials merge in a second kernel.
#include "common.h"

#define AD_D 128
#define AD_MAXG 8

// Partial layout: po [B, Hk, NSPLIT, G, D] f32; pml [B, Hk, NSPLIT, G, 2] f32.
template <int G, bool WIDE>
__global__ void __launch_bounds__(256) attn_decode_kernel(
    u16* __restrict__ out, float* __restrict__ po, float* __restrict__ pml,
    const u16* __restrict__ q, const u16* __restrict__ kc, const u16* __restrict__ vc,
    const i32* __restrict__ bt, const i32* __restrict__ len,
    float scale, int Hk, int page_size, int max_pages, int nsplit, i64 qs,
    int win) {
  const int b = blockIdx.x, kvh = blockIdx.y, split = blockIdx.z;
  const int wid = threadIdx.x >> 6, lane = threadIdx.x & 63;
  const int tg = lane >> 4, dl = lane & 15;
  const int Hq = Hk * G;

  const int L = len[b];
  // sliding window (Mistral): only the last `win` tokens are visible
  const int s0 = (win > 0 && L > win) ? L - win : 0;
  const int chunk = (L - s0 + nsplit - 1) / nsplit;
  const int t0 = s0 + split * chunk;
  const int t1 = min(L, t0 + chunk);

  // Q (8 dims per lane per head), pre-scaled
  float qr[G][8];
#pragma unroll
  for (int g = 0; g < G; ++g) {
    const u16* qp = q + (size_t)b * qs + (kvh * G + g) * AD_D + dl * 8;
    s16x8 qv = *reinterpret_cast<const s16x8*>(qp);
#pragma unroll
    for (int j = 0; j < 8; ++j) qr[g][j] = bf2f((u16)qv[j]) * scale;
  }

  float m[G], l[G], acc[G][8];
#pragma unroll
  for (int g = 0; g < G; ++g) {
    m[g] = AF_NEG_INF;
    l[g] = 0.f;
#pragma unroll
    for (int j = 0; j < 8; ++j) acc[g][j] = 0.f;
  }

  const i32* btrow = bt + (size_t)b * max_pages;
  if (!WIDE) {
    // one 4-token tile per wave iteration: wins at LOW grid occupancy
    // (<= 2 blocks/CU) where the wider loop loses throughput (measured)
    for (int tb = t0 + wid * 4; tb < t1; tb += 16) {
      const int t = tb + tg;
      const bool valid = t < t1;
      const int tc = valid ? t : (t1 - 1);
      const i64 page = btrow[tc / page_size];
      const size_t base = (((size_t)page * Hk + kvh) * page_size +
                           (tc % page_size)) * AD_D + dl * 8;
      const s16x8 kv8 = *reinterpret_cast<const s16x8*>(kc + base);
      const s16x8 vv8 = *reinterpret_cast<const s16x8*>(vc + base);
      float kf[8], vf[8];
#pragma unroll
      for (int j = 0; j < 8; ++j) {
        kf[j] = bf2f((u16)kv8[j]);
        vf[j] = bf2f((u16)vv8[j]);
      }
#pragma unroll
      for (int g = 0; g < G; ++g) {
        float d = 0.f;
#pragma unroll
        for (int j = 0; j < 8; ++j) d += qr[g][j] * kf[j];
        float s = group16_sum_f32(d);
        if (!valid) s = AF_NEG_INF;
        float tm = fmaxf(s, __shfl_xor(s, 16, 64));
        tm = fmaxf(tm, __shfl_xor(tm, 32, 64));
        const float mn = fmaxf(m[g], tm);
        const float corr = (m[g] <= AF_NEG_INF) ? 0.f : __expf(m[g] - mn);
        const float p = (s <= AF_NEG_INF) ? 0.f : __expf(s - mn);
        float psum = p + __shfl_xor(p, 16, 64);
        psum += __shfl_xor(psum, 32, 64);
        l[g] = l[g] * corr + psum;
        m[g] = mn;
#pragma unroll
        for (int j = 0; j < 8; ++j)
          acc[g][j] = acc[g][j] * corr + p * vf[j];
      }
    }
  } else {
  // wave w handles tokens {i*32 + w*4 + tg} and {i*32 + 16 + w*4 + tg}:
  // TWO 4-token tiles per iteration with all four 16-B K/V loads issued
  // before any compute — the kernel is gather-latency-bound, so doubling
  // the in-flight bytes per wave is the lever (guide: memory-level
  // parallelism, not arithmetic, sets gather throughput)
  auto tile_base = [&](int t) {
    const i64 page = btrow[t / page_size];
    return (((size_t)page * Hk + kvh) * page_size + (t % page_size)) * AD_D +
           dl * 8;
  };
  for (int tb = t0 + wid * 4; tb < t1; tb += 32) {
    const int ta = tb + tg, tb2 = tb + 16 + tg;
    const bool va = ta < t1, vb = tb2 < t1;
    // unconditional clamped loads: a branch around the second tile's
    // loads stops the compiler issuing them early, which defeats the
    // whole point (measured -25% on short chunks)
    const size_t base_a = tile_base(va ? ta : (t1 - 1));
    const size_t base_b = tile_base(vb ? tb2 : (t1 - 1));
    const s16x8 ka8 = *reinterpret_cast<const s16x8*>(kc + base_a);
    const s16x8 va8 = *reinterpret_cast<const s16x8*>(vc + base_a);
    const s16x8 kb8 = *reinterpret_cast<const s16x8*>(kc + base_b);
    const s16x8 vb8 = *reinterpret_cast<const s16x8*>(vc + base_b);
    float kfa[8], vfa[8], kfb[8], vfb[8];
#pragma unroll
    for (int j = 0; j < 8; ++j) {
      kfa[j] = bf2f((u16)ka8[j]);
      vfa[j] = bf2f((u16)va8[j]);
      kfb[j] = bf2f((u16)kb8[j]);
      vfb[j] = bf2f((u16)vb8[j]);
    }
#pragma unroll
    for (int half = 0; half < 2; ++half) {
      const bool valid = half ? vb : va;
      const float* kf = half ? kfb : kfa;
      const float* vf = half ? vfb : vfa;
#pragma unroll
      for (int g = 0; g < G; ++g) {
        float d = 0.f;
#pragma unroll
        for (int j = 0; j < 8; ++j) d += qr[g][j] * kf[j];
        // 16-lane reduce -> s uniform within token group
        float s = group16_sum_f32(d);
        if (!valid) s = AF_NEG_INF;
        // tile max across the 4 token groups
        float tm = fmaxf(s, __shfl_xor(s, 16, 64));
        tm = fmaxf(tm, __shfl_xor(tm, 32, 64));
        if (tm <= AF_NEG_INF) continue;  // whole tile invalid (tail)
        const float mn = fmaxf(m[g], tm);
        const float corr = (m[g] <= AF_NEG_INF) ? 0.f : __expf(m[g] - mn);
        const float p = (s <= AF_NEG_INF) ? 0.f : __expf(s - mn);
        float psum = p + __shfl_xor(p, 16, 64);
        psum += __shfl_xor(psum, 32, 64);
        l[g] = l[g] * corr + psum;
        m[g] = mn;
#pragma unroll
        for (int j = 0; j < 8; ++j) acc[g][j] = acc[g][j] * corr + p * vf[j];
      }
    }
  }
  }
  // fold the 4 token-groups' partial o (same dims, disjoint tokens)
#pragma unroll
  for (int g = 0; g < G; ++g)
#pragma unroll
    for (int j = 0; j < 8; ++j) {
      acc[g][j] += __shfl_xor(acc[g][j], 16, 64);
      acc[g][j] += __shfl_xor(acc[g][j], 32, 64);
    }

  // ---- merge 4 wave partials through LDS ----
  __shared__ float s_ml[4][G][2];
  __shared__ float s_o[4][G][AD_D];
  if (lane == 0) {
#pragma unroll
    for (int g = 0; g < G; ++g) { s_ml[wid][g][0] = m[g]; s_ml[wid][g][1] = l[g]; }
  }
  if (tg == 0) {
#pragma unroll
    for (int g = 0; g < G; ++g)
#pragma unroll
      for (int j = 0; j < 8; ++j) s_o[wid][g][dl * 8 + j] = acc[g][j];
  }
  __syncthreads();
  if (wid != 0) return;

#pragma unroll
  for (int g = 0; g < G; ++g) {
    float M = AF_NEG_INF;
#pragma unroll
    for (int w = 0; w < 4; ++w) M = fmaxf(M, s_ml[w][g][0]);
    float L2 = 0.f, o8[8];
#pragma unroll
    for (int j = 0; j < 8; ++j) o8[j] = 0.f;
#pragma unroll
    for (int w = 0; w < 4; ++w) {
      const float c = (s_ml[w][g][0] <= AF_NEG_INF) ? 0.f : __expf(s_ml[w][g][0] - M);
      L2 += s_ml[w][g][1] * c;
#pragma unroll
      for (int j = 0; j < 8; ++j) o8[j] += s_o[w][g][dl * 8 + j] * c;
    }
    if (tg != 0) continue;
    if (nsplit == 1) {
      const float inv = (L2 > 0.f) ? 1.f / L2 : 0.f;
      u16* op = out + (size_t)b * (Hq * AD_D) + (kvh * G + g) * AD_D + dl * 8;
      s16x8 ov;
#pragma unroll
      for (int j = 0; j < 8; ++j) ov[j] = (short)f2bf(o8[j] * inv);
      *reinterpret_cast<s16x8*>(op) = ov;
    } else {
      const size_t pbase = ((((size_t)b * Hk + kvh) * nsplit + split) * G + g);
      float* od = po + pbase * AD_D + dl * 8;
#pragma unroll
      for (int j = 0; j < 8; ++j) od[j] = o8[j];
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
  u16* op = out + (size_t)b * (Hq * AD_D) + h * AD_D + d0;
  op[0] = f2bf(o0 * inv);
  op[1] = f2bf(o1 * inv);
}

AF_EXPORT int af_attn_decode(void* out, void* po, void* pml, const void* q,
                             const void* kc, const void* vc, const void* bt,
                             const void* len, float scale, int B, int Hq, int Hk,
                             int D, int page_size, int max_pages, int nsplit,
                             i64 qs, int win, void* stream) {
  if (D != AD_D) return 9002;
  const int G = Hq / Hk;
  if (G < 1 || G > AD_MAXG || G * Hk != Hq) return 9003;
  if (B == 0) return 0;
  dim3 grid(B, Hk, nsplit), blk(256);
  hipStream_t st = (hipStream_t)stream;
  const bool wide = (size_t)B * Hk * nsplit >= 1024;  // >= 4 blocks/CU
#define AF_LAUNCH(GG)                                                            \
  do {                                                                           \
    if (wide)                                                                    \
      attn_decode_kernel<GG, true><<<grid, blk, 0, st>>>(                        \
          (u16*)out, (float*)po, (float*)pml, (const u16*)q, (const u16*)kc,     \
          (const u16*)vc, (const i32*)bt, (const i32*)len, scale, Hk,            \
          page_size, max_pages, nsplit, qs, win);                                \
    else                                                                         \
      attn_decode_kernel<GG, false><<<grid, blk, 0, st>>>(                       \
          (u16*)out, (float*)po, (float*)pml, (const u16*)q, (const u16*)kc,     \
          (const u16*)vc, (const i32*)bt, (const i32*)len, scale, Hk,            \
          page_size, max_pages, nsplit, qs, win);                                \
  } while (0)
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
