// Token sampling over the LM-head logits.
//
// Greedy and temperature sampling share one argmax machine: temperature
// sampling uses the Gumbel-max trick (argmax(logits/T + g), g ~ Gumbel(0,1))
// so no sort or cumulative sum over the 128k vocab is needed and the whole
// step is hipGraph-capturable.  Randomness is a counter-based hash of
// (seed, step, b, v); the step counter lives in device memory and is bumped
// by a kernel inside the graph, so replays draw fresh samples.
//
// logits: [B, V] bf16;  temp: [B] f32 (0 => greedy);  out: [B] i32
// partial scratch: pv [B, SPLITS] f32, pi [B, SPLITS] i32
#include "common.h"

#define SMP_SPLITS 16

__device__ __forceinline__ float u32_to_unit(uint32_t x) {
  return (x >> 8) * (1.f / 16777216.f) + (0.5f / 16777216.f);
}

// Gumbel(0,1) sample from u in (0,1).  The inner fast-math __logf(u) can
// round to EXACTLY 0 for u near 1, making -__logf(0) = +inf — which lets
// any token (including grammar-masked -1e30 rows) win the argmax.  Floor
// the inner result at 1e-30 (gumbel <= ~69, finite).
__device__ __forceinline__ float gumbel_of(float u) {
  const float nl = fmaxf(-__logf(u), 1e-30f);
  return -__logf(nl);
}

__device__ __forceinline__ uint32_t hash3(uint32_t a, uint32_t b, uint32_t c) {
  // Wang/xxhash-style avalanche mix
  uint32_t h = a * 0x9E3779B1u + b * 0x85EBCA77u + c * 0xC2B2AE3Du + 0x27220A95u;
  h ^= h >> 15; h *= 0x2C1B3C6Du;
  h ^= h >> 12; h *= 0x297A2D39u;
  h ^= h >> 15;
  return h;
}

__global__ void __launch_bounds__(256) sample_partial_kernel(
    float* __restrict__ pv, i32* __restrict__ pi, const u16* __restrict__ logits,
    const float* __restrict__ temp, const uint32_t* __restrict__ step,
    uint32_t seed, int V) {
  const int b = blockIdx.x, split = blockIdx.y;
  const int chunk = (V + SMP_SPLITS - 1) / SMP_SPLITS;
  const int v0 = split * chunk, v1 = min(V, v0 + chunk);
  const float t = temp[b];
  const float invt = (t > 0.f) ? 1.f / t : 0.f;
  const uint32_t st = *step;

  float best = AF_NEG_INF;
  int bidx = 0;
  const u16* row = logits + (size_t)b * V;
  for (int v = v0 + threadIdx.x; v < v1; v += 256) {
    float x = bf2f(row[v]);
    if (t > 0.f) {
      const float u = u32_to_unit(hash3(seed ^ st, (uint32_t)b, (uint32_t)v));
      x = x * invt + gumbel_of(u);
    }
    if (x > best) { best = x; bidx = v; }
  }
  // wave then block argmax
#pragma unroll
  for (int o = 32; o > 0; o >>= 1) {
    const float ov = __shfl_xor(best, o, 64);
    const int oi = __shfl_xor(bidx, o, 64);
    if (ov > best || (ov == best && oi < bidx)) { best = ov; bidx = oi; }
  }
  __shared__ float sv[4];
  __shared__ int si[4];
  const int wid = threadIdx.x >> 6;
  if ((threadIdx.x & 63) == 0) { sv[wid] = best; si[wid] = bidx; }
  __syncthreads();
  if (threadIdx.x == 0) {
#pragma unroll
    for (int w = 1; w < 4; ++w)
      if (sv[w] > best || (sv[w] == best && si[w] < bidx)) { best = sv[w]; bidx = si[w]; }
    pv[b * SMP_SPLITS + split] = best;
    pi[b * SMP_SPLITS + split] = bidx;
  }
}

__global__ void __launch_bounds__(64) sample_combine_kernel(
    i32* __restrict__ out, const float* __restrict__ pv, const i32* __restrict__ pi) {
  const int b = blockIdx.x;
  if (threadIdx.x != 0) return;
  float best = AF_NEG_INF;
  int bidx = 0;
  for (int s = 0; s < SMP_SPLITS; ++s) {
    const float v = pv[b * SMP_SPLITS + s];
    if (v > best || (v == best && pi[b * SMP_SPLITS + s] < bidx)) {
      best = v;
      bidx = pi[b * SMP_SPLITS + s];
    }
  }
  out[b] = bidx;
}

__global__ void step_inc_kernel(uint32_t* step) {
  if (threadIdx.x == 0 && blockIdx.x == 0) *step += 1u;
}

// ---- top-k / top-p restriction --------------------------------------------
// Sort-free: a 256-bin histogram of logits over [rowmax-SPAN, rowmax]
// (bin 255 = max) gives, per row, the threshold above which the top-k count
// or top-p probability mass is reached; the Gumbel-max draw then ignores
// logits below the threshold.  Boundary-bin granularity is SPAN/256 ~ 0.08
// logits — the standard approximation for GPU nucleus sampling.
#define TKP_BINS 256
#define TKP_SPAN 20.0f

// pass A: row max (reuses partial scratch pv/pi from the argmax machinery)
__global__ void __launch_bounds__(256) row_max_kernel(
    float* __restrict__ rmax, const u16* __restrict__ logits, int V) {
  const int b = blockIdx.x;
  float best = AF_NEG_INF;
  const u16* row = logits + (size_t)b * V;
  for (int v = threadIdx.x; v < V; v += 256) best = fmaxf(best, bf2f(row[v]));
#pragma unroll
  for (int o = 32; o > 0; o >>= 1) best = fmaxf(best, __shfl_xor(best, o, 64));
  __shared__ float sv[4];
  if ((threadIdx.x & 63) == 0) sv[threadIdx.x >> 6] = best;
  __syncthreads();
  if (threadIdx.x == 0)
    rmax[b] = fmaxf(fmaxf(sv[0], sv[1]), fmaxf(sv[2], sv[3]));
}

// pass B: histogram of counts and exp-mass per bin (temperature applied)
__global__ void __launch_bounds__(256) tkp_hist_kernel(
    i32* __restrict__ hist_n, float* __restrict__ hist_m,
    const u16* __restrict__ logits, const float* __restrict__ rmax,
    const float* __restrict__ temp, int V) {
  const int b = blockIdx.x;
  __shared__ i32 hn[TKP_BINS];
  __shared__ float hm[TKP_BINS];
  for (int i = threadIdx.x; i < TKP_BINS; i += 256) { hn[i] = 0; hm[i] = 0.f; }
  __syncthreads();
  const float t = temp[b];
  const float invt = (t > 0.f) ? 1.f / t : 1.f;
  const float mx = rmax[b] * invt;
  const u16* row = logits + (size_t)b * V;
  for (int v = threadIdx.x; v < V; v += 256) {
    const float x = bf2f(row[v]) * invt;
    int bin = (int)((x - (mx - TKP_SPAN)) * (TKP_BINS / TKP_SPAN));
    if (bin < 0) continue;  // below span: negligible mass
    if (bin > TKP_BINS - 1) bin = TKP_BINS - 1;
    atomicAdd(&hn[bin], 1);
    atomicAdd(&hm[bin], __expf(x - mx));
  }
  __syncthreads();
  for (int i = threadIdx.x; i < TKP_BINS; i += 256) {
    hist_n[b * TKP_BINS + i] = hn[i];
    hist_m[b * TKP_BINS + i] = hm[i];
  }
}

// pass C: per row, walk bins from the top until top-k count / top-p mass is
// covered -> threshold (in temperature-scaled logit space)
__global__ void __launch_bounds__(64) tkp_threshold_kernel(
    float* __restrict__ thresh, const i32* __restrict__ hist_n,
    const float* __restrict__ hist_m, const float* __restrict__ rmax,
    const float* __restrict__ temp, const i32* __restrict__ topk,
    const float* __restrict__ topp, int V) {
  const int b = blockIdx.x;
  if (threadIdx.x != 0) return;
  const float t = temp[b];
  const float invt = (t > 0.f) ? 1.f / t : 1.f;
  const float mx = rmax[b] * invt;
  float total = 0.f;
  for (int i = 0; i < TKP_BINS; ++i) total += hist_m[b * TKP_BINS + i];
  const int k = (topk[b] > 0) ? topk[b] : V;
  const float p = (topp[b] > 0.f && topp[b] < 1.f) ? topp[b] : 1.f;
  if (k == 1) {  // exact: only the argmax survives
    thresh[b] = mx;
    return;
  }
  int cnt = 0;
  float mass = 0.f;
  float th = mx - TKP_SPAN;
  for (int i = TKP_BINS - 1; i >= 0; --i) {
    cnt += hist_n[b * TKP_BINS + i];
    mass += hist_m[b * TKP_BINS + i];
    if (cnt >= k || mass >= p * total) {
      th = mx - TKP_SPAN + (float)i * (TKP_SPAN / TKP_BINS);
      break;
    }
  }
  thresh[b] = th;
}

// pass D: Gumbel-argmax over {x/T >= thresh}
__global__ void __launch_bounds__(256) tkp_sample_kernel(
    float* __restrict__ pv, i32* __restrict__ pi, const u16* __restrict__ logits,
    const float* __restrict__ temp, const float* __restrict__ thresh,
    const uint32_t* __restrict__ step, uint32_t seed, int V) {
  const int b = blockIdx.x, split = blockIdx.y;
  const int chunk = (V + SMP_SPLITS - 1) / SMP_SPLITS;
  const int v0 = split * chunk, v1 = min(V, v0 + chunk);
  const float t = temp[b];
  const float invt = (t > 0.f) ? 1.f / t : 1.f;
  const float th = thresh[b];
  const uint32_t st = *step;
  float best = AF_NEG_INF;
  int bidx = 0;
  const u16* row = logits + (size_t)b * V;
  for (int v = v0 + threadIdx.x; v < v1; v += 256) {
    float x = bf2f(row[v]) * invt;
    if (x < th) continue;
    if (t > 0.f) {
      const float u = u32_to_unit(hash3(seed ^ st, (uint32_t)b, (uint32_t)v));
      x += gumbel_of(u);
    }
    if (x > best) { best = x; bidx = v; }
  }
#pragma unroll
  for (int o = 32; o > 0; o >>= 1) {
    const float ov = __shfl_xor(best, o, 64);
    const int oi = __shfl_xor(bidx, o, 64);
    if (ov > best || (ov == best && oi < bidx)) { best = ov; bidx = oi; }
  }
  __shared__ float sv[4];
  __shared__ int si[4];
  if ((threadIdx.x & 63) == 0) { sv[threadIdx.x >> 6] = best; si[threadIdx.x >> 6] = bidx; }
  __syncthreads();
  if (threadIdx.x == 0) {
#pragma unroll
    for (int w = 1; w < 4; ++w)
      if (sv[w] > best || (sv[w] == best && si[w] < bidx)) { best = sv[w]; bidx = si[w]; }
    pv[b * SMP_SPLITS + split] = best;
    pi[b * SMP_SPLITS + split] = bidx;
  }
}

// scratch layout for af_sample_topkp: caller provides rmax [B], thresh [B],
// hist_n [B,256] i32, hist_m [B,256] f32 in addition to pv/pi.
AF_EXPORT int af_sample_topkp(void* out, void* pv, void* pi, void* rmax,
                              void* thresh, void* hist_n, void* hist_m,
                              const void* logits, const void* temp,
                              const void* topk, const void* topp, void* step,
                              uint32_t seed, int B, int V, void* stream) {
  if (B == 0) return 0;
  hipStream_t st = (hipStream_t)stream;
  row_max_kernel<<<B, 256, 0, st>>>((float*)rmax, (const u16*)logits, V);
  tkp_hist_kernel<<<B, 256, 0, st>>>((i32*)hist_n, (float*)hist_m,
                                     (const u16*)logits, (const float*)rmax,
                                     (const float*)temp, V);
  tkp_threshold_kernel<<<B, 64, 0, st>>>(
      (float*)thresh, (const i32*)hist_n, (const float*)hist_m,
      (const float*)rmax, (const float*)temp, (const i32*)topk,
      (const float*)topp, V);
  dim3 g(B, SMP_SPLITS);
  tkp_sample_kernel<<<g, 256, 0, st>>>(
      (float*)pv, (i32*)pi, (const u16*)logits, (const float*)temp,
      (const float*)thresh, (const uint32_t*)step, seed, V);
  sample_combine_kernel<<<B, 64, 0, st>>>((i32*)out, (const float*)pv,
                                          (const i32*)pi);
  step_inc_kernel<<<1, 64, 0, st>>>((uint32_t*)step);
  return af_last_err();
}

AF_EXPORT int af_sample(void* out, void* pv, void* pi, const void* logits,
                        const void* temp, void* step, uint32_t seed,
                        int B, int V, void* stream) {
  if (B == 0) return 0;
  hipStream_t st = (hipStream_t)stream;
  dim3 g1(B, SMP_SPLITS);
  sample_partial_kernel<<<g1, 256, 0, st>>>(
      (float*)pv, (i32*)pi, (const u16*)logits, (const float*)temp,
      (const uint32_t*)step, seed, V);
  sample_combine_kernel<<<B, 64, 0, st>>>((i32*)out, (const float*)pv, (const i32*)pi);
  step_inc_kernel<<<1, 64, 0, st>>>((uint32_t*)step);
  return af_last_err();
}

// Gather last-token logits rows: hidden [T,H] -> rows at idx [B] -> out [B,H]
__global__ void __launch_bounds__(256) gather_rows_kernel(
    u16* __restrict__ out, const u16* __restrict__ x, const i32* __restrict__ idx,
    int H) {
  const int b = blockIdx.x;
  const u16* src = x + (size_t)idx[b] * H;
  u16* dst = out + (size_t)b * H;
  for (int i = threadIdx.x * 8; i < H; i += 256 * 8)
    *reinterpret_cast<s16x8*>(dst + i) = *reinterpret_cast<const s16x8*>(src + i);
}

AF_EXPORT int af_gather_rows(void* out, const void* x, const void* idx, int B,
                             int H, void* stream) {
  if (H % 8) return 9001;
  if (B == 0) return 0;
  gather_rows_kernel<<<B, 256, 0, (hipStream_t)stream>>>(
      (u16*)out, (const u16*)x, (const i32*)idx, H);
  return af_last_err();
}
