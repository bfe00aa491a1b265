// Paged-KV-cache scatter: append freshly computed K/V rows into the paged
// cache at slots chosen by the engine's page allocator.
//
// k, v      : [T, Hk*D] bf16 (post-RoPE)
// kc, vc    : [npages, Hk, page_size, D] bf16
// slot[t]   : global slot = page_index*page_size + in_page_offset (i64), -1 skips
#include "common.h"

__global__ void __launch_bounds__(256) reshape_and_cache_kernel(
    const u16* __restrict__ k, const u16* __restrict__ v,
    u16* __restrict__ kc, u16* __restrict__ vc, const i64* __restrict__ slot,
    int T, int Hk, int D, int page_size) {
  const int per_tok = Hk * D / 8;
  for (int t = blockIdx.x; t < T; t += gridDim.x) {
    const i64 s = slot[t];
    if (s < 0) continue;
    const i64 page = s / page_size, off = s % page_size;
    const u16* ksrc = k + (size_t)t * Hk * D;
    const u16* vsrc = v + (size_t)t * Hk * D;
    for (int i = threadIdx.x; i < per_tok; i += blockDim.x) {
      const int h = (i * 8) / D, d = (i * 8) % D;
      const size_t dst = (((size_t)page * Hk + h) * page_size + off) * D + d;
      *reinterpret_cast<s16x8*>(kc + dst) = *reinterpret_cast<const s16x8*>(ksrc + i * 8);
      *reinterpret_cast<s16x8*>(vc + dst) = *reinterpret_cast<const s16x8*>(vsrc + i * 8);
    }
  }
}

AF_EXPORT int af_reshape_and_cache(const void* k, const void* v, void* kc, void* vc,
                                   const void* slot, int T, int Hk, int D,
                                   int page_size, void* stream) {
  if (D % 8) return 9001;
  if (T == 0) return 0;
  int blocks = T < 2048 ? T : 2048;
  reshape_and_cache_kernel<<<blocks, 256, 0, (hipStream_t)stream>>>(
      (const u16*)k, (const u16*)v, (u16*)kc, (u16*)vc, (const i64*)slot,
      T, Hk, D, page_size);
  return af_last_err();
}

// Gather rows of an i32 embedding-free path is not needed; but the engine uses
// an embedding gather for input ids -> hidden states.
// emb: [V, H] bf16, ids: [T] i32, out: [T, H] bf16.  ss (optional) gets the
// per-row sum of squares, seeding the fused-norm decode path.
__global__ void __launch_bounds__(256) embedding_kernel(
    u16* __restrict__ out, const u16* __restrict__ emb, const i32* __restrict__ ids,
    float* __restrict__ ss, int T, int H) {
  const int per_row = H / 8;
  __shared__ float red[4];
  for (int t = blockIdx.x; t < T; t += gridDim.x) {
    const u16* src = emb + (size_t)ids[t] * H;
    u16* dst = out + (size_t)t * H;
    float local = 0.f;
    for (int i = threadIdx.x; i < per_row; i += blockDim.x) {
      s16x8 v = *reinterpret_cast<const s16x8*>(src + i * 8);
      *reinterpret_cast<s16x8*>(dst + i * 8) = v;
      if (ss) {
#pragma unroll
        for (int j = 0; j < 8; ++j) {
          const float f = bf2f((u16)v[j]);
          local += f * f;
        }
      }
    }
    if (ss) {
      local = wave_sum_f32(local);
      if ((threadIdx.x & 63) == 0) red[threadIdx.x >> 6] = local;
      __syncthreads();
      if (threadIdx.x < 8)  // [T][8] column-block layout; total in slot 0
        ss[t * 8 + threadIdx.x] =
            threadIdx.x == 0 ? red[0] + red[1] + red[2] + red[3] : 0.f;
      __syncthreads();
    }
  }
}

AF_EXPORT int af_embedding(void* out, const void* emb, const void* ids,
                           void* ss, int T, int H, void* stream) {
  if (H % 8) return 9001;
  if (T == 0) return 0;
  int blocks = T < 2048 ? T : 2048;
  embedding_kernel<<<blocks, 256, 0, (hipStream_t)stream>>>(
      (u16*)out, (const u16*)emb, (const i32*)ids, (float*)ss, T, H);
  return af_last_err();
}
