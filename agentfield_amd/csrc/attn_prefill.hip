// Paged causal prefill attention (flash-style, GQA) with MFMA 16x16x32 bf16.
//
// q      : chunk rows at stride qs (view into the fused QKV buffer), head h
//          at offset h*D, D=128, packed by sequence (cu_seqlens of the CHUNK).
// kc/vc  : the paged KV cache [npages, Hk, page_size, D] — rope_cache has
//          already appended the chunk, so attention reads history + chunk
//          uniformly through the block table (chunked prefill falls out).
// qstart : per-seq absolute position of the chunk's first row (0 for a
//          full-prompt prefill; >0 when continuing a chunked prefill).
// out    : [chunk_T, Hq*D] bf16 contiguous.
// Host precomputes a flat tile map (tile_seq[i], tile_q0[i]): workgroup i on
// grid.x handles chunk rows [q0, q0+ROWS) of sequence tile_seq[i];
// grid.y = kv head.
//
// Workgroup = 256 threads = 4 waves.  Each wave owns a 16-row q-tile of one
// query head in the GQA group:
//   G>=4 : all waves share one 16-row tile; wave w covers heads g=w, w+4, ...
//   G<4  : 4/G row-tiles per workgroup; wave w -> (g = w%G, tile = w/G)
// K/V tiles (KVBLK=64 tokens) are staged in LDS once per workgroup and shared
// by the 4 waves; 64-token tiles halve the barrier count per token vs 32.
//
// CDNA4 specifics (see /opt/skills/guides/cdna_hip_programming.md):
//  - K LDS tile is XOR-swizzled (byte ^= (row&7)<<4) so the B-fragment
//    ds_read_b128 is bank-conflict-free (guide §6 G4: row-major [64][128]
//    would be a 32-way conflict).
//  - V is stored transposed [D][KVBLK+pad] with token pairs packed as u32
//    ds_writes (2 tokens per 4 B slot) so staging is 8 b32 writes per two
//    16 B loads; pad 64->72 spreads banks on the PV-fragment reads.
//  - P (scores) round-trips through a small per-wave LDS tile to convert the
//    MFMA C-layout into the A-fragment layout.
//  - fragment layouts (guide §3, measured, validated by ops.mfma_probe):
//      A: lane holds A[m=lane&15][k=(lane>>4)*8+j], j=0..7
//      B: lane holds B[k=(lane>>4)*8+j][n=lane&15]
//      C: lane reg r holds C[m=(lane>>4)*4+r][n=lane&15]
#include "common.h"

#define AP_D 128
#define AP_KVBLK 64
#define AP_VPAD 72  // 64 tokens padded to 72 (residual 2-way conflicts are free, guide m136)

typedef __attribute__((ext_vector_type(8))) __bf16 bf16x8;

__device__ __forceinline__ bf16x8 as_bf16x8(s16x8 v) {
  union { s16x8 s; bf16x8 b; } u;
  u.s = v;
  return u.b;
}

__device__ __forceinline__ s16x8 lds_read8(const u16* p) {
  return *reinterpret_cast<const s16x8*>(p);
}

template <int G>
__global__ void __launch_bounds__(256) attn_prefill_kernel(
    u16* __restrict__ out, const u16* __restrict__ q,
    const u16* __restrict__ kc, const u16* __restrict__ vc,
    const i32* __restrict__ bt, const i32* __restrict__ qstart,
    const i32* __restrict__ cu_seqlens,
    const i32* __restrict__ tile_seq, const i32* __restrict__ tile_q0,
    float scale, int Hq, int Hk, i64 qs, int page_size, int max_pages,
    int win) {
  const int tile = blockIdx.x, kvh = blockIdx.y;
  const int wid = threadIdx.x >> 6, lane = threadIdx.x & 63;
  const int seq = tile_seq[tile];
  const int seq_start = cu_seqlens[seq];
  const int len = cu_seqlens[seq + 1] - seq_start;  // chunk rows
  const int hist = qstart[seq];                     // cached tokens before chunk
  const i32* btrow = bt + (size_t)seq * max_pages;

  constexpr int ROWTILES = (G >= 4) ? 1 : (4 / G);   // 16-row tiles per WG
  constexpr int ROWS = 16 * ROWTILES;
  constexpr int HLOOP = (G >= 4) ? (G / 4) : 1;      // heads per wave

  const int my_tile = (G >= 4) ? 0 : (wid / G);
  const int g0 = (G >= 4) ? wid : (wid % G);
  const int q0 = tile_q0[tile] + my_tile * 16;        // abs q row of wave tile

  __shared__ u16 k_lds[AP_KVBLK * AP_D];              // swizzled rows
  __shared__ u16 v_lds[AP_D][AP_VPAD];                // transposed, tok-paired
  __shared__ u16 p_lds[4][16][AP_VPAD];               // per-wave P tile

  // ---- load Q fragments (stay in registers for all KV tiles) ----
  const int qrow_frag = min(q0 + (lane & 15), len - 1);
  s16x8 q_frag[HLOOP][4];
#pragma unroll
  for (int hl = 0; hl < HLOOP; ++hl) {
    const int qh = kvh * G + g0 + hl * 4;
    const u16* qp = q + (size_t)(seq_start + qrow_frag) * qs + qh * AP_D +
                    (lane >> 4) * 8;
#pragma unroll
    for (int c = 0; c < 4; ++c) q_frag[hl][c] = lds_read8(qp + c * 32);
  }

  float m[HLOOP][4], l[HLOOP][4];
  f32x4 o_acc[HLOOP][8];  // 8 d-tiles of 16
#pragma unroll
  for (int hl = 0; hl < HLOOP; ++hl) {
#pragma unroll
    for (int r = 0; r < 4; ++r) { m[hl][r] = AF_NEG_INF; l[hl][r] = 0.f; }
#pragma unroll
    for (int dt = 0; dt < 8; ++dt) o_acc[hl][dt] = f32x4{0.f, 0.f, 0.f, 0.f};
  }

  // tokens visible to this WG: history + chunk rows up to the causal bound,
  // and (sliding window) nothing below qrow_min - win + 1
  const int kv_end = min(hist + len, hist + tile_q0[tile] + ROWS);
  const int n_kv_tiles = (kv_end + AP_KVBLK - 1) / AP_KVBLK;
  const int kv_total = hist + len;
  const int kt_first =
      (win > 0) ? max(0, hist + tile_q0[tile] - win + 1) / AP_KVBLK : 0;

  for (int kt = kt_first; kt < n_kv_tiles; ++kt) {
    const int kv0 = kt * AP_KVBLK;
    __syncthreads();  // previous iteration's frag reads done
    // ---- stage K (swizzled) from the paged cache ----
    for (int i = threadIdx.x; i < AP_KVBLK * (AP_D / 8); i += 256) {
      const int tok = i / (AP_D / 8);
      const int d8 = (i % (AP_D / 8)) * 8;
      const int tg = min(kv0 + tok, kv_total - 1);
      const i64 page = btrow[tg / page_size];
      const size_t src =
          (((size_t)page * Hk + kvh) * page_size + (tg % page_size)) * AP_D + d8;
      s16x8 kv8 = lds_read8(kc + src);
      const int byte = tok * (AP_D * 2) + ((d8 * 2) ^ ((tok & 7) << 4));
      *reinterpret_cast<s16x8*>(reinterpret_cast<char*>(k_lds) + byte) = kv8;
    }
    // ---- stage V transposed: token pairs -> u32 writes (8 per 2 loads).
    // Lane decomposition: consecutive lanes take consecutive token pairs at
    // the SAME d-slice, so the 8 v_lds writes land in consecutive words
    // (bank-conflict-free).  The d8-major variant put 16 lanes on one bank
    // (d8 stride x 36-word rows = 0 mod 32): PMC measured 1.7e8
    // SQ_LDS_BANK_CONFLICT per dispatch (profiles/pmc_r01_counters.txt).
    for (int i = threadIdx.x; i < (AP_KVBLK / 2) * (AP_D / 8); i += 256) {
      const int tp = i % (AP_KVBLK / 2);      // token pair (consecutive lanes)
      const int d8 = (i / (AP_KVBLK / 2)) * 8;
      const int t0g = kv0 + tp * 2;
      const int c0 = min(t0g, kv_total - 1), c1 = min(t0g + 1, kv_total - 1);
      const size_t s0 = (((size_t)btrow[c0 / page_size] * Hk + kvh) * page_size +
                         (c0 % page_size)) * AP_D + d8;
      const size_t s1 = (((size_t)btrow[c1 / page_size] * Hk + kvh) * page_size +
                         (c1 % page_size)) * AP_D + d8;
      s16x8 a = (t0g < kv_total) ? lds_read8(vc + s0)
                                 : s16x8{0, 0, 0, 0, 0, 0, 0, 0};
      s16x8 b = (t0g + 1 < kv_total) ? lds_read8(vc + s1)
                                     : s16x8{0, 0, 0, 0, 0, 0, 0, 0};
#pragma unroll
      for (int j = 0; j < 8; ++j) {
        const uint32_t packed = (uint32_t)(unsigned short)a[j] |
                                ((uint32_t)(unsigned short)b[j] << 16);
        *reinterpret_cast<uint32_t*>(&v_lds[d8 + j][tp * 2]) = packed;
      }
    }
    __syncthreads();

#pragma unroll
    for (int hl = 0; hl < HLOOP; ++hl) {
      // ---- QK^T: four 16-token sub-tiles ----
      f32x4 s_acc[4];
#pragma unroll
      for (int st = 0; st < 4; ++st) {
        s_acc[st] = f32x4{0, 0, 0, 0};
        const int tok = st * 16 + (lane & 15);
#pragma unroll
        for (int c = 0; c < 4; ++c) {
          const int byte = tok * (AP_D * 2) +
                           (((c * 32 + (lane >> 4) * 8) * 2) ^ ((tok & 7) << 4));
          s16x8 kf = *reinterpret_cast<const s16x8*>(
              reinterpret_cast<const char*>(k_lds) + byte);
          s_acc[st] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
              as_bf16x8(q_frag[hl][c]), as_bf16x8(kf), s_acc[st], 0, 0, 0);
        }
      }
      // ---- mask + online softmax (C-layout: reg r = qrow (lane>>4)*4+r) ----
#pragma unroll
      for (int r = 0; r < 4; ++r) {
        const int qrow = hist + q0 + (lane >> 4) * 4 + r;  // absolute position
        float sr[4];
        float tmax = AF_NEG_INF;
#pragma unroll
        for (int st = 0; st < 4; ++st) {
          sr[st] = s_acc[st][r] * scale;
          const int kpos = kv0 + st * 16 + (lane & 15);
          if (kpos > qrow || (win > 0 && kpos <= qrow - win))
            sr[st] = AF_NEG_INF;
          tmax = fmaxf(tmax, sr[st]);
        }
        tmax = group16_max_f32(tmax);
        const float mn = fmaxf(m[hl][r], tmax);
        const float corr = (m[hl][r] <= AF_NEG_INF) ? 0.f : __expf(m[hl][r] - mn);
        float psum = 0.f;
#pragma unroll
        for (int st = 0; st < 4; ++st) {
          sr[st] = (sr[st] <= AF_NEG_INF) ? 0.f : __expf(sr[st] - mn);
          psum += sr[st];
          s_acc[st][r] = sr[st];  // reuse acc regs to carry P
        }
        l[hl][r] = l[hl][r] * corr + group16_sum_f32(psum);
        m[hl][r] = mn;
#pragma unroll
        for (int dt = 0; dt < 8; ++dt) o_acc[hl][dt][r] *= corr;
      }
      // ---- P -> per-wave LDS (C-layout -> A-fragment layout) ----
#pragma unroll
      for (int st = 0; st < 4; ++st)
#pragma unroll
        for (int r = 0; r < 4; ++r)
          p_lds[wid][(lane >> 4) * 4 + r][st * 16 + (lane & 15)] =
              f2bf(s_acc[st][r]);
      asm volatile("s_waitcnt lgkmcnt(0)" ::: "memory");
      s16x8 p_frag0 = lds_read8(&p_lds[wid][lane & 15][(lane >> 4) * 8]);
      s16x8 p_frag1 = lds_read8(&p_lds[wid][lane & 15][32 + (lane >> 4) * 8]);
      // ---- PV: o_acc[dt] += P(16x64) @ V(64x16), two K=32 chunks ----
#pragma unroll
      for (int dt = 0; dt < 8; ++dt) {
        const u16* vrow = &v_lds[dt * 16 + (lane & 15)][0];
        s16x8 vf0 = lds_read8(vrow + (lane >> 4) * 8);
        s16x8 vf1 = lds_read8(vrow + 32 + (lane >> 4) * 8);
        o_acc[hl][dt] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
            as_bf16x8(p_frag0), as_bf16x8(vf0), o_acc[hl][dt], 0, 0, 0);
        o_acc[hl][dt] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
            as_bf16x8(p_frag1), as_bf16x8(vf1), o_acc[hl][dt], 0, 0, 0);
      }
    }
  }

  // ---- epilogue: out = o_acc / l ----
#pragma unroll
  for (int hl = 0; hl < HLOOP; ++hl) {
    const int qh = kvh * G + g0 + hl * 4;
#pragma unroll
    for (int r = 0; r < 4; ++r) {
      const int qrow = q0 + (lane >> 4) * 4 + r;
      if (qrow >= len) continue;
      const float inv = (l[hl][r] > 0.f) ? 1.f / l[hl][r] : 0.f;
      u16* op = out + (size_t)(seq_start + qrow) * (Hq * AP_D) + qh * AP_D +
                (lane & 15);
#pragma unroll
      for (int dt = 0; dt < 8; ++dt) op[dt * 16] = f2bf(o_acc[hl][dt][r] * inv);
    }
  }
}

AF_EXPORT int af_attn_prefill(void* out, const void* q, const void* kc,
                              const void* vc, const void* bt, const void* qstart,
                              const void* cu_seqlens, const void* tile_seq,
                              const void* tile_q0, float scale, int ntiles,
                              int Hq, int Hk, int D, i64 qs, int page_size,
                              int max_pages, int win, void* stream) {
  if (D != AP_D) return 9002;
  const int G = Hq / Hk;
  if (G * Hk != Hq) return 9003;
  if (ntiles == 0) return 0;
  dim3 grid(ntiles, Hk), blk(256);
  hipStream_t st = (hipStream_t)stream;
#define AF_LAUNCH(GG)                                                           \
  attn_prefill_kernel<GG><<<grid, blk, 0, st>>>(                                \
      (u16*)out, (const u16*)q, (const u16*)kc, (const u16*)vc,                 \
      (const i32*)bt, (const i32*)qstart,                                       \
      (const i32*)cu_seqlens, (const i32*)tile_seq, (const i32*)tile_q0,        \
      scale, Hq, Hk, qs, page_size, max_pages, win)
  switch (G) {
    case 1: AF_LAUNCH(1); break;
    case 2: AF_LAUNCH(2); break;
    case 4: AF_LAUNCH(4); break;
    case 8: AF_LAUNCH(8); break;
    default: return 9003;
  }
#undef AF_LAUNCH
  return af_last_err();
}
