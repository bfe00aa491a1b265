// Causal varlen prefill attention (flash-style, GQA) with MFMA 16x16x32 bf16.
//
// q/k/v : [total_T, H*, D=128] bf16 packed by sequence (cu_seqlens-style),
//         post-RoPE.  out: [total_T, Hq, D] bf16.
// Host precomputes a flat tile map (tile_seq[i], tile_q0[i]): workgroup i on
// grid.x handles q rows [q0, q0+ROWS) of sequence tile_seq[i]; grid.y = kv head.
//
// Workgroup = 256 threads = 4 waves.  Each wave owns a 16-row q-tile of one
// query head in the GQA group:
//   G>=4 : all waves share one 16-row tile; wave w covers heads g=w, w+4, ...
//   G<4  : 4/G row-tiles per workgroup; wave w -> (g = w%G, tile = w/G)
// K/V tiles (KVBLK=32 tokens) are staged in LDS once per workgroup and shared.
//
// CDNA4 specifics (see /opt/skills/guides/cdna_hip_programming.md):
//  - K LDS tile is XOR-swizzled (byte ^= (row&7)<<4) so the B-fragment
//    ds_read_b128 is bank-conflict-free (guide §6 G4: row-major [32][128]
//    would be a 32-way conflict).
//  - V is stored transposed [D][KVBLK+pad] so the PV B-fragment read is a
//    contiguous 16 B ds_read; pad 32->40 spreads banks.
//  - P (scores) round-trips through a small per-wave LDS tile to convert the
//    MFMA C-layout into the A-fragment layout.
//  - fragment layouts (guide §3, measured):
//      A: lane holds A[m=lane&15][k=(lane>>4)*8+j], j=0..7
//      B: lane holds B[k=(lane>>4)*8+j][n=lane&15]
//      C: lane reg r holds C[m=(lane>>4)*4+r][n=lane&15]
#include "common.h"

#define AP_D 128
#define AP_KVBLK 32
#define AP_VPAD 40  // 32 tokens padded to 40 (80 B row stride)

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
    u16* __restrict__ out, const u16* __restrict__ q, const u16* __restrict__ k,
    const u16* __restrict__ v, const i32* __restrict__ cu_seqlens,
    const i32* __restrict__ tile_seq, const i32* __restrict__ tile_q0,
    float scale, int Hq, int Hk, i64 qs, i64 ks, i64 vs) {
  const int tile = blockIdx.x, kvh = blockIdx.y;
  const int wid = threadIdx.x >> 6, lane = threadIdx.x & 63;
  const int seq = tile_seq[tile];
  const int seq_start = cu_seqlens[seq];
  const int len = cu_seqlens[seq + 1] - seq_start;

  constexpr int ROWTILES = (G >= 4) ? 1 : (4 / G);   // 16-row tiles per WG
  constexpr int ROWS = 16 * ROWTILES;
  constexpr int HLOOP = (G >= 4) ? (G / 4) : 1;      // heads per wave

  const int my_tile = (G >= 4) ? 0 : (wid / G);
  const int g0 = (G >= 4) ? wid : (wid % G);
  const int q0 = tile_q0[tile] + my_tile * 16;        // abs q row of wave tile
  if (q0 >= len) {
    // whole wave's tile out of range; still must participate in staging
    // barriers, so fall through with masked rows (q0 clamp below).
  }

  __shared__ u16 k_lds[AP_KVBLK * AP_D];              // swizzled rows
  __shared__ u16 v_lds[AP_D][AP_VPAD];                // transposed
  __shared__ u16 p_lds[4][16][AP_VPAD];               // per-wave P tile

  // ---- load Q fragments (stay in registers for all KV tiles) ----
  // lane holds Q[q0 + (lane&15)][c*32 + (lane>>4)*8 .. +8] for c=0..3
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

  const int kv_end = min(len, tile_q0[tile] + ROWS);  // causal upper bound (max over WG waves)
  const int n_kv_tiles = (kv_end + AP_KVBLK - 1) / AP_KVBLK;

  for (int kt = 0; kt < n_kv_tiles; ++kt) {
    const int kv0 = kt * AP_KVBLK;
    __syncthreads();  // previous iteration's frag reads done
    // ---- stage K (swizzled) and V (transposed) ----
    // 32 tokens x 128 dims / 8 per chunk = 512 chunks; 256 threads x 2 iters
    for (int i = threadIdx.x; i < AP_KVBLK * (AP_D / 8); i += 256) {
      const int tok = i / (AP_D / 8);
      const int d8 = (i % (AP_D / 8)) * 8;
      const int tg = kv0 + tok;
      const size_t kvrow = (size_t)(seq_start + min(tg, len - 1));
      {  // K: row-major swizzled, garbage beyond len is masked later
        s16x8 kv8 = lds_read8(k + kvrow * ks + kvh * AP_D + d8);
        const int byte = tok * (AP_D * 2) + ((d8 * 2) ^ ((tok & 7) << 4));
        *reinterpret_cast<s16x8*>(reinterpret_cast<char*>(k_lds) + byte) = kv8;
      }
      {  // V: transposed, zero-filled beyond len (0 * P avoids NaN)
        s16x8 vv8 = (tg < len) ? lds_read8(v + kvrow * vs + kvh * AP_D + d8)
                               : s16x8{0, 0, 0, 0, 0, 0, 0, 0};
#pragma unroll
        for (int j = 0; j < 8; ++j) v_lds[d8 + j][tok] = (u16)vv8[j];
      }
    }
    __syncthreads();

#pragma unroll
    for (int hl = 0; hl < HLOOP; ++hl) {
      // ---- QK^T: two 16-token sub-tiles ----
      f32x4 s_acc[2] = {f32x4{0, 0, 0, 0}, f32x4{0, 0, 0, 0}};
#pragma unroll
      for (int st = 0; st < 2; ++st) {
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
      // ---- mask + online softmax (rows spread: reg r = qrow (lane>>4)*4+r) ----
      float p[2][4];
#pragma unroll
      for (int r = 0; r < 4; ++r) {
        const int qrow = q0 + (lane >> 4) * 4 + r;
        float s0 = s_acc[0][r] * scale, s1 = s_acc[1][r] * scale;
        if (kv0 + (lane & 15) > qrow) s0 = AF_NEG_INF;
        if (kv0 + 16 + (lane & 15) > qrow) s1 = AF_NEG_INF;
        const float tmax = group16_max_f32(fmaxf(s0, s1));
        const float mn = fmaxf(m[hl][r], tmax);
        const float corr = (m[hl][r] <= AF_NEG_INF) ? 0.f : __expf(m[hl][r] - mn);
        p[0][r] = (s0 <= AF_NEG_INF) ? 0.f : __expf(s0 - mn);
        p[1][r] = (s1 <= AF_NEG_INF) ? 0.f : __expf(s1 - mn);
        l[hl][r] = l[hl][r] * corr + group16_sum_f32(p[0][r] + p[1][r]);
        m[hl][r] = mn;
#pragma unroll
        for (int dt = 0; dt < 8; ++dt) o_acc[hl][dt][r] *= corr;
      }
      // ---- P -> per-wave LDS (C-layout -> A-fragment layout) ----
#pragma unroll
      for (int st = 0; st < 2; ++st)
#pragma unroll
        for (int r = 0; r < 4; ++r)
          p_lds[wid][(lane >> 4) * 4 + r][st * 16 + (lane & 15)] = f2bf(p[st][r]);
      asm volatile("s_waitcnt lgkmcnt(0)" ::: "memory");
      s16x8 p_frag = lds_read8(&p_lds[wid][lane & 15][(lane >> 4) * 8]);
      // ---- PV: o_acc[dt] += P(16x32) @ V(32x16) ----
#pragma unroll
      for (int dt = 0; dt < 8; ++dt) {
        s16x8 vf = lds_read8(&v_lds[dt * 16 + (lane & 15)][0] + (lane >> 4) * 8);
        o_acc[hl][dt] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
            as_bf16x8(p_frag), as_bf16x8(vf), o_acc[hl][dt], 0, 0, 0);
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
      u16* op = out + ((size_t)(seq_start + qrow) * Hq + qh) * AP_D + (lane & 15);
#pragma unroll
      for (int dt = 0; dt < 8; ++dt) op[dt * 16] = f2bf(o_acc[hl][dt][r] * inv);
    }
  }
}

AF_EXPORT int af_attn_prefill(void* out, const void* q, const void* k, const void* v,
                              const void* cu_seqlens, const void* tile_seq,
                              const void* tile_q0, float scale, int ntiles,
                              int Hq, int Hk, int D, i64 qs, i64 ks, i64 vs,
                              void* stream) {
  if (D != AP_D) return 9002;
  const int G = Hq / Hk;
  if (G * Hk != Hq) return 9003;
  if (ntiles == 0) return 0;
  dim3 grid(ntiles, Hk), blk(256);
  hipStream_t st = (hipStream_t)stream;
#define AF_LAUNCH(GG)                                                           \
  attn_prefill_kernel<GG><<<grid, blk, 0, st>>>(                                \
      (u16*)out, (const u16*)q, (const u16*)k, (const u16*)v,                   \
      (const i32*)cu_seqlens, (const i32*)tile_seq, (const i32*)tile_q0,        \
      scale, Hq, Hk, qs, ks, vs)
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
