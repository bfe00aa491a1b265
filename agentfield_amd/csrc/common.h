// Common device helpers for the agentfield_amd CDNA4 (gfx950) kernel library.
//
// Design notes (MI355X-first, see /opt/skills/guides/cdna_hip_programming.md):
//  - wavefront = 64 lanes; all cross-lane idioms use 64-wide shuffles
//  - bf16 handled as raw u16 bits; fp32 accumulate everywhere
//  - vectorized loads via ext_vector_type shorts (8 bf16 = 16 B per lane)
#pragma once
#include <hip/hip_runtime.h>
#include <stdint.h>

#define AF_EXPORT extern "C" __attribute__((visibility("default")))

typedef uint16_t u16;
typedef int32_t i32;
typedef int64_t i64;

typedef __attribute__((ext_vector_type(2))) float f32x2;
typedef __attribute__((ext_vector_type(4))) float f32x4;
typedef __attribute__((ext_vector_type(2))) short s16x2;
typedef __attribute__((ext_vector_type(4))) short s16x4;
typedef __attribute__((ext_vector_type(8))) short s16x8;  // 8 bf16 = 16 B
typedef __attribute__((ext_vector_type(2))) unsigned short u16x2;

// ---- bf16 <-> fp32 (round-to-nearest-even, matches torch semantics) ----
__device__ __forceinline__ float bf2f(u16 v) {
  union { uint32_t u; float f; } c;
  c.u = (uint32_t)v << 16;
  return c.f;
}

__device__ __forceinline__ u16 f2bf(float f) {
  union { float f; uint32_t u; } c;
  c.f = f;
  if ((c.u & 0x7fffffffu) > 0x7f800000u) return (u16)((c.u >> 16) | 0x0040);  // NaN
  uint32_t lsb = (c.u >> 16) & 1u;
  return (u16)((c.u + 0x7fffu + lsb) >> 16);
}

// ---- wave (64-lane) reductions: every lane ends with the result ----
__device__ __forceinline__ float wave_sum_f32(float v) {
#pragma unroll
  for (int o = 32; o > 0; o >>= 1) v += __shfl_xor(v, o, 64);
  return v;
}

__device__ __forceinline__ float wave_max_f32(float v) {
#pragma unroll
  for (int o = 32; o > 0; o >>= 1) v = fmaxf(v, __shfl_xor(v, o, 64));
  return v;
}

// Reduction within 16-lane groups (used for MFMA 16x16 row reductions).
__device__ __forceinline__ float group16_max_f32(float v) {
#pragma unroll
  for (int o = 8; o > 0; o >>= 1) v = fmaxf(v, __shfl_xor(v, o, 64));
  return v;
}

__device__ __forceinline__ float group16_sum_f32(float v) {
#pragma unroll
  for (int o = 8; o > 0; o >>= 1) v += __shfl_xor(v, o, 64);
  return v;
}

#define AF_NEG_INF (-3.0e38f)

// XCD-aware bijective blockIdx swizzle (8 XCDs on MI355X).
// Maps round-robin dispatch onto contiguous per-XCD chunks for L2 locality.
__device__ __forceinline__ int xcd_swizzle(int bid, int nwg) {
  const int NXCD = 8;
  if (nwg <= NXCD) return bid;
  int q = nwg / NXCD, r = nwg % NXCD;
  int xcd = bid % NXCD, idx = bid / NXCD;
  int base = (xcd < r) ? xcd * (q + 1) : r * (q + 1) + (xcd - r) * q;
  return base + idx;
}

// Supertile variant: map the 32 CONCURRENT workgroups of one XCD (1 WG/CU,
// 32 CUs/XCD) onto an 8x4 tile block instead of one 32-wide tile row.  The
// concurrent set then touches 8 A-stripes + 4 B-stripes per K-slice instead
// of 1 + 32, cutting L2-miss traffic ~2.7x for K-synchronized walks.
// Bijective when nwg % 256 == 0 and the grid divides 8x4; callers fall back
// to xcd_swizzle otherwise.
__device__ __forceinline__ void xcd_tile_map(int bid, int tiles_m,
                                             int tiles_n, int order,
                                             int* tm, int* tn) {
  const int nwg = tiles_m * tiles_n;
  // order encodes the supertile aspect: 2 = 8x4, 3 = 4x8, 4 = 16x2
  const int sm = (order == 3) ? 4 : (order == 4) ? 16 : 8;
  const int sn = 32 / sm;
  if (order >= 2 && tiles_m % sm == 0 && tiles_n % sn == 0 &&
      (nwg & 255) == 0) {
    const int xcd = bid & 7, idx = bid >> 3;
    const int within = idx & 31, super_seq = idx >> 5;
    const int sid = super_seq * 8 + xcd;      // supertile of sm x sn tiles
    const int nsup_n = tiles_n / sn;
    *tm = (sid / nsup_n) * sm + within % sm;
    *tn = (sid % nsup_n) * sn + within / sm;
    return;
  }
  const int b = xcd_swizzle(bid, nwg);
  *tm = b / tiles_n;
  *tn = b % tiles_n;
}

static inline int af_last_err() { return (int)hipGetLastError(); }
