// MFMA layout probe: single-wave 16x16x32 bf16 matmul using the assumed
// fragment layouts (guide §3).  Host checks D == A@B against torch fp32 with
// asymmetric random inputs (guide G9: symmetric inputs can't catch transposes).
// A: [16,32] row-major bf16, B: [32,16] row-major bf16, D: [16,16] row-major f32.
#include "common.h"

typedef __attribute__((ext_vector_type(8))) __bf16 bf16x8p;

__global__ void mfma_probe_kernel(float* __restrict__ D, const u16* __restrict__ A,
                                  const u16* __restrict__ B) {
  const int lane = threadIdx.x & 63;
  union { s16x8 s; bf16x8p b; } a, b8;
#pragma unroll
  for (int j = 0; j < 8; ++j) {
    // A[m = lane&15][k = (lane>>4)*8 + j]
    a.s[j] = (short)A[(lane & 15) * 32 + (lane >> 4) * 8 + j];
    // B[k = (lane>>4)*8 + j][n = lane&15]
    b8.s[j] = (short)B[((lane >> 4) * 8 + j) * 16 + (lane & 15)];
  }
  f32x4 acc = {0, 0, 0, 0};
  acc = __builtin_amdgcn_mfma_f32_16x16x32_bf16(a.b, b8.b, acc, 0, 0, 0);
#pragma unroll
  for (int r = 0; r < 4; ++r)
    // C[m = (lane>>4)*4 + r][n = lane&15]
    D[((lane >> 4) * 4 + r) * 16 + (lane & 15)] = acc[r];
}

AF_EXPORT int af_mfma_probe(void* D, const void* A, const void* B, void* stream) {
  mfma_probe_kernel<<<1, 64, 0, (hipStream_t)stream>>>(
      (float*)D, (const u16*)A, (const u16*)B);
  return af_last_err();
}

// MX-fp8 layout probe: single-wave 16x16x128 e4m3 matmul via the scaled
// MFMA (hardware-fused dequant).  Layouts mapped EMPIRICALLY on gfx950
// (tools/mx_probe*.py, block-one-hot + single-byte scale perturbation):
//   C: reg r -> C[m = (lane>>4)*4 + r][n = lane&15]  (shape-determined)
//   scale: byte 0 (op_sel 0) of scale lane (m|n, s = lane>>4) covers
//     hw K-block s (32 consecutive hw k) of that row/col; E8M0.
//   data: lane (m, g = lane>>4), byte j maps to
//     hw k = 32*(2*(j>>4) + (g>>1)) + 16*(g&1) + (j&15)
//   i.e. a lane's 32 bytes SPAN TWO hw blocks (halves j<16 and j>=16) —
//   the naive "lane g = block g" guess mispairs data with scales (it
//   cancels for uniform scales, which is why data-only probes passed).
// Loading with the inverse mapping below makes memory-k == hw-k, so a
// per-32-memory-block scale vector works: sa[row][s] at lane (row, s).
// A: [16,128] row-major e4m3 bytes, sa: [16,4] E8M0 per (row, k-block);
// B: [128,16] col-read,         sb: [16,4] per (col, k-block); D [16,16] f32.
#include <hip/hip_fp8.h>

typedef __attribute__((ext_vector_type(8))) int i32x8p;

__device__ __forceinline__ int mx_hw_k(int g, int j) {
  return 32 * (2 * (j >> 4) + (g >> 1)) + 16 * (g & 1) + (j & 15);
}

__global__ void mfma_mx_probe_kernel(float* __restrict__ D,
                                     const unsigned char* __restrict__ A,
                                     const unsigned char* __restrict__ B,
                                     const int* __restrict__ SA,
                                     const int* __restrict__ SB) {
  const int lane = threadIdx.x & 63;
  const int m = lane & 15, kb = lane >> 4;
  union { unsigned char u8[32]; i32x8p v; } a, b;
#pragma unroll
  for (int j = 0; j < 32; ++j) {
    const int k = mx_hw_k(kb, j);
    a.u8[j] = A[m * 128 + k];
    b.u8[j] = B[k * 16 + m];
  }
  const int sa = SA[lane];  // full i32 scale operand, host-crafted per lane
  const int sb = SB[lane];
  f32x4 acc = {0, 0, 0, 0};
  acc = __builtin_amdgcn_mfma_scale_f32_16x16x128_f8f6f4(
      a.v, b.v, acc, 0, 0, 0, sa, 0, sb);
#pragma unroll
  for (int r = 0; r < 4; ++r)
    D[((lane >> 4) * 4 + r) * 16 + (lane & 15)] = acc[r];
}

AF_EXPORT int af_mfma_mx_probe(void* D, const void* A, const void* B,
                               const void* SA, const void* SB, void* stream) {
  mfma_mx_probe_kernel<<<1, 64, 0, (hipStream_t)stream>>>(
      (float*)D, (const unsigned char*)A, (const unsigned char*)B,
      (const int*)SA, (const int*)SB);
  return af_last_err();
}

// MX-fp8 32x32x64 layout probe (analog of the 16x16x128 probe): assumed
//   A: lane (m = lane&31, g = lane>>5) byte j -> hw k with the same
//      half-interleave: k = 16*(2*(j>>4) + ???)  — start from the naive
//      guess k = 32*g + j and let the host diagnose via scales.
//   C: reg r -> C[row = (r&3) + 8*(r>>2) + 4*(lane>>5)][col = lane&31]
//   scale: lane (m, s = lane>>5) byte0 covers hw K-block s (2 blocks)
__global__ void mfma_mx32_probe_kernel(float* __restrict__ D,
                                       const unsigned char* __restrict__ A,
                                       const unsigned char* __restrict__ B,
                                       const int* __restrict__ SA,
                                       const int* __restrict__ SB,
                                       int layout) {
  const int lane = threadIdx.x & 63;
  const int m = lane & 31, g = lane >> 5;
  union { unsigned char u8[32]; i32x8p v; } a, b;
#pragma unroll
  for (int j = 0; j < 32; ++j) {
    int k;
    if (layout == 0) k = g * 32 + j;                       // naive
    else k = 16 * (2 * (j >> 4) + g) + (j & 15);           // interleave
    a.u8[j] = A[m * 64 + k];
    b.u8[j] = B[k * 32 + m];
  }
  typedef __attribute__((ext_vector_type(16))) float f32x16p;
  f32x16p acc;
#pragma unroll
  for (int r = 0; r < 16; ++r) acc[r] = 0.f;
  acc = __builtin_amdgcn_mfma_scale_f32_32x32x64_f8f6f4(
      a.v, b.v, acc, 0, 0, 0, SA[lane], 0, SB[lane]);
#pragma unroll
  for (int r = 0; r < 16; ++r) {
    const int row = (r & 3) + 8 * (r >> 2) + 4 * (lane >> 5);
    D[row * 32 + (lane & 31)] = acc[r];
  }
}

AF_EXPORT int af_mfma_mx32_probe(void* D, const void* A, const void* B,
                                 const void* SA, const void* SB, int layout,
                                 void* stream) {
  mfma_mx32_probe_kernel<<<1, 64, 0, (hipStream_t)stream>>>(
      (float*)D, (const unsigned char*)A, (const unsigned char*)B,
      (const int*)SA, (const int*)SB, layout);
  return af_last_err();
}

// global_load_lds sub-dword semantics probe: each lane loads `size`
// bytes from src + lane*size with a wave-uniform LDS base; dump the LDS
// region so the host can see the actual per-lane LDS stride.
__global__ void lds_stride_probe_kernel(unsigned char* __restrict__ out,
                                        const unsigned char* __restrict__ src,
                                        int size) {
  __shared__ unsigned char lds[1024];
  for (int i = threadIdx.x; i < 1024; i += 64) lds[i] = 0xEE;
  __syncthreads();
  const unsigned char* s = src + threadIdx.x * size;
  if (size == 1)
    __builtin_amdgcn_global_load_lds(
        (const __attribute__((address_space(1))) uint32_t*)s,
        (__attribute__((address_space(3))) uint32_t*)lds, 1, 0, 0);
  else if (size == 2)
    __builtin_amdgcn_global_load_lds(
        (const __attribute__((address_space(1))) uint32_t*)s,
        (__attribute__((address_space(3))) uint32_t*)lds, 2, 0, 0);
  else
    __builtin_amdgcn_global_load_lds(
        (const __attribute__((address_space(1))) uint32_t*)s,
        (__attribute__((address_space(3))) uint32_t*)lds, 4, 0, 0);
  asm volatile("s_waitcnt vmcnt(0)" ::: "memory");
  __syncthreads();
  for (int i = threadIdx.x; i < 1024; i += 64) out[i] = lds[i];
}

AF_EXPORT int af_lds_stride_probe(void* out, const void* src, int size,
                                  void* stream) {
  lds_stride_probe_kernel<<<1, 64, 0, (hipStream_t)stream>>>(
      (unsigned char*)out, (const unsigned char*)src, size);
  return af_last_err();
}

// Tiny helper so tests can verify the ctypes plumbing end-to-end without MFMA.
__global__ void axpy_kernel(float* y, const float* x, float a, int n) {
  int i = blockIdx.x * blockDim.x + threadIdx.x;
  if (i < n) y[i] += a * x[i];
}

AF_EXPORT int af_axpy(void* y, const void* x, float a, int n, void* stream) {
  axpy_kernel<<<(n + 255) / 256, 256, 0, (hipStream_t)stream>>>(
      (float*)y, (const float*)x, a, n);
  return af_last_err();
}

AF_EXPORT int af_device_sync() { return (int)hipDeviceSynchronize(); }
