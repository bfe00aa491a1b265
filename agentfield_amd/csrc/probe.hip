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
