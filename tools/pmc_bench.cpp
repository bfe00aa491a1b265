// Torch-free PMC profiling driver: exercises the hot HIP kernels through the
// libafops C API (rocprofv3's counter sampler aborts on several torch
// kernels, so this driver uses raw hipMalloc + zeroed buffers — numerics are
// irrelevant for counter collection).
//
//   hipcc -O2 --offload-arch=gfx950 tools/pmc_bench.cpp \
//     -L agentfield_amd -lafops -o /tmp/pmc_bench
#include <hip/hip_runtime.h>

#include <cstdio>
#include <cstdlib>
#include <vector>

extern "C" {
int af_attn_prefill(void*, const void*, const void*, const void*, const void*,
                    const void*, const void*, const void*, const void*, float,
                    int, int, int, int, long, int, int, void*);
int af_attn_decode(void*, void*, void*, const void*, const void*, const void*,
                   const void*, const void*, float, int, int, int, int, int,
                   int, int, long, void*);
int af_gemm_skinny(void*, void*, void*, const void*, const void*, int, int,
                   int, int, int, const void*, float, void*, void*);
int af_gemm_bf16(void*, const void*, const void*, int, int, int, void*);
int af_rmsnorm(void*, void*, const void*, const void*, const void*, float,
               int, int, void*);
}

#define CK(x) do { hipError_t e = (x); if (e != hipSuccess) { \
  fprintf(stderr, "HIP error %d at line %d\n", e, __LINE__); exit(1); } } while (0)

static void* dz(size_t bytes) {
  void* p = nullptr;
  CK(hipMalloc(&p, bytes));
  CK(hipMemset(p, 0, bytes));
  return p;
}

static void* d_i32(const std::vector<int>& v) {
  void* p = nullptr;
  CK(hipMalloc(&p, v.size() * 4));
  CK(hipMemcpy(p, v.data(), v.size() * 4, hipMemcpyHostToDevice));
  return p;
}

int main() {
  const int D = 128, Hq = 32, Hk = 8, page = 16;
  hipStream_t s = nullptr;

  // ---- paged prefill attention: B=4, S=2048 ----
  {
    const int B = 4, S = 2048, T = B * S;
    const int maxp = S / page, npages = 1 + B * maxp;
    void* q = dz((size_t)T * Hq * D * 2);
    void* kc = dz((size_t)npages * Hk * page * D * 2);
    void* vc = dz((size_t)npages * Hk * page * D * 2);
    void* out = dz((size_t)T * Hq * D * 2);
    std::vector<int> bt(B * maxp), cu(B + 1), qs(B, 0), ts, tq;
    for (int i = 0; i < B * maxp; ++i) bt[i] = 1 + i;
    for (int i = 0; i <= B; ++i) cu[i] = i * S;
    for (int b = 0; b < B; ++b)
      for (int q0 = 0; q0 < S; q0 += 16) { ts.push_back(b); tq.push_back(q0); }
    void *d_bt = d_i32(bt), *d_cu = d_i32(cu), *d_qs = d_i32(qs),
         *d_ts = d_i32(ts), *d_tq = d_i32(tq);
    for (int it = 0; it < 3; ++it)
      af_attn_prefill(out, q, kc, vc, d_bt, d_qs, d_cu, d_ts, d_tq,
                      0.0883f, (int)ts.size(), Hq, Hk, D, (long)Hq * D, page,
                      maxp, s);
    CK(hipDeviceSynchronize());
    printf("prefill ok\n");
  }

  // ---- skinny GEMM: M=16, qkv 6144x4096 ----
  {
    const int M = 16, N = 6144, K = 4096, SK = 10;
    void* x = dz((size_t)M * K * 2);
    void* w = dz((size_t)N * K * 2);
    void* out = dz((size_t)M * N * 2);
    void* part = dz((size_t)SK * M * N * 4);
    for (int it = 0; it < 3; ++it)
      af_gemm_skinny(out, part, nullptr, x, w, M, N, K, SK, 0, nullptr, 0.f,
                     nullptr, s);
    CK(hipDeviceSynchronize());
    printf("skinny ok\n");
  }

  // ---- decode attention: B=64, L=1024, nsplit=2 ----
  {
    const int B = 64, L = 1024, maxp = L / page, npages = 1 + B * maxp,
              ns = 2, G = Hq / Hk;
    void* q = dz((size_t)B * Hq * D * 2);
    void* kc = dz((size_t)npages * Hk * page * D * 2);
    void* vc = dz((size_t)npages * Hk * page * D * 2);
    void* out = dz((size_t)B * Hq * D * 2);
    void* po = dz((size_t)B * Hk * ns * G * D * 4);
    void* pml = dz((size_t)B * Hk * ns * G * 2 * 4);
    std::vector<int> bt(B * maxp), len(B, L);
    for (int i = 0; i < B * maxp; ++i) bt[i] = 1 + i;
    void *d_bt = d_i32(bt), *d_len = d_i32(len);
    for (int it = 0; it < 3; ++it)
      af_attn_decode(out, po, pml, q, kc, vc, d_bt, d_len, 0.0883f, B, Hq, Hk,
                     D, page, maxp, ns, (long)Hq * D, s);
    CK(hipDeviceSynchronize());
    printf("decode ok\n");
  }

  // ---- tiled MFMA GEMM 4096^3 ----
  {
    const int N = 4096;
    void* a = dz((size_t)N * N * 2);
    void* w = dz((size_t)N * N * 2);
    void* c = dz((size_t)N * N * 2);
    for (int it = 0; it < 3; ++it) af_gemm_bf16(c, a, w, N, N, N, s);
    CK(hipDeviceSynchronize());
    printf("gemm ok\n");
  }

  // ---- fused rmsnorm T=8192 ----
  {
    const int T = 8192, H = 4096;
    void* x = dz((size_t)T * H * 2);
    void* res = dz((size_t)T * H * 2);
    void* w = dz((size_t)H * 2);
    void* out = dz((size_t)T * H * 2);
    for (int it = 0; it < 3; ++it)
      af_rmsnorm(out, res, x, res, w, 1e-5f, T, H, s);
    CK(hipDeviceSynchronize());
    printf("rmsnorm ok\n");
  }
  printf("pmc bench done\n");
  return 0;
}
