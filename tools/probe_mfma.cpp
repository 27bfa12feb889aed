// Standalone gfx950 probe: verifies the MFMA fragment layouts the
// attention kernels assume, plus permlane32_swap semantics.
// Build: hipcc --offload-arch=gfx950 -O2 -std=c++17 tools/probe_mfma.cpp -o gpurun_out/probe_mfma
// Run on the GPU box; prints PASS/FAIL per check.
#include <hip/hip_runtime.h>

#include <cmath>
#include <cstdio>
#include <cstdlib>
#include <vector>

typedef __attribute__((ext_vector_type(8))) short short8v;
typedef __attribute__((ext_vector_type(16))) float f32x16;
typedef __attribute__((ext_vector_type(4))) float f32x4;

static unsigned short f2bf(float f) {
  union { float f; unsigned u; } x; x.f = f;
  unsigned r = 0x7fffu + ((x.u >> 16) & 1u);
  return (unsigned short)((x.u + r) >> 16);
}
static float bf2f(unsigned short u) {
  union { float f; unsigned u; } x; x.u = ((unsigned)u) << 16;
  return x.f;
}

// ---- probe 1: mfma_f32_32x32x16_bf16 with assumed layouts
// A row-major [32][16], B row-major [16][32], C row-major [32][32].
// Assumed: A lane l holds A[l&31][(l>>5)*8 + j]; B lane l holds
// B[(l>>5)*8 + j][l&31]; C lane l reg r -> row (r&3)+8*(r>>2)+4*(l>>5),
// col l&31.
__global__ void probe32(const unsigned short* A, const unsigned short* B,
                        float* C) {
  int l = threadIdx.x;
  short8v a = *reinterpret_cast<const short8v*>(A + (l & 31) * 16 + (l >> 5) * 8);
  short8v b;
  for (int j = 0; j < 8; ++j) b[j] = B[((l >> 5) * 8 + j) * 32 + (l & 31)];
  f32x16 c = {};
  c = __builtin_amdgcn_mfma_f32_32x32x16_bf16(a, b, c, 0, 0, 0);
  for (int r = 0; r < 16; ++r)
    C[((r & 3) + 8 * (r >> 2) + 4 * (l >> 5)) * 32 + (l & 31)] = c[r];
}

// ---- probe 2: permlane32_swap
__global__ void probe_permlane(int* out) {
  int l = threadIdx.x;
  int a = l;            // vdst
  int b = 1000 + l;     // src
  auto r = __builtin_amdgcn_permlane32_swap(a, b, false, false);
  out[l] = r[0];
  out[64 + l] = r[1];
}

// ---- probe 3: 16x16x32 known-good layout for cross-check
__global__ void probe16(const unsigned short* A, const unsigned short* B,
                        float* C) {
  int l = threadIdx.x;
  short8v a = *reinterpret_cast<const short8v*>(A + (l & 15) * 32 + (l >> 4) * 8);
  short8v b;
  for (int j = 0; j < 8; ++j) b[j] = B[((l >> 4) * 8 + j) * 16 + (l & 15)];
  f32x4 c = {};
  c = __builtin_amdgcn_mfma_f32_16x16x32_bf16(a, b, c, 0, 0, 0);
  for (int r = 0; r < 4; ++r)
    C[((l >> 4) * 4 + r) * 16 + (l & 15)] = c[r];
}

int main() {
  srand(7);
  int fails = 0;
  // ---------- 32x32x16
  {
    int M = 32, N = 32, K = 16;
    std::vector<unsigned short> A(M * K), B(K * N);
    std::vector<float> Af(M * K), Bf(K * N), ref(M * N, 0.f), got(M * N);
    for (int i = 0; i < M * K; ++i) { Af[i] = (rand() % 1000 - 500) / 250.0f; A[i] = f2bf(Af[i]); Af[i] = bf2f(A[i]); }
    for (int i = 0; i < K * N; ++i) { Bf[i] = (rand() % 1000 - 500) / 250.0f; B[i] = f2bf(Bf[i]); Bf[i] = bf2f(B[i]); }
    for (int m = 0; m < M; ++m)
      for (int k = 0; k < K; ++k)
        for (int n = 0; n < N; ++n) ref[m * N + n] += Af[m * K + k] * Bf[k * N + n];
    unsigned short *dA, *dB; float* dC;
    hipMalloc(&dA, A.size() * 2); hipMalloc(&dB, B.size() * 2); hipMalloc(&dC, got.size() * 4);
    hipMemcpy(dA, A.data(), A.size() * 2, hipMemcpyHostToDevice);
    hipMemcpy(dB, B.data(), B.size() * 2, hipMemcpyHostToDevice);
    probe32<<<1, 64>>>(dA, dB, dC);
    hipMemcpy(got.data(), dC, got.size() * 4, hipMemcpyDeviceToHost);
    float maxerr = 0;
    for (int i = 0; i < M * N; ++i) maxerr = fmaxf(maxerr, fabsf(got[i] - ref[i]));
    printf("mfma_32x32x16 layout: maxerr=%g %s\n", maxerr, maxerr < 0.05 ? "PASS" : "FAIL");
    fails += maxerr >= 0.05;
    hipFree(dA); hipFree(dB); hipFree(dC);
  }
  // ---------- 16x16x32
  {
    int M = 16, N = 16, K = 32;
    std::vector<unsigned short> A(M * K), B(K * N);
    std::vector<float> Af(M * K), Bf(K * N), ref(M * N, 0.f), got(M * N);
    for (int i = 0; i < M * K; ++i) { Af[i] = (rand() % 1000 - 500) / 250.0f; A[i] = f2bf(Af[i]); Af[i] = bf2f(A[i]); }
    for (int i = 0; i < K * N; ++i) { Bf[i] = (rand() % 1000 - 500) / 250.0f; B[i] = f2bf(Bf[i]); Bf[i] = bf2f(B[i]); }
    for (int m = 0; m < M; ++m)
      for (int k = 0; k < K; ++k)
        for (int n = 0; n < N; ++n) ref[m * N + n] += Af[m * K + k] * Bf[k * N + n];
    unsigned short *dA, *dB; float* dC;
    hipMalloc(&dA, A.size() * 2); hipMalloc(&dB, B.size() * 2); hipMalloc(&dC, got.size() * 4);
    hipMemcpy(dA, A.data(), A.size() * 2, hipMemcpyHostToDevice);
    hipMemcpy(dB, B.data(), B.size() * 2, hipMemcpyHostToDevice);
    probe16<<<1, 64>>>(dA, dB, dC);
    hipMemcpy(got.data(), dC, got.size() * 4, hipMemcpyDeviceToHost);
    float maxerr = 0;
    for (int i = 0; i < M * N; ++i) maxerr = fmaxf(maxerr, fabsf(got[i] - ref[i]));
    printf("mfma_16x16x32 layout: maxerr=%g %s\n", maxerr, maxerr < 0.05 ? "PASS" : "FAIL");
    fails += maxerr >= 0.05;
    hipFree(dA); hipFree(dB); hipFree(dC);
  }
  // ---------- permlane32_swap
  {
    int* d; hipMalloc(&d, 128 * 4);
    probe_permlane<<<1, 64>>>(d);
    std::vector<int> h(128);
    hipMemcpy(h.data(), d, 128 * 4, hipMemcpyDeviceToHost);
    // expected: r0 (vdst): lanes 0-31 keep a=l, lanes 32-63 get src lanes 0-31 (1000+l-32)
    //           r1 (src):  lanes 0-31 get vdst lanes 32-63 (l+32), lanes 32-63 keep 1000+l
    bool ok = true;
    for (int l = 0; l < 64; ++l) {
      int r0 = h[l], r1 = h[64 + l];
      int e0 = l < 32 ? l : 1000 + (l - 32);
      int e1 = l < 32 ? (l + 32) : 1000 + l;
      if (r0 != e0 || r1 != e1) ok = false;
    }
    printf("permlane32_swap: %s", ok ? "PASS (vdst.hi<->src.lo)\n" : "model-A FAIL; dump:\n");
    if (!ok)
      for (int l = 0; l < 64; l += 8)
        printf("  l=%d r0=%d r1=%d\n", l, h[l], h[64 + l]);
    fails += !ok;
    hipFree(d);
  }
  printf(fails ? "PROBE FAILURES: %d\n" : "ALL PROBES PASS\n", fails);
  return fails;
}
