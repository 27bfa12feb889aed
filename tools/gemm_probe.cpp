// Standalone A/B probe for the hand-written NT GEMM (no torch): variants
// run interleaved in ONE process (guide §5.4 rule 24), random data
// (rule 25), bitwise cross-check between variants (same math order =>
// identical bits).
//   hipcc --offload-arch=gfx950 -O3 -std=c++17 tools/gemm_probe.cpp -o tools/gemm_probe.bin
//   ./tools/gemm_probe.bin bench 24576 4096 4096 [rounds]
//   ./tools/gemm_probe.bin one <grid> <vs> 24576 4096 4096 <iters>   # for rocprofv3
#include <cstdio>
#include <cstdlib>
#include <cstring>
#include <vector>
#include <chrono>

#include "../datatunerx_amd/ops/hip/gemm.hip"

#define CK(x) do { hipError_t e = (x); if (e != hipSuccess) { \
  fprintf(stderr, "HIP error %s at %s:%d\n", hipGetErrorString(e), \
          __FILE__, __LINE__); exit(1); } } while (0)

__global__ void fill_rand(unsigned short* p, long n, unsigned seed) {
  long i = (long)blockIdx.x * blockDim.x + threadIdx.x;
  if (i >= n) return;
  unsigned long long z = seed + (unsigned long long)i * 0x9E3779B97F4A7C15ull;
  z ^= z >> 30; z *= 0xBF58476D1CE4E5B9ull; z ^= z >> 27;
  float f = ((z >> 40) & 0xFFFFFF) / 8388608.0f * 2.f - 1.f;  // [-1,1)
  union { float f; unsigned u; } u; u.f = f;
  p[i] = (unsigned short)(u.u >> 16);
}

template <int GRID, int PIPE, int MF = 0>
void run(const unsigned short* A, const unsigned short* B,
         unsigned short* C, long M, int N, int K) {
  const int mb_n = (int)((M + 255) / 256), nb_n = N / 256;
  hipLaunchKernelGGL((gemm_nt_kernel<false, GRID, PIPE, MF>),
                     dim3(mb_n * nb_n),
                     dim3(512), 0, 0, A, B, nullptr, C, M, N, K, mb_n);
}

typedef void (*runfn)(const unsigned short*, const unsigned short*,
                      unsigned short*, long, int, int);
#define NVAR 6
static runfn FNS[NVAR] = {run<1, 2, 0>, run<1, 6, 0>, run<1, 7, 0>,
                          run<2, 2, 0>, run<2, 6, 0>, run<2, 7, 0>};
static const char* NAMES[NVAR] = {"g1p2", "g1p6-novmcnt-RACY",
                                  "g1p7-nobar-RACY",
                                  "g2p2", "g2p6-novmcnt-RACY",
                                  "g2p7-nobar-RACY"};

int main(int argc, char** argv) {
  if (argc < 5) { fprintf(stderr, "usage: see header\n"); return 1; }
  const bool one = !strcmp(argv[1], "one");
  int ai = one ? 3 : 2;  // one <grid> M N K [iters]
  long M = atol(argv[ai]); int N = atoi(argv[ai + 1]), K = atoi(argv[ai + 2]);
  int rounds = argc > ai + 3 ? atoi(argv[ai + 3]) : 6;

  unsigned short *A, *B, *C;
  CK(hipMalloc(&A, M * (long)K * 2));
  CK(hipMalloc(&B, (long)N * K * 2));
  CK(hipMalloc(&C, M * (long)N * 2));
  hipLaunchKernelGGL(fill_rand, dim3((M * K + 255) / 256), dim3(256), 0, 0,
                     A, M * (long)K, 1u);
  hipLaunchKernelGGL(fill_rand, dim3(((long)N * K + 255) / 256), dim3(256),
                     0, 0, B, (long)N * K, 2u);
  CK(hipDeviceSynchronize());
  const double fl = 2.0 * M * N * K;

  if (one) {
    int g = atoi(argv[2]);
    for (int i = 0; i < rounds; ++i) FNS[g](A, B, C, M, N, K);
    CK(hipDeviceSynchronize());
    printf("done %s\n", NAMES[g]);
    return 0;
  }

  const bool cluster_ok = (M / 256) % 8 == 0 && (N / 256) % 4 == 0
                          && M % 256 == 0;
  // bitwise cross-check of variants against variant 0
  std::vector<unsigned short> ref(4096), got(4096);
  FNS[3](A, B, C, M, N, K);
  CK(hipDeviceSynchronize());
  CK(hipMemcpy(ref.data(), C + M * (long)N / 2, 8192, hipMemcpyDeviceToHost));
  for (int v = 0; v < NVAR; ++v) {
    if (v == 3 || (v < 3 && !cluster_ok)) continue;
    CK(hipMemset(C + M * (long)N / 2, 0, 8192));
    FNS[v](A, B, C, M, N, K);
    CK(hipDeviceSynchronize());
    CK(hipMemcpy(got.data(), C + M * (long)N / 2, 8192,
                 hipMemcpyDeviceToHost));
    // MF1 variants have a different accumulation split (32x32x16 vs
    // 16x16x32) so compare with tolerance, not bitwise
    float mx = 0, sc = 0;
    for (int i = 0; i < 4096; ++i) {
      union { unsigned u; float f; } a, b;
      a.u = (unsigned)ref[i] << 16; b.u = (unsigned)got[i] << 16;
      float d = a.f - b.f; if (d < 0) d = -d; if (d > mx) mx = d;
      float m = a.f < 0 ? -a.f : a.f; if (m > sc) sc = m;
    }
    if (mx > 0.02f * (sc > 1 ? sc : 1))
      printf("MISMATCH variant %s vs base: maxdiff %f scale %f\n",
             NAMES[v], mx, sc);
  }

  double best[NVAR] = {1e30, 1e30, 1e30, 1e30, 1e30, 1e30};
  for (int r = 0; r < rounds; ++r)
    for (int v = 0; v < NVAR; ++v) {
      if (v < 3 && !cluster_ok) continue;
      CK(hipDeviceSynchronize());
      auto t0 = std::chrono::steady_clock::now();
      for (int i = 0; i < 3; ++i) FNS[v](A, B, C, M, N, K);
      CK(hipDeviceSynchronize());
      double dt = std::chrono::duration<double>(
                      std::chrono::steady_clock::now() - t0).count() / 3;
      if (dt < best[v]) best[v] = dt;
    }
  for (int v = 0; v < NVAR; ++v)
    if (best[v] < 1e29)
      printf("%s  M=%ld N=%d K=%d  %7.3f ms  %7.1f TF/s\n", NAMES[v], M, N,
             K, best[v] * 1e3, fl / best[v] / 1e12);
  return 0;
}
