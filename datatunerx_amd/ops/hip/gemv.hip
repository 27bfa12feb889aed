// Decode-path GEMV for gfx950: y[1,N] = x[1,K] @ W[N,K]^T, bf16 in/out,
// fp32 accumulate. One wave per 2 output rows (ILP), lanes stream the
// W rows with coalesced 16-byte loads at the HBM rate; x is re-read per
// row straight from L1/L2 (8-22 KB, hot). The guide's "GEMV / M <= 16
// decode weights" row: no LDS round trip, deep unroll, late waits.
// hipBLASLt's M=1 kernels leave ~3x on the table for these shapes.
#include "dtx_common.h"

__global__ __launch_bounds__(DTX_BLOCK)
void gemv_bf16_kernel(const unsigned short* __restrict__ X,
                      const unsigned short* __restrict__ W,
                      unsigned short* __restrict__ Y,
                      int N, int K) {
  const int lane = threadIdx.x & 63;
  const int wid = threadIdx.x >> 6;
  const long wstride = (long)gridDim.x * 8;      // rows per grid pass
  for (long n0 = (long)blockIdx.x * 8 + wid * 2; n0 < N; n0 += wstride) {
    float acc0 = 0.f, acc1 = 0.f;
    const bool has1 = n0 + 1 < N;
    const unsigned short* w0 = W + n0 * K;
    const unsigned short* w1 = w0 + (has1 ? K : 0);
    for (int k = lane * 8; k < K; k += 64 * 8) {
      float xv[8], a[8], b[8];
      load_bf16x8(X + k, xv);
      load_bf16x8(w0 + k, a);
      load_bf16x8(w1 + k, b);
#pragma unroll
      for (int i = 0; i < 8; ++i) {
        acc0 += xv[i] * a[i];
        acc1 += xv[i] * b[i];
      }
    }
    acc0 = wave_reduce_sum(acc0);
    acc1 = wave_reduce_sum(acc1);
    if (lane == 0) Y[n0] = f2bf(acc0);
    if (lane == 1 && has1) Y[n0 + 1] = f2bf(acc1);
  }
}

void launch_gemv_bf16(const void* x, const void* w, void* y, int N, int K,
                      hipStream_t st) {
  long blocks = DTX_CDIV((long)N, 8);
  int grid = (int)(blocks < 2048 ? (blocks < 1 ? 1 : blocks) : 2048);
  gemv_bf16_kernel<<<grid, DTX_BLOCK, 0, st>>>(
      (const unsigned short*)x, (const unsigned short*)w,
      (unsigned short*)y, N, K);
}
