// Fused LoRA low-rank path (the hand-written hot path the north star
// names: "fused LoRA A/B + base GEMM" — the base GEMM runs on hipBLASLt,
// these kernels fuse everything low-rank around it).
//
// Layouts (HF PEFT adapter contract): A [r,K], B [N,r].
//   contract:   t[M,r]  = X[M,K] @ W[r,K]^T           (W LDS-resident)
//   expand_add: Y[M,N] += s * T[M,r] @ W[N,r]^T       (W transposed into
//               LDS; fused in-place epilogue on Y)
//   wgrad:      dW[r,K] = s * T[M,r]^T @ X[M,K]       (deterministic
//               split-M partials + reduce, no atomics)
// All skinny/memory-bound: stream X/Y at HBM rate, stage the tiny
// operand in LDS. R is a compile-time bound so accumulators stay in
// VGPRs (runtime-indexed arrays spill to scratch — guide §5.4 rule 20).
#include "dtx_common.h"

// ---------------------------------------------------------------- contract
// Block: 4 waves; wave handles one row m (grid-stride). W chunked in LDS
// ([R][chunk] bf16); multi-chunk accumulates into fp32 out.
template <int R>
__global__ __launch_bounds__(DTX_BLOCK)
void lora_contract_kernel(const unsigned short* __restrict__ X,
                          const unsigned short* __restrict__ W,
                          float* __restrict__ out,
                          long M, int K, int r, int chunk) {
  extern __shared__ __attribute__((aligned(16))) unsigned short wlds[];
  const int lane = threadIdx.x & (WAVE - 1);
  const int wid = threadIdx.x / WAVE;
  for (int k0 = 0; k0 < K; k0 += chunk) {
    const int kc = min(chunk, K - k0);
    for (int idx = threadIdx.x * 8; idx < r * kc; idx += DTX_BLOCK * 8) {
      int j = idx / kc, k = idx - j * kc;
      *reinterpret_cast<short8v*>(&wlds[j * chunk + k]) =
          *reinterpret_cast<const short8v*>(&W[(long)j * K + k0 + k]);
    }
    __syncthreads();
    for (long m = blockIdx.x * 4 + wid; m < M; m += (long)gridDim.x * 4) {
      const unsigned short* xr = X + m * K + k0;
      float acc[R];
#pragma unroll
      for (int j = 0; j < R; ++j) acc[j] = 0.f;
      for (int k = lane * 8; k < kc; k += WAVE * 8) {
        float xv[8];
        load_bf16x8(xr + k, xv);
#pragma unroll
        for (int j = 0; j < R; ++j) {
          if (j < r) {
            float wv[8];
            load_bf16x8(&wlds[j * chunk + k], wv);
#pragma unroll
            for (int i = 0; i < 8; ++i) acc[j] += xv[i] * wv[i];
          }
        }
      }
#pragma unroll
      for (int j = 0; j < R; ++j) {
        if (j < r) {
          float tot = wave_reduce_sum(acc[j]);
          if (lane == 0) {
            if (k0 == 0) out[m * r + j] = tot;
            else out[m * r + j] += tot;
          }
        }
      }
    }
    __syncthreads();
  }
}

// -------------------------------------------------------------- expand_add
// Y[M,N] += s * T[M,r] @ W[N,r]^T. W transposed into LDS [r][chunk of N].
template <int R>
__global__ __launch_bounds__(DTX_BLOCK)
void lora_expand_add_kernel(unsigned short* __restrict__ Y,
                            const float* __restrict__ T,
                            const unsigned short* __restrict__ W,
                            long M, int N, int r, float s, int chunk) {
  extern __shared__ __attribute__((aligned(16))) unsigned short wlds[];
  const int lane = threadIdx.x & (WAVE - 1);
  const int wid = threadIdx.x / WAVE;
  for (int n0 = 0; n0 < N; n0 += chunk) {
    const int nc = min(chunk, N - n0);
    for (int n = threadIdx.x; n < nc; n += DTX_BLOCK) {
      const unsigned short* wr = W + (long)(n0 + n) * r;
#pragma unroll
      for (int j = 0; j < R; ++j)
        if (j < r) wlds[j * chunk + n] = wr[j];
    }
    __syncthreads();
    for (long m = blockIdx.x * 4 + wid; m < M; m += (long)gridDim.x * 4) {
      float tv[R];
      const float* tr = T + m * r;
#pragma unroll
      for (int j = 0; j < R; ++j) tv[j] = (j < r) ? s * tr[j] : 0.f;
      unsigned short* yr = Y + m * N + n0;
      for (int n = lane * 8; n < nc; n += WAVE * 8) {
        float y[8];
        load_bf16x8(yr + n, y);
#pragma unroll
        for (int j = 0; j < R; ++j) {
          if (j < r) {
            float wv[8];
            load_bf16x8(&wlds[j * chunk + n], wv);
#pragma unroll
            for (int i = 0; i < 8; ++i) y[i] += tv[j] * wv[i];
          }
        }
        store_bf16x8(yr + n, y);
      }
    }
    __syncthreads();
  }
}

// ------------------------------------------------------------------ wgrad
// part[ms][j][K] = s * sum_{m in split ms} T[m,j] * X[m,k]
template <int RCH>
__global__ __launch_bounds__(DTX_BLOCK)
void lora_wgrad_kernel(const float* __restrict__ T,
                       const unsigned short* __restrict__ X,
                       float* __restrict__ part,
                       long M, int K, int r, int j0, int splitm, float s) {
  const int col = blockIdx.x * 2048 + threadIdx.x * 8;
  const int ms = blockIdx.y;
  if (col >= K) return;
  float acc[RCH][8];
#pragma unroll
  for (int j = 0; j < RCH; ++j)
#pragma unroll
    for (int i = 0; i < 8; ++i) acc[j][i] = 0.f;
  const long m_begin = (M * ms) / splitm;
  const long m_end = (M * (ms + 1)) / splitm;
  for (long m = m_begin; m < m_end; ++m) {
    float xv[8];
    load_bf16x8(X + m * K + col, xv);
    const float* tr = T + m * r + j0;
#pragma unroll
    for (int j = 0; j < RCH; ++j) {
      if (j0 + j < r) {
        float t = tr[j];
#pragma unroll
        for (int i = 0; i < 8; ++i) acc[j][i] += t * xv[i];
      }
    }
  }
  float* pb = part + ((long)ms * r) * K;
#pragma unroll
  for (int j = 0; j < RCH; ++j) {
    if (j0 + j < r) {
#pragma unroll
      for (int i = 0; i < 8; ++i)
        pb[(long)(j0 + j) * K + col + i] = s * acc[j][i];
    }
  }
}

// ------------------------------------------------------------- launchers
void launch_reduce_partials(const float* part, float* out, int P, long L,
                            hipStream_t s);

static int lora_chunk(int Kdim, int r) {
  int budget = (32768 / r) & ~7;           // 64 KiB of bf16 LDS
  return Kdim < budget ? Kdim : budget;
}

void launch_lora_contract(const void* X, const void* W, float* out, long M,
                          int K, int r, hipStream_t s) {
  const int chunk = lora_chunk(K, r);
  size_t lds = (size_t)r * chunk * 2;
  long gw = DTX_CDIV(M, 4);
  int grid = (int)(gw < 1024 ? (gw < 1 ? 1 : gw) : 1024);
#define CASE(RR) lora_contract_kernel<RR><<<grid, DTX_BLOCK, lds, s>>>( \
      (const unsigned short*)X, (const unsigned short*)W, out, M, K, r, chunk)
  if (r <= 8) CASE(8);
  else if (r <= 16) CASE(16);
  else if (r <= 32) CASE(32);
  else CASE(64);
#undef CASE
}

void launch_lora_expand_add(void* Y, const float* T, const void* W, long M,
                            int N, int r, float scale, hipStream_t s) {
  const int chunk = lora_chunk(N, r);
  size_t lds = (size_t)r * chunk * 2;
  long gw = DTX_CDIV(M, 4);
  int grid = (int)(gw < 1024 ? (gw < 1 ? 1 : gw) : 1024);
#define CASE(RR) lora_expand_add_kernel<RR><<<grid, DTX_BLOCK, lds, s>>>( \
      (unsigned short*)Y, T, (const unsigned short*)W, M, N, r, scale, chunk)
  if (r <= 8) CASE(8);
  else if (r <= 16) CASE(16);
  else if (r <= 32) CASE(32);
  else CASE(64);
#undef CASE
}

int lora_wgrad_splitm(int K) {
  int kblocks = DTX_CDIV(K, 2048);
  int sm = 512 / kblocks;
  return sm < 1 ? 1 : sm;
}

void launch_lora_wgrad(const float* T, const void* X, float* part,
                       float* out, long M, int K, int r, float s,
                       hipStream_t st) {
  const int splitm = lora_wgrad_splitm(K);
  dim3 grid(DTX_CDIV(K, 2048), splitm);
  for (int j0 = 0; j0 < r; j0 += 16) {
    int rch = r - j0;
#define CASE(RC) lora_wgrad_kernel<RC><<<grid, DTX_BLOCK, 0, st>>>( \
        T, (const unsigned short*)X, part, M, K, r, j0, splitm, s)
    if (rch <= 4) CASE(4);
    else if (rch <= 8) CASE(8);
    else CASE(16);
#undef CASE
  }
  launch_reduce_partials(part, out, splitm, (long)r * K, st);
}
