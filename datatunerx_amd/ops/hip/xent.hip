// Fused log-softmax cross-entropy with ignore_index masking.
// logits [N,V] bf16 (V % 8 == 0), targets i64, fp32 reduce.
// fwd: loss[i] = lse_i - x[i,t_i] (0 if ignored); saves lse for bwd.
// bwd: dlogits = scale_i * (exp(x - lse) - onehot)  (scale_i = 0 ignored).
// Replaces the HF loss path the reference uses (train.py:256-264 forces
// an fp32 lm_head; here the reduction is fp32 while logits stay bf16).
#include "dtx_common.h"

__device__ __forceinline__ float block_reduce_max(float x, float* scratch) {
  const int lane = threadIdx.x & (WAVE - 1);
  const int wid = threadIdx.x / WAVE;
  x = wave_reduce_max(x);
  if (lane == 0) scratch[wid] = x;
  __syncthreads();
  float r = fmaxf(fmaxf(scratch[0], scratch[1]),
                  fmaxf(scratch[2], scratch[3]));
  __syncthreads();
  return r;
}

__global__ __launch_bounds__(DTX_BLOCK)
void xent_fwd_kernel(const unsigned short* __restrict__ logits,
                     const long* __restrict__ targets,
                     float* __restrict__ loss, float* __restrict__ lse_out,
                     long N, int V, long ignore_index) {
  __shared__ float scratch[4];
  const int groups = V / 8;
  for (long row = blockIdx.x; row < N; row += gridDim.x) {
    const unsigned short* xr = logits + row * V;
    float mx = -3.4e38f;
    for (int g = threadIdx.x; g < groups; g += DTX_BLOCK) {
      float v[8];
      load_bf16x8(xr + g * 8, v);
#pragma unroll
      for (int i = 0; i < 8; ++i) mx = fmaxf(mx, v[i]);
    }
    const float rowmax = block_reduce_max(mx, scratch);
    float se = 0.f;
    for (int g = threadIdx.x; g < groups; g += DTX_BLOCK) {
      float v[8];
      load_bf16x8(xr + g * 8, v);
#pragma unroll
      for (int i = 0; i < 8; ++i) se += __expf(v[i] - rowmax);
    }
    const float sum = block_reduce_sum(se, scratch);
    if (threadIdx.x == 0) {
      const float lse = rowmax + __logf(sum);
      lse_out[row] = lse;
      const long t = targets[row];
      loss[row] = (t == ignore_index)
                      ? 0.f
                      : lse - bf2f(xr[(int)t]);
    }
    __syncthreads();
  }
}

__global__ __launch_bounds__(DTX_BLOCK)
void xent_bwd_kernel(const unsigned short* __restrict__ logits,
                     const long* __restrict__ targets,
                     const float* __restrict__ lse,
                     const float* __restrict__ dloss,
                     unsigned short* __restrict__ dlogits,
                     long N, int V, long ignore_index) {
  const int groups = V / 8;
  for (long row = blockIdx.x; row < N; row += gridDim.x) {
    const unsigned short* xr = logits + row * V;
    unsigned short* dr = dlogits + row * V;
    const long t = targets[row];
    const float scale = (t == ignore_index) ? 0.f : dloss[row];
    const float l = lse[row];
    for (int g = threadIdx.x; g < groups; g += DTX_BLOCK) {
      float v[8], o[8];
      load_bf16x8(xr + g * 8, v);
      const int base = g * 8;
#pragma unroll
      for (int i = 0; i < 8; ++i) {
        float p = __expf(v[i] - l);
        if ((long)(base + i) == t) p -= 1.f;
        o[i] = scale * p;
      }
      store_bf16x8(dr + g * 8, o);
    }
  }
}

void launch_xent_fwd(const void* logits, const long* targets, float* loss,
                     float* lse, long N, int V, long ignore,
                     hipStream_t s) {
  int grid = (int)(N < 2048 ? (N < 1 ? 1 : N) : 2048);
  xent_fwd_kernel<<<grid, DTX_BLOCK, 0, s>>>(
      (const unsigned short*)logits, targets, loss, lse, N, V, ignore);
}

void launch_xent_bwd(const void* logits, const long* targets,
                     const float* lse, const float* dloss, void* dlogits,
                     long N, int V, long ignore, hipStream_t s) {
  int grid = (int)(N < 2048 ? (N < 1 ? 1 : N) : 2048);
  xent_bwd_kernel<<<grid, DTX_BLOCK, 0, s>>>(
      (const unsigned short*)logits, targets, lse, dloss,
      (unsigned short*)dlogits, N, V, ignore);
}

// ---- chunked-vocab fused lm_head+CE support ----
// The [M,V] logits tensor is never materialized: per vocab chunk
// [N, Vc] the caller runs the chunk GEMM (cache-resident) and these
// kernels fold it into running online-logsumexp state / the dX sweep.

// online LSE merge: m/l are the running rowwise (max, sumexp) pair;
// tgt picks up the target logit when it falls inside this chunk.
__global__ __launch_bounds__(DTX_BLOCK)
void xent_lse_merge_kernel(const unsigned short* __restrict__ logits,
                           const long* __restrict__ targets,
                           float* __restrict__ m_run,
                           float* __restrict__ l_run,
                           float* __restrict__ tgt,
                           long N, int Vc, long v0, long ignore_index) {
  __shared__ float scratch[4];
  const int groups = Vc / 8;
  for (long row = blockIdx.x; row < N; row += gridDim.x) {
    const unsigned short* xr = logits + row * Vc;
    float mx = -3.4e38f;
    for (int g = threadIdx.x; g < groups; g += DTX_BLOCK) {
      float v[8];
      load_bf16x8(xr + g * 8, v);
#pragma unroll
      for (int i = 0; i < 8; ++i) mx = fmaxf(mx, v[i]);
    }
    const float cmax = block_reduce_max(mx, scratch);
    float se = 0.f;
    for (int g = threadIdx.x; g < groups; g += DTX_BLOCK) {
      float v[8];
      load_bf16x8(xr + g * 8, v);
#pragma unroll
      for (int i = 0; i < 8; ++i) se += __expf(v[i] - cmax);
    }
    const float csum = block_reduce_sum(se, scratch);
    if (threadIdx.x == 0) {
      const float m_old = m_run[row];
      const float m_new = fmaxf(m_old, cmax);
      l_run[row] = l_run[row] * __expf(m_old - m_new)
                   + csum * __expf(cmax - m_new);
      m_run[row] = m_new;
      const long t = targets[row];
      if (t != ignore_index && t >= v0 && t < v0 + Vc)
        tgt[row] = bf2f(xr[(int)(t - v0)]);
    }
    __syncthreads();
  }
}

// dlogits for one chunk given the final lse: dl = exp(x - lse) - onehot
// (0 for ignored rows). UNscaled: the caller multiplies the scalar
// grad_out/n_valid in its backward.
__global__ __launch_bounds__(DTX_BLOCK)
void xent_dlogits_kernel(const unsigned short* __restrict__ logits,
                         const long* __restrict__ targets,
                         const float* __restrict__ lse,
                         unsigned short* __restrict__ dl,
                         long N, int Vc, long v0, long ignore_index) {
  const int groups = Vc / 8;
  for (long row = blockIdx.x; row < N; row += gridDim.x) {
    const unsigned short* xr = logits + row * Vc;
    unsigned short* dr = dl + row * Vc;
    const long t = targets[row];
    if (t == ignore_index) {
      for (int g = threadIdx.x; g < groups; g += DTX_BLOCK)
        *reinterpret_cast<short8v*>(dr + g * 8) =
            short8v{0, 0, 0, 0, 0, 0, 0, 0};
      continue;
    }
    const float ls = lse[row];
    const long tl = t - v0;
    for (int g = threadIdx.x; g < groups; g += DTX_BLOCK) {
      float v[8];
      load_bf16x8(xr + g * 8, v);
      float o[8];
#pragma unroll
      for (int i = 0; i < 8; ++i) {
        float p = __expf(v[i] - ls);
        if ((long)g * 8 + i == tl) p -= 1.f;
        o[i] = p;
      }
      store_bf16x8(dr + g * 8, o);
    }
  }
}

void launch_xent_lse_merge(const void* logits, const long* targets,
                           float* m_run, float* l_run, float* tgt, long N,
                           int Vc, long v0, long ignore_index,
                           hipStream_t stream) {
  const int blocks = (int)min(N, (long)2048);
  hipLaunchKernelGGL(xent_lse_merge_kernel, dim3(blocks), dim3(DTX_BLOCK),
                     0, stream, (const unsigned short*)logits, targets,
                     m_run, l_run, tgt, N, Vc, v0, ignore_index);
}

void launch_xent_dlogits(const void* logits, const long* targets,
                         const float* lse, void* dl, long N, int Vc,
                         long v0, long ignore_index, hipStream_t stream) {
  const int blocks = (int)min(N, (long)2048);
  hipLaunchKernelGGL(xent_dlogits_kernel, dim3(blocks), dim3(DTX_BLOCK),
                     0, stream, (const unsigned short*)logits, targets,
                     lse, (unsigned short*)dl, N, Vc, v0, ignore_index);
}
