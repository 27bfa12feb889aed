// Fused AdamW (single pass: m, v, decoupled decay, master update, bf16
// param write) + deterministic two-stage L2 norm.
// Matches torch.optim.AdamW semantics (ops/reference.py adamw_step).
#include "dtx_common.h"

// p_out: params in their own dtype (bf16 on GPU; f32 master is
// authoritative). grad fp32 (trainer accumulates micro-batch grads fp32).
__global__ __launch_bounds__(DTX_BLOCK)
void adamw_kernel_bf16(unsigned short* __restrict__ p_out,
                       float* __restrict__ master,
                       const float* __restrict__ grad,
                       float* __restrict__ m, float* __restrict__ v,
                       long n4, float lr, float beta1, float beta2,
                       float eps, float wd, float inv_bc1, float inv_bc2) {
  long i = (long)blockIdx.x * DTX_BLOCK + threadIdx.x;
  long stride = (long)gridDim.x * DTX_BLOCK;
  for (; i < n4; i += stride) {
    float4v g = *reinterpret_cast<const float4v*>(grad + i * 4);
    float4v mm = *reinterpret_cast<float4v*>(m + i * 4);
    float4v vv = *reinterpret_cast<float4v*>(v + i * 4);
    float4v p = *reinterpret_cast<float4v*>(master + i * 4);
    short4v pb;
#pragma unroll
    for (int k = 0; k < 4; ++k) {
      mm[k] = beta1 * mm[k] + (1.f - beta1) * g[k];
      vv[k] = beta2 * vv[k] + (1.f - beta2) * g[k] * g[k];
      p[k] = p[k] * (1.f - lr * wd);
      float denom = sqrtf(vv[k] * inv_bc2) + eps;
      p[k] -= lr * inv_bc1 * mm[k] / denom;
      pb[k] = (short)f2bf(p[k]);
    }
    *reinterpret_cast<float4v*>(m + i * 4) = mm;
    *reinterpret_cast<float4v*>(v + i * 4) = vv;
    *reinterpret_cast<float4v*>(master + i * 4) = p;
    *reinterpret_cast<short4v*>(p_out + i * 4) = pb;
  }
}

// tail for n % 4 != 0 handled by padding on the host side (flat buffers
// are allocated in multiples of 4).

__global__ __launch_bounds__(DTX_BLOCK)
void l2_partial_kernel(const float* __restrict__ x, float* __restrict__ part,
                       long n) {
  __shared__ float scratch[4];
  float ss = 0.f;
  for (long i = (long)blockIdx.x * DTX_BLOCK + threadIdx.x; i < n;
       i += (long)gridDim.x * DTX_BLOCK) {
    float v = x[i];
    ss += v * v;
  }
  float tot = block_reduce_sum(ss, scratch);
  if (threadIdx.x == 0) part[blockIdx.x] = tot;
}

__global__ __launch_bounds__(DTX_BLOCK)
void l2_final_kernel(const float* __restrict__ part, float* __restrict__ out,
                     int nparts) {
  __shared__ float scratch[4];
  float ss = 0.f;
  for (int i = threadIdx.x; i < nparts; i += DTX_BLOCK) ss += part[i];
  float tot = block_reduce_sum(ss, scratch);
  if (threadIdx.x == 0) out[0] = sqrtf(tot);
}

void launch_adamw(void* p_bf16, float* master, const float* grad, float* m,
                  float* v, long n, float lr, float b1, float b2, float eps,
                  float wd, int step, hipStream_t s) {
  long n4 = n / 4;
  int grid = (int)(n4 < 2048 * DTX_BLOCK ? DTX_CDIV(n4, DTX_BLOCK) : 2048);
  if (grid < 1) grid = 1;
  float bc1 = 1.f - powf(b1, (float)step);
  float bc2 = 1.f - powf(b2, (float)step);
  adamw_kernel_bf16<<<grid, DTX_BLOCK, 0, s>>>(
      (unsigned short*)p_bf16, master, grad, m, v, n4, lr, b1, b2, eps,
      wd, 1.f / bc1, 1.f / bc2);
}

void launch_l2_norm(const float* x, float* workspace, float* out, long n,
                    hipStream_t s) {
  int grid = 1024;
  l2_partial_kernel<<<grid, DTX_BLOCK, 0, s>>>(x, workspace, n);
  l2_final_kernel<<<1, DTX_BLOCK, 0, s>>>(workspace, out, grid);
}
