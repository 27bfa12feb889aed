// Memory-bound fused kernels: RMSNorm fwd/bwd, RoPE, SwiGLU fwd/bwd.
// All bf16 I/O, fp32 math, vectorized 16 B/lane (guide G13).
// Per-thread row caches are compile-time-indexed (templated NG) so they
// stay in VGPRs (runtime-indexed ext_vector arrays spill to scratch).
// Numerics contract: datatunerx_amd/ops/reference.py.
#include "dtx_common.h"

// ------------------------------------------------------------ RMSNorm fwd
// x [M,H] bf16, w [H] bf16 -> y [M,H] bf16, inv [M] f32. One block per
// row (grid-stride). NG = ceil(H/8/256) unrolled groups per thread.
template <int NG>
__global__ __launch_bounds__(DTX_BLOCK)
void rmsnorm_fwd_kernel(const unsigned short* __restrict__ x,
                        const unsigned short* __restrict__ w,
                        unsigned short* __restrict__ y,
                        float* __restrict__ inv_out,
                        int M, int H, float eps) {
  __shared__ float scratch[4];
  const int groups = H / 8;
  const int tid = threadIdx.x;
  float xs[NG][8];
  for (int row = blockIdx.x; row < M; row += gridDim.x) {
    const unsigned short* xr = x + (long)row * H;
    float ss = 0.f;
#pragma unroll
    for (int n = 0; n < NG; ++n) {
      const int g = tid + n * DTX_BLOCK;
      if (g < groups) {
        load_bf16x8(xr + g * 8, xs[n]);
#pragma unroll
        for (int i = 0; i < 8; ++i) ss += xs[n][i] * xs[n][i];
      }
    }
    float total = block_reduce_sum(ss, scratch);
    float inv = rsqrtf(total / (float)H + eps);
    if (tid == 0) inv_out[row] = inv;
    unsigned short* yr = y + (long)row * H;
#pragma unroll
    for (int n = 0; n < NG; ++n) {
      const int g = tid + n * DTX_BLOCK;
      if (g < groups) {
        float wv[8], out[8];
        load_bf16x8(w + g * 8, wv);
#pragma unroll
        for (int i = 0; i < 8; ++i) out[i] = xs[n][i] * inv * wv[i];
        store_bf16x8(yr + g * 8, out);
      }
    }
    __syncthreads();
  }
}

// ------------------------------------------------------------ RMSNorm bwd
// dx = inv*g - inv^3/H * (g.x) * x   with g = dy*w
// dw partials: [gridDim][H] f32, reduced by reduce_partials_kernel
// (deterministic — no atomics).
template <int NG>
__global__ __launch_bounds__(DTX_BLOCK)
void rmsnorm_bwd_kernel(const unsigned short* __restrict__ dy,
                        const unsigned short* __restrict__ x,
                        const unsigned short* __restrict__ w,
                        const float* __restrict__ inv_in,
                        const unsigned short* __restrict__ dres,
                        unsigned short* __restrict__ dx,
                        float* __restrict__ dw_part,
                        int M, int H) {
  __shared__ float scratch[4];
  const int groups = H / 8;
  const int tid = threadIdx.x;
  float wv[NG][8], dwacc[NG][8], xs[NG][8], dys[NG][8];
#pragma unroll
  for (int n = 0; n < NG; ++n) {
    const int g = tid + n * DTX_BLOCK;
    if (g < groups) load_bf16x8(w + g * 8, wv[n]);
#pragma unroll
    for (int i = 0; i < 8; ++i) dwacc[n][i] = 0.f;
  }
  for (int row = blockIdx.x; row < M; row += gridDim.x) {
    const unsigned short* xr = x + (long)row * H;
    const unsigned short* dyr = dy + (long)row * H;
    const float inv = inv_in[row];
    float dot = 0.f;
#pragma unroll
    for (int n = 0; n < NG; ++n) {
      const int g = tid + n * DTX_BLOCK;
      if (g < groups) {
        load_bf16x8(xr + g * 8, xs[n]);
        load_bf16x8(dyr + g * 8, dys[n]);
#pragma unroll
        for (int i = 0; i < 8; ++i)
          dot += dys[n][i] * wv[n][i] * xs[n][i];
      }
    }
    float tot = block_reduce_sum(dot, scratch);
    const float c = inv * inv * inv / (float)H * tot;
    unsigned short* dxr = dx + (long)row * H;
    const unsigned short* drr =
        dres ? dres + (long)row * H : nullptr;
#pragma unroll
    for (int n = 0; n < NG; ++n) {
      const int g = tid + n * DTX_BLOCK;
      if (g < groups) {
        float out[8];
        float dr[8];
        if (drr) load_bf16x8(drr + g * 8, dr);
#pragma unroll
        for (int i = 0; i < 8; ++i) {
          float gi = dys[n][i] * wv[n][i];
          out[i] = inv * gi - c * xs[n][i];
          if (drr) out[i] += dr[i];   // fused residual-branch gradient
          dwacc[n][i] += dys[n][i] * xs[n][i] * inv;
        }
        store_bf16x8(dxr + g * 8, out);
      }
    }
    __syncthreads();
  }
  float* dwp = dw_part + (long)blockIdx.x * H;
#pragma unroll
  for (int n = 0; n < NG; ++n) {
    const int g = tid + n * DTX_BLOCK;
    if (g < groups) {
#pragma unroll
      for (int i = 0; i < 8; ++i) dwp[g * 8 + i] = dwacc[n][i];
    }
  }
}

// Deterministic partial reduce: out[l] = sum_p part[p*L + l].
// float4 loads, 4 planes in flight (the scalar version was
// latency-bound at ~0.5 TB/s; this one streams).
__global__ __launch_bounds__(DTX_BLOCK)
void reduce_partials_kernel(const float* __restrict__ part,
                            float* __restrict__ out, int P, long L) {
  typedef __attribute__((ext_vector_type(4))) float f4;
  const long nvec = L / 4;
  long i = (long)blockIdx.x * DTX_BLOCK + threadIdx.x;
  const long stride = (long)gridDim.x * DTX_BLOCK;
  for (; i < nvec; i += stride) {
    f4 s0 = {0.f, 0.f, 0.f, 0.f};
    f4 s1 = s0, s2 = s0, s3 = s0;
    f4 s4 = s0, s5 = s0, s6 = s0, s7 = s0;
    int p = 0;
    for (; p + 8 <= P; p += 8) {       // 8 planes in flight: the
      const long b = i * 4;            // plane-major layout makes each
      s0 += *reinterpret_cast<const f4*>(&part[(long)p * L + b]);
      s1 += *reinterpret_cast<const f4*>(&part[(long)(p + 1) * L + b]);
      s2 += *reinterpret_cast<const f4*>(&part[(long)(p + 2) * L + b]);
      s3 += *reinterpret_cast<const f4*>(&part[(long)(p + 3) * L + b]);
      s4 += *reinterpret_cast<const f4*>(&part[(long)(p + 4) * L + b]);
      s5 += *reinterpret_cast<const f4*>(&part[(long)(p + 5) * L + b]);
      s6 += *reinterpret_cast<const f4*>(&part[(long)(p + 6) * L + b]);
      s7 += *reinterpret_cast<const f4*>(&part[(long)(p + 7) * L + b]);
    }
    for (; p + 4 <= P; p += 4) {
      const long b = i * 4;
      s0 += *reinterpret_cast<const f4*>(&part[(long)p * L + b]);
      s1 += *reinterpret_cast<const f4*>(&part[(long)(p + 1) * L + b]);
      s2 += *reinterpret_cast<const f4*>(&part[(long)(p + 2) * L + b]);
      s3 += *reinterpret_cast<const f4*>(&part[(long)(p + 3) * L + b]);
    }
    for (; p < P; ++p)
      s0 += *reinterpret_cast<const f4*>(&part[(long)p * L + i * 4]);
    *reinterpret_cast<f4*>(&out[i * 4]) =
        ((s0 + s1) + (s2 + s3)) + ((s4 + s5) + (s6 + s7));
  }
  // ragged tail (L % 4)
  const long tail0 = nvec * 4;
  for (long t = tail0 + blockIdx.x * DTX_BLOCK + threadIdx.x; t < L;
       t += stride) {
    float s = 0.f;
    for (int p = 0; p < P; ++p) s += part[(long)p * L + t];
    out[t] = s;
  }
}

// ----------------------------------------------------------------- RoPE
// x [B,S,H,D] bf16; cos/sin [Smax, D/2] f32; rotate-half convention.
// Each thread: 4 (x1,x2) pairs. backward => sin sign flip.
__global__ __launch_bounds__(DTX_BLOCK)
void rope_kernel(const unsigned short* __restrict__ x,
                 const float* __restrict__ cosb,
                 const float* __restrict__ sinb,
                 unsigned short* __restrict__ y,
                 const int* __restrict__ pos_dev,  // graph-mode position
                 long total_quads,  // B*S*H*(D/2/4)
                 int S, int H, int D, int pos0, int backward,
                 int pos_per_b) {
  // pos_per_b: pos_dev is a PER-ROW [B] vector (batched ragged decode)
  if (pos_dev && !pos_per_b) pos0 += *pos_dev;
  const int qpr = D / 8;                      // 4-pair groups per head-row
  long idx = (long)blockIdx.x * DTX_BLOCK + threadIdx.x;
  long stride = (long)gridDim.x * DTX_BLOCK;
  for (; idx < total_quads; idx += stride) {
    long row = idx / qpr;                     // (b*S + s)*H + h
    int q = (int)(idx % qpr);
    int s_pos = (int)((row / H) % S);
    if (pos_dev && pos_per_b)
      s_pos += pos_dev[row / ((long)S * H)];
    const long base = row * D;
    short4v x1 = *reinterpret_cast<const short4v*>(x + base + q * 4);
    short4v x2 = *reinterpret_cast<const short4v*>(x + base + D / 2 + q * 4);
    float4v c = *reinterpret_cast<const float4v*>(
        cosb + (long)(pos0 + s_pos) * (D / 2) + q * 4);
    float4v s = *reinterpret_cast<const float4v*>(
        sinb + (long)(pos0 + s_pos) * (D / 2) + q * 4);
    short4v y1, y2;
#pragma unroll
    for (int i = 0; i < 4; ++i) {
      float a = bf2f((unsigned short)x1[i]);
      float b = bf2f((unsigned short)x2[i]);
      float sv = backward ? -s[i] : s[i];
      y1[i] = (short)f2bf(a * c[i] - b * sv);
      y2[i] = (short)f2bf(b * c[i] + a * sv);
    }
    *reinterpret_cast<short4v*>(y + base + q * 4) = y1;
    *reinterpret_cast<short4v*>(y + base + D / 2 + q * 4) = y2;
  }
}

// --------------------------------------------------------------- SwiGLU
__global__ __launch_bounds__(DTX_BLOCK)
void swiglu_fwd_kernel(const unsigned short* __restrict__ gate,
                       const unsigned short* __restrict__ up,
                       unsigned short* __restrict__ out, long n8) {
  long idx = (long)blockIdx.x * DTX_BLOCK + threadIdx.x;
  long stride = (long)gridDim.x * DTX_BLOCK;
  for (; idx < n8; idx += stride) {
    float g[8], u[8], o[8];
    load_bf16x8(gate + idx * 8, g);
    load_bf16x8(up + idx * 8, u);
#pragma unroll
    for (int i = 0; i < 8; ++i) {
      // v_rcp_f32 (1 ulp) — the IEEE division chain is ~10 VALU ops
      float sig = __builtin_amdgcn_rcpf(1.f + __expf(-g[i]));
      o[i] = g[i] * sig * u[i];
    }
    store_bf16x8(out + idx * 8, o);
  }
}

__global__ __launch_bounds__(DTX_BLOCK)
void swiglu_bwd_kernel(const unsigned short* __restrict__ dout,
                       const unsigned short* __restrict__ gate,
                       const unsigned short* __restrict__ up,
                       unsigned short* __restrict__ dgate,
                       unsigned short* __restrict__ dup, long n8) {
  long idx = (long)blockIdx.x * DTX_BLOCK + threadIdx.x;
  long stride = (long)gridDim.x * DTX_BLOCK;
  for (; idx < n8; idx += stride) {
    float d[8], g[8], u[8], dg[8], du[8];
    load_bf16x8(dout + idx * 8, d);
    load_bf16x8(gate + idx * 8, g);
    load_bf16x8(up + idx * 8, u);
#pragma unroll
    for (int i = 0; i < 8; ++i) {
      float sig = __builtin_amdgcn_rcpf(1.f + __expf(-g[i]));
      float silu = g[i] * sig;
      dg[i] = d[i] * u[i] * (sig + silu * (1.f - sig));
      du[i] = d[i] * silu;
    }
    store_bf16x8(dgate + idx * 8, dg);
    store_bf16x8(dup + idx * 8, du);
  }
}

// ------------------------------------------------------------- launchers
static int ew_grid(long work_items) {
  long g = DTX_CDIV(work_items, DTX_BLOCK);
  return (int)(g < 2048 ? (g < 1 ? 1 : g) : 2048);
}

void launch_rmsnorm_fwd(const void* x, const void* w, void* y, float* inv,
                        int M, int H, float eps, hipStream_t s) {
  int grid = M < 4096 ? (M < 1 ? 1 : M) : 4096;
  const int ng = DTX_CDIV(H / 8, DTX_BLOCK);
#define CASE(N) rmsnorm_fwd_kernel<N><<<grid, DTX_BLOCK, 0, s>>>( \
      (const unsigned short*)x, (const unsigned short*)w, \
      (unsigned short*)y, inv, M, H, eps)
  switch (ng) {
    case 1: CASE(1); break;
    case 2: CASE(2); break;
    case 3: CASE(3); break;
    case 4: CASE(4); break;
    case 5: CASE(5); break;
    case 6: CASE(6); break;
    default: CASE(8); break;  // H up to 16384
  }
#undef CASE
}

int rmsnorm_bwd_nblocks(int M) { return M < 1024 ? (M < 1 ? 1 : M) : 1024; }

void launch_rmsnorm_bwd(const void* dy, const void* x, const void* w,
                        const float* inv, const void* dres, void* dx,
                        float* dw_part, float* dw, int M, int H,
                        hipStream_t s) {
  int grid = rmsnorm_bwd_nblocks(M);
  const int ng = DTX_CDIV(H / 8, DTX_BLOCK);
#define CASE(N) rmsnorm_bwd_kernel<N><<<grid, DTX_BLOCK, 0, s>>>( \
      (const unsigned short*)dy, (const unsigned short*)x, \
      (const unsigned short*)w, inv, (const unsigned short*)dres, \
      (unsigned short*)dx, dw_part, M, H)
  switch (ng) {
    case 1: CASE(1); break;
    case 2: CASE(2); break;
    case 3: CASE(3); break;
    case 4: CASE(4); break;
    case 5: CASE(5); break;
    case 6: CASE(6); break;
    default: CASE(8); break;
  }
#undef CASE
  reduce_partials_kernel<<<ew_grid(H), DTX_BLOCK, 0, s>>>(dw_part, dw,
                                                          grid, H);
}

void launch_reduce_partials(const float* part, float* out, int P, long L,
                            hipStream_t s) {
  reduce_partials_kernel<<<ew_grid(L), DTX_BLOCK, 0, s>>>(part, out, P, L);
}

void launch_rope(const void* x, const float* cosb, const float* sinb,
                 void* y, long B, int S, int H, int D, int pos0,
                 int backward, const int* pos_dev, int pos_per_b,
                 hipStream_t s) {
  long quads = B * S * H * (D / 8);
  rope_kernel<<<ew_grid(quads), DTX_BLOCK, 0, s>>>(
      (const unsigned short*)x, cosb, sinb, (unsigned short*)y, pos_dev, quads,
      S, H, D, pos0, backward, pos_per_b);
}

void launch_swiglu_fwd(const void* g, const void* u, void* o, long n,
                       hipStream_t s) {
  swiglu_fwd_kernel<<<ew_grid(n / 8), DTX_BLOCK, 0, s>>>(
      (const unsigned short*)g, (const unsigned short*)u,
      (unsigned short*)o, n / 8);
}

void launch_swiglu_bwd(const void* d, const void* g, const void* u,
                       void* dg, void* du, long n, hipStream_t s) {
  swiglu_bwd_kernel<<<ew_grid(n / 8), DTX_BLOCK, 0, s>>>(
      (const unsigned short*)d, (const unsigned short*)g,
      (const unsigned short*)u, (unsigned short*)dg,
      (unsigned short*)du, n / 8);
}
