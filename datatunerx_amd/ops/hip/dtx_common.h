// Shared helpers for the gfx950 (CDNA4) kernels.
// Conventions: wave = 64 lanes; block = 256 threads unless stated;
// bf16 handled as raw ushort bits, vectorized as short4/short8 (16 B/lane
// loads — guide G13: hipcc does not auto-vectorize bf16 scalar loads).
#pragma once

#include <hip/hip_runtime.h>

#define WAVE 64
#define DTX_BLOCK 256

typedef __attribute__((ext_vector_type(2))) short short2v;
typedef __attribute__((ext_vector_type(4))) short short4v;
typedef __attribute__((ext_vector_type(8))) short short8v;
typedef __attribute__((ext_vector_type(4))) float float4v;
typedef __attribute__((ext_vector_type(8))) float float8v;
typedef __attribute__((ext_vector_type(16))) float float16v;

__device__ __forceinline__ float bf2f(unsigned short u) {
  union { float f; unsigned int i; } x;
  x.i = ((unsigned int)u) << 16;
  return x.f;
}

__device__ __forceinline__ unsigned short f2bf(float f) {
  union { float f; unsigned int i; } x;
  x.f = f;
  unsigned int u = x.i;
  if ((u & 0x7fffffffu) > 0x7f800000u) return 0x7fc0;  // NaN
  unsigned int rounding = 0x7fffu + ((u >> 16) & 1u);
  return (unsigned short)((u + rounding) >> 16);
}

// load 8 bf16 (16B) and convert to 8 floats
__device__ __forceinline__ void load_bf16x8(const unsigned short* p,
                                            float* out) {
  short8v v = *reinterpret_cast<const short8v*>(p);
#pragma unroll
  for (int i = 0; i < 8; ++i) out[i] = bf2f((unsigned short)v[i]);
}

__device__ __forceinline__ void store_bf16x8(unsigned short* p,
                                             const float* in) {
  short8v v;
#pragma unroll
  for (int i = 0; i < 8; ++i) v[i] = (short)f2bf(in[i]);
  *reinterpret_cast<short8v*>(p) = v;
}

// round two floats to bf16 (RNE) with one v_cvt_pk_bf16_f32 — ~3x
// cheaper than the scalar f2bf path (no branch, no integer rounding)
__device__ __forceinline__ unsigned dtx_cvt_pk_bf16(float lo, float hi) {
  unsigned r;
  asm volatile("v_cvt_pk_bf16_f32 %0, %1, %2\n\ts_nop 1"
               : "=v"(r) : "v"(lo), "v"(hi));
  return r;
}

// Counter-based dropout RNG (splitmix64 on the element-group index):
// the SAME (seed, linear-offset) pair yields the same mask bits in
// every kernel that consumes it (contract / wgrad / expand), so the
// dropout mask is never materialized in HBM. One hash covers 4
// elements (16 random bits each; keep-prob quantized to 1/65536).
__device__ __forceinline__ void dtx_dropout4(unsigned long long seed,
                                             unsigned long long group,
                                             unsigned thr16, float inv_keep,
                                             float mv[4]) {
  unsigned long long z = seed + group * 0x9E3779B97F4A7C15ull;
  z ^= z >> 30; z *= 0xBF58476D1CE4E5B9ull;
  z ^= z >> 27; z *= 0x94D049BB133111EBull;
  z ^= z >> 31;
#pragma unroll
  for (int i = 0; i < 4; ++i)
    mv[i] = ((unsigned)(z >> (16 * i)) & 0xFFFFu) < thr16 ? inv_keep : 0.f;
}

// 8-element convenience over an 8-aligned linear offset
__device__ __forceinline__ void dtx_dropout8(unsigned long long seed,
                                             long off, unsigned thr16,
                                             float inv_keep, float mv[8]) {
  dtx_dropout4(seed, (unsigned long long)(off >> 2), thr16, inv_keep, mv);
  dtx_dropout4(seed, (unsigned long long)(off >> 2) + 1, thr16, inv_keep,
               mv + 4);
}

// ---- wave/block reductions (64-wide wave) ----
__device__ __forceinline__ float wave_reduce_sum(float x) {
#pragma unroll
  for (int off = 32; off > 0; off >>= 1)
    x += __shfl_xor(x, off, WAVE);
  return x;
}

__device__ __forceinline__ float wave_reduce_max(float x) {
#pragma unroll
  for (int off = 32; off > 0; off >>= 1)
    x = fmaxf(x, __shfl_xor(x, off, WAVE));
  return x;
}

// Block-wide sum for 256-thread blocks; `scratch` must hold >= 4 floats.
// Result valid on all threads.
__device__ __forceinline__ float block_reduce_sum(float x, float* scratch) {
  const int lane = threadIdx.x & (WAVE - 1);
  const int wid = threadIdx.x / WAVE;
  x = wave_reduce_sum(x);
  if (lane == 0) scratch[wid] = x;
  __syncthreads();
  float r = (scratch[0] + scratch[1]) + (scratch[2] + scratch[3]);
  __syncthreads();
  return r;
}

#define DTX_CDIV(a, b) (((a) + (b) - 1) / (b))
