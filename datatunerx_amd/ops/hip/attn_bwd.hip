// Flash-attention backward for gfx950, FA2-style recompute from saved
// lse. Three kernels:
//   1. preprocess: delta = rowsum(dO * O)
//   2. dK/dV: per KV-tile block, loop q-tiles; S^T recomputed as
//      mfma(K, Q^T) so P^T lands directly in the contraction layout for
//      dV = P^T dO and dK = dS^T Q (no cross-operand transposes beyond
//      the LDS staging of Q/dO both row-major and transposed).
//   3. dQ: per Q-tile block, loop kv-tiles; dQ = dS K.
// GQA: kernel 2 writes per-Q-head partials [B,Hq,Skv,D]; the host sums
// over the group (deterministic; no atomics anywhere).
// Numerics contract: ops/reference.py attn_bwd.
#include "dtx_common.h"

typedef __attribute__((ext_vector_type(4))) float f32x4;
#define MFMA_B16(a, b, c) __builtin_amdgcn_mfma_f32_16x16x32_bf16(a, b, c, 0, 0, 0)
#define NEG_INF (-3.0e38f)

__device__ __forceinline__ short8v s8load_or_zero(
    const unsigned short* p, bool ok) {
  short8v v;
  if (ok) {
    v = *reinterpret_cast<const short8v*>(p);
  } else {
#pragma unroll
    for (int i = 0; i < 8; ++i) v[i] = 0;
  }
  return v;
}

// ------------------------------------------------------------ preprocess
// delta[rows] = sum_d dO[row,d] * O[row,d]; wave per row, D%32==0.
__global__ __launch_bounds__(DTX_BLOCK)
void attn_delta_kernel(const unsigned short* __restrict__ dO,
                       const unsigned short* __restrict__ O,
                       float* __restrict__ delta, long rows, int D) {
  const int lane = threadIdx.x & 63;
  const int wid = threadIdx.x >> 6;
  const long wstride = (long)gridDim.x * 4;
  for (long row = (long)blockIdx.x * 4 + wid; row < rows; row += wstride) {
    float acc = 0.f;
    for (int d = lane * 2; d < D; d += 128) {
      float a0 = bf2f(dO[row * D + d]), a1 = bf2f(dO[row * D + d + 1]);
      float b0 = bf2f(O[row * D + d]), b1 = bf2f(O[row * D + d + 1]);
      acc += a0 * b0 + a1 * b1;
    }
    acc = wave_reduce_sum(acc);
    if (lane == 0) delta[row] = acc;
  }
}

// --------------------------------------------------------------- dK / dV
template <int D>
struct BwdKVLds {
  unsigned short Qr[32][D + 8];     // Q rows (B-frag for S^T)
  unsigned short dOr[32][D + 8];    // dO rows (B-frag for dP^T)
  unsigned short QT[D][32 + 8];     // Q^T (B-frag for dK)
  unsigned short dOT[D][32 + 8];    // dO^T (B-frag for dV)
  unsigned short PT[4][16][32 + 8];   // per-wave P^T (A-frag stage)
  unsigned short DST[4][16][32 + 8];  // per-wave dS^T (A-frag stage)
  float lse[32];
  float delta[32];
};

template <int D>
__global__ __launch_bounds__(256, 2)
void attn_bwd_dkdv_kernel(const unsigned short* __restrict__ Q,
                          const unsigned short* __restrict__ K,
                          const unsigned short* __restrict__ V,
                          const unsigned short* __restrict__ dO,
                          const float* __restrict__ lse,
                          const float* __restrict__ delta,
                          unsigned short* __restrict__ dKout,  // [B,Hq,Skv,D]
                          unsigned short* __restrict__ dVout,  // [B,Hq,Skv,D]
                          int B, int Hq, int Hkv, int S, int Skv,
                          float scale, int causal) {
  constexpr int DC = D / 32;
  constexpr int NC2 = D / 16;
  __shared__ BwdKVLds<D> lds;
  const int lane = threadIdx.x & 63;
  const int wid = threadIdx.x >> 6;
  const int l15 = lane & 15, l4 = lane >> 4;

  const int bh = blockIdx.y;
  const int b = bh / Hq, hq = bh % Hq;
  const int hkv = hq / (Hq / Hkv);
  const int kv0 = blockIdx.x * 64;
  const int kvw = kv0 + wid * 16;          // wave's kv rows
  const long qbase = (((long)b * Hq + hq) * S) * D;
  const long kbase = (((long)b * Hkv + hkv) * Skv) * D;
  const long obase = (((long)b * Hq + hq) * Skv) * D;  // dK/dV per-hq
  const long lbase = ((long)b * Hq + hq) * S;
  const int diag = Skv - S;

  // resident K and V fragments (A-layout): row kvw+l15, k = kc*32+l4*8
  short8v kfrag[DC], vfrag[DC];
  {
    const int kr = kvw + l15;
    const bool ok = kr < Skv;
#pragma unroll
    for (int kc = 0; kc < DC; ++kc) {
      kfrag[kc] = s8load_or_zero(K + kbase + (long)kr * D + kc * 32 + l4 * 8, ok);
      vfrag[kc] = s8load_or_zero(V + kbase + (long)kr * D + kc * 32 + l4 * 8, ok);
    }
  }
  f32x4 dk_acc[NC2], dv_acc[NC2];
#pragma unroll
  for (int c = 0; c < NC2; ++c) {
    dk_acc[c] = f32x4{0.f, 0.f, 0.f, 0.f};
    dv_acc[c] = f32x4{0.f, 0.f, 0.f, 0.f};
  }

  const int q_lo = causal ? max(0, kv0 - diag) : 0;
  const int qt0 = (q_lo / 32) * 32;

  for (int qt = qt0; qt < S; qt += 32) {
    // ---- stage Q/dO rows + transposes + lse/delta
    {
      const int gpr = D / 8;
      for (int idx = threadIdx.x; idx < 32 * gpr; idx += 256) {
        const int row = idx / gpr, g = idx - row * gpr;
        const bool ok = qt + row < S;
        short8v qv = s8load_or_zero(Q + qbase + (long)(qt + row) * D + g * 8, ok);
        short8v dv = s8load_or_zero(dO + qbase + (long)(qt + row) * D + g * 8, ok);
        *reinterpret_cast<short8v*>(&lds.Qr[row][g * 8]) = qv;
        *reinterpret_cast<short8v*>(&lds.dOr[row][g * 8]) = dv;
#pragma unroll
        for (int i = 0; i < 8; ++i) {
          lds.QT[g * 8 + i][row] = (unsigned short)qv[i];
          lds.dOT[g * 8 + i][row] = (unsigned short)dv[i];
        }
      }
      if (threadIdx.x < 32) {
        const int row = threadIdx.x;
        const bool ok = qt + row < S;
        lds.lse[row] = ok ? lse[lbase + qt + row] : 0.f;
        lds.delta[row] = ok ? delta[lbase + qt + row] : 0.f;
      }
    }
    __syncthreads();

    // ---- S^T = K Q^T ; dP^T = V dO^T   (C-layout: kv=(l4*4+r), q=c*16+l15)
    f32x4 st_acc[2], dpt_acc[2];
#pragma unroll
    for (int c = 0; c < 2; ++c) {
      st_acc[c] = f32x4{0.f, 0.f, 0.f, 0.f};
      dpt_acc[c] = f32x4{0.f, 0.f, 0.f, 0.f};
    }
#pragma unroll
    for (int kc = 0; kc < DC; ++kc) {
#pragma unroll
      for (int c = 0; c < 2; ++c) {
        short8v qf = *reinterpret_cast<const short8v*>(
            &lds.Qr[c * 16 + l15][kc * 32 + l4 * 8]);
        short8v df = *reinterpret_cast<const short8v*>(
            &lds.dOr[c * 16 + l15][kc * 32 + l4 * 8]);
        st_acc[c] = MFMA_B16(kfrag[kc], qf, st_acc[c]);
        dpt_acc[c] = MFMA_B16(vfrag[kc], df, dpt_acc[c]);
      }
    }

    // ---- P^T and dS^T (elementwise in C-layout), stage to LDS
#pragma unroll
    for (int c = 0; c < 2; ++c) {
      const int qg = qt + c * 16 + l15;
      const float l_q = lds.lse[c * 16 + l15];
      const float d_q = lds.delta[c * 16 + l15];
#pragma unroll
      for (int r = 0; r < 4; ++r) {
        const int kvg = kvw + l4 * 4 + r;
        bool dead = (kvg >= Skv) | (qg >= S) |
                    (causal && (kvg > qg + diag));
        float pt = dead ? 0.f : __expf(st_acc[c][r] * scale - l_q);
        float dst = pt * (dpt_acc[c][r] - d_q) * scale;
        lds.PT[wid][l4 * 4 + r][c * 16 + l15] = f2bf(pt);
        lds.DST[wid][l4 * 4 + r][c * 16 + l15] = f2bf(dst);
      }
    }

    // ---- dV += P^T dO ; dK += dS^T Q  (contraction over 32 q)
    short8v ptf = *reinterpret_cast<const short8v*>(
        &lds.PT[wid][l15][l4 * 8]);
    short8v dstf = *reinterpret_cast<const short8v*>(
        &lds.DST[wid][l15][l4 * 8]);
#pragma unroll
    for (int c2 = 0; c2 < NC2; ++c2) {
      short8v dotf = *reinterpret_cast<const short8v*>(
          &lds.dOT[c2 * 16 + l15][l4 * 8]);
      short8v qtf = *reinterpret_cast<const short8v*>(
          &lds.QT[c2 * 16 + l15][l4 * 8]);
      dv_acc[c2] = MFMA_B16(ptf, dotf, dv_acc[c2]);
      dk_acc[c2] = MFMA_B16(dstf, qtf, dk_acc[c2]);
    }
    __syncthreads();
  }

  // ---- store dK/dV (per-hq layout [B,Hq,Skv,D]; host reduces GQA groups)
#pragma unroll
  for (int r = 0; r < 4; ++r) {
    const int kvg = kvw + l4 * 4 + r;
    if (kvg < Skv) {
      unsigned short* dkrow = dKout + obase + (long)kvg * D;
      unsigned short* dvrow = dVout + obase + (long)kvg * D;
#pragma unroll
      for (int c2 = 0; c2 < NC2; ++c2) {
        dkrow[c2 * 16 + l15] = f2bf(dk_acc[c2][r]);
        dvrow[c2 * 16 + l15] = f2bf(dv_acc[c2][r]);
      }
    }
  }
}

// -------------------------------------------------------------------- dQ
template <int D>
struct BwdQLds {
  unsigned short Kr[32][D + 8];       // K rows (B-frag for S)
  unsigned short Vr[32][D + 8];       // V rows (B-frag for dP)
  unsigned short KT[D][32 + 8];       // K^T (B-frag for dQ)
  unsigned short DS[4][16][32 + 8];   // per-wave dS (A-frag stage)
};

template <int D>
__global__ __launch_bounds__(256, 2)
void attn_bwd_dq_kernel(const unsigned short* __restrict__ Q,
                        const unsigned short* __restrict__ K,
                        const unsigned short* __restrict__ V,
                        const unsigned short* __restrict__ dO,
                        const float* __restrict__ lse,
                        const float* __restrict__ delta,
                        unsigned short* __restrict__ dQout,
                        int B, int Hq, int Hkv, int S, int Skv,
                        float scale, int causal) {
  constexpr int DC = D / 32;
  constexpr int NC2 = D / 16;
  __shared__ BwdQLds<D> lds;
  const int lane = threadIdx.x & 63;
  const int wid = threadIdx.x >> 6;
  const int l15 = lane & 15, l4 = lane >> 4;

  const int bh = blockIdx.y;
  const int b = bh / Hq, hq = bh % Hq;
  const int hkv = hq / (Hq / Hkv);
  const int q0 = blockIdx.x * 64;
  const int qw = q0 + wid * 16;
  const long qbase = (((long)b * Hq + hq) * S) * D;
  const long kbase = (((long)b * Hkv + hkv) * Skv) * D;
  const long lbase = ((long)b * Hq + hq) * S;
  const int diag = Skv - S;

  // resident Q and dO fragments (A-layout) + per-row lse/delta
  short8v qfrag[DC], dofrag[DC];
  {
    const int qr = qw + l15;
    const bool ok = qr < S;
#pragma unroll
    for (int kc = 0; kc < DC; ++kc) {
      qfrag[kc] = s8load_or_zero(Q + qbase + (long)qr * D + kc * 32 + l4 * 8, ok);
      dofrag[kc] = s8load_or_zero(dO + qbase + (long)qr * D + kc * 32 + l4 * 8, ok);
    }
  }
  float lse_r[4], delta_r[4];
#pragma unroll
  for (int r = 0; r < 4; ++r) {
    const int qg = qw + l4 * 4 + r;
    lse_r[r] = (qg < S) ? lse[lbase + qg] : 0.f;
    delta_r[r] = (qg < S) ? delta[lbase + qg] : 0.f;
  }
  f32x4 dq_acc[NC2];
#pragma unroll
  for (int c = 0; c < NC2; ++c) dq_acc[c] = f32x4{0.f, 0.f, 0.f, 0.f};

  const int q_hi = min(q0 + 63, S - 1);
  const int kv_hi = causal ? min(Skv - 1, q_hi + diag) : (Skv - 1);

  for (int kt = 0; kt <= kv_hi; kt += 32) {
    // ---- stage K/V rows + K^T
    {
      const int gpr = D / 8;
      for (int idx = threadIdx.x; idx < 32 * gpr; idx += 256) {
        const int row = idx / gpr, g = idx - row * gpr;
        const bool ok = kt + row < Skv;
        short8v kv8 = s8load_or_zero(K + kbase + (long)(kt + row) * D + g * 8, ok);
        short8v vv8 = s8load_or_zero(V + kbase + (long)(kt + row) * D + g * 8, ok);
        *reinterpret_cast<short8v*>(&lds.Kr[row][g * 8]) = kv8;
        *reinterpret_cast<short8v*>(&lds.Vr[row][g * 8]) = vv8;
#pragma unroll
        for (int i = 0; i < 8; ++i) lds.KT[g * 8 + i][row] = (unsigned short)kv8[i];
      }
    }
    __syncthreads();

    // ---- S = Q K^T ; dP = dO V^T  (C-layout: q=(l4*4+r), kv=c*16+l15)
    f32x4 s_acc[2], dp_acc[2];
#pragma unroll
    for (int c = 0; c < 2; ++c) {
      s_acc[c] = f32x4{0.f, 0.f, 0.f, 0.f};
      dp_acc[c] = f32x4{0.f, 0.f, 0.f, 0.f};
    }
#pragma unroll
    for (int kc = 0; kc < DC; ++kc) {
#pragma unroll
      for (int c = 0; c < 2; ++c) {
        short8v kf = *reinterpret_cast<const short8v*>(
            &lds.Kr[c * 16 + l15][kc * 32 + l4 * 8]);
        short8v vf = *reinterpret_cast<const short8v*>(
            &lds.Vr[c * 16 + l15][kc * 32 + l4 * 8]);
        s_acc[c] = MFMA_B16(qfrag[kc], kf, s_acc[c]);
        dp_acc[c] = MFMA_B16(dofrag[kc], vf, dp_acc[c]);
      }
    }

    // ---- dS = P (dP - delta) scale, stage per-wave
#pragma unroll
    for (int c = 0; c < 2; ++c) {
      const int kvg = kt + c * 16 + l15;
#pragma unroll
      for (int r = 0; r < 4; ++r) {
        const int qg = qw + l4 * 4 + r;
        bool dead = (kvg >= Skv) | (qg >= S) |
                    (causal && (kvg > qg + diag));
        float p = dead ? 0.f : __expf(s_acc[c][r] * scale - lse_r[r]);
        float ds = p * (dp_acc[c][r] - delta_r[r]) * scale;
        lds.DS[wid][l4 * 4 + r][c * 16 + l15] = f2bf(ds);
      }
    }

    // ---- dQ += dS K  (contraction over 32 kv)
    short8v dsf = *reinterpret_cast<const short8v*>(
        &lds.DS[wid][l15][l4 * 8]);
#pragma unroll
    for (int c2 = 0; c2 < NC2; ++c2) {
      short8v ktf = *reinterpret_cast<const short8v*>(
          &lds.KT[c2 * 16 + l15][l4 * 8]);
      dq_acc[c2] = MFMA_B16(dsf, ktf, dq_acc[c2]);
    }
    __syncthreads();
  }

  // ---- store dQ
#pragma unroll
  for (int r = 0; r < 4; ++r) {
    const int qg = qw + l4 * 4 + r;
    if (qg < S) {
      unsigned short* dqrow = dQout + qbase + (long)qg * D;
#pragma unroll
      for (int c2 = 0; c2 < NC2; ++c2)
        dqrow[c2 * 16 + l15] = f2bf(dq_acc[c2][r]);
    }
  }
}

// ------------------------------------------------------------- launchers
void launch_attn_delta(const void* dO, const void* O, float* delta,
                       long rows, int D, hipStream_t s) {
  long w = DTX_CDIV(rows, 4);
  int grid = (int)(w < 2048 ? (w < 1 ? 1 : w) : 2048);
  attn_delta_kernel<<<grid, DTX_BLOCK, 0, s>>>(
      (const unsigned short*)dO, (const unsigned short*)O, delta, rows, D);
}

void launch_attn_bwd(const void* q, const void* k, const void* v,
                     const void* dO, const float* lse, const float* delta,
                     void* dq, void* dk, void* dv,
                     int B, int Hq, int Hkv, int S, int Skv, int D,
                     float scale, int causal, hipStream_t st) {
#define LAUNCH(DD)                                                        \
  do {                                                                    \
    dim3 gkv(DTX_CDIV(Skv, 64), B * Hq);                                  \
    attn_bwd_dkdv_kernel<DD><<<gkv, 256, 0, st>>>(                        \
        (const unsigned short*)q, (const unsigned short*)k,               \
        (const unsigned short*)v, (const unsigned short*)dO, lse, delta,  \
        (unsigned short*)dk, (unsigned short*)dv,                         \
        B, Hq, Hkv, S, Skv, scale, causal);                               \
    dim3 gq(DTX_CDIV(S, 64), B * Hq);                                     \
    attn_bwd_dq_kernel<DD><<<gq, 256, 0, st>>>(                           \
        (const unsigned short*)q, (const unsigned short*)k,               \
        (const unsigned short*)v, (const unsigned short*)dO, lse, delta,  \
        (unsigned short*)dq, B, Hq, Hkv, S, Skv, scale, causal);          \
  } while (0)
  if (D == 128) LAUNCH(128);
  else if (D == 64) LAUNCH(64);
#undef LAUNCH
}
