// Flash-attention forward for gfx950 (CDNA4), bf16 I/O, fp32 softmax.
//
// Structure (v1, correctness-first with MFMA throughput):
//   grid = (ceil(S/64), B*Hq); block = 256 threads = 4 waves.
//   Each wave owns 16 q-rows (A-fragment resident in VGPRs, pre-loaded
//   once); KV tiles of 64 staged in LDS per block: K row-major (padded),
//   V transposed (so the PV B-fragment is a contiguous ds_read_b128).
//   S-tile = mfma_f32_16x16x32_bf16(Q, K^T) over D; online softmax in
//   the MFMA C-layout (row = (lane>>4)*4+reg, col = lane&15) with the
//   16-lane xor-shuffle row reduce; P staged per-wave in LDS to convert
//   C-layout -> A-layout for the PV mfma.
// Causal + GQA + ragged S handled by masking. lse (= m + log l) saved
// for the backward. Numerics contract: ops/reference.py attn_fwd.
#include "dtx_common.h"

typedef __attribute__((ext_vector_type(4))) float f32x4;

#define MFMA_B16(a, b, c) __builtin_amdgcn_mfma_f32_16x16x32_bf16(a, b, c, 0, 0, 0)
#define NEG_INF (-3.0e38f)

template <int D>
struct AttnFwdLds {
  unsigned short K[64][D + 8];
  unsigned short VT[D][64 + 8];
  unsigned short P[4][16][64 + 8];
};

template <int D>
__global__ __launch_bounds__(256, 2)
void attn_fwd_kernel(const unsigned short* __restrict__ Q,
                     const unsigned short* __restrict__ Kp,
                     const unsigned short* __restrict__ Vp,
                     unsigned short* __restrict__ O,
                     float* __restrict__ lse_out,
                     int B, int Hq, int Hkv, int S, int Skv,
                     float scale, int causal) {
  constexpr int KVB = 64;
  constexpr int DC = D / 32;       // QK^T k-chunks
  constexpr int NC2 = D / 16;      // PV output column tiles
  __shared__ AttnFwdLds<D> lds;

  const int lane = threadIdx.x & 63;
  const int wid = threadIdx.x >> 6;
  const int l15 = lane & 15;
  const int l4 = lane >> 4;

  const int bh = blockIdx.y;
  const int b = bh / Hq, hq = bh % Hq;
  const int hkv = hq / (Hq / Hkv);
  const int q0 = blockIdx.x * 64;          // block's first q row
  const int qw = q0 + wid * 16;            // wave's first q row

  const long qbase = (((long)b * Hq + hq) * S) * D;
  const long kbase = (((long)b * Hkv + hkv) * Skv) * D;
  const int diag = Skv - S;                // causal diagonal offset

  // ---- load Q fragments (A-layout): row l15, k = kc*32 + l4*8 + j
  short8v qfrag[DC];
  {
    const int qrow = qw + l15;
#pragma unroll
    for (int kc = 0; kc < DC; ++kc) {
      if (qrow < S) {
        qfrag[kc] = *reinterpret_cast<const short8v*>(
            Q + qbase + (long)qrow * D + kc * 32 + l4 * 8);
      } else {
#pragma unroll
        for (int i = 0; i < 8; ++i) qfrag[kc][i] = 0;
      }
    }
  }

  f32x4 o_acc[NC2];
#pragma unroll
  for (int c = 0; c < NC2; ++c) o_acc[c] = f32x4{0.f, 0.f, 0.f, 0.f};
  float m_run[4], l_run[4];
#pragma unroll
  for (int r = 0; r < 4; ++r) { m_run[r] = NEG_INF; l_run[r] = 0.f; }

  // kv tiles this block must visit
  const int q_hi = min(q0 + 63, S - 1);
  const int kv_hi = causal ? min(Skv - 1, q_hi + diag) : (Skv - 1);
  const int ntiles = kv_hi / KVB + 1;

  for (int t = 0; t < ntiles; ++t) {
    const int kv0 = t * KVB;
    // ---- stage K tile [64][D] and V^T tile [D][64]
    {
      const int gpr = D / 8;                      // 16B groups per row
      for (int idx = threadIdx.x; idx < KVB * gpr; idx += 256) {
        const int row = idx / gpr, g = idx - row * gpr;
        short8v kv8 = {0, 0, 0, 0, 0, 0, 0, 0};
        short8v vv8 = {0, 0, 0, 0, 0, 0, 0, 0};
        if (kv0 + row < Skv) {
          kv8 = *reinterpret_cast<const short8v*>(
              Kp + kbase + (long)(kv0 + row) * D + g * 8);
          vv8 = *reinterpret_cast<const short8v*>(
              Vp + kbase + (long)(kv0 + row) * D + g * 8);
        }
        *reinterpret_cast<short8v*>(&lds.K[row][g * 8]) = kv8;
#pragma unroll
        for (int i = 0; i < 8; ++i)
          lds.VT[g * 8 + i][row] = (unsigned short)vv8[i];
      }
    }
    __syncthreads();

    // ---- S tile: s_acc[c] = Q @ K^T (C-layout)
    f32x4 s_acc[4];
#pragma unroll
    for (int c = 0; c < 4; ++c) s_acc[c] = f32x4{0.f, 0.f, 0.f, 0.f};
#pragma unroll
    for (int kc = 0; kc < DC; ++kc) {
#pragma unroll
      for (int c = 0; c < 4; ++c) {
        short8v kf = *reinterpret_cast<const short8v*>(
            &lds.K[c * 16 + l15][kc * 32 + l4 * 8]);
        s_acc[c] = MFMA_B16(qfrag[kc], kf, s_acc[c]);
      }
    }

    // ---- scale + mask (value at q-row (l4*4+reg), kv-col (c*16+l15))
#pragma unroll
    for (int c = 0; c < 4; ++c) {
      const int kvg = kv0 + c * 16 + l15;
#pragma unroll
      for (int r = 0; r < 4; ++r) {
        const int qg = qw + l4 * 4 + r;
        bool dead = (kvg >= Skv) | (qg >= S) |
                    (causal && (kvg > qg + diag));
        s_acc[c][r] = dead ? NEG_INF : s_acc[c][r] * scale;
      }
    }

    // ---- online softmax (per q-row r owned at reg r)
    float p[4][4];
    float alpha[4];
#pragma unroll
    for (int r = 0; r < 4; ++r) {
      float mx = fmaxf(fmaxf(s_acc[0][r], s_acc[1][r]),
                       fmaxf(s_acc[2][r], s_acc[3][r]));
#pragma unroll
      for (int off = 1; off < 16; off <<= 1)
        mx = fmaxf(mx, __shfl_xor(mx, off, 64));
      const float m_new = fmaxf(m_run[r], mx);
      alpha[r] = (m_new == m_run[r]) ? 1.f : __expf(m_run[r] - m_new);
      float rowsum = 0.f;
      if (m_new == NEG_INF) {          // fully-masked row (padding)
#pragma unroll
        for (int c = 0; c < 4; ++c) p[c][r] = 0.f;
      } else {
#pragma unroll
        for (int c = 0; c < 4; ++c) {
          p[c][r] = __expf(s_acc[c][r] - m_new);
          rowsum += p[c][r];
        }
      }
#pragma unroll
      for (int off = 1; off < 16; off <<= 1)
        rowsum += __shfl_xor(rowsum, off, 64);
      l_run[r] = l_run[r] * alpha[r] + rowsum;
      m_run[r] = m_new;
    }
    // rescale O accumulator
#pragma unroll
    for (int c2 = 0; c2 < NC2; ++c2)
#pragma unroll
      for (int r = 0; r < 4; ++r) o_acc[c2][r] *= alpha[r];

    // ---- stage P (C-layout -> LDS), then PV with A-layout reads
#pragma unroll
    for (int c = 0; c < 4; ++c)
#pragma unroll
      for (int r = 0; r < 4; ++r)
        lds.P[wid][l4 * 4 + r][c * 16 + l15] = f2bf(p[c][r]);
    // wave-private LDS: no __syncthreads needed (compiler orders ds ops)
#pragma unroll
    for (int ks = 0; ks < 2; ++ks) {
      short8v pf = *reinterpret_cast<const short8v*>(
          &lds.P[wid][l15][ks * 32 + l4 * 8]);
#pragma unroll
      for (int c2 = 0; c2 < NC2; ++c2) {
        short8v vf = *reinterpret_cast<const short8v*>(
            &lds.VT[c2 * 16 + l15][ks * 32 + l4 * 8]);
        o_acc[c2] = MFMA_B16(pf, vf, o_acc[c2]);
      }
    }
    __syncthreads();
  }

  // ---- epilogue: normalize, store O (bf16) and lse (f32)
#pragma unroll
  for (int r = 0; r < 4; ++r) {
    const int qg = qw + l4 * 4 + r;
    if (qg < S) {
      const float rcp = l_run[r] > 0.f ? 1.f / l_run[r] : 0.f;
      unsigned short* orow = O + qbase + (long)qg * D;
#pragma unroll
      for (int c2 = 0; c2 < NC2; ++c2)
        orow[c2 * 16 + l15] = f2bf(o_acc[c2][r] * rcp);
      if (l15 == 0)
        lse_out[((long)b * Hq + hq) * S + qg] =
            m_run[r] + __logf(fmaxf(l_run[r], 1e-30f));
    }
  }
}

void launch_attn_fwd(const void* q, const void* k, const void* v, void* o,
                     float* lse, int B, int Hq, int Hkv, int S, int Skv,
                     int D, float scale, int causal, hipStream_t st) {
  dim3 grid(DTX_CDIV(S, 64), B * Hq);
  if (D == 128) {
    attn_fwd_kernel<128><<<grid, 256, 0, st>>>(
        (const unsigned short*)q, (const unsigned short*)k,
        (const unsigned short*)v, (unsigned short*)o, lse,
        B, Hq, Hkv, S, Skv, scale, causal);
  } else if (D == 64) {
    attn_fwd_kernel<64><<<grid, 256, 0, st>>>(
        (const unsigned short*)q, (const unsigned short*)k,
        (const unsigned short*)v, (unsigned short*)o, lse,
        B, Hq, Hkv, S, Skv, scale, causal);
  }
}
