// Flash-attention forward for gfx950 (CDNA4), bf16 I/O, fp32 softmax.
//
// v2 — 8-wave 32x32-MFMA structure (guide Appendix B "8-warp 32x32
// ladder"): block = 512 threads = 8 waves, each wave owns 32 q-rows
// (QBLK=32, block covers 256), KV tiles of 64 staged in LDS.
//
//  - Layout is BSHD (q/k: [B,S,H,D]) — no transpose copies in the model;
//    V arrives PRE-TRANSPOSED as VT [B,Hkv,D,Skv] (transpose_sd kernel).
//  - Swapped QK^T: S^T[kv][q] = mfma(A=K-frag, B=Q-frag) puts a whole
//    P-row (over kv) in each lane's registers -> softmax (row max, exp2,
//    row sum) is lane-local; the cross-half reduce is one shfl_xor(32).
//  - P(C-layout) -> PV A-fragments via v_cvt_pk_bf16_f32 +
//    permlane32_swap (16 cvt + 8 swaps per tile, no LDS round trip).
//  - PV unswapped: O[q][d] = mfma(A=P, B=V^T-frag from the VT tile); the
//    per-q-row rescale factor is fetched with one ds_bpermute per
//    accumulator row.
//  - K LDS tile [64][D+8] and VT tile [D][64+8]: +8 padding makes the
//    column-subtile ds_read_b128 fragment reads bank-conflict-free
//    (row strides 68/36 dwords; distinct (4r mod 64) starts per group).
//
// Numerics contract: ops/reference.py attn_fwd (fp32 softmax, exp2
// domain internally, natural-log lse out). Causal + GQA + ragged S by
// masking; interior tiles skip the mask; waves fully above the causal
// diagonal skip compute.
#include "dtx_common.h"


typedef __attribute__((ext_vector_type(16))) float f32x16;
typedef __attribute__((ext_vector_type(4))) unsigned int u32x4;
typedef __attribute__((ext_vector_type(2))) unsigned int uint2_t;

#define MFMA32(a, b, c) __builtin_amdgcn_mfma_f32_32x32x16_bf16(a, b, c, 0, 0, 0)
#define NEG_INF (-3.0e38f)
#define LOG2E 1.4426950408889634f
#define LN2 0.6931471805599453f

__device__ __forceinline__ unsigned cvt_pk_bf16(float lo, float hi) {
  unsigned r;
  // s_nop 1 covers the VALU-write -> v_permlane read hazard window
  // (guide T21) for the swap that consumes this value next.
  asm volatile("v_cvt_pk_bf16_f32 %0, %1, %2\n\ts_nop 1"
               : "=v"(r) : "v"(lo), "v"(hi));
  return r;
}

// C-layout row index for f32x16 accumulator register r (32x32 MFMA).
__device__ __forceinline__ int crow(int r, int hi) {
  return (r & 3) + 8 * (r >> 2) + 4 * hi;
}

// MFMA B-fragment from a ROW-major [k][n] LDS image via the gfx950
// hardware transpose read (mapping verified by tools/probe_tr.cpp;
// see attn_bwd.hip tr_bfrag for the derivation).
typedef __attribute__((ext_vector_type(4))) __bf16 bf16x4f;
#define DTX_AS3F __attribute__((address_space(3)))
template <int STRIDE>
__device__ __forceinline__ short8v fw_tr_bfrag(
    const DTX_AS3F unsigned short* img, int k0, int n0, int lane) {
  const int row = k0 + ((lane >> 2) & 3);
  const int col = n0 + ((lane >> 4) & 1) * 16 + (lane & 3) * 4;
  const DTX_AS3F unsigned short* p = img + row * STRIDE + col;
  union { bf16x4f v[2]; short8v s; } u;
  u.v[0] = __builtin_amdgcn_ds_read_tr16_b64_v4bf16(
      (DTX_AS3F bf16x4f*)p);
  u.v[1] = __builtin_amdgcn_ds_read_tr16_b64_v4bf16(
      (DTX_AS3F bf16x4f*)(p + 4 * STRIDE));
  return u.s;
}

template <int D>
struct AttnFwdLds {
  // double-buffered stages: ONE barrier per 128 kv rows instead of the
  // [sync; write; sync] full stop (same restructure as the bwd kernels)
  unsigned short K[2][128][D + 8];
  unsigned short V[2][128][D + 8];  // row-major; PV B-frags via tr-read
};

// WAVES: q-rows per workgroup = 32*WAVES. 8 (256 rows) shares each
// staged K/V tile across more q-rows; 4 (128 rows) halves the causal
// wave-skew inside the workgroup (utilization 82.5% -> 91.7% at
// S=1024) and lets TWO workgroups co-reside per CU. Measured A/B picks
// the launcher default.
template <int D, int WAVES = 8>
__global__ __launch_bounds__(64 * WAVES, 1)
void attn_fwd2_kernel(const unsigned short* __restrict__ Q,
                      const unsigned short* __restrict__ Kp,
                      const unsigned short* __restrict__ Vp,
                      unsigned short* __restrict__ O,
                      float* __restrict__ lse_out,
                      int B, int Hq, int Hkv, int S, int Skv,
                      float scale, int causal) {
  constexpr int KVB = 64;
  constexpr int DC16 = D / 16;       // QK^T d-chunks (k-dim 16 each)
  constexpr int ND32 = D / 32;       // PV output d-subtiles
  __shared__ AttnFwdLds<D> lds;

  const int lane = threadIdx.x & 63;
  const int wid = threadIdx.x >> 6;
  const int l31 = lane & 31;
  const int hi = lane >> 5;

  const int bh = blockIdx.y;
  const int b = bh / Hq, hq = bh % Hq;
  const int hkv = hq / (Hq / Hkv);
  constexpr int TPB = 64 * WAVES;
  constexpr int QROWS = 32 * WAVES;
  const int q0 = blockIdx.x * QROWS;
  const int qw = q0 + wid * 32;            // wave's first q row

  const int qrowstr = Hq * D;              // BSHD row stride
  const int krowstr = Hkv * D;
  const long qbase = (long)b * S * qrowstr + (long)hq * D;
  const long kbase = (long)b * Skv * krowstr + (long)hkv * D;
  const long lbase = ((long)b * Hq + hq) * S;
  const int diag = Skv - S;                // causal diagonal offset

  // ---- Q fragments (B-layout for swapped QK^T): lane holds
  // Q[qw+l31][kc*16 + hi*8 + j]
  short8v qfrag[DC16];
  {
    const int qrow = qw + l31;
#pragma unroll
    for (int kc = 0; kc < DC16; ++kc) {
      if (qrow < S) {
        qfrag[kc] = *reinterpret_cast<const short8v*>(
            Q + qbase + (long)qrow * qrowstr + kc * 16 + hi * 8);
      } else {
        qfrag[kc] = short8v{0, 0, 0, 0, 0, 0, 0, 0};
      }
    }
  }

  f32x16 o_acc[ND32];
#pragma unroll
  for (int c = 0; c < ND32; ++c)
#pragma unroll
    for (int r = 0; r < 16; ++r) o_acc[c][r] = 0.f;
  float m_run = NEG_INF, l_run = 0.f;      // per-lane: q-row qw + l31
  const float kscale = scale * LOG2E;      // softmax in exp2 domain

  const int q_hi_blk = min(q0 + QROWS - 1, S - 1);
  const int kv_hi = causal ? min(Skv - 1, q_hi_blk + diag) : (Skv - 1);
  const int nstages = kv_hi / 128 + 1;     // 128 kv rows per stage

  // T14 async-stage split: the next stage's K/VT global loads are
  // issued while the current stage computes; LDS writes after the
  // barrier.  KIT/VIT iterations cover the 128-row K and VT tiles.
  constexpr int KIT = (128 * D / 8) / TPB;
  short8v stg[KIT * 2];
  auto issue_stage = [&](int st2) {
    const int kvs = st2 * 128;
#pragma unroll
    for (int it = 0; it < KIT; ++it) {
      const int idx = threadIdx.x + it * TPB;
      const int row = idx / (D / 8), g = idx % (D / 8);
      short8v k8 = {0, 0, 0, 0, 0, 0, 0, 0};
      short8v v8 = {0, 0, 0, 0, 0, 0, 0, 0};
      if (kvs + row < Skv) {
        const long off = kbase + (long)(kvs + row) * krowstr + g * 8;
        k8 = *reinterpret_cast<const short8v*>(Kp + off);
        v8 = *reinterpret_cast<const short8v*>(Vp + off);
      }
      stg[it * 2] = k8;
      stg[it * 2 + 1] = v8;
    }
  };
  auto write_stage = [&](int slot) {
#pragma unroll
    for (int it = 0; it < KIT; ++it) {
      const int idx = threadIdx.x + it * TPB;
      const int row = idx / (D / 8), g = idx % (D / 8);
      *reinterpret_cast<short8v*>(&lds.K[slot][row][g * 8]) =
          stg[it * 2];
      *reinterpret_cast<short8v*>(&lds.V[slot][row][g * 8]) =
          stg[it * 2 + 1];
    }
  };
  issue_stage(0);
  write_stage(0);
  __syncthreads();
  int cur = 0;
  for (int st2 = 0; st2 < nstages; ++st2) {
    if (st2 + 1 < nstages) issue_stage(st2 + 1);
    const int kvs = st2 * 128;

    for (int half = 0; half < 2; ++half) {
    const int kv0 = kvs + half * KVB;
    const int koff = half * KVB;             // LDS row offset
    const bool wave_dead = (kv0 > kv_hi) ||
                           (causal && (kv0 > qw + 31 + diag));
    if (!wave_dead) {
      // ---- S^T tiles: s[ss] = K[kv0+ss*32..][*] x Q^T  (C-layout:
      // row kv = crow(r,hi)+32*ss, col q = qw+l31)
      f32x16 s0v, s1v;
#pragma unroll
      for (int r = 0; r < 16; ++r) { s0v[r] = 0.f; s1v[r] = 0.f; }
#pragma unroll
      for (int kc = 0; kc < DC16; ++kc) {
        short8v k0 = *reinterpret_cast<const short8v*>(
            &lds.K[cur][koff + l31][kc * 16 + hi * 8]);
        short8v k1 = *reinterpret_cast<const short8v*>(
            &lds.K[cur][koff + 32 + l31][kc * 16 + hi * 8]);
        s0v = MFMA32(k0, qfrag[kc], s0v);
        s1v = MFMA32(k1, qfrag[kc], s1v);
      }

      // ---- mask on boundary tiles (raw scores; kscale is folded into
      // the exp2 later: p = exp2(fma(s, kscale, -m)), saving the
      // separate 32-mult scale pass on every tile — guide "exp2+fma
      // fold")
      const bool interior =
          (kv0 + KVB <= Skv) &&
          (!causal || (kv0 + KVB - 1 <= qw + diag));
      const int qg = qw + l31;
      if (!interior) {
#pragma unroll
        for (int r = 0; r < 16; ++r) {
          const int kv_a = kv0 + crow(r, hi);
          const int kv_b = kv_a + 32;
          bool dead_a = (kv_a >= Skv) | (causal && (kv_a > qg + diag));
          bool dead_b = (kv_b >= Skv) | (causal && (kv_b > qg + diag));
          if (dead_a) s0v[r] = NEG_INF;
          if (dead_b) s1v[r] = NEG_INF;
        }
      }

      // ---- online softmax, lane-local over 32 regs + one cross-half
      // (max computed on RAW scores; kscale > 0 commutes with max)
      float mxr = s0v[0];
#pragma unroll
      for (int r = 1; r < 16; ++r) mxr = fmaxf(mxr, s0v[r]);
#pragma unroll
      for (int r = 0; r < 16; ++r) mxr = fmaxf(mxr, s1v[r]);
      float mx = fmaxf(mxr, __shfl_xor(mxr, 32, 64));
      mx = mx <= NEG_INF ? NEG_INF : mx * kscale;
      // defer-max (guide ladder): if no lane's tile max exceeds the
      // running max by more than 8 (exp2 domain -> P bounded by 2^8),
      // keep m_run and SKIP the O-rescale (16 bpermutes + 64 mults).
      // First live tile has m_run = NEG_INF so defer is never taken
      // there; NaN (dead vs dead) compares false -> normal path.
      const bool defer = __all(mx - m_run <= 8.f);
      const float m_new = defer ? m_run : fmaxf(m_run, mx);
      const float alpha =
          (defer || m_new == NEG_INF) ? 1.f
                                      : __builtin_exp2f(m_run - m_new);
      float rowsum = 0.f;
      if (m_new == NEG_INF) {
#pragma unroll
        for (int r = 0; r < 16; ++r) { s0v[r] = 0.f; s1v[r] = 0.f; }
      } else {
#pragma unroll
        for (int r = 0; r < 16; ++r) {
          s0v[r] = __builtin_exp2f(fmaf(s0v[r], kscale, -m_new));
          s1v[r] = __builtin_exp2f(fmaf(s1v[r], kscale, -m_new));
          rowsum += s0v[r] + s1v[r];
        }
      }
      rowsum += __shfl_xor(rowsum, 32, 64);
      l_run = l_run * alpha + rowsum;
      m_run = m_new;

      // ---- P (C-layout f32) -> bf16 A-fragments [q][kv-slice]
      short8v pa[4];
#pragma unroll
      for (int ss = 0; ss < 2; ++ss) {
        const f32x16& p = ss ? s1v : s0v;
        unsigned c0 = cvt_pk_bf16(p[0], p[1]);
        unsigned c1 = cvt_pk_bf16(p[2], p[3]);
        unsigned c2 = cvt_pk_bf16(p[4], p[5]);
        unsigned c3 = cvt_pk_bf16(p[6], p[7]);
        unsigned c4 = cvt_pk_bf16(p[8], p[9]);
        unsigned c5 = cvt_pk_bf16(p[10], p[11]);
        unsigned c6 = cvt_pk_bf16(p[12], p[13]);
        unsigned c7 = cvt_pk_bf16(p[14], p[15]);
        auto r02 = __builtin_amdgcn_permlane32_swap(c0, c2, false, false);
        auto r13 = __builtin_amdgcn_permlane32_swap(c1, c3, false, false);
        auto r46 = __builtin_amdgcn_permlane32_swap(c4, c6, false, false);
        auto r57 = __builtin_amdgcn_permlane32_swap(c5, c7, false, false);
        u32x4 lo{(unsigned)r02[0], (unsigned)r13[0],
                 (unsigned)r02[1], (unsigned)r13[1]};
        u32x4 hi4{(unsigned)r46[0], (unsigned)r57[0],
                  (unsigned)r46[1], (unsigned)r57[1]};
        pa[2 * ss + 0] = *reinterpret_cast<short8v*>(&lo);
        pa[2 * ss + 1] = *reinterpret_cast<short8v*>(&hi4);
      }

      // ---- rescale O rows by alpha[q-row] (one bpermute per row)
      if (!defer && (st2 > 0 || half > 0)) {
#pragma unroll
        for (int r = 0; r < 16; ++r) {
          const float a_r = __shfl(alpha, crow(r, hi), 64);
#pragma unroll
          for (int c = 0; c < ND32; ++c) o_acc[c][r] *= a_r;
        }
      }

      // ---- PV: O[q][d] += P[q][kv] x V-frag (tr-read, row-major V).
      // ks OUTER so consecutive MFMAs hit DIFFERENT accumulators
      // (o_acc[0..3]) — back-to-back MFMAs on one accumulator pay the
      // full RAW latency (guide: SQ_WAIT_INST_ANY).
      const DTX_AS3F unsigned short* v3 =
          (const DTX_AS3F unsigned short*)&lds.V[cur][0][0];
#pragma unroll
      for (int ks = 0; ks < 4; ++ks) {
#pragma unroll
        for (int c = 0; c < ND32; ++c) {
          short8v vf = fw_tr_bfrag<D + 8>(v3, koff + ks * 16 + hi * 8,
                                          c * 32, lane);
          o_acc[c] = MFMA32(pa[ks], vf, o_acc[c]);
        }
        __builtin_amdgcn_sched_barrier(0);  // no cross-ks frag hoist
      }
    }
    }  // half
    if (st2 + 1 < nstages) write_stage(cur ^ 1);
    __syncthreads();
    cur ^= 1;
  }

  // ---- epilogue: normalize rows, store O per wave IMMEDIATELY (an
  // LDS-restaged barrier epilogue was measured 20% SLOWER end-to-end:
  // the block-wide sync serializes the causal wave skew that per-wave
  // stores overlap). cvt_pk packs row-pairs so the convert is one VALU
  // op per two elements instead of the branchy scalar f2bf.
  const float rcp = l_run > 0.f ? 1.f / l_run : 0.f;
#pragma unroll
  for (int r = 0; r < 16; r += 2) {
    const int qa = qw + crow(r, hi);
    const int qb = qw + crow(r + 1, hi);
    const float n_a = __shfl(rcp, crow(r, hi), 64);
    const float n_b = __shfl(rcp, crow(r + 1, hi), 64);
    unsigned short* orow_a = O + qbase + (long)qa * qrowstr;
    unsigned short* orow_b = O + qbase + (long)qb * qrowstr;
#pragma unroll
    for (int c = 0; c < ND32; ++c) {
      const unsigned pk = dtx_cvt_pk_bf16(o_acc[c][r] * n_a,
                                          o_acc[c][r + 1] * n_b);
      if (qa < S) orow_a[c * 32 + l31] = (unsigned short)(pk & 0xffff);
      if (qb < S) orow_b[c * 32 + l31] = (unsigned short)(pk >> 16);
    }
  }
  if (hi == 0) {
    const int qg = qw + l31;
    if (qg < S)
      lse_out[lbase + qg] =
          m_run * LN2 + __logf(fmaxf(l_run, 1e-30f));
  }
}

// ---------------------------------------------------------------- V^T
// [B,S,H,D] -> [B,H,D,S] tile transpose (64x64 tiles through LDS).
__global__ __launch_bounds__(256)
void transpose_sd_kernel(const unsigned short* __restrict__ X,
                         unsigned short* __restrict__ XT,
                         int B, int S, int H, int D) {
  __shared__ unsigned short tile[64][72];
  const int s0 = blockIdx.x * 64;
  const int d0 = blockIdx.y * 64;
  const int bh = blockIdx.z;
  const int b = bh / H, h = bh % H;
  const long xbase = (long)b * S * H * D + (long)h * D;
  const long tbase = ((long)b * H + h) * (long)D * S;
  const int rowstr = H * D;

#pragma unroll
  for (int i = 0; i < 2; ++i) {
    const int row = (threadIdx.x >> 3) + i * 32;     // s offset
    const int g = threadIdx.x & 7;                   // d group
    short8v v8 = {0, 0, 0, 0, 0, 0, 0, 0};
    if (s0 + row < S)
      v8 = *reinterpret_cast<const short8v*>(
          X + xbase + (long)(s0 + row) * rowstr + d0 + g * 8);
#pragma unroll
    for (int j = 0; j < 8; ++j) tile[g * 8 + j][row] = (unsigned short)v8[j];
  }
  __syncthreads();
#pragma unroll
  for (int i = 0; i < 2; ++i) {
    const int drow = (threadIdx.x >> 3) + i * 32;    // d offset
    const int g = threadIdx.x & 7;                   // s group
    if (s0 + g * 8 + 8 <= S) {
      *reinterpret_cast<short8v*>(XT + tbase + (long)(d0 + drow) * S +
                                  s0 + g * 8) =
          *reinterpret_cast<const short8v*>(&tile[drow][g * 8]);
    } else {
#pragma unroll
      for (int j = 0; j < 8; ++j)
        if (s0 + g * 8 + j < S)
          XT[tbase + (long)(d0 + drow) * S + s0 + g * 8 + j] =
              tile[drow][g * 8 + j];
    }
  }
}

void launch_transpose_sd(const void* x, void* xt, int B, int S, int H,
                         int D, hipStream_t st) {
  dim3 grid(DTX_CDIV(S, 64), DTX_CDIV(D, 64), B * H);
  transpose_sd_kernel<<<grid, 256, 0, st>>>(
      (const unsigned short*)x, (unsigned short*)xt, B, S, H, D);
}

void launch_attn_fwd(const void* q, const void* k, const void* v, void* o,
                     float* lse, int B, int Hq, int Hkv, int S, int Skv,
                     int D, float scale, int causal, hipStream_t st) {
  // 4-wave workgroups measured 45% SLOWER than 8-wave on the training
  // shape (a second 256-thread WG does not co-reside at 256 VGPRs, so
  // half the CU idles); keep 8-wave and leave the variant for probes.
  const bool w4 = getenv("DTX_ATTN_W4") != nullptr;
  if (w4) {
    dim3 grid(DTX_CDIV(S, 128), B * Hq);
    if (D == 128) {
      attn_fwd2_kernel<128, 4><<<grid, 256, 0, st>>>(
          (const unsigned short*)q, (const unsigned short*)k,
          (const unsigned short*)v, (unsigned short*)o, lse,
          B, Hq, Hkv, S, Skv, scale, causal);
    } else if (D == 64) {
      attn_fwd2_kernel<64, 4><<<grid, 256, 0, st>>>(
          (const unsigned short*)q, (const unsigned short*)k,
          (const unsigned short*)v, (unsigned short*)o, lse,
          B, Hq, Hkv, S, Skv, scale, causal);
    }
    return;
  }
  dim3 grid(DTX_CDIV(S, 256), B * Hq);
  if (D == 128) {
    attn_fwd2_kernel<128, 8><<<grid, 512, 0, st>>>(
        (const unsigned short*)q, (const unsigned short*)k,
        (const unsigned short*)v, (unsigned short*)o, lse,
        B, Hq, Hkv, S, Skv, scale, causal);
  } else if (D == 64) {
    attn_fwd2_kernel<64, 8><<<grid, 512, 0, st>>>(
        (const unsigned short*)q, (const unsigned short*)k,
        (const unsigned short*)v, (unsigned short*)o, lse,
        B, Hq, Hkv, S, Skv, scale, causal);
  }
}

// ------------------------------------------------------- decode (S=1)
// Flash-decoding for the serving engine: one token's attention against
// a long KV cache. Split-KV partials (m, l, o[D]) per 512-row chunk,
// combined by a second small kernel. The training kernel would leave
// 255/256 of each block idle at S=1; this one streams K/V at the HBM
// rate with every lane busy.
//   grid (NS, B*Hq), block 256 = 4 waves; 16 lanes per kv row (8 els
//   each), 4 row slots per wave.
// len_dev (optional): current cache length read on DEVICE so a
// hipGraph-captured decode step can grow the window without re-capture
// (Skv stays the CAPACITY for addressing; len_dev bounds the rows).
template <int D>
__global__ __launch_bounds__(DTX_BLOCK)
void attn_decode_kernel(const unsigned short* __restrict__ Q,
                        const unsigned short* __restrict__ Kp,
                        const unsigned short* __restrict__ Vp,
                        const int* __restrict__ len_dev,
                        float* __restrict__ part,   // [B*Hq, NS, D+2]
                        int B, int Hq, int Hkv, int Skv, int chunk,
                        float scale, int len_stride) {
  constexpr int LPR = D / 8;                   // lanes per row
  constexpr int SLOTS = 64 / LPR;              // row slots per wave
  const int lane = threadIdx.x & 63;
  const int wid = threadIdx.x >> 6;
  const int slot = lane / LPR;
  const int g = lane % LPR;

  const int bh = blockIdx.y;
  const int b = bh / Hq, hq = bh % Hq;
  const int hkv = hq / (Hq / Hkv);
  const int krowstr = Hkv * D;
  const long kbase = (long)b * Skv * krowstr + (long)hkv * D;
  const int len = len_dev ? len_dev[b * len_stride] : Skv;
  const int kv0 = blockIdx.x * chunk;
  const int kv_end = min(len, kv0 + chunk);

  // q slice for this lane (f32)
  float qv[8];
  {
    const long qoff = ((long)b * Hq + hq) * D + g * 8;
    load_bf16x8(Q + qoff, qv);
#pragma unroll
    for (int i = 0; i < 8; ++i) qv[i] *= scale;
  }

  float m_run = NEG_INF, l_run = 0.f;
  float o_acc[8];
#pragma unroll
  for (int i = 0; i < 8; ++i) o_acc[i] = 0.f;

  const int rows_per_iter = 4 * SLOTS;         // per block
  for (int r = kv0 + wid * SLOTS + slot; r < kv_end; r += rows_per_iter) {
    const long off = kbase + (long)r * krowstr + g * 8;
    float kv8[8];
    load_bf16x8(Kp + off, kv8);
    float dot = 0.f;
#pragma unroll
    for (int i = 0; i < 8; ++i) dot += qv[i] * kv8[i];
#pragma unroll
    for (int o = LPR / 2; o > 0; o >>= 1)
      dot += __shfl_xor(dot, o, 64);           // reduce within the row
    const float m_new = fmaxf(m_run, dot);
    const float alpha = __builtin_exp2f((m_run - m_new) * LOG2E);
    const float pr = __builtin_exp2f((dot - m_new) * LOG2E);
    l_run = l_run * alpha + pr;
    m_run = m_new;
    float vv[8];
    load_bf16x8(Vp + off, vv);
#pragma unroll
    for (int i = 0; i < 8; ++i) o_acc[i] = o_acc[i] * alpha + pr * vv[i];
  }

  // combine the SLOTS row-streams of this wave, then the 4 waves (LDS)
  __shared__ float red[4 * SLOTS][D + 2];
  const int my = wid * SLOTS + slot;
#pragma unroll
  for (int i = 0; i < 8; ++i) red[my][g * 8 + i] = o_acc[i];
  if (g == 0) {
    red[my][D] = m_run;
    red[my][D + 1] = l_run;
  }
  __syncthreads();
  if (threadIdx.x < 64) {                      // one wave combines
    const int gg = threadIdx.x % LPR;
    float m_all = NEG_INF;
    for (int s = 0; s < 4 * SLOTS; ++s) m_all = fmaxf(m_all, red[s][D]);
    float l_all = 0.f;
    float oc[8];
#pragma unroll
    for (int i = 0; i < 8; ++i) oc[i] = 0.f;
    for (int s = 0; s < 4 * SLOTS; ++s) {
      const float w = (red[s][D] == NEG_INF)
          ? 0.f : __builtin_exp2f((red[s][D] - m_all) * LOG2E);
      l_all += w * red[s][D + 1];
#pragma unroll
      for (int i = 0; i < 8; ++i) oc[i] += w * red[s][gg * 8 + i];
    }
    if (threadIdx.x < LPR) {                   // lanes 0..LPR-1 write
      float* pb = part + ((long)bh * gridDim.x + blockIdx.x) * (D + 2);
#pragma unroll
      for (int i = 0; i < 8; ++i) pb[gg * 8 + i] = oc[i];
      if (gg == 0) {
        pb[D] = m_all;
        pb[D + 1] = l_all;
      }
    }
  }
}

// combine NS chunk partials -> o (bf16 BSHD S=1) + lse
template <int D>
__global__ __launch_bounds__(64)
void attn_decode_combine_kernel(const float* __restrict__ part,
                                unsigned short* __restrict__ O,
                                float* __restrict__ lse_out,
                                int B, int Hq, int NS) {
  constexpr int LPR = D / 8;
  const int bh = blockIdx.x;
  const int lane = threadIdx.x;
  const float* pb = part + (long)bh * NS * (D + 2);
  float m_all = NEG_INF;
  for (int s = 0; s < NS; ++s) m_all = fmaxf(m_all, pb[s * (D + 2) + D]);
  float l_all = 0.f;
  float oc[8];
#pragma unroll
  for (int i = 0; i < 8; ++i) oc[i] = 0.f;
  const int g = lane % LPR;
  for (int s = 0; s < NS; ++s) {
    const float ms = pb[s * (D + 2) + D];
    const float w = (ms == NEG_INF)
        ? 0.f : __builtin_exp2f((ms - m_all) * LOG2E);
    l_all += w * pb[s * (D + 2) + D + 1];
    if (lane < LPR) {
#pragma unroll
      for (int i = 0; i < 8; ++i)
        oc[i] += w * pb[s * (D + 2) + g * 8 + i];
    }
  }
  if (lane < LPR) {
    const float rcp = l_all > 0.f ? 1.f / l_all : 0.f;
    unsigned short* orow = O + (long)bh * D;
#pragma unroll
    for (int i = 0; i < 8; ++i) orow[g * 8 + i] = f2bf(oc[i] * rcp);
    if (g == 0)
      lse_out[bh] = m_all + __logf(fmaxf(l_all, 1e-30f));
  }
}

int attn_decode_nsplit(int Skv) {
  int ns = DTX_CDIV(Skv, 512);
  return ns < 1 ? 1 : ns;
}

void launch_attn_decode(const void* q, const void* k, const void* v,
                        const int* len_dev, float* part, void* o,
                        float* lse, int B, int Hq, int Hkv, int Skv,
                        int D, float scale, int len_stride,
                        hipStream_t st) {
  const int ns = attn_decode_nsplit(Skv);
  const int chunk = DTX_CDIV(Skv, ns);
  dim3 grid(ns, B * Hq);
  if (D == 128) {
    attn_decode_kernel<128><<<grid, DTX_BLOCK, 0, st>>>(
        (const unsigned short*)q, (const unsigned short*)k,
        (const unsigned short*)v, len_dev, part, B, Hq, Hkv, Skv, chunk,
        scale, len_stride);
    attn_decode_combine_kernel<128><<<B * Hq, 64, 0, st>>>(
        part, (unsigned short*)o, lse, B, Hq, ns);
  } else if (D == 64) {
    attn_decode_kernel<64><<<grid, DTX_BLOCK, 0, st>>>(
        (const unsigned short*)q, (const unsigned short*)k,
        (const unsigned short*)v, len_dev, part, B, Hq, Hkv, Skv, chunk,
        scale, len_stride);
    attn_decode_combine_kernel<64><<<B * Hq, 64, 0, st>>>(
        part, (unsigned short*)o, lse, B, Hq, ns);
  }
}
