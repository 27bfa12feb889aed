// Flash-attention backward for gfx950 (CDNA4), bf16 I/O, fp32 math.
//
// v2 — 8-wave 32x32-MFMA structure matching attn_fwd.hip, BSHD layout.
// Recompute strategy: P = exp2(S*scale*log2e - lse*log2e) from the saved
// lse (no S x S materialization). Host pre-transposes Q, K, dO once per
// call with transpose_sd ([B,S,H,D] -> [B,H,D,S]) so every LDS staging
// load is a coalesced 16-byte row read.
//
// dkdv kernel (one block = 256 kv rows, 8 waves x 32):
//   wave-resident K fragments; V re-read from LDS; per q-tile (32 rows):
//     S  [q][kv] = mfma(A=Q-rows,  B=K-frag)      (K-frag doubles as B)
//     dP [q][kv] = mfma(A=dO-rows, B=V-frag)
//     dS = P o (dP - delta) * scale
//     dV[kv][d] += mfma(A=conv(P),  B=dO^T-rows)  (conv = cvt_pk+permlane
//     dK[kv][d] += mfma(A=conv(dS), B=Q^T-rows)    C-layout -> A-frag)
// dq kernel (one block = 256 q rows, 8 waves x 32):
//   wave-resident Q/dO fragments; per kv-tile (64 rows):
//     S^T, dP^T as in the forward (swapped), then
//     dQ[q][d] += mfma(A=conv(dS^T), B=K^T-rows)
//
// Numerics contract: ops/reference.py attn_bwd.
#include "dtx_common.h"

typedef __attribute__((ext_vector_type(16))) float f32x16;
typedef __attribute__((ext_vector_type(4))) unsigned int u32x4;

#define MFMA32(a, b, c) __builtin_amdgcn_mfma_f32_32x32x16_bf16(a, b, c, 0, 0, 0)
#define NEG_INF (-3.0e38f)
#define LOG2E 1.4426950408889634f

__device__ __forceinline__ unsigned bw_cvt_pk_bf16(float lo, float hi) {
  unsigned r;
  asm volatile("v_cvt_pk_bf16_f32 %0, %1, %2\n\ts_nop 1"
               : "=v"(r) : "v"(lo), "v"(hi));
  return r;
}

__device__ __forceinline__ int bw_crow(int r, int hi) {
  return (r & 3) + 8 * (r >> 2) + 4 * hi;
}

// MFMA B-fragment gathered from a ROW-major [k][n] LDS image with
// ds_read_b64_tr_b16 (gfx950 hardware 4x4 transpose read) — replaces
// the separately-staged transposed image. Verified mapping
// (tools/probe_tr.cpp): with 16-lane-group addresses
// R_i = &img[k0 + (i>>2)][n0 + 4*(i&3)], lane l = 4a+c receives
// component j = img[k0 + j][n0 + (l&15)] — exactly B[k][n=l&31] when
// n0 = 32-col base + 16*((l>>4)&1). Two reads cover the lane's 8 ks.
typedef __attribute__((ext_vector_type(4))) __bf16 bf16x4;
#define DTX_AS3 __attribute__((address_space(3)))
// img must be an address_space(3) pointer so all address math stays in
// 32-bit LDS offsets (a generic pointer here costs 64-bit address
// arithmetic per read and ~+120 VGPRs in the dkdv kernel).
template <int STRIDE>
__device__ __forceinline__ short8v tr_bfrag(
    const DTX_AS3 unsigned short* img, int k0, int n0, int lane) {
  const int row = k0 + ((lane >> 2) & 3);
  const int col = n0 + ((lane >> 4) & 1) * 16 + (lane & 3) * 4;
  const DTX_AS3 unsigned short* p = img + row * STRIDE + col;
  union { bf16x4 v[2]; short8v s; } u;
  u.v[0] = __builtin_amdgcn_ds_read_tr16_b64_v4bf16(
      (DTX_AS3 bf16x4*)p);
  u.v[1] = __builtin_amdgcn_ds_read_tr16_b64_v4bf16(
      (DTX_AS3 bf16x4*)(p + 4 * STRIDE));
  return u.s;
}

// C-layout f32x16 (rows R on regs, cols on lanes) -> two bf16 A/B
// fragments with k = R (frag0: R 0..15, frag1: R 16..31).
__device__ __forceinline__ void conv_c_to_frag(const f32x16& p,
                                               short8v& f0, short8v& f1) {
  unsigned c0 = bw_cvt_pk_bf16(p[0], p[1]);
  unsigned c1 = bw_cvt_pk_bf16(p[2], p[3]);
  unsigned c2 = bw_cvt_pk_bf16(p[4], p[5]);
  unsigned c3 = bw_cvt_pk_bf16(p[6], p[7]);
  unsigned c4 = bw_cvt_pk_bf16(p[8], p[9]);
  unsigned c5 = bw_cvt_pk_bf16(p[10], p[11]);
  unsigned c6 = bw_cvt_pk_bf16(p[12], p[13]);
  unsigned c7 = bw_cvt_pk_bf16(p[14], p[15]);
  auto r02 = __builtin_amdgcn_permlane32_swap(c0, c2, false, false);
  auto r13 = __builtin_amdgcn_permlane32_swap(c1, c3, false, false);
  auto r46 = __builtin_amdgcn_permlane32_swap(c4, c6, false, false);
  auto r57 = __builtin_amdgcn_permlane32_swap(c5, c7, false, false);
  u32x4 lo{(unsigned)r02[0], (unsigned)r13[0],
           (unsigned)r02[1], (unsigned)r13[1]};
  u32x4 hi4{(unsigned)r46[0], (unsigned)r57[0],
            (unsigned)r46[1], (unsigned)r57[1]};
  f0 = *reinterpret_cast<short8v*>(&lo);
  f1 = *reinterpret_cast<short8v*>(&hi4);
}

// ------------------------------------------------------------- delta
// delta[b,h,s] = sum_d dO[b,s,h,d] * O[b,s,h,d]   (BSHD in, [B,H,S] out)
// Vectorized: a wave covers 64*8/D rows per iteration with 16B loads
// (the scalar version ran at ~40% of the HBM roofline).
template <int D>
__global__ __launch_bounds__(DTX_BLOCK)
void attn_delta2_kernel(const unsigned short* __restrict__ dO,
                        const unsigned short* __restrict__ O,
                        float* __restrict__ delta,
                        long nrows, int H, int S) {
  constexpr int LPR = D / 8;                  // lanes per row
  constexpr int RPW = 64 / LPR;               // rows per wave
  const int lane = threadIdx.x & 63;
  const int wid = threadIdx.x >> 6;
  const int sub = lane / LPR;                 // row slot in wave
  const int g = lane % LPR;                   // 16B group in row
  for (long r0 = ((long)blockIdx.x * 4 + wid) * RPW; r0 < nrows;
       r0 += (long)gridDim.x * 4 * RPW) {
    const long row = r0 + sub;
    float acc = 0.f;
    if (row < nrows) {
      const long base = row * D + g * 8;
      float a[8], b8[8];
      load_bf16x8(dO + base, a);
      load_bf16x8(O + base, b8);
#pragma unroll
      for (int i = 0; i < 8; ++i) acc += a[i] * b8[i];
    }
    // reduce within each LPR-lane group
#pragma unroll
    for (int off = LPR / 2; off > 0; off >>= 1)
      acc += __shfl_xor(acc, off, 64);
    if (g == 0 && row < nrows) {
      const int h = (int)(row % H);
      const long bs = row / H;
      const int s = (int)(bs % S);
      const long b = bs / S;
      delta[((long)b * H + h) * S + s] = acc;
    }
  }
}

// ------------------------------------------------------------- dk/dv
// 8 waves (512 threads), TWO waves per SIMD: waves 0-3 accumulate dV
// and waves 4-7 accumulate dK for the same 4x32 kv rows. Splitting the
// roles keeps each wave's accumulator set at 64 VGPRs (one acc array,
// ~225 total) instead of the fused kernel's 128 (344 total -> one wave
// per SIMD, 50% of wave time parked with nothing co-resident to hide
// it). The price is S recomputed by both roles (+25% MFMA issue on a
// pipe that idles ~85% of the time).
template <int D>
struct DkdvLds {
  // double-buffered staged q rows (two 32-row tiles per slot): one
  // barrier per stage instead of the [sync; write; sync] full stop
  unsigned short Qr[2][64][D + 8];
  unsigned short dOr[2][64][D + 8];
  // no transposed images: dV/dK B-fragments come from these row-major
  // tiles via ds_read_b64_tr_b16 (tr_bfrag) — halves the LDS footprint
  // (4 blocks/CU co-resident) and the staged global traffic
};

template <int D>
__global__ __launch_bounds__(512, 1)
void attn_bwd_dkdv2_kernel(const unsigned short* __restrict__ Q,
                           const unsigned short* __restrict__ Kp,
                           const unsigned short* __restrict__ Vp,
                           const unsigned short* __restrict__ dO,
                           const float* __restrict__ lse_in,
                           const float* __restrict__ delta_in,
                           unsigned short* __restrict__ dK,
                           unsigned short* __restrict__ dV,
                           int B, int Hq, int Hkv, int S, int Skv,
                           float scale, int causal) {
  constexpr int DC16 = D / 16;
  constexpr int ND32 = D / 32;
  __shared__ DkdvLds<D> lds;
  // separate object: float arrays INSIDE DkdvLds made hipcc scalarize
  // every b128 fragment read of the struct (see profiles/README.md)
  __shared__ float lsed[2][64 + 64];

  const int lane = threadIdx.x & 63;
  const int wid = threadIdx.x >> 6;
  const int l31 = lane & 31;
  const int hi = lane >> 5;
  const int rep = Hq / Hkv;
  const bool role_dv = wid < 4;            // waves 0-3 dV, 4-7 dK

  const int bh = blockIdx.y;
  const int b = bh / Hkv, hkv = bh % Hkv;
  const int kv0b = blockIdx.x * 128;
  const int kw = kv0b + (wid & 3) * 32;    // wave's first kv row

  const int qrowstr = Hq * D, krowstr = Hkv * D;
  const long kbase = (long)b * Skv * krowstr + (long)hkv * D;
  const int diag = Skv - S;
  const float kscale = scale * LOG2E;

  // wave-resident K fragments (+ V for the dK waves, which need
  // dP = dO V^T): lane holds X[kw+l31][kc*16+hi*8+j]
  short8v kfrag[DC16], vfrag[DC16];
  {
    const int krow = kw + l31;
#pragma unroll
    for (int kc = 0; kc < DC16; ++kc) {
      kfrag[kc] = short8v{0, 0, 0, 0, 0, 0, 0, 0};
      vfrag[kc] = short8v{0, 0, 0, 0, 0, 0, 0, 0};
      if (krow < Skv) {
        const long off = kbase + (long)krow * krowstr + kc * 16 + hi * 8;
        kfrag[kc] = *reinterpret_cast<const short8v*>(Kp + off);
        if (!role_dv)
          vfrag[kc] = *reinterpret_cast<const short8v*>(Vp + off);
      }
    }
  }

  f32x16 acc[ND32];                        // dV or dK by role
#pragma unroll
  for (int c = 0; c < ND32; ++c)
#pragma unroll
    for (int r = 0; r < 16; ++r) acc[c][r] = 0.f;

  for (int gi = 0; gi < rep; ++gi) {
    const int hq = hkv * rep + gi;
    const long qbase = (long)b * S * qrowstr + (long)hq * D;
    const long lbase = ((long)b * Hq + hq) * S;

    // q rows that can see this block's kv rows: q + diag >= kv0b
    const int qs_lo = causal ? max(0, (kv0b - diag) / 64) : 0;
    const int qs_hi = (S + 63) / 64;
    // T14 async-stage split: stage qs+1's global loads are issued while
    // stage qs computes; the LDS write happens after the barrier.
    constexpr int RIT = (64 * (D / 8)) / 512;   // row-staging iters
    short8v stg[RIT * 2];
    auto issue_stage = [&](int qs) {
#pragma unroll
      for (int it = 0; it < RIT; ++it) {
        const int idx = threadIdx.x + it * 512;
        const int row = idx / (D / 8), g = idx % (D / 8);
        const int q0s = qs * 64;
        short8v q8 = {0, 0, 0, 0, 0, 0, 0, 0};
        short8v d8 = {0, 0, 0, 0, 0, 0, 0, 0};
        if (q0s + row < S) {
          const long off = qbase + (long)(q0s + row) * qrowstr + g * 8;
          q8 = *reinterpret_cast<const short8v*>(Q + off);
          d8 = *reinterpret_cast<const short8v*>(dO + off);
        }
        stg[it * 2] = q8;
        stg[it * 2 + 1] = d8;
      }
    };
    auto write_stage = [&](int slot) {
#pragma unroll
      for (int it = 0; it < RIT; ++it) {
        const int idx = threadIdx.x + it * 512;
        const int row = idx / (D / 8), g = idx % (D / 8);
        *reinterpret_cast<short8v*>(&lds.Qr[slot][row][g * 8]) =
            stg[it * 2];
        *reinterpret_cast<short8v*>(&lds.dOr[slot][row][g * 8]) =
            stg[it * 2 + 1];
      }
    };
    auto write_lsed = [&](int slot, int qs) {
      if (threadIdx.x < 64) {
        const int qg = qs * 64 + threadIdx.x;
        lsed[slot][threadIdx.x] = qg < S ? lse_in[lbase + qg] : 0.f;
        lsed[slot][64 + threadIdx.x] =
            qg < S ? delta_in[lbase + qg] : 0.f;
      }
    };
    int cur = 0;
    if (qs_lo < qs_hi) {
      // prologue: first stage lands in slot 0 before the loop
      issue_stage(qs_lo);
      write_stage(0);
      write_lsed(0, qs_lo);
      __syncthreads();
    }
    for (int qs = qs_lo; qs < qs_hi; ++qs) {
      const int q0s = qs * 64;
      if (qs + 1 < qs_hi) issue_stage(qs + 1);

#pragma unroll
      for (int qh = 0; qh < 2; ++qh) {
      const int q0 = q0s + qh * 32;
      const int qoff = qh * 32;                  // LDS row offset
      // wave skip: its kv rows all above this q-tile's diagonal
      const bool live_tile =
          !(q0 >= S || (causal && (q0 + 31 + diag < kw)));
      if (live_tile) {

      // ---- S[q][kv] (both roles) and dP[q][kv] (dK waves only)
      // C-layout: q rows on regs, kv on lanes = the wave's kw + l31
      f32x16 sv, dpv;
#pragma unroll
      for (int r = 0; r < 16; ++r) { sv[r] = 0.f; dpv[r] = 0.f; }
      if (role_dv) {
#pragma unroll
        for (int kc = 0; kc < DC16; ++kc) {
          short8v qa = *reinterpret_cast<const short8v*>(
              &lds.Qr[cur][qoff + l31][kc * 16 + hi * 8]);
          sv = MFMA32(qa, kfrag[kc], sv);
        }
      } else {
#pragma unroll
        for (int kc = 0; kc < DC16; ++kc) {
          short8v qa = *reinterpret_cast<const short8v*>(
              &lds.Qr[cur][qoff + l31][kc * 16 + hi * 8]);
          short8v da = *reinterpret_cast<const short8v*>(
              &lds.dOr[cur][qoff + l31][kc * 16 + hi * 8]);
          sv = MFMA32(qa, kfrag[kc], sv);
          dpv = MFMA32(da, vfrag[kc], dpv);
        }
      }

      // ---- P = exp2(s*kscale - lse2); dS = P o (dP - delta) * scale
      const int kvg = kw + l31;
      const bool interior = (kv0b + 128 <= Skv) && (q0 + 32 <= S) &&
                            (!causal || (kw + 31 <= q0 + diag));
      // P overwrites sv; dS overwrites dpv (keeps the VGPR count spill-free)
#pragma unroll
      for (int r = 0; r < 16; ++r) {
        const int qg = q0 + bw_crow(r, hi);
        const float lse2 = lsed[cur][qoff + bw_crow(r, hi)] * LOG2E;
        const float dlt = lsed[cur][64 + qoff + bw_crow(r, hi)];
        float p;
        if (interior) {
          p = __builtin_exp2f(sv[r] * kscale - lse2);
        } else {
          const bool dead = (kvg >= Skv) | (qg >= S) |
                            (causal && (kvg > qg + diag));
          p = dead ? 0.f : __builtin_exp2f(sv[r] * kscale - lse2);
        }
        sv[r] = p;
        dpv[r] = p * (dpv[r] - dlt) * scale;
      }

      // ---- fragments (k = q) and accumulate this role's output
      short8v f0, f1;
      if (role_dv) conv_c_to_frag(sv, f0, f1);
      else conv_c_to_frag(dpv, f0, f1);
      const DTX_AS3 unsigned short* img3 = role_dv
          ? (const DTX_AS3 unsigned short*)&lds.dOr[cur][0][0]
          : (const DTX_AS3 unsigned short*)&lds.Qr[cur][0][0];
#pragma unroll
      for (int c = 0; c < ND32; ++c) {
        // B[k=q][n=d] straight from the row-major tile (tr-read)
        short8v b0 = tr_bfrag<D + 8>(img3, qoff + hi * 8, c * 32, lane);
        short8v b1 = tr_bfrag<D + 8>(img3, qoff + 16 + hi * 8, c * 32,
                                     lane);
        acc[c] = MFMA32(f0, b0, acc[c]);
        acc[c] = MFMA32(f1, b1, acc[c]);
        __builtin_amdgcn_sched_barrier(0);    // short (no cross-c hoist)
      }
      }  // live_tile
      }  // qh
      if (qs + 1 < qs_hi) {
        write_stage(cur ^ 1);
        write_lsed(cur ^ 1, qs + 1);
      }
      // one barrier per stage: publishes slot cur^1 AND guards the
      // next iteration's write into the slot every wave just read
      __syncthreads();
      cur ^= 1;
    }
  }

  // ---- epilogue: this role's C-layout [kv regs][d lanes] -> BSHD
  unsigned short* out = role_dv ? dV : dK;
#pragma unroll
  for (int r = 0; r < 16; ++r) {
    const int kvg = kw + bw_crow(r, hi);
    if (kvg < Skv) {
      unsigned short* orow = out + kbase + (long)kvg * krowstr;
#pragma unroll
      for (int c = 0; c < ND32; ++c)
        orow[c * 32 + l31] = f2bf(acc[c][r]);
    }
  }
}

// ---------------------------------------------------------------- dq
template <int D>
struct DqLds {
  // double-buffered (one barrier per 128-row stage; see DkdvLds)
  unsigned short K[2][128][D + 8];
  unsigned short V[2][128][D + 8];
  // dQ's K B-fragments come from the row-major K tile via
  // ds_read_b64_tr_b16 (tr_bfrag) — no transposed KT image
};

template <int D>
__global__ __launch_bounds__(512, 1)
void attn_bwd_dq2_kernel(const unsigned short* __restrict__ Q,
                         const unsigned short* __restrict__ Kp,
                         const unsigned short* __restrict__ Vp,
                         const unsigned short* __restrict__ dO,
                         const float* __restrict__ lse_in,
                         const float* __restrict__ delta_in,
                         unsigned short* __restrict__ dQ,
                         int B, int Hq, int Hkv, int S, int Skv,
                         float scale, int causal) {
  constexpr int KVB = 64;
  constexpr int DC16 = D / 16;
  constexpr int ND32 = D / 32;
  __shared__ DqLds<D> lds;

  const int lane = threadIdx.x & 63;
  const int wid = threadIdx.x >> 6;
  const int l31 = lane & 31;
  const int hi = lane >> 5;

  const int bh = blockIdx.y;
  const int b = bh / Hq, hq = bh % Hq;
  const int hkv = hq / (Hq / Hkv);
  const int q0 = blockIdx.x * 256;
  const int qw = q0 + wid * 32;

  const int qrowstr = Hq * D, krowstr = Hkv * D;
  const long qbase = (long)b * S * qrowstr + (long)hq * D;
  const long kbase = (long)b * Skv * krowstr + (long)hkv * D;
  const long lbase = ((long)b * Hq + hq) * S;
  const int diag = Skv - S;
  const float kscale = scale * LOG2E;

  // wave-resident Q and dO fragments (B-layout: lane q = qw + l31)
  short8v qfrag[DC16], dofrag[DC16];
  float lse2 = 0.f, dlt = 0.f;
  {
    const int qrow = qw + l31;
    if (qrow < S) {
#pragma unroll
      for (int kc = 0; kc < DC16; ++kc) {
        const long off = qbase + (long)qrow * qrowstr + kc * 16 + hi * 8;
        qfrag[kc] = *reinterpret_cast<const short8v*>(Q + off);
        dofrag[kc] = *reinterpret_cast<const short8v*>(dO + off);
      }
      lse2 = lse_in[lbase + qrow] * LOG2E;
      dlt = delta_in[lbase + qrow];
    } else {
#pragma unroll
      for (int kc = 0; kc < DC16; ++kc) {
        qfrag[kc] = short8v{0, 0, 0, 0, 0, 0, 0, 0};
        dofrag[kc] = short8v{0, 0, 0, 0, 0, 0, 0, 0};
      }
    }
  }

  f32x16 dq_acc[ND32];
#pragma unroll
  for (int c = 0; c < ND32; ++c)
#pragma unroll
    for (int r = 0; r < 16; ++r) dq_acc[c][r] = 0.f;

  const int q_hi_blk = min(q0 + 255, S - 1);
  const int kv_hi = causal ? min(Skv - 1, q_hi_blk + diag) : (Skv - 1);
  const int nstages = kv_hi / 128 + 1;     // 128 kv rows per stage

  // T14 async-stage split (as in fwd/dkdv)
  constexpr int KIT = (128 * (D / 8)) / 512;
  short8v stg[KIT * 2];
  auto issue_stage = [&](int st2) {
    const int kvs = st2 * 128;
#pragma unroll
    for (int it = 0; it < KIT; ++it) {
      const int idx = threadIdx.x + it * 512;
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
      const int idx = threadIdx.x + it * 512;
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

    for (int kh = 0; kh < 2; ++kh) {
    const int kv0 = kvs + kh * KVB;
    const int koff = kh * KVB;               // LDS row/col offset
    const bool wave_dead = (kv0 > kv_hi) ||
                           (causal && (kv0 > qw + 31 + diag));
    if (!wave_dead) {
      const int qg = qw + l31;
      const bool interior = (kv0 + KVB <= Skv) &&
                            (!causal || (kv0 + KVB - 1 <= qw + diag));
#pragma unroll
      for (int ss = 0; ss < 2; ++ss) {
        // S^T and dP^T for kv subtile ss (C-layout: kv rows on regs,
        // q on lanes)
        f32x16 st, dpt;
#pragma unroll
        for (int r = 0; r < 16; ++r) { st[r] = 0.f; dpt[r] = 0.f; }
#pragma unroll
        for (int kc = 0; kc < DC16; ++kc) {
          short8v ka = *reinterpret_cast<const short8v*>(
              &lds.K[cur][koff + ss * 32 + l31][kc * 16 + hi * 8]);
          short8v va = *reinterpret_cast<const short8v*>(
              &lds.V[cur][koff + ss * 32 + l31][kc * 16 + hi * 8]);
          st = MFMA32(ka, qfrag[kc], st);
          dpt = MFMA32(va, dofrag[kc], dpt);
        }
        f32x16 dst;
#pragma unroll
        for (int r = 0; r < 16; ++r) {
          float p;
          if (interior) {
            p = __builtin_exp2f(st[r] * kscale - lse2);
          } else {
            const int kvg = kv0 + ss * 32 + bw_crow(r, hi);
            const bool dead = (kvg >= Skv) | (qg >= S) |
                              (causal && (kvg > qg + diag));
            p = dead ? 0.f : __builtin_exp2f(st[r] * kscale - lse2);
          }
          dst[r] = p * (dpt[r] - dlt) * scale;
        }
        short8v f0, f1;
        conv_c_to_frag(dst, f0, f1);
        const DTX_AS3 unsigned short* k3 =
            (const DTX_AS3 unsigned short*)&lds.K[cur][0][0];
        // two passes over c so consecutive MFMAs hit different
        // accumulators (dq_acc[0..3]) instead of pairing on one
#pragma unroll
        for (int c = 0; c < ND32; ++c) {
          short8v kt0 = tr_bfrag<D + 8>(k3, koff + ss * 32 + hi * 8,
                                        c * 32, lane);
          dq_acc[c] = MFMA32(f0, kt0, dq_acc[c]);
        }
        __builtin_amdgcn_sched_barrier(0);
#pragma unroll
        for (int c = 0; c < ND32; ++c) {
          short8v kt1 = tr_bfrag<D + 8>(k3, koff + ss * 32 + 16 + hi * 8,
                                        c * 32, lane);
          dq_acc[c] = MFMA32(f1, kt1, dq_acc[c]);
        }
        __builtin_amdgcn_sched_barrier(0);  // no cross-stage frag hoist
      }
    }
    }  // kh
    if (st2 + 1 < nstages) write_stage(cur ^ 1);
    __syncthreads();
    cur ^= 1;
  }

  // ---- epilogue: dQ C-layout [q regs][d lanes] -> BSHD stores
#pragma unroll
  for (int r = 0; r < 16; ++r) {
    const int qg = qw + bw_crow(r, hi);
    if (qg < S) {
      unsigned short* qrow = dQ + qbase + (long)qg * qrowstr;
#pragma unroll
      for (int c = 0; c < ND32; ++c)
        qrow[c * 32 + l31] = f2bf(dq_acc[c][r]);
    }
  }
}

// ------------------------------------------------------------- launchers
void launch_attn_delta(const void* dO, const void* O, float* delta,
                       long nrows, int H, int S, int D, hipStream_t st) {
  const int rpw = 64 / (D / 8);
  long gw = DTX_CDIV(nrows, 4 * rpw);
  int grid = (int)(gw < 2048 ? (gw < 1 ? 1 : gw) : 2048);
  if (D == 128) {
    attn_delta2_kernel<128><<<grid, DTX_BLOCK, 0, st>>>(
        (const unsigned short*)dO, (const unsigned short*)O, delta,
        nrows, H, S);
  } else if (D == 64) {
    attn_delta2_kernel<64><<<grid, DTX_BLOCK, 0, st>>>(
        (const unsigned short*)dO, (const unsigned short*)O, delta,
        nrows, H, S);
  }
}

void launch_attn_bwd_dkdv(const void* q, const void* k, const void* v,
                          const void* dO, const float* lse,
                          const float* delta, void* dk, void* dv, int B,
                          int Hq, int Hkv, int S, int Skv, int D,
                          float scale, int causal, hipStream_t st) {
  dim3 grid(DTX_CDIV(Skv, 128), B * Hkv);
  if (D == 128) {
    attn_bwd_dkdv2_kernel<128><<<grid, 512, 0, st>>>(
        (const unsigned short*)q, (const unsigned short*)k,
        (const unsigned short*)v, (const unsigned short*)dO,
        lse, delta, (unsigned short*)dk, (unsigned short*)dv,
        B, Hq, Hkv, S, Skv, scale, causal);
  } else if (D == 64) {
    attn_bwd_dkdv2_kernel<64><<<grid, 512, 0, st>>>(
        (const unsigned short*)q, (const unsigned short*)k,
        (const unsigned short*)v, (const unsigned short*)dO,
        lse, delta, (unsigned short*)dk, (unsigned short*)dv,
        B, Hq, Hkv, S, Skv, scale, causal);
  }
}

void launch_attn_bwd_dq(const void* q, const void* k, const void* v,
                        const void* dO, const float* lse,
                        const float* delta, void* dq, int B, int Hq,
                        int Hkv, int S, int Skv, int D, float scale,
                        int causal, hipStream_t st) {
  dim3 grid(DTX_CDIV(S, 256), B * Hq);
  if (D == 128) {
    attn_bwd_dq2_kernel<128><<<grid, 512, 0, st>>>(
        (const unsigned short*)q, (const unsigned short*)k,
        (const unsigned short*)v,
        (const unsigned short*)dO, lse, delta, (unsigned short*)dq,
        B, Hq, Hkv, S, Skv, scale, causal);
  } else if (D == 64) {
    attn_bwd_dq2_kernel<64><<<grid, 512, 0, st>>>(
        (const unsigned short*)q, (const unsigned short*)k,
        (const unsigned short*)v,
        (const unsigned short*)dO, lse, delta, (unsigned short*)dq,
        B, Hq, Hkv, S, Skv, scale, causal);
  }
}
