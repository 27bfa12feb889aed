// Hand-written CDNA4 (gfx950) bf16 GEMM for the frozen-base projections —
// the kernel SURVEY.md §2.4 row 1 names first ("base GEMMs ... HIP MFMA
// tiled GEMM"; reference exercise site: HF LlamaForCausalLM fwd/bwd,
// /root/reference/cmd/tuning/train.py:236-242).
//
//   C[M,N] = A[M,K] @ B[N,K]^T (+ optional bf16 source add)
//
// Everything is row-major bf16 with the CONTRACTION contiguous in both
// operands ("NT"). The dgrad pass dX = dY @ W reuses this same kernel
// with a cached W^T copy: base weights are frozen under LoRA, and 288 GB
// of HBM3E per MI355X makes a persistent transposed copy free — so both
// hot GEMMs of the training step run the one fast layout instead of a
// strided-B variant.
//
// Structure (guide: cdna_hip_programming.md §5 "256^2 8-phase template"):
//   - 256x256 output tile, K-step 64 as two 32-deep k-halves.
//   - 512 threads = 8 waves in a 2(M) x 4(N) grid; per-wave output
//     128x64 = 8x4 fragments of mfma_f32_16x16x32_bf16 (f32 acc).
//   - LDS ring: 4 A half-slots + 4 B half-slots of 16 KiB ([256 rows] x
//     [32 bf16]), 128 KiB total — compute tile kt while tile kt+1 lands.
//   - Staging by buffer_load_dwordx4 ... lds (LDS-DMA, inline asm: hipcc
//     never auto-emits it and must not count it), one half-tile per
//     phase, counted s_waitcnt vmcnt(4) at odd-phase ends ONLY (loads
//     span barriers — T3+T4), raw s_barrier (never __syncthreads: with
//     LDS-DMA in flight its fence drains vmcnt to 0).
//   - Per-phase s_setprio(1) around the 16-MFMA cluster (T5).
//   - XCD-aware bijective block remap, nb-major so each XCD's contiguous
//     chunk re-reads the same 2 MB B-panel through its private L2 (T1).
//
// LDS swizzle (bank-conflict-free, derived for this [256][32] half
// layout): a half-row is 64 B = 4 16-B sub-slots; data sub-slot s of row
// r is stored at position s ^ ((-(r>>2))&3). Each ds_read_b128 lane
// group then touches 16 distinct (row%4, slot) pairs = all 64 banks
// exactly once (verified for all four hardware lane groups). The
// LDS-DMA destination is lane-linear, so the swizzle is applied to the
// per-lane GLOBAL source address (guide §5.4 rule 21); it permutes
// 16-B chunks only WITHIN one row's 64-B k-half, so global requests
// still cover whole 64-B lines (no FETCH cost).
#include "dtx_common.h"

typedef __attribute__((ext_vector_type(4))) unsigned uint4v;
typedef __attribute__((ext_vector_type(2))) unsigned uint2v;

#define MFMA16(a, b, c) __builtin_amdgcn_mfma_f32_16x16x32_bf16(a, b, c, 0, 0, 0)
#define MFMA32(a, b, c) __builtin_amdgcn_mfma_f32_32x32x16_bf16(a, b, c, 0, 0, 0)
typedef __attribute__((ext_vector_type(16))) float f32x16;

// Buffer resource descriptor (T8/T20): built from readfirstlane'd scalars
// so hipcc proves uniformity (no waterfall loops), num_records = bytes so
// out-of-bounds rows of the M-tail clamp to zero loads / dropped stores.
__device__ __forceinline__ uint4v dtx_srd(const void* base,
                                          unsigned long long bytes) {
  unsigned long long b = (unsigned long long)base;
  uint4v r;
  r.x = __builtin_amdgcn_readfirstlane((unsigned)b);
  r.y = __builtin_amdgcn_readfirstlane((unsigned)(b >> 32));
  r.z = __builtin_amdgcn_readfirstlane((unsigned)bytes);
  r.w = 0x00020000u;  // dfmt/nfmt raw buffer config
  return r;
}

// LDS-DMA: stage 64 lanes x 16 B at global (srd + voff) into LDS at
// wave-uniform byte base `lds_dst` + lane*16. M0 written in the same
// statement (s_nop 0 = the required wait state after the M0 write).
__device__ __forceinline__ void glds16(uint4v srd, unsigned voff,
                                       unsigned lds_dst) {
  asm volatile(
      "s_mov_b32 m0, %1\n\t"
      "s_nop 0\n\t"
      "buffer_load_dwordx4 %0, %2, 0 offen lds"
      :: "v"(voff), "s"(lds_dst), "s"(srd) : "memory");
}

__device__ __forceinline__ void dtx_vmcnt4() {
  asm volatile("s_waitcnt vmcnt(4)" ::: "memory");
}
__device__ __forceinline__ void dtx_vmcnt0() {
  asm volatile("s_waitcnt vmcnt(0)" ::: "memory");
}
__device__ __forceinline__ void dtx_bar() {
  __builtin_amdgcn_s_barrier();
  asm volatile("" ::: "memory");
}

#define GEMM_BM 256
#define GEMM_BN 256
#define GEMM_BK 64
#define HALF_BYTES (256 * 64)          // one [256][32] bf16 half-slot

// f((r>>2)&3) of the swizzle: 0->0, 1->3, 2->2, 3->1  == (-x)&3
__device__ __forceinline__ unsigned swz_f(unsigned x) { return (0u - x) & 3u; }

// GRID: 0 = nb-major linear chunks per XCD; 1 = 8mb x 4nb clusters
//       inside each XCD chunk (A-panel reuse in L2/L3 as well as B;
//       needs mb_n % 8 == 0 and nb_n % 4 == 0); 2 = 8-row mb-groups
//       sweeping all nb (general shapes, L3-bounded window).
// PIPE: 0 = 4-slot ring, 2 barriers/phase, vmcnt(4) at p1/p3 (lockstep
//       phases); 1 = 4-slot ring, barriers ONLY at the two publication
//       points (p1/p3 end) — waves de-lockstep so one wave's loads
//       overlap its SIMD partner's MFMAs; 2 = 5-slot ring (all 160 KiB
//       LDS), stage TWO tiles ahead, ONE vmcnt(8) + two barriers per
//       tile (max latency slack); 3 = ring5 + sparse barriers +
//       vmcnt(4) at both odd phases (publication one phase earlier) +
//       fragment PRELOAD: phase p+1's ds_reads issue before phase p's
//       MFMA cluster (incl. across the tile seam), so LDS latency and
//       read issue hide inside the MFMA stream; 5 = B on the 4-slot
//       ring, A on the 5-slot ring with the A stages shifted one phase
//       pair earlier — every slot reuse then crosses the p3 barrier, so
//       ONE barrier + ONE vmcnt(2) per tile; 4 = PIPE3 +
//       sched_group_barrier [MFMA,MFMA,ds_read] interleave (the reads
//       are EMITTED inside the MFMA cluster so the matrix pipe is never
//       starved during the read window) + static young-half s_setprio
//       instead of per-phase flips (T5 static form).
// MF:   0 = mfma_f32_16x16x32_bf16; 1 = mfma_f32_32x32x16_bf16 (the
//       higher-ceiling intrinsic: 2382 vs 2075 TF ubench — and half the
//       MFMA + fragment-read instructions per phase).
template <bool HAS_SRC, int GRID = 0, int PIPE = 0, int MF = 0>
__global__ __launch_bounds__(512, 2)
void gemm_nt_kernel(const unsigned short* __restrict__ A,
                    const unsigned short* __restrict__ B,
                    const unsigned short* __restrict__ SRC,
                    unsigned short* __restrict__ C,
                    long M, int N, int K, int mb_n) {
  constexpr int RING = (PIPE >= 2) ? 5 : 4;   // PIPE5: A-ring only
  constexpr bool PRELOAD = (PIPE == 3) || (PIPE == 4);
  constexpr unsigned B_RING = RING * HALF_BYTES;
  __shared__ __attribute__((aligned(16))) unsigned char
      lds[2 * RING * HALF_BYTES];

  const int tid = threadIdx.x;
  const int lane = tid & 63;
  const int wid_div = tid >> 6;                       // divergent to hipcc
  const int wid = __builtin_amdgcn_readfirstlane(wid_div);
  const int wm = wid >> 2, wn = wid & 3;

  // ---- XCD-aware bijective remap; nb-major chunks (B-panel L2 reuse)
  const int nbn = N >> 8;
  const int nwg = gridDim.x;
  const int q = nwg >> 3, r8 = nwg & 7;
  const int xcd = blockIdx.x & 7, pos = blockIdx.x >> 3;
  const int wgid = (xcd < r8 ? xcd * (q + 1) : r8 * (q + 1) + (xcd - r8) * q)
                   + pos;
  int mb, nb;
  if (GRID == 1) {
    // 32-block clusters of 8(mb) x 4(nb); clusters enumerated nb-major
    const int mbc = mb_n >> 3;
    const int cl = wgid >> 5, ci = wgid & 31;
    mb = (cl % mbc) * 8 + (ci & 7);
    nb = (cl / mbc) * 4 + (ci >> 3);
  } else if (GRID == 2) {
    // 8-row mb-groups sweeping ALL nb columns, dispatch order (no XCD
    // remap): the ~256 concurrently-resident blocks then span <= 2
    // groups, so the A-panel (16 rows x K) and B re-reads stay
    // L3-resident chip-wide instead of streaming A from HBM once per
    // nb column (PMC: FETCH dropped ~4x on the mb24 shapes).
    const int g = blockIdx.x / (nbn * 8);
    const int rem = blockIdx.x % (nbn * 8);
    const int rows = min(8, mb_n - g * 8);
    nb = rem / rows;
    mb = g * 8 + rem % rows;
  } else {
    mb = wgid % mb_n;
    nb = wgid / mb_n;
  }
  const long m0 = (long)mb * GEMM_BM;
  const int n0 = nb * GEMM_BN;

  // SRDs are rebased per block (A at row m0, B at row n0, C/SRC at row
  // m0) so every voffset fits 32 bits even for multi-GB logits tensors;
  // num_records clamps the M-tail (OOB loads read 0, OOB stores drop).
  const long ldab = (long)K * 2;                       // A/B row bytes
  const uint4v srdA = dtx_srd(A + m0 * K, (unsigned long long)(M - m0) * ldab);
  const uint4v srdB = dtx_srd(B + (long)n0 * K,
                              (unsigned long long)(N - n0) * ldab);

  // ---- per-lane staging constants (swizzle on the SOURCE address)
  // stage: lane l covers (row = base + l>>2, sub-slot pos q = l&3);
  // the data sub-slot is s = q ^ f(row bits 2..3). row = l>>2 and every
  // stage base row is a multiple of 16, so row bits 2..3 == (l>>4)&3.
  const unsigned st_row = lane >> 2;                   // 0..15
  const unsigned st_s = (lane & 3) ^ swz_f((lane >> 4) & 3);
  // fragment read: lane l reads (row = frag_base + (l&15),
  // slot s = l>>4) at position s ^ f(((l&15)>>2)&3).
  const unsigned fr_off = (lane & 15) * 64
      + ((unsigned)(lane >> 4) ^ swz_f(((lane & 15) >> 2) & 3)) * 16;

  // A stage: wave wid covers rows wid*32 + piece*16 + st_row of the
  // 256-row tile. Global voffset (bytes):
  //   (m0 + rows)*ldab + kbyte + st_s*16
  const long a_row0 = wid * 32 + st_row;               // rows local to SRD
  const long b_row0 = wid * 32 + st_row;
  // lds dst base for this wave's slice of a half-slot
  const unsigned st_lds = wid * 2048;

  const int KT = K / GEMM_BK;

  float4v acc[8][4];          // MF0 accumulators (8x4 16x16 frags)
  f32x16 acc32[4][2];         // MF1 accumulators (4x2 32x32 frags)
  if (MF == 0) {
#pragma unroll
    for (int i = 0; i < 8; ++i)
#pragma unroll
      for (int j = 0; j < 4; ++j) acc[i][j] = float4v{0.f, 0.f, 0.f, 0.f};
  } else {
#pragma unroll
    for (int i = 0; i < 4; ++i)
#pragma unroll
      for (int j = 0; j < 2; ++j)
#pragma unroll
        for (int e = 0; e < 16; ++e) acc32[i][j][e] = 0.f;
  }
  // MF1 fragment addressing: lane covers row (l&31) of its 32-row frag
  // at 16-B sub-slot (ks*2 + (l>>5)) ^ f(((l&31)>>2)&3) of the 64-B row.
  const unsigned f31 = lane & 31;
  const unsigned mf1_posx = swz_f((f31 >> 2) & 3);
  const unsigned mf1_row = f31 * 64;
  const unsigned mf1_hi = lane >> 5;

  // ---- staging helpers: one half = 2 glds per wave (its 2 KiB slice).
  // `slot` is the ring slot byte offset, `kt`/`kh` pick the source k.
#define STAGE_A(slot, kt, kh)                                               \
  {                                                                         \
    const long kb = (long)(kt) * 128 + (kh) * 64 + st_s * 16;               \
    glds16(srdA, (unsigned)(a_row0 * ldab + kb), (slot) + st_lds);          \
    glds16(srdA, (unsigned)((a_row0 + 16) * ldab + kb),                     \
           (slot) + st_lds + 1024);                                         \
  }
#define STAGE_B(slot, kt, kh)                                               \
  {                                                                         \
    const long kb = (long)(kt) * 128 + (kh) * 64 + st_s * 16;               \
    glds16(srdB, (unsigned)(b_row0 * ldab + kb), B_RING + (slot) + st_lds); \
    glds16(srdB, (unsigned)((b_row0 + 16) * ldab + kb),                     \
           B_RING + (slot) + st_lds + 1024);                                \
  }
#define SLOT4(h) ((unsigned)((h) & 3) * HALF_BYTES)
#define SLOT5(h) ((unsigned)((h) % 5) * HALF_BYTES)

  // ---- prologue
  if (PIPE == 5) {
    // tile0's four halves + tile1's A-kh0 (the one half the in-loop
    // schedule stages at kt-1.p2); vmcnt(2) leaves exactly that A-kh0
    // in flight = the steady-state invariant at tile entry
    STAGE_B(SLOT4(0), 0, 0); STAGE_B(SLOT4(1), 0, 1);
    STAGE_A(SLOT5(0), 0, 0); STAGE_A(SLOT5(1), 0, 1);
    if (KT > 1) {
      STAGE_A(SLOT5(2), 1, 0);
      asm volatile("s_waitcnt vmcnt(2)" ::: "memory");
    } else {
      dtx_vmcnt0();
    }
  } else if (PIPE >= 2) {
    // stage tiles 0 and 1 (slots h%5), wait for tile 0's 8 ops
    STAGE_B(SLOT5(0), 0, 0); STAGE_A(SLOT5(0), 0, 0);
    STAGE_B(SLOT5(1), 0, 1); STAGE_A(SLOT5(1), 0, 1);
    STAGE_B(SLOT5(2), 1, 0); STAGE_A(SLOT5(2), 1, 0);
    STAGE_B(SLOT5(3), 1, 1); STAGE_A(SLOT5(3), 1, 1);
    asm volatile("s_waitcnt vmcnt(8)" ::: "memory");
  } else {
    // stage tile 0 (kh0 pair first), wait for the kh0 pair
    STAGE_B(SLOT4(0), 0, 0);
    STAGE_A(SLOT4(0), 0, 0);
    STAGE_B(SLOT4(1), 0, 1);
    STAGE_A(SLOT4(1), 0, 1);
    dtx_vmcnt4();
  }
  dtx_bar();

  // ---- K loop: 4 phases per tile; phase p: kh = p>>1, m-half = p&1.
  // B fragments are read at even phases and reused at the odd phase.
  // PIPE 0/1 (4-ring): tile kt stages kt+1, one half per phase
  //   (B-kh0@p0, A-kh0@p1, B-kh1@p2, A-kh1@p3); vmcnt(4) at p1/p3 ends
  //   retires exactly the halves the next two phases read.
  // PIPE 2 (5-ring): tile kt stages kt+2; ONE vmcnt(8) at p3 end
  //   retires ALL four halves of tile kt+1 (staged during kt-1);
  //   barriers at p1/p3 ends gate slot reuse (WAR) and publication.
  short8v bfr[4];
  short8v afr2[2][4], bfr2[2][4];      // PIPE3/4 double-buffered fragments
  // PIPE5 slot bookkeeping: B slots (2kt+kh)&3, A slots (2kt+kh)%5
  if (PIPE == 4) {
    // T5 static form: the second-dispatched half loses VALU arbitration
    // on every segment; one wave-uniform setprio before the loop.
    if (__builtin_amdgcn_readfirstlane(threadIdx.x) >= 256)
      __builtin_amdgcn_s_setprio(1);
  }
  // ring-5 slot indices maintained incrementally ((2kt+i) % 5)
  int h5 = 0;                          // (2*kt) % 5

#define RD_A(dst, as, mh)                                                   \
  _Pragma("unroll") for (int fm = 0; fm < 4; ++fm)                          \
      dst[fm] = *reinterpret_cast<const short8v*>(                          \
          &lds[(as) + wm * 8192 + ((mh) * 4 + fm) * 1024 + fr_off]);
#define RD_B(dst, bs)                                                       \
  _Pragma("unroll") for (int fn = 0; fn < 4; ++fn)                          \
      dst[fn] = *reinterpret_cast<const short8v*>(                          \
          &lds[(bs) + wn * 4096 + fn * 1024 + fr_off]);
#define MFMA_PHASE(af, bf, mh)                                              \
  _Pragma("unroll") for (int fm = 0; fm < 4; ++fm)                          \
      _Pragma("unroll") for (int fn = 0; fn < 4; ++fn)                      \
          acc[(mh) * 4 + fm][fn] =                                          \
              MFMA16(af[fm], bf[fn], acc[(mh) * 4 + fm][fn]);

  if (PRELOAD) {
    // preload tile 0 phase 0 fragments
    RD_A(afr2[0], 0u, 0);
    RD_B(bfr2[0], B_RING + 0u);
    for (int kt = 0; kt < KT; ++kt) {
      const unsigned as0 = (unsigned)h5 * HALF_BYTES;
      const int h5b = h5 + 1 - (h5 + 1 >= 5 ? 5 : 0);
      const unsigned as1 = (unsigned)h5b * HALF_BYTES;
      const int h5s0 = h5 + 4 - (h5 + 4 >= 5 ? 5 : 0);
      const int h5s1 = h5s0 + 1 - (h5s0 + 1 >= 5 ? 5 : 0);
      const unsigned ss0 = (unsigned)h5s0 * HALF_BYTES;
      const unsigned ss1 = (unsigned)h5s1 * HALF_BYTES;
      const bool pre = kt + 2 < KT;
      const bool last = kt + 1 == KT;
      h5 = h5 + 2 - (h5 + 2 >= 5 ? 5 : 0);
      const unsigned as0n = (unsigned)h5 * HALF_BYTES;  // next tile kh0
#pragma unroll
      for (int p = 0; p < 4; ++p) {
        if (PIPE == 3 && pre) {
          if (p == 0) STAGE_B(ss0, kt + 2, 0)
          else if (p == 1) STAGE_A(ss0, kt + 2, 0)
          else if (p == 2) STAGE_B(ss1, kt + 2, 1)
          else STAGE_A(ss1, kt + 2, 1)
        }
        // preload next phase's fragments (phase 3 preloads the NEXT
        // tile's phase 0: its halves were published at this tile's
        // p1-end vmcnt, and its slot's next writer stages at kt+1.p2 —
        // after the read).
        int nreads = 0;
        if (p == 0) {
          RD_A(afr2[1], as0, 1);
          nreads = 4;
        } else if (p == 1) {
          RD_A(afr2[0], as1, 0);
          RD_B(bfr2[1], as1 + B_RING);
          nreads = 8;
        } else if (p == 2) {
          RD_A(afr2[1], as1, 1);
          nreads = 4;
        } else if (!last) {
          RD_A(afr2[0], as0n, 0);
          RD_B(bfr2[0], as0n + B_RING);
          nreads = 8;
        }
        if (PIPE == 3) __builtin_amdgcn_s_setprio(1);
        if (p == 0) { MFMA_PHASE(afr2[0], bfr2[0], 0); }
        else if (p == 1) { MFMA_PHASE(afr2[1], bfr2[0], 1); }
        else if (p == 2) { MFMA_PHASE(afr2[0], bfr2[1], 0); }
        else { MFMA_PHASE(afr2[1], bfr2[1], 1); }
        if (PIPE == 3) __builtin_amdgcn_s_setprio(0);
        if (PIPE == 4) {
          // emit the region as [2 MFMA, 1 ds_read] x nreads, remaining
          // MFMAs after (guide T19 masks: MFMA=0x8, DS_READ=0x100)
          if (nreads == 8) {
#pragma unroll
            for (int i = 0; i < 8; ++i) {
              __builtin_amdgcn_sched_group_barrier(0x8, 2, 0);
              __builtin_amdgcn_sched_group_barrier(0x100, 1, 0);
            }
          } else if (nreads == 4) {
#pragma unroll
            for (int i = 0; i < 4; ++i) {
              __builtin_amdgcn_sched_group_barrier(0x8, 3, 0);
              __builtin_amdgcn_sched_group_barrier(0x100, 1, 0);
            }
            __builtin_amdgcn_sched_group_barrier(0x8, 4, 0);
          }
        }
        if (PIPE == 4 && pre) {
          if (p == 0) STAGE_B(ss0, kt + 2, 0)
          else if (p == 1) STAGE_A(ss0, kt + 2, 0)
          else if (p == 2) STAGE_B(ss1, kt + 2, 1)
          else STAGE_A(ss1, kt + 2, 1)
        }
        if (p & 1) {
          if (pre) dtx_vmcnt4(); else dtx_vmcnt0();
          dtx_bar();
        }
      }
    }
  } else
  for (int kt = 0; kt < KT; ++kt) {

    const unsigned as0 = (PIPE == 2 || PIPE == 5)
                             ? (unsigned)h5 * HALF_BYTES
                             : SLOT4(2 * kt);
    const int h5b = h5 + 1 - (h5 + 1 >= 5 ? 5 : 0);
    const unsigned as1 = (PIPE == 2 || PIPE == 5)
                             ? (unsigned)h5b * HALF_BYTES
                             : SLOT4(2 * kt + 1);
    // PIPE5: B lives on its own 4-slot ring
    const unsigned bs0 = SLOT4(2 * kt), bs1 = SLOT4(2 * kt + 1);
    // stage destination slots (kt+1 for 4-ring, kt+2 for 5-ring)
    const int h5b2 = h5b + 2 - (h5b + 2 >= 5 ? 5 : 0); // (2kt+3) % 5
    const int h5s0 = h5 + 4 - (h5 + 4 >= 5 ? 5 : 0);   // (2kt+4) % 5
    const int h5s1 = h5s0 + 1 - (h5s0 + 1 >= 5 ? 5 : 0);
    const unsigned ss0 = (PIPE == 2) ? (unsigned)h5s0 * HALF_BYTES
                                     : SLOT4(2 * kt + 2);
    const unsigned ss1 = (PIPE == 2) ? (unsigned)h5s1 * HALF_BYTES
                                     : SLOT4(2 * kt + 3);
    const int kt_s = (PIPE == 2) ? kt + 2 : kt + 1;
    const bool pre = kt_s < KT;
    h5 = h5 + 2 - (h5 + 2 >= 5 ? 5 : 0);

#pragma unroll
    for (int p = 0; p < 4; ++p) {
      const unsigned mh = p & 1;
      const unsigned as = (p < 2) ? as0 : as1;
      const unsigned bs = (PIPE == 5 ? ((p < 2) ? bs0 : bs1)
                                     : ((p < 2) ? as0 : as1)) + B_RING;
      // ds_read register subtile for this phase
      short8v afr[4];
      short8v bfr32[2];
      if (MF == 0) {
        if (mh == 0) {
#pragma unroll
          for (int fn = 0; fn < 4; ++fn)
            bfr[fn] = *reinterpret_cast<const short8v*>(
                &lds[bs + wn * 4096 + fn * 1024 + fr_off]);
        }
#pragma unroll
        for (int fm = 0; fm < 4; ++fm)
          afr[fm] = *reinterpret_cast<const short8v*>(
              &lds[as + wm * 8192 + (mh * 4 + fm) * 1024 + fr_off]);
      } else {
        // MF1: phase p = kstep; A 4 frags (32 rows), B 2 frags
        const unsigned pos = ((mh * 2 + mf1_hi) ^ mf1_posx) * 16;
#pragma unroll
        for (int fm = 0; fm < 4; ++fm)
          afr[fm] = *reinterpret_cast<const short8v*>(
              &lds[as + wm * 8192 + fm * 2048 + mf1_row + pos]);
#pragma unroll
        for (int fn = 0; fn < 2; ++fn)
          bfr32[fn] = *reinterpret_cast<const short8v*>(
              &lds[bs + wn * 4096 + fn * 2048 + mf1_row + pos]);
      }
      // issue the stage for this phase (one half per phase)
      if (PIPE == 5) {
        // p0: B-kh0(kt+1) + A-kh1(kt+1); p1: B-kh1(kt+1);
        // p2: A-kh0(kt+2). Every overwritten slot's last read is
        // separated from the stage by the kt-1.p3 barrier.
        if (p == 0 && kt + 1 < KT) {
          STAGE_B(SLOT4(2 * kt + 2), kt + 1, 0)
          STAGE_A((unsigned)h5b2 * HALF_BYTES, kt + 1, 1)
        } else if (p == 1 && kt + 1 < KT) {
          STAGE_B(SLOT4(2 * kt + 3), kt + 1, 1)
        } else if (p == 2 && kt + 2 < KT) {
          STAGE_A((unsigned)h5s0 * HALF_BYTES, kt + 2, 0)
        }
      } else if (pre) {
        if (p == 0) STAGE_B(ss0, kt_s, 0)
        else if (p == 1) STAGE_A(ss0, kt_s, 0)
        else if (p == 2) STAGE_B(ss1, kt_s, 1)
        else STAGE_A(ss1, kt_s, 1)
      }
      if (PIPE == 0) dtx_bar();
      __builtin_amdgcn_s_setprio(1);
      if (MF == 0) {
#pragma unroll
        for (int fm = 0; fm < 4; ++fm)
#pragma unroll
          for (int fn = 0; fn < 4; ++fn)
            acc[mh * 4 + fm][fn] =
                MFMA16(afr[fm], bfr[fn], acc[mh * 4 + fm][fn]);
      } else {
#pragma unroll
        for (int fm = 0; fm < 4; ++fm)
#pragma unroll
          for (int fn = 0; fn < 2; ++fn)
            acc32[fm][fn] = MFMA32(afr[fm], bfr32[fn], acc32[fm][fn]);
      }
      __builtin_amdgcn_s_setprio(0);
      // publication waits: counted only (loads span barriers); the last
      // tile(s) issue no stages, so their in-flight count is too low
      // for the counted wait to retire what the next phases read —
      // drain fully there (end of the loop anyway).
      if (PIPE == 6 || PIPE == 7) {
        // TIMING PROBES ONLY (results are racy/wrong): 6 = PIPE2
        // without the counted vmcnt (isolates the wait cost);
        // 7 = PIPE2 without the barriers (isolates barrier parking).
        if (PIPE == 6 && (p & 1)) dtx_bar();
        if (PIPE == 7 && (p & 1)) {
          if (p == 3) { if (pre) { asm volatile(
              "s_waitcnt vmcnt(8)" ::: "memory"); } else dtx_vmcnt0(); }
        }
      } else if (PIPE == 5) {
        if (p == 3) {
          if (kt + 2 < KT) {
            asm volatile("s_waitcnt vmcnt(2)" ::: "memory");
          } else {
            dtx_vmcnt0();
          }
          dtx_bar();
        }
      } else if (p & 1) {
        if (PIPE == 2) {
          if (p == 3) { if (pre) { asm volatile(
              "s_waitcnt vmcnt(8)" ::: "memory"); } else dtx_vmcnt0(); }
          dtx_bar();
        } else {
          if (!pre) dtx_vmcnt0(); else dtx_vmcnt4();
          dtx_bar();
        }
      }
    }
  }
  dtx_vmcnt0();
  dtx_bar();

  // ---- epilogue: stage f32 through LDS ([256][128] f32 = 128 KiB per
  // round, 2 rounds of 128 columns), re-read coalesced, cvt_pk to bf16,
  // (optional source add), buffer stores (OOB rows clamp on num_records).
  const long ldc = (long)N * 2;
  const unsigned long long c_bytes = (unsigned long long)(M - m0) * ldc;
  const __amdgpu_buffer_rsrc_t rsC = __builtin_amdgcn_make_buffer_rsrc(
      (void*)(C + m0 * N), (short)0, (int)(c_bytes > 0xffffffffull
                                           ? 0xffffffffu : c_bytes),
      0x00020000);
  const __amdgpu_buffer_rsrc_t rsS = __builtin_amdgcn_make_buffer_rsrc(
      (void*)(SRC ? SRC + m0 * N : C), (short)0,
      (int)(HAS_SRC ? (c_bytes > 0xffffffffull ? 0xffffffffu : c_bytes) : 0),
      0x00020000);
  float* fl = reinterpret_cast<float*>(lds);

#pragma unroll
  for (int rnd = 0; rnd < 2; ++rnd) {
    if ((wn >> 1) == rnd) {
      if (MF == 0) {
        const int colb = (wn & 1) * 64 + (lane & 15);
#pragma unroll
        for (int fm = 0; fm < 8; ++fm)
#pragma unroll
          for (int fn = 0; fn < 4; ++fn) {
            const int row = wm * 128 + fm * 16 + (lane >> 4) * 4;
            const int col = colb + fn * 16;
#pragma unroll
            for (int j = 0; j < 4; ++j)
              fl[(row + j) * 128 + col] = acc[fm][fn][j];
          }
      } else {
        // 32x32 C layout: col = lane&31, row = (reg&3)+8*(reg>>2)+4*hi
        const int colb = (wn & 1) * 64 + (lane & 31);
#pragma unroll
        for (int fm = 0; fm < 4; ++fm)
#pragma unroll
          for (int fn = 0; fn < 2; ++fn) {
            const int rowb = wm * 128 + fm * 32 + (lane >> 5) * 4;
            const int col = colb + fn * 32;
#pragma unroll
            for (int r = 0; r < 16; ++r)
              fl[(rowb + (r & 3) + 8 * (r >> 2)) * 128 + col]
                  = acc32[fm][fn][r];
          }
      }
    }
    dtx_bar();
    // read-out: 512 threads, 16 passes; thread t handles row t>>5 (+16
    // per pass), cols (t&31)*4 .. +3 of this round's 128-col block.
#pragma unroll
    for (int pass = 0; pass < 16; ++pass) {
      const int row = pass * 16 + (tid >> 5);
      const int col = (tid & 31) * 4;
      const float4v v = *reinterpret_cast<const float4v*>(
          &fl[row * 128 + col]);
      const unsigned voff =
          (unsigned)(row * ldc + (n0 + rnd * 128 + col) * 2);
      unsigned lo, hi;
      if (HAS_SRC) {
        uint2v sv = __builtin_amdgcn_raw_buffer_load_b64(rsS, voff, 0, 0);
        lo = dtx_cvt_pk_bf16(v[0] + bf2f((unsigned short)(sv.x & 0xffff)),
                             v[1] + bf2f((unsigned short)(sv.x >> 16)));
        hi = dtx_cvt_pk_bf16(v[2] + bf2f((unsigned short)(sv.y & 0xffff)),
                             v[3] + bf2f((unsigned short)(sv.y >> 16)));
      } else {
        lo = dtx_cvt_pk_bf16(v[0], v[1]);
        hi = dtx_cvt_pk_bf16(v[2], v[3]);
      }
      uint2v out; out.x = lo; out.y = hi;
      __builtin_amdgcn_raw_buffer_store_b64(out, rsC, voff, 0, 0);
    }
    dtx_bar();
  }
#undef STAGE_A
#undef STAGE_B
#undef SLOT4
#undef SLOT5
}

void launch_gemm_nt(const void* A, const void* B, const void* SRC, void* C,
                    long M, int N, int K, hipStream_t stream) {
  const int mb_n = (int)((M + GEMM_BM - 1) / GEMM_BM);
  const int nb_n = N / GEMM_BN;
  dim3 grid(mb_n * nb_n), block(512);
  const bool cluster = (mb_n % 8 == 0) && (nb_n % 4 == 0) && (M % 256 == 0);
  // best measured config (tools/gemm_probe.bin A/B): PIPE=2 (5-slot
  // ring + sparse barriers), GRID=1 clusters when divisible else
  // GRID=2 snake groups.
#define LAUNCH(HS, G)                                                     \
  hipLaunchKernelGGL((gemm_nt_kernel<HS, G, 2, 0>), grid, block, 0,       \
                     stream, (const unsigned short*)A,                    \
                     (const unsigned short*)B, (const unsigned short*)SRC,\
                     (unsigned short*)C, M, N, K, mb_n)
  if (SRC) { if (cluster) LAUNCH(true, 1); else LAUNCH(true, 2); }
  else     { if (cluster) LAUNCH(false, 1); else LAUNCH(false, 2); }
#undef LAUNCH
}
