// Hand-written CDNA4 bf16 GEMM (gfx950) — 256x256x64 tile, 8-wave,
// glds-staged, 8-phase counted-vmcnt schedule (plan K6 upgrade: replace
// hipBLASLt for the model's dominant shapes; reference counterpart:
// XLA-generated matmuls from src/layers.py:55-57 / src/model.py:24-49).
//
// Geometry (guide cdna_hip_programming.md §5 "256² 8-phase template"):
//   tile BM x BN = 256 x 256, K-step BK = 64, 8 waves (2M x 4N), 512 thr
//   per-wave output 128 x 64 = acc f32x4[8][4] (128 VGPRs)
//   MFMA: v_mfma_f32_16x16x32_bf16; 64 per wave per K-tile, in 4
//   "quadrant" phases of 16
//   LDS 128 KiB = 2 buffers x (A 32K + B 32K); each operand K-tile is
//   staged as 2 halves of 128 rows x 64 k (16 KiB), 2 glds_dwordx4 per
//   wave per half (8 waves cooperate)
//   st_16x32 LDS swizzle: byte ^= ((byte>>9)&1)<<5 within each 1 KiB
//   subtile == colbyte ^= 32 when (row & 4). glds writes lane-linear, so
//   the swizzle is applied to the per-lane SOURCE address and to the
//   ds_read address (both-sides involution, guide §5.4 rule 21).
//
// Schedule per K-tile u — ONE raw s_barrier per phase, counted vmcnt
// only (never __syncthreads, which would drain the in-flight glds).
// Each phase is a single scheduling region {setprio(1); 16 MFMA; next
// phase's fragment ds_reads; this phase's stage glds; setprio(0)} with
// sched_group_barrier directives weaving the reads/stages INTO the MFMA
// stream (without them hipcc front-loads the reads and the read segment
// runs fully serial to the MFMA segment — measured +26% kernel time):
//   p0: MFMA q0 | read A strip1      | glds A-lo(u+1)
//   p1: MFMA q1 | read A strip2      | glds A-hi(u+1)
//   p2: MFMA q2 | read A strip3      | glds B-lo(u+2)+B-hi(u+2); vmcnt(4)
//   p3: MFMA q3 | read B(u+1)+A strip0(u+1) (guarded by p2's vmcnt+bar)
// Slot-lifetime: A(u+1) overwrites A(u-1) (last read 2 regions back);
// B(u+2) overwrites B(u) (last read 3 regions back). The p2 vmcnt(4)
// leaves only B(u+2)'s glds in flight, so A(u+1)/B(u+1) have landed
// before p3's prefetch reads; at the loop tail (stages skipped) the
// wait degrades to vmcnt(0) because the 4 newest in-flight loads would
// otherwise be the very tile about to be read. The prologue stages
// A(0), B(0), B(1) and guards tile 0 with vmcnt(4) (= B(1) in flight).
//
// Operand layouts ("NT"): A[M,K] row-major, B[N,K] row-major, C[M,N]
// row-major = A @ B^T — the natural layout of torch Linear fwd
// (y = x @ W^T) AND of dgrad against a transposed weight copy
// (dx = dy @ W = dy @ (W^T)^T). M, N, K must be multiples of 256/256/128.
#include "common.h"

using bf16x8g = __attribute__((ext_vector_type(8))) __bf16;

DEVINL f32x4 mfma16(bf16x8g a, bf16x8g b, f32x4 c) {
  return __builtin_amdgcn_mfma_f32_16x16x32_bf16(a, b, c, 0, 0, 0);
}

#define GLDS(gsrc, lds_off) \
  __builtin_amdgcn_global_load_lds((const __attribute__((address_space(1))) unsigned*)(gsrc), \
                                   (__attribute__((address_space(3))) unsigned*)(lds_off), 16, 0, 0)

// LDS map (bytes): buf p (p=0,1) at p*65536; A at +0 (lo 0..16K, hi
// 16..32K), B at +32768 (lo/hi).
#define LDS_A(p) ((p) * 65536)
#define LDS_B(p) ((p) * 65536 + 32768)

// swizzled in-half byte offset for element (row, colbyte).
// T2 XOR swizzle (row&7)<<4: a 16-lane ds_read_b128 group reading 16
// consecutive rows at one column lands on 8 distinct bank slots (2-way)
// instead of 4-way with a single toggled bit; involution within each
// 128-byte row so the glds source-side permutation stays 16B-contiguous.
DEVINL int swz(int row, int colbyte) {
  return row * 128 + (colbyte ^ ((row & 7) << 4));
}

// ---------------------------------------------------------------------
// Staging: half h (0/1) of operand tile u. 512 threads cover 16 KiB via
// 16 wave-instructions (wave w: bytes [(2w+j)*1024, +1024)).
// Per-lane global source honors the inverse swizzle.
// rows0 = first global row of the half; ldb = row stride in BYTES.
// ---------------------------------------------------------------------
DEVINL void stage_half(const u16* __restrict__ g, long rows0, long ldb,
                       long kbyte0, char* lds_base /*half base*/) {
  const int w = wave_id();
  const int l = lane_id();
#pragma unroll
  for (int j = 0; j < 2; ++j) {
    int q = (w * 2 + j) * 1024 + l * 16;       // dest byte in half image
    int row = q >> 7;
    int colb = (q & 127) ^ ((row & 7) << 4);   // inverse swizzle on SOURCE
    const char* src = (const char*)g + (rows0 + row) * ldb + kbyte0 + colb;
    GLDS(src, lds_base + (w * 2 + j) * 1024);
  }
}

// SAFE=1: drain-everything debug schedule (vmcnt(0) before every
// barrier) — used to bisect schedule races from layout bugs.
// ORDER: 0 = column-major tile walk (B-panel reuse), 1 = grouped walk
// (GROUP tile_m rows per super-column: A panels L2-resident too).
// ABL=1: skip all in-loop ds_reads (fragments loaded once, kept live —
// measures the MFMA+stage+barrier structure's ceiling). ABL=2: also skip
// the stage glds (pure MFMA+barrier ceiling). Timing-only (wrong math).
template <int SAFE, int GROUP, int PH, int ABL = 0>
__launch_bounds__(512, 1)
__global__ void gemm_nt_kernel(const u16* __restrict__ A,
                               const u16* __restrict__ B,
                               u16* __restrict__ C,
                               int M, int N, int K, int swizzle_xcd) {
  extern __shared__ __attribute__((aligned(16))) char smem[];
  const int l = lane_id();
  const int w = wave_id();
  const int wave_m = w >> 2;       // 0..1
  const int wave_n = w & 3;        // 0..3
  const int tiles_m = M >> 8;
  const int tiles_n = N >> 8;

  int bid = blockIdx.x;
  if (swizzle_xcd) {
    // bijective XCD remap: contiguous chunk of the grid per XCD (T1)
    const int nwg = tiles_m * tiles_n;
    const int qq = nwg >> 3, rr = nwg & 7;
    const int xcd = bid & 7, idx = bid >> 3;
    bid = (xcd < rr ? xcd * (qq + 1) : rr * (qq + 1) + (xcd - rr) * qq) + idx;
  }
  // column-major walk: consecutive blocks share the B panel; GROUP>0
  // walks n-fastest within GROUP-row bands so A panels stay L2-warm too
  long bm, bn;
  if (GROUP > 0) {
    const int band = GROUP * tiles_n;      // blocks per band
    const int b0 = bid / band, r0 = bid % band;
    bm = (long)(b0 * GROUP + r0 % GROUP) << 8;
    bn = (long)(r0 / GROUP) << 8;
  } else {
    bm = (long)(bid % tiles_m) << 8;
    bn = (long)(bid / tiles_m) << 8;
  }

  const long ldab = (long)K * 2;   // A/B row stride bytes
  const int NT = K >> 6;           // K-tiles

  f32x4 acc[8][4] = {};

  // ---- prologue: stage A(0), B(0), B(1) --------------------------------
  stage_half(A, bm + 0, ldab, 0, smem + LDS_A(0) + 0);
  stage_half(A, bm + 128, ldab, 0, smem + LDS_A(0) + 16384);
  stage_half(B, bn + 0, ldab, 0, smem + LDS_B(0) + 0);
  stage_half(B, bn + 128, ldab, 0, smem + LDS_B(0) + 16384);
  if (NT > 1) {
    stage_half(B, bn + 0, ldab, 128, smem + LDS_B(1) + 0);
    stage_half(B, bn + 128, ldab, 128, smem + LDS_B(1) + 16384);
  }
  // A(0)+B(0) (the first 8 glds) must land before tile 0's ds_reads;
  // B(1) (the last 4) may stay in flight. Counted wait + raw barrier —
  // the steady-state guard for tile u+1 is phase 3's vmcnt(4)+barrier.
  if (SAFE || NT == 1) asm volatile("s_waitcnt vmcnt(0)" ::: "memory");
  else asm volatile("s_waitcnt vmcnt(4)" ::: "memory");
  __builtin_amdgcn_s_barrier();
  asm volatile("" ::: "memory");

  // per-lane read offsets
  const int frag_colb = (l >> 4) * 16;        // k-chunk byte within kstep
  const int a_row = l & 15;                   // row within 16-row fragment
  // B fragments: n_local = wave_n*64 + nr*16 + (l&15)
  const int b_nloc = wave_n * 64 + (l & 15);
  char* const lds = smem;

  bf16x8g bfr[4][2];   // B frags [nr][kstep], live across the tile
  bf16x8g afr[4][2];   // A strip frags: two [mr][ks] pairs (double-buffer)

// slot: afr register slot; the fragment is rows (q*32 + (slot&1)*16)
#define READ_AP(pp, q, ks, slot)                                              \
  do { if (ABL == 0)                                                          \
    afr[slot][ks] = *(const bf16x8g*)(lds + LDS_A(pp) + wave_m * 16384 +      \
        swz((q) * 32 + ((slot) & 1) * 16 + a_row, (ks) * 64 + frag_colb));    \
  } while (0)
#define READ_A(q, ks, slot) READ_AP(par, q, ks, slot)
#define READ_BP(pp, nr, ks)                                                   \
  do { if (ABL == 0)                                                          \
    bfr[nr][ks] = *(const bf16x8g*)(lds + LDS_B(pp) +                         \
        ((b_nloc + (nr) * 16) >> 7) * 16384 +                                 \
        swz((b_nloc + (nr) * 16) & 127, (ks) * 64 + frag_colb));              \
  } while (0)
#define READ_B(nr, ks) READ_BP(par, nr, ks)
#define MFMA_QUAD2(q, base)                                                   \
  _Pragma("unroll") for (int ks = 0; ks < 2; ++ks)                            \
    _Pragma("unroll") for (int mr = 0; mr < 2; ++mr)                          \
      _Pragma("unroll") for (int nr = 0; nr < 4; ++nr)                        \
        acc[(q) * 2 + mr][nr] = mfma16(afr[(base) + mr][ks], bfr[nr][ks],     \
                                       acc[(q) * 2 + mr][nr])
// Raw s_barrier is NOT a compiler memory fence: without the empty
// "memory"-clobber asm on both sides hipcc may hoist a glds / ds_read
// across it, re-staging an LDS slot other waves still read (observed as
// a rare nondeterministic ~1e-3-per-block corruption). The fences order
// the compiler only; the hardware wait discipline stays counted-vmcnt.
#define BAR() do { if (SAFE) asm volatile("s_waitcnt vmcnt(0)" ::: "memory"); \
                  asm volatile("" ::: "memory");                              \
                  __builtin_amdgcn_s_barrier();                               \
                  asm volatile("" ::: "memory"); } while (0)
#define PRIO(x) __builtin_amdgcn_s_setprio(x)
// compile-time interleave of the post-MFMA prefetch reads into the MFMA
// cluster (T19): without it hipcc front-loads the ds_reads, serializing
// the read segment against the MFMA segment (measured: reads = +25%% of
// kernel time at zero overlap). Masks: MFMA=0x8, DS_READ=0x100.
#define SGB(mask, n) __builtin_amdgcn_sched_group_barrier(mask, n, 0)
#define INTERLEAVE(nds)                                                       \
  _Pragma("unroll") for (int gg = 0; gg < 4; ++gg) {                          \
    SGB(0x8, 4); SGB(0x100, nds);                                             \
  }
// variant with the phase's stage glds (VMEM 0x10) woven in as well
#define INTERLEAVE2(nds)                                                      \
  _Pragma("unroll") for (int gg = 0; gg < 4; ++gg) {                          \
    SGB(0x8, 2); SGB(0x100, nds); SGB(0x8, 2); SGB(0x10, 1);                  \
  }
#define VMCNT(n) asm volatile("s_waitcnt vmcnt(" #n ")" ::: "memory")

  if constexpr (ABL > 0) {
#pragma unroll
    for (int nr = 0; nr < 4; ++nr)
#pragma unroll
      for (int ks = 0; ks < 2; ++ks)
        bfr[nr][ks] = *(const bf16x8g*)(lds + LDS_B(0) + nr * 2048 + ks * 64);
#pragma unroll
    for (int sl = 0; sl < 4; ++sl)
#pragma unroll
      for (int ks = 0; ks < 2; ++ks)
        afr[sl][ks] = *(const bf16x8g*)(lds + LDS_A(0) + sl * 2048 + ks * 64);
  }
  if constexpr (PH == 4 && ABL == 0) {
    // tile 0's fragments (the loop reads tile u+1's at u.p3)
#pragma unroll
    for (int nr = 0; nr < 4; ++nr) { READ_BP(0, nr, 0); READ_BP(0, nr, 1); }
    READ_AP(0, 0, 0, 0); READ_AP(0, 0, 1, 0);
    READ_AP(0, 0, 0, 1); READ_AP(0, 0, 1, 1);
  }

  for (int u = 0; u < NT; ++u) {
    const int par = u & 1;
    const int nxt = par ^ 1;
    const long kb1 = (long)(u + 1) << 7;   // byte col of K-tile u+1
    const long kb2 = (long)(u + 2) << 7;

    if constexpr (PH == 2) {
      // ---- 2 phases per tile: longer MFMA clusters, half the barriers.
      // Same vmcnt discipline (stage order and per-wave glds counts are
      // identical to the 4-phase schedule, just paired).
      // phase A: B(u) + A strips 0,1; stage A(u+1) (both halves)
#pragma unroll
      for (int nr = 0; nr < 4; ++nr) { READ_B(nr, 0); READ_B(nr, 1); }
      READ_A(0, 0, 0); READ_A(0, 1, 0); READ_A(0, 0, 1); READ_A(0, 1, 1);
      READ_A(1, 0, 2); READ_A(1, 1, 2); READ_A(1, 0, 3); READ_A(1, 1, 3);
      if (u + 1 < NT) {
        stage_half(A, bm + 0, ldab, kb1, lds + LDS_A(nxt) + 0);
        stage_half(A, bm + 128, ldab, kb1, lds + LDS_A(nxt) + 16384);
      }
      BAR();
      PRIO(1);
#pragma unroll
      for (int ks = 0; ks < 2; ++ks)
#pragma unroll
        for (int mr = 0; mr < 4; ++mr)
#pragma unroll
          for (int nr = 0; nr < 4; ++nr)
            acc[mr][nr] = mfma16(afr[mr][ks], bfr[nr][ks], acc[mr][nr]);
      PRIO(0);
      BAR();
      // phase B: A strips 2,3; stage B(u+2)
      READ_A(2, 0, 0); READ_A(2, 1, 0); READ_A(2, 0, 1); READ_A(2, 1, 1);
      READ_A(3, 0, 2); READ_A(3, 1, 2); READ_A(3, 0, 3); READ_A(3, 1, 3);
      if (u + 2 < NT) {
        stage_half(B, bn + 0, ldab, kb2, lds + LDS_B(par) + 0);
        stage_half(B, bn + 128, ldab, kb2, lds + LDS_B(par) + 16384);
        VMCNT(4);
      } else {
        VMCNT(0);
      }
      BAR();
      PRIO(1);
#pragma unroll
      for (int ks = 0; ks < 2; ++ks)
#pragma unroll
        for (int mr = 0; mr < 4; ++mr)
#pragma unroll
          for (int nr = 0; nr < 4; ++nr)
            acc[4 + mr][nr] = mfma16(afr[mr][ks], bfr[nr][ks], acc[4 + mr][nr]);
      PRIO(0);
      BAR();
      continue;
    }

#define STAGE_IF(cond, ...) do { if (ABL < 2 && (cond)) stage_half(__VA_ARGS__); } while (0)
    // ---- 4 phases per tile, POST-MFMA fragment prefetch: phase q's
    // MFMA consumes afr pair (q&1); the reads for phase q+1 are issued
    // right after the MFMA cluster into the other pair, so their LDS
    // latency hides under the barrier + next phase's stage segment.
    // B(u+1)+strip0(u+1) reads go after p3's vmcnt(4)+MFMA (guarded).
    // phase 0: stage A-lo(u+1); MFMA q0 [afr pair0]; read strip1->pair1
    // ONE barrier per phase: every phase's reads/stages now live inside
    // its MFMA region, and all slot-reuse distances span >= 2 regions, so
    // the second barrier of the classic template guards nothing.
    PRIO(1);
    MFMA_QUAD2(0, 0);
    READ_A(1, 0, 2); READ_A(1, 1, 2); READ_A(1, 0, 3); READ_A(1, 1, 3);
    STAGE_IF(u + 1 < NT, A, bm + 0, ldab, kb1, lds + LDS_A(nxt) + 0);
    INTERLEAVE2(1);
    PRIO(0);
    BAR();
    // phase 1
    PRIO(1);
    MFMA_QUAD2(1, 2);
    READ_A(2, 0, 0); READ_A(2, 1, 0); READ_A(2, 0, 1); READ_A(2, 1, 1);
    STAGE_IF(u + 1 < NT, A, bm + 128, ldab, kb1, lds + LDS_A(nxt) + 16384);
    INTERLEAVE2(1);
    PRIO(0);
    BAR();
    // phase 2: both B(u+2) halves staged here, then the tile guard
    PRIO(1);
    MFMA_QUAD2(2, 0);
    READ_A(3, 0, 2); READ_A(3, 1, 2); READ_A(3, 0, 3); READ_A(3, 1, 3);
    STAGE_IF(u + 2 < NT, B, bn + 0, ldab, kb2, lds + LDS_B(par) + 0);
    STAGE_IF(u + 2 < NT, B, bn + 128, ldab, kb2, lds + LDS_B(par) + 16384);
    INTERLEAVE2(1);
    PRIO(0);
    if (ABL < 2 && u + 2 < NT) {
      VMCNT(4);  // leaves exactly B(u+2)'s 4 glds; A(u+1)/B(u+1) landed
    } else {
      // tail: B(u+2) skipped, so the 4 newest in-flight glds would be
      // A(u+1) itself — drain fully before tile u+1 reads it
      VMCNT(0);
    }
    BAR();
    // phase 3: MFMA q3 + next tile's fragments (guarded by the vmcnt+bar)
    PRIO(1);
    MFMA_QUAD2(3, 2);
    {
      // unconditional (single basic block, so the interleave directives
      // can act); at the last tile read this tile's slots again (dead)
      const int snxt = (u + 1 < NT) ? nxt : par;
#pragma unroll
      for (int nr = 0; nr < 4; ++nr) { READ_BP(snxt, nr, 0); READ_BP(snxt, nr, 1); }
      READ_AP(snxt, 0, 0, 0); READ_AP(snxt, 0, 1, 0);
      READ_AP(snxt, 0, 0, 1); READ_AP(snxt, 0, 1, 1);
      INTERLEAVE(3);
    }
    PRIO(0);
    BAR();
  }
#undef READ_A
#undef READ_B
#undef MFMA_QUAD

  // ---- epilogue: bf16 C write (D layout: col=l&15, row=4*(l>>4)+r) ----
  const long ldc = N;
  const long crow0 = bm + wave_m * 128 + (l >> 4) * 4;
  const long ccol0 = bn + wave_n * 64 + (l & 15);
#pragma unroll
  for (int mr = 0; mr < 8; ++mr) {
#pragma unroll
    for (int r = 0; r < 4; ++r) {
      const long row = crow0 + mr * 16 + r;
#pragma unroll
      for (int nr = 0; nr < 4; ++nr)
        C[row * ldc + ccol0 + nr * 16] = f2b(acc[mr][nr][r]);
    }
  }
}


// ---------------------------------------------------------------------
// 1-wave/SIMD variant: 4 waves (256 threads), each owning a 128x128
// sub-tile — accumulators spill into the unified AGPR file (gfx950:
// 512 regs/lane at 1 wave/SIMD). Motivation (measured): at 8 waves the
// MFMA skeleton runs at the issue floor but the per-tile LDS fragment
// reads (24 per 64 MFMAs per wave) add ~47% that never hides; this
// shape needs only 32 reads per 128 MFMAs (0.25/MFMA) and has no
// co-resident wave competing for the SIMD.
// ---------------------------------------------------------------------
DEVINL void stage_half4(const u16* __restrict__ g, long rows0, long ldb,
                        long kbyte0, char* lds_base) {
  const int w = wave_id();
  const int l = lane_id();
#pragma unroll
  for (int j = 0; j < 4; ++j) {
    int q = (w * 4 + j) * 1024 + l * 16;
    int row = q >> 7;
    int colb = (q & 127) ^ ((row & 7) << 4);
    const char* src = (const char*)g + (rows0 + row) * ldb + kbyte0 + colb;
    GLDS(src, lds_base + (w * 4 + j) * 1024);
  }
}

template <int GROUP>
__launch_bounds__(256, 1)
__global__ void gemm_nt_1w_kernel(const u16* __restrict__ A,
                                  const u16* __restrict__ B,
                                  u16* __restrict__ C,
                                  int M, int N, int K, int swizzle_xcd) {
  extern __shared__ __attribute__((aligned(16))) char smem[];
  const int l = lane_id();
  const int w = wave_id();
  const int wave_m = w >> 1;       // 0..1 (128-row half)
  const int wave_n = w & 1;        // 0..1 (128-col half)
  const int tiles_m = M >> 8;
  const int tiles_n = N >> 8;

  int bid = blockIdx.x;
  if (swizzle_xcd) {
    const int nwg = tiles_m * tiles_n;
    const int qq = nwg >> 3, rr = nwg & 7;
    const int xcd = bid & 7, idx = bid >> 3;
    bid = (xcd < rr ? xcd * (qq + 1) : rr * (qq + 1) + (xcd - rr) * qq) + idx;
  }
  long bm, bn;
  if (GROUP > 0) {
    const int band = GROUP * tiles_n;
    const int b0 = bid / band, r0 = bid % band;
    bm = (long)(b0 * GROUP + r0 % GROUP) << 8;
    bn = (long)(r0 / GROUP) << 8;
  } else {
    bm = (long)(bid % tiles_m) << 8;
    bn = (long)(bid / tiles_m) << 8;
  }

  const long ldab = (long)K * 2;
  const int NT = K >> 6;

  f32x4 acc[8][8] = {};   // 256 regs -> AGPR half of the unified file

  stage_half4(A, bm + 0, ldab, 0, smem + LDS_A(0) + 0);
  stage_half4(A, bm + 128, ldab, 0, smem + LDS_A(0) + 16384);
  stage_half4(B, bn + 0, ldab, 0, smem + LDS_B(0) + 0);
  stage_half4(B, bn + 128, ldab, 0, smem + LDS_B(0) + 16384);
  if (NT > 1) {
    stage_half4(B, bn + 0, ldab, 128, smem + LDS_B(1) + 0);
    stage_half4(B, bn + 128, ldab, 128, smem + LDS_B(1) + 16384);
  }
  // first 16 glds (A(0)+B(0)) must land; B(1)'s 8 may fly
  if (NT == 1) asm volatile("s_waitcnt vmcnt(0)" ::: "memory");
  else asm volatile("s_waitcnt vmcnt(8)" ::: "memory");
  __builtin_amdgcn_s_barrier();
  asm volatile("" ::: "memory");

  const int frag_colb = (l >> 4) * 16;
  const int a_row = l & 15;
  const int b_nloc0 = wave_n * 128 + (l & 15);
  char* const lds = smem;

  bf16x8g bfr[8][2];   // B frags [nr 0..7][ks]
  bf16x8g afr[4][2];   // two pairs (double-buffer across phases)

#define RD1_A(pp, q, ks, slot)                                                \
  afr[slot][ks] = *(const bf16x8g*)(lds + LDS_A(pp) + wave_m * 16384 +        \
      swz((q) * 32 + ((slot) & 1) * 16 + a_row, (ks) * 64 + frag_colb))
#define RD1_B(pp, nr, ks)                                                     \
  bfr[nr][ks] = *(const bf16x8g*)(lds + LDS_B(pp) +                           \
      ((b_nloc0 + (nr) * 16) >> 7) * 16384 +                                  \
      swz((b_nloc0 + (nr) * 16) & 127, (ks) * 64 + frag_colb))
#define MFMA_Q1(q, base)                                                      \
  _Pragma("unroll") for (int ks = 0; ks < 2; ++ks)                            \
    _Pragma("unroll") for (int mr = 0; mr < 2; ++mr)                          \
      _Pragma("unroll") for (int nr = 0; nr < 8; ++nr)                        \
        acc[(q) * 2 + mr][nr] = mfma16(afr[(base) + mr][ks], bfr[nr][ks],     \
                                       acc[(q) * 2 + mr][nr])
#define BAR1() do { asm volatile("" ::: "memory");                            \
                  __builtin_amdgcn_s_barrier();                               \
                  asm volatile("" ::: "memory"); } while (0)
#define ILV1(nds)                                                             \
  _Pragma("unroll") for (int gg = 0; gg < 8; ++gg) {                          \
    SGB(0x8, 2); SGB(0x100, nds); SGB(0x8, 2); SGB(0x10, 1);                  \
  }

  // tile 0 fragments
#pragma unroll
  for (int nr = 0; nr < 8; ++nr) { RD1_B(0, nr, 0); RD1_B(0, nr, 1); }
  RD1_A(0, 0, 0, 0); RD1_A(0, 0, 1, 0); RD1_A(0, 0, 0, 1); RD1_A(0, 0, 1, 1);

  for (int u = 0; u < NT; ++u) {
    const int par = u & 1;
    const int nxt = par ^ 1;
    const long kb1 = (long)(u + 1) << 7;
    const long kb2 = (long)(u + 2) << 7;

    // phase 0: 32 MFMA (q0) + strip1 reads + stage A-lo(u+1)
    PRIO(1);
    MFMA_Q1(0, 0);
    RD1_A(par, 1, 0, 2); RD1_A(par, 1, 1, 2); RD1_A(par, 1, 0, 3); RD1_A(par, 1, 1, 3);
    if (u + 1 < NT) stage_half4(A, bm + 0, ldab, kb1, lds + LDS_A(nxt) + 0);
    ILV1(1);
    PRIO(0);
    BAR1();
    // phase 1
    PRIO(1);
    MFMA_Q1(1, 2);
    RD1_A(par, 2, 0, 0); RD1_A(par, 2, 1, 0); RD1_A(par, 2, 0, 1); RD1_A(par, 2, 1, 1);
    if (u + 1 < NT) stage_half4(A, bm + 128, ldab, kb1, lds + LDS_A(nxt) + 16384);
    ILV1(1);
    PRIO(0);
    BAR1();
    // phase 2: B(u+2) both halves + the tile guard
    PRIO(1);
    MFMA_Q1(2, 0);
    RD1_A(par, 3, 0, 2); RD1_A(par, 3, 1, 2); RD1_A(par, 3, 0, 3); RD1_A(par, 3, 1, 3);
    if (u + 2 < NT) {
      stage_half4(B, bn + 0, ldab, kb2, lds + LDS_B(par) + 0);
      stage_half4(B, bn + 128, ldab, kb2, lds + LDS_B(par) + 16384);
    }
    ILV1(1);
    PRIO(0);
    if (u + 2 < NT) { VMCNT(8); } else { VMCNT(0); }
    BAR1();
    // phase 3: q3 + next tile's fragments
    PRIO(1);
    MFMA_Q1(3, 2);
    {
      const int snxt = (u + 1 < NT) ? nxt : par;
#pragma unroll
      for (int nr = 0; nr < 8; ++nr) { RD1_B(snxt, nr, 0); RD1_B(snxt, nr, 1); }
      RD1_A(snxt, 0, 0, 0); RD1_A(snxt, 0, 1, 0);
      RD1_A(snxt, 0, 0, 1); RD1_A(snxt, 0, 1, 1);
      ILV1(3);
    }
    PRIO(0);
    BAR1();
  }
#undef RD1_A
#undef RD1_B
#undef MFMA_Q1
#undef ILV1

  // epilogue: scalar bf16 stores (as the 8-wave kernel)
  const long ldc = N;
  const long crow0 = bm + wave_m * 128 + (l >> 4) * 4;
  const long ccol0 = bn + wave_n * 128 + (l & 15);
#pragma unroll
  for (int mr = 0; mr < 8; ++mr) {
#pragma unroll
    for (int r = 0; r < 4; ++r) {
      const long row = crow0 + mr * 16 + r;
#pragma unroll
      for (int nr = 0; nr < 8; ++nr)
        C[row * ldc + ccol0 + nr * 16] = f2b(acc[mr][nr][r]);
    }
  }
}

// ---------------------------------------------------------------------
// 32x32x16-MFMA sibling: same tile/LDS/staging/schedule, half the MFMA
// instructions (32768 FLOP per issue slot vs 16384) — tests whether the
// 16x16 variant is issue-bound. Wave tile 128x64 = 4 m_reps x 2 n_reps
// of 32x32 fragments; K-tile = 4 ksteps of 16.
// ---------------------------------------------------------------------
using bf16x8m = bf16x8g;
DEVINL f32x16 mfma32(bf16x8m a, bf16x8m b, f32x16 c) {
  return __builtin_amdgcn_mfma_f32_32x32x16_bf16(a, b, c, 0, 0, 0);
}

template <int SAFE, int GROUP>
__launch_bounds__(512, 1)
__global__ void gemm_nt32_kernel(const u16* __restrict__ A,
                                 const u16* __restrict__ B,
                                 u16* __restrict__ C,
                                 int M, int N, int K, int swizzle_xcd) {
  extern __shared__ __attribute__((aligned(16))) char smem[];
  const int l = lane_id();
  const int w = wave_id();
  const int wave_m = w >> 2;
  const int wave_n = w & 3;
  const int tiles_m = M >> 8;
  const int tiles_n = N >> 8;

  int bid = blockIdx.x;
  if (swizzle_xcd) {
    const int nwg = tiles_m * tiles_n;
    const int qq = nwg >> 3, rr = nwg & 7;
    const int xcd = bid & 7, idx = bid >> 3;
    bid = (xcd < rr ? xcd * (qq + 1) : rr * (qq + 1) + (xcd - rr) * qq) + idx;
  }
  long bm, bn;
  if (GROUP > 0) {
    const int band = GROUP * tiles_n;
    const int b0 = bid / band, r0 = bid % band;
    bm = (long)(b0 * GROUP + r0 % GROUP) << 8;
    bn = (long)(r0 / GROUP) << 8;
  } else {
    bm = (long)(bid % tiles_m) << 8;
    bn = (long)(bid / tiles_m) << 8;
  }

  const long ldab = (long)K * 2;
  const int NT = K >> 6;

  f32x16 acc[4][2] = {};

  stage_half(A, bm + 0, ldab, 0, smem + LDS_A(0) + 0);
  stage_half(A, bm + 128, ldab, 0, smem + LDS_A(0) + 16384);
  stage_half(B, bn + 0, ldab, 0, smem + LDS_B(0) + 0);
  stage_half(B, bn + 128, ldab, 0, smem + LDS_B(0) + 16384);
  if (NT > 1) {
    stage_half(B, bn + 0, ldab, 128, smem + LDS_B(1) + 0);
    stage_half(B, bn + 128, ldab, 128, smem + LDS_B(1) + 16384);
  }
  if (SAFE || NT == 1) asm volatile("s_waitcnt vmcnt(0)" ::: "memory");
  else asm volatile("s_waitcnt vmcnt(4)" ::: "memory");
  __builtin_amdgcn_s_barrier();
  asm volatile("" ::: "memory");

  // 32x32 fragment addressing: A row l&31, k-chunk byte 16*(l>>5)
  const int a_row = l & 31;
  const int kchunk = (l >> 5) * 16;
  const int b_nloc = wave_n * 64 + (l & 31);
  char* const lds = smem;

  bf16x8m bfr[2][4];    // [nr][kstep]
  bf16x8m afr[2][4];    // two buffers x 4 ksteps (one m_rep each)

#define RD_A32(pp, q, ks, buf)                                                \
  afr[buf][ks] = *(const bf16x8m*)(lds + LDS_A(pp) + wave_m * 16384 +         \
      swz((q) * 32 + a_row, (ks) * 32 + kchunk))
#define RD_B32(pp, nr, ks)                                                    \
  bfr[nr][ks] = *(const bf16x8m*)(lds + LDS_B(pp) +                           \
      ((b_nloc + (nr) * 32) >> 7) * 16384 +                                   \
      swz((b_nloc + (nr) * 32) & 127, (ks) * 32 + kchunk))
#define MFMA_M32(q, buf)                                                      \
  _Pragma("unroll") for (int ks = 0; ks < 4; ++ks)                            \
    _Pragma("unroll") for (int nr = 0; nr < 2; ++nr)                          \
      acc[q][nr] = mfma32(afr[buf][ks], bfr[nr][ks], acc[q][nr])
#define BAR32() do { if (SAFE) asm volatile("s_waitcnt vmcnt(0)" ::: "memory"); \
                  asm volatile("" ::: "memory");                              \
                  __builtin_amdgcn_s_barrier();                               \
                  asm volatile("" ::: "memory"); } while (0)

  // tile 0 fragments
#pragma unroll
  for (int nr = 0; nr < 2; ++nr)
#pragma unroll
    for (int ks = 0; ks < 4; ++ks) RD_B32(0, nr, ks);
  RD_A32(0, 0, 0, 0); RD_A32(0, 0, 1, 0); RD_A32(0, 0, 2, 0); RD_A32(0, 0, 3, 0);

  for (int u = 0; u < NT; ++u) {
    const int par = u & 1;
    const int nxt = par ^ 1;
    const long kb1 = (long)(u + 1) << 7;
    const long kb2 = (long)(u + 2) << 7;

    // p0: m_rep 0
    if (u + 1 < NT) stage_half(A, bm + 0, ldab, kb1, lds + LDS_A(nxt) + 0);
    BAR32();
    PRIO(1); MFMA_M32(0, 0); PRIO(0);
    RD_A32(par, 1, 0, 1); RD_A32(par, 1, 1, 1); RD_A32(par, 1, 2, 1); RD_A32(par, 1, 3, 1);
    BAR32();
    // p1: m_rep 1
    if (u + 1 < NT) stage_half(A, bm + 128, ldab, kb1, lds + LDS_A(nxt) + 16384);
    BAR32();
    PRIO(1); MFMA_M32(1, 1); PRIO(0);
    RD_A32(par, 2, 0, 0); RD_A32(par, 2, 1, 0); RD_A32(par, 2, 2, 0); RD_A32(par, 2, 3, 0);
    BAR32();
    // p2: m_rep 2
    if (u + 2 < NT) stage_half(B, bn + 0, ldab, kb2, lds + LDS_B(par) + 0);
    BAR32();
    PRIO(1); MFMA_M32(2, 0); PRIO(0);
    RD_A32(par, 3, 0, 1); RD_A32(par, 3, 1, 1); RD_A32(par, 3, 2, 1); RD_A32(par, 3, 3, 1);
    BAR32();
    // p3: m_rep 3; then prefetch tile u+1 fragments
    if (u + 2 < NT) {
      stage_half(B, bn + 128, ldab, kb2, lds + LDS_B(par) + 16384);
      VMCNT(4);
    } else {
      VMCNT(0);
    }
    BAR32();
    PRIO(1); MFMA_M32(3, 1); PRIO(0);
    if (u + 1 < NT) {
#pragma unroll
      for (int nr = 0; nr < 2; ++nr)
#pragma unroll
        for (int ks = 0; ks < 4; ++ks) RD_B32(nxt, nr, ks);
      RD_A32(nxt, 0, 0, 0); RD_A32(nxt, 0, 1, 0); RD_A32(nxt, 0, 2, 0); RD_A32(nxt, 0, 3, 0);
    }
    BAR32();
  }
#undef RD_A32
#undef RD_B32
#undef MFMA_M32
#undef BAR32

  // epilogue: D map col = l&31, row = (r&3) + 8*(r>>2) + 4*(l>>5)
  const long ldc = N;
  const long ccol0 = bn + wave_n * 64 + (l & 31);
#pragma unroll
  for (int mrep = 0; mrep < 4; ++mrep) {
#pragma unroll
    for (int r = 0; r < 16; ++r) {
      const long row = bm + wave_m * 128 + mrep * 32 +
                       (r & 3) + 8 * (r >> 2) + 4 * (l >> 5);
#pragma unroll
      for (int nr = 0; nr < 2; ++nr)
        C[row * ldc + ccol0 + nr * 32] = f2b(acc[mrep][nr][r]);
    }
  }
}

// host-side launcher (shared by bindings and the standalone probe)
template <int SAFE, int GROUP, int PH, int ABL = 0>
static hipError_t launch_gemm_nt_t(const u16* A, const u16* B, u16* C,
                                   int M, int N, int K, hipStream_t stream,
                                   int swizzle_xcd) {
  if (M % 256 || N % 256 || K % 64) return hipErrorInvalidValue;
  if (GROUP > 0 && (M >> 8) % GROUP) return hipErrorInvalidValue;
  static int lds_set = 0;
  if (!lds_set) {
    (void)hipFuncSetAttribute(
        reinterpret_cast<const void*>(&gemm_nt_kernel<SAFE, GROUP, PH, ABL>),
        hipFuncAttributeMaxDynamicSharedMemorySize, 131072);
    lds_set = 1;
  }
  const int grid = (M >> 8) * (N >> 8);
  hipLaunchKernelGGL((gemm_nt_kernel<SAFE, GROUP, PH, ABL>), dim3(grid),
                     dim3(512), 131072, stream, A, B, C, M, N, K, swizzle_xcd);
  return hipGetLastError();
}

template <int SAFE, int GROUP>
static hipError_t launch_gemm_nt32_t(const u16* A, const u16* B, u16* C,
                                     int M, int N, int K, hipStream_t stream,
                                     int swizzle_xcd) {
  if (M % 256 || N % 256 || K % 64) return hipErrorInvalidValue;
  if (GROUP > 0 && (M >> 8) % GROUP) return hipErrorInvalidValue;
  static int lds_set = 0;
  if (!lds_set) {
    (void)hipFuncSetAttribute(
        reinterpret_cast<const void*>(&gemm_nt32_kernel<SAFE, GROUP>),
        hipFuncAttributeMaxDynamicSharedMemorySize, 131072);
    lds_set = 1;
  }
  const int grid = (M >> 8) * (N >> 8);
  hipLaunchKernelGGL((gemm_nt32_kernel<SAFE, GROUP>), dim3(grid), dim3(512),
                     131072, stream, A, B, C, M, N, K, swizzle_xcd);
  return hipGetLastError();
}

static inline hipError_t launch_gemm_nt(const u16* A, const u16* B, u16* C,
                                        int M, int N, int K,
                                        hipStream_t stream,
                                        int swizzle_xcd = 1, int safe = 0,
                                        int group = 0, int ph = 2) {
  if (ph == 32) {
    if (group == 8) return launch_gemm_nt32_t<0, 8>(A, B, C, M, N, K, stream, swizzle_xcd);
    return launch_gemm_nt32_t<0, 0>(A, B, C, M, N, K, stream, swizzle_xcd);
  }
  if (safe) {
    if (group == 8) return launch_gemm_nt_t<1, 8, 4>(A, B, C, M, N, K, stream, swizzle_xcd);
    return launch_gemm_nt_t<1, 0, 4>(A, B, C, M, N, K, stream, swizzle_xcd);
  }
  if (ph == 10) {
    if (M % 256 || N % 256 || K % 64) return hipErrorInvalidValue;
    static int lds1 = 0;
    if (!lds1) {
      (void)hipFuncSetAttribute(
          reinterpret_cast<const void*>(&gemm_nt_1w_kernel<8>),
          hipFuncAttributeMaxDynamicSharedMemorySize, 131072);
      lds1 = 1;
    }
    hipLaunchKernelGGL((gemm_nt_1w_kernel<8>), dim3((M >> 8) * (N >> 8)),
                       dim3(256), 131072, stream, A, B, C, M, N, K, swizzle_xcd);
    return hipGetLastError();
  }
  if (ph == 101)
    return launch_gemm_nt_t<0, 8, 4, 1>(A, B, C, M, N, K, stream, swizzle_xcd);
  if (ph == 102)
    return launch_gemm_nt_t<0, 8, 4, 2>(A, B, C, M, N, K, stream, swizzle_xcd);
  if (ph == 2) {
    if (group == 8) return launch_gemm_nt_t<0, 8, 2>(A, B, C, M, N, K, stream, swizzle_xcd);
    return launch_gemm_nt_t<0, 0, 2>(A, B, C, M, N, K, stream, swizzle_xcd);
  }
  if (group == 8) return launch_gemm_nt_t<0, 8, 4>(A, B, C, M, N, K, stream, swizzle_xcd);
  if (group == 16) return launch_gemm_nt_t<0, 16, 4>(A, B, C, M, N, K, stream, swizzle_xcd);
  return launch_gemm_nt_t<0, 0, 4>(A, B, C, M, N, K, stream, swizzle_xcd);
}
