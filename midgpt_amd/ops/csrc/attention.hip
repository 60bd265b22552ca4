// Flash-style causal attention, forward + backward (plans K1/K2).
// MI355X-native: MFMA v_mfma_f32_32x32x16_bf16 tiles, LDS-staged K/V with
// XOR swizzles (guide T2/G4), online fp32 softmax held IN REGISTERS via the
// swapped-QK^T layout + permlane32_swap half exchanges (guide T12), fp32
// LSE saved for the recompute-based backward.
//
// Reference numerics: softmax(mask(QK^T)/sqrt(C)) in fp32, bf16 elsewhere
// (reference src/model.py:71-79). The T x T score matrix is never
// materialized.
//
// Geometry (fwd): one workgroup = NW waves (8 at T%256==0, else 4), each
// wave owning 32 q rows; KV tiles of 32 rows double-buffered in LDS with
// T14 load-early/write-late staging; grid = B*H*(T/(NW*32)), ordered so
// one (b,h)'s q-blocks share an XCD's L2.
// Geometry (bwd): TWO kernels — dkv (workgroup = NW*32 k rows, iterates
// q tiles >= its diagonal; dK/dV accumulate in registers) and dq
// (workgroup = NW*32 q rows, iterates k tiles; dQ in registers, no
// atomics), both with double-buffered T14 staging. A delta kernel
// (rowsum(dO*O)) runs first.
#include "common.h"
#include "mfma.h"

// ---------------------------------------------------------------------------
// LDS swizzles
// ---------------------------------------------------------------------------
// Row-major [R][C] bf16 tile, rows of C*2 bytes, read by ds_read_b128 at
// per-lane rows: XOR byte bits 4..7 with row (C=128: row&15 -> conflict-free;
// C=64 rows are 128 B: row&7).
template <int C> DEVINL int swz_rm(int row, int byte_in_row) {
  constexpr int M = (C == 128) ? 15 : 7;
  return byte_in_row ^ ((row & M) << 4);
}
// Transposed [C][32] bf16 tile (rows of 64 B): XOR byte bits 4..5 with
// (row>>2)&3 (see analysis: removes the 4-way conflict of the 64-B stride).
DEVINL int swz_tr(int row, int byte_in_row) {
  return byte_in_row ^ (((row >> 2) & 3) << 4);
}

// ---------------------------------------------------------------------------
// Cooperative staging (T14 async split: issue global loads EARLY, hold the
// tile in registers through the compute phase, ds_write late — the HBM
// latency hides under the MFMA phase instead of stalling the wave at a
// vmcnt before its LDS writes).
// ---------------------------------------------------------------------------
// Single-row loader -> row-major swizzled image only.
template <int C, int NT>
struct RmStage {
  static constexpr int NV = (32 * C + NT * 8 - 1) / (NT * 8);
  u16x8 v[NV];
  DEVINL void load(const u16* __restrict__ g) {
#pragma unroll
    for (int i = 0; i < NV; ++i) {
      const int idx = threadIdx.x * 8 + i * NT * 8;
      if (idx < 32 * C)
        v[i] = *(const u16x8*)(g + (long)(idx / C) * C + idx % C);
    }
  }
  DEVINL void write(u16* lds) const {
#pragma unroll
    for (int i = 0; i < NV; ++i) {
      const int idx = threadIdx.x * 8 + i * NT * 8;
      if (idx < 32 * C) {
        const int row = idx / C, col = idx % C;
        *(u16x8*)((char*)lds + row * C * 2 + swz_rm<C>(row, col * 2)) = v[i];
      }
    }
  }
};
// Row-pair loader -> transposed image (u16x2 pair writes), and optionally
// also the row-major image (both from ONE set of global loads).
template <int C, int NT>
struct TrStage {
  static constexpr int NP = (32 * C + NT * 16 - 1) / (NT * 16);
  u16x8 a[NP], b[NP];  // rows 2p and 2p+1 at an 8-col span
  DEVINL void load(const u16* __restrict__ g) {
#pragma unroll
    for (int i = 0; i < NP; ++i) {
      const int idx = threadIdx.x * 16 + i * NT * 16;
      if (idx >= 32 * C) continue;
      const int row = 2 * (idx / (2 * C));
      const int col = (idx / 2) % C;
      a[i] = *(const u16x8*)(g + (long)row * C + col);
      b[i] = *(const u16x8*)(g + (long)(row + 1) * C + col);
    }
  }
  DEVINL void write_tr(u16* lds) const {
#pragma unroll
    for (int i = 0; i < NP; ++i) {
      const int idx = threadIdx.x * 16 + i * NT * 16;
      if (idx >= 32 * C) continue;
      const int row = 2 * (idx / (2 * C));
      const int col = (idx / 2) % C;
#pragma unroll
      for (int j = 0; j < 8; ++j) {
        const int r = col + j;  // LDS row = c
        u16x2 pv; pv.x = a[i][j]; pv.y = b[i][j];
        *(u16x2*)((char*)lds + r * 64 + swz_tr(r, row * 2)) = pv;
      }
    }
  }
  DEVINL void write_rm(u16* lds) const {
#pragma unroll
    for (int i = 0; i < NP; ++i) {
      const int idx = threadIdx.x * 16 + i * NT * 16;
      if (idx >= 32 * C) continue;
      const int row = 2 * (idx / (2 * C));
      const int col = (idx / 2) % C;
      *(u16x8*)((char*)lds + row * C * 2 + swz_rm<C>(row, col * 2)) = a[i];
      *(u16x8*)((char*)lds + (row + 1) * C * 2 + swz_rm<C>(row + 1, col * 2)) = b[i];
    }
  }
};
// Synchronous convenience wrappers (prologue use).
template <int C, int NT>
DEVINL void stage_rm(const u16* __restrict__ g, u16* lds) {
  RmStage<C, NT> st; st.load(g); st.write(lds);
}
template <int C, int NT>
DEVINL void stage_tr(const u16* __restrict__ g, u16* lds) {
  TrStage<C, NT> st; st.load(g); st.write_tr(lds);
}

// Read one A/B fragment (bf16x8) from a swizzled row-major [R][C] tile:
// lane l -> row (l&31)+row0, bytes 16*chunk32? caller passes byte base.
template <int C>
DEVINL bf16x8_t read_rm_frag(const u16* lds, int row, int cbyte) {
  return *(const bf16x8_t*)((const char*)lds + row * C * 2 + swz_rm<C>(row, cbyte));
}
DEVINL bf16x8_t read_tr_frag(const u16* lds, int row, int kbyte) {
  return *(const bf16x8_t*)((const char*)lds + row * 64 + swz_tr(row, kbyte));
}

DEVINL float shfl32(float v, int src) { return __shfl(v, src, 32); }

// ===========================================================================
// Forward. SPW strips per wave: with SPW=2 each wave owns MIRRORED strips
// (w and NW*SPW-1-w), so causal work per wave is exactly equal and the
// staging barriers park nobody (the measured 50% SQ_WAIT_ANY at SPW=1 came
// from low-diagonal waves waiting on high-diagonal ones).
// ===========================================================================
template <int C, int NW, int SPW, int MINW>
__global__ __launch_bounds__(NW * 64, MINW) void attn_fwd_kernel(const u16* __restrict__ q,
                                const u16* __restrict__ k,
                                const u16* __restrict__ v,
                                u16* __restrict__ o, float* __restrict__ lse,
                                int B, int H, int T) {
  constexpr int NCB = C / 32;   // 32-col c-blocks
  constexpr int NCH = C / 16;   // 16-deep mfma chunks
  constexpr int NSTRIP = NW * SPW;
  const float scale = rsqrtf((float)C);
  // qb-outermost grid order: all q-blocks of one (b,h) land on the same
  // XCD (b%8 dispatch) for K/V L2 reuse (guide T1).
  const long bh = blockIdx.x % ((long)B * H);
  const int qb = blockIdx.x / (B * H);
  const int q0 = qb * (NSTRIP * 32);
  const int lane = lane_id();
  const int w = wave_id();
  int strip[SPW];
  strip[0] = w;
  if (SPW == 2) strip[1] = NSTRIP - 1 - w;

  extern __shared__ __attribute__((aligned(16))) char smem[];
  u16* ldsK = (u16*)smem;                       // [2][32*C]
  u16* ldsVt = (u16*)(smem + 2 * 32 * C * 2);   // [2][C*32]
  float* obuf = (float*)smem;                   // epilogue reuse: [NW][32*32]

  const u16* qg = q + (bh * T) * C;
  const u16* kg = k + (bh * T) * C;
  const u16* vg = v + (bh * T) * C;

  // Q B-fragments in registers for the LAST (busiest) strip only; earlier
  // strips re-read their Q rows from global per tile (L2-resident, few
  // tiles) to stay under the 256-VGPR cap at 2 waves/SIMD.
  bf16x8_t qf[NCH];
  f32x16 oacc[SPW][NCB];
  float m[SPW], lsum[SPW];
  {
    const u16* qrow = qg + (long)(q0 + 32 * strip[SPW - 1] + (lane & 31)) * C;
#pragma unroll
    for (int ch = 0; ch < NCH; ++ch)
      qf[ch] = *(const bf16x8_t*)(qrow + 16 * ch + 8 * (lane >> 5));
  }
#pragma unroll
  for (int sp = 0; sp < SPW; ++sp) {
#pragma unroll
    for (int cb = 0; cb < NCB; ++cb) oacc[sp][cb] = (f32x16)(0.f);
    m[sp] = -1e30f;
    lsum[sp] = 0.f;
  }

  const int nkt = (q0 + NSTRIP * 32) / 32;
  stage_rm<C, NW * 64>(kg, ldsK);
  stage_tr<C, NW * 64>(vg, ldsVt);
  __syncthreads();

  RmStage<C, NW * 64> kst;
  TrStage<C, NW * 64> vst;
  for (int kt = 0; kt < nkt; ++kt) {
    const int buf = kt & 1;
    const bool pre = kt + 1 < nkt;
    if (pre) {  // T14: issue next tile's loads before this tile's compute
      kst.load(kg + (long)(kt + 1) * 32 * C);
      vst.load(vg + (long)(kt + 1) * 32 * C);
    }
    const int k0 = kt * 32;
#pragma unroll
    for (int sp = 0; sp < SPW; ++sp) {
      const int qw0 = q0 + 32 * strip[sp];
      const int myq = qw0 + (lane & 31);
      if (k0 > qw0 + 31) continue;  // wave-uniform per strip
      // S = K x Q^T  (swapped: D rows = k, cols = q)
      f32x16 s = (f32x16)(0.f);
      const u16* kb = ldsK + buf * 32 * C;
      const u16* qrow = qg + (long)myq * C;
#pragma unroll
      for (int ch = 0; ch < NCH; ++ch) {
        bf16x8_t a = read_rm_frag<C>(kb, lane & 31, 16 * ch * 2 + 16 * (lane >> 5));
        bf16x8_t qfr = (sp == SPW - 1) ? qf[ch]
            : *(const bf16x8_t*)(qrow + 16 * ch + 8 * (lane >> 5));
        s = mfma_32x32x16_bf16(a, qfr, s);
      }
      // mask + tile row-max (per q = lane&31; halves merged via xor 32)
      float sv[16];
      float mt = -1e30f;
#pragma unroll
      for (int r = 0; r < 16; ++r) {
        sv[r] = (k0 + mfma_d_row(lane, r) > myq) ? -1e30f : s[r];
        mt = fmaxf(mt, sv[r]);
      }
      mt = fmaxf(mt, __shfl_xor(mt, 32));
      const float mn = fmaxf(m[sp], mt);
      const float alpha = __expf((m[sp] - mn) * scale);
      const bool need_rescale = !__all(mt <= m[sp]);  // EXACT: alpha==1 otherwise
      m[sp] = mn;
      float p[16], psum = 0.f;
#pragma unroll
      for (int r = 0; r < 16; ++r) {
        p[r] = __expf((sv[r] - mn) * scale);
        psum += p[r];
      }
      psum += __shfl_xor(psum, 32);
      lsum[sp] = lsum[sp] * alpha + psum;
      // rescale O by alpha[q_of_reg] — skipped (exactly) when no row's max
      // grew this tile, which is most tiles once the running max settles
      if (need_rescale) {
        float arow[16];
#pragma unroll
        for (int r = 0; r < 16; ++r) arow[r] = shfl32(alpha, mfma_d_row(lane, r));
#pragma unroll
        for (int cb = 0; cb < NCB; ++cb)
#pragma unroll
          for (int r = 0; r < 16; ++r) oacc[sp][cb][r] *= arow[r];
      }
      // P -> bf16 A-fragments; PV
      bf16x8_t pf0 = dlayout_to_afrag(p);
      bf16x8_t pf1 = dlayout_to_afrag(p + 8);
      const u16* vb = ldsVt + buf * C * 32;
#pragma unroll
      for (int cb = 0; cb < NCB; ++cb) {
        bf16x8_t b0 = read_tr_frag(vb, 32 * cb + (lane & 31), 16 * (lane >> 5));
        bf16x8_t b1 = read_tr_frag(vb, 32 * cb + (lane & 31), 32 + 16 * (lane >> 5));
        oacc[sp][cb] = mfma_32x32x16_bf16(pf0, b0, oacc[sp][cb]);
        oacc[sp][cb] = mfma_32x32x16_bf16(pf1, b1, oacc[sp][cb]);
      }
    }
    if (pre) {  // T14: LDS writes after compute (loads have landed by now)
      kst.write(ldsK + (1 - buf) * 32 * C);
      vst.write_tr(ldsVt + (1 - buf) * C * 32);
    }
    __syncthreads();
  }

  // epilogue: normalize, bounce through LDS, wide stores (per strip)
#pragma unroll
  for (int sp = 0; sp < SPW; ++sp) {
    const int qw0 = q0 + 32 * strip[sp];
    const int myq = qw0 + (lane & 31);
    const float rec = 1.f / lsum[sp];
    float rrow[16];
#pragma unroll
    for (int r = 0; r < 16; ++r) rrow[r] = shfl32(rec, mfma_d_row(lane, r));
    if (lane < 32) lse[bh * T + myq] = m[sp] * scale + __logf(lsum[sp]);
    float* ob = obuf + w * 32 * 32;
    u16* og = o + (bh * T + qw0) * C;
#pragma unroll
    for (int cb = 0; cb < NCB; ++cb) {
#pragma unroll
      for (int r = 0; r < 16; ++r) {
        const int row = mfma_d_row(lane, r);
        *(float*)((char*)ob + row * 128 + (((lane & 31) * 4) ^ ((row & 7) << 4))) =
            oacc[sp][cb][r] * rrow[r];
      }
      __builtin_amdgcn_s_waitcnt(0);  // lgkmcnt: LDS writes land (same wave)
      const int row = lane & 31;
      const int c16 = 16 * (lane >> 5);
      float tmp[16];
#pragma unroll
      for (int i = 0; i < 4; ++i) {
        f32x4 t = *(const f32x4*)((char*)ob + row * 128 + (((c16 + 4 * i) * 4) ^ ((row & 7) << 4)));
        tmp[4 * i] = t[0]; tmp[4 * i + 1] = t[1]; tmp[4 * i + 2] = t[2]; tmp[4 * i + 3] = t[3];
      }
      u16x8 out0, out1;
#pragma unroll
      for (int j = 0; j < 8; ++j) { out0[j] = f2b(tmp[j]); out1[j] = f2b(tmp[8 + j]); }
      *(u16x8*)(og + (long)row * C + 32 * cb + c16) = out0;
      *(u16x8*)(og + (long)row * C + 32 * cb + c16 + 8) = out1;
      __builtin_amdgcn_s_waitcnt(0);
    }
  }
}

// ===========================================================================
// Forward, DEPTH-2 prefetch variant (SPW=1): four LDS buffers, loads for
// tile kt+2 issued while tile kt computes — each load gets ~2 compute
// phases + a barrier to land instead of T14's single phase (the kernels
// are SQ_WAIT memory-bound at 2 waves/SIMD; this deepens the cover at
// +~12 VGPR). 4-tile unrolled loop keeps every stage object and buffer
// index static (rule 20: runtime-indexed vector arrays go to scratch).
// ===========================================================================
template <int C, int NW>
__global__ __launch_bounds__(NW * 64, 2) void attn_fwd_d2_kernel(
    const u16* __restrict__ q, const u16* __restrict__ k,
    const u16* __restrict__ v, u16* __restrict__ o, float* __restrict__ lse,
    int B, int H, int T) {
  constexpr int NCB = C / 32;
  constexpr int NCH = C / 16;
  const float scale = rsqrtf((float)C);
  const long bh = blockIdx.x % ((long)B * H);
  const int qb = blockIdx.x / (B * H);
  const int q0 = qb * (NW * 32);
  const int lane = lane_id();
  const int w = wave_id();
  const int qw0 = q0 + 32 * w;
  const int myq = qw0 + (lane & 31);

  extern __shared__ __attribute__((aligned(16))) char smem[];
  u16* ldsK = (u16*)smem;                       // [4][32*C]
  u16* ldsVt = (u16*)(smem + 4 * 32 * C * 2);   // [4][C*32]
  float* obuf = (float*)smem;                   // epilogue reuse

  const u16* qg = q + (bh * T) * C;
  const u16* kg = k + (bh * T) * C;
  const u16* vg = v + (bh * T) * C;

  bf16x8_t qf[NCH];
  {
    const u16* qrow = qg + (long)myq * C;
#pragma unroll
    for (int ch = 0; ch < NCH; ++ch)
      qf[ch] = *(const bf16x8_t*)(qrow + 16 * ch + 8 * (lane >> 5));
  }
  f32x16 oacc[NCB];
#pragma unroll
  for (int cb = 0; cb < NCB; ++cb) oacc[cb] = (f32x16)(0.f);
  float m = -1e30f, lsum = 0.f;

  const int nkt = (q0 + NW * 32) / 32;  // multiple of NW (>= 8): % 4 == 0

  auto tile_compute = [&](int kt, const u16* kb, const u16* vb) {
    const int k0 = kt * 32;
    if (k0 > qw0 + 31) return;  // wave-uniform
    f32x16 s = (f32x16)(0.f);
#pragma unroll
    for (int ch = 0; ch < NCH; ++ch) {
      bf16x8_t a = read_rm_frag<C>(kb, lane & 31, 16 * ch * 2 + 16 * (lane >> 5));
      s = mfma_32x32x16_bf16(a, qf[ch], s);
    }
    float sv[16];
    float mt = -1e30f;
#pragma unroll
    for (int r = 0; r < 16; ++r) {
      sv[r] = (k0 + mfma_d_row(lane, r) > myq) ? -1e30f : s[r];
      mt = fmaxf(mt, sv[r]);
    }
    mt = fmaxf(mt, __shfl_xor(mt, 32));
    const float mn = fmaxf(m, mt);
    const float alpha = __expf((m - mn) * scale);
    const bool need_rescale = !__all(mt <= m);
    m = mn;
    float pp[16], psum = 0.f;
#pragma unroll
    for (int r = 0; r < 16; ++r) {
      pp[r] = __expf((sv[r] - mn) * scale);
      psum += pp[r];
    }
    psum += __shfl_xor(psum, 32);
    lsum = lsum * alpha + psum;
    if (need_rescale) {
      float arow[16];
#pragma unroll
      for (int r = 0; r < 16; ++r) arow[r] = shfl32(alpha, mfma_d_row(lane, r));
#pragma unroll
      for (int cb = 0; cb < NCB; ++cb)
#pragma unroll
        for (int r = 0; r < 16; ++r) oacc[cb][r] *= arow[r];
    }
    bf16x8_t pf0 = dlayout_to_afrag(pp);
    bf16x8_t pf1 = dlayout_to_afrag(pp + 8);
#pragma unroll
    for (int cb = 0; cb < NCB; ++cb) {
      bf16x8_t b0 = read_tr_frag(vb, 32 * cb + (lane & 31), 16 * (lane >> 5));
      bf16x8_t b1 = read_tr_frag(vb, 32 * cb + (lane & 31), 32 + 16 * (lane >> 5));
      oacc[cb] = mfma_32x32x16_bf16(pf0, b0, oacc[cb]);
      oacc[cb] = mfma_32x32x16_bf16(pf1, b1, oacc[cb]);
    }
  };

  // prologue: tile 0 staged synchronously; tile 1's loads issued (stO)
  stage_rm<C, NW * 64>(kg, ldsK);
  stage_tr<C, NW * 64>(vg, ldsVt);
  RmStage<C, NW * 64> kE, kO;
  TrStage<C, NW * 64> vE, vO;
  if (1 < nkt) { kO.load(kg + (long)32 * C); vO.load(vg + (long)32 * C); }
  __syncthreads();

#define FWD_D2_STEP(J, ST_LD, ST_WR, LBUF, WBUF)                              \
  do {                                                                        \
    if (kt + (J) + 2 < nkt) {                                                 \
      k##ST_LD.load(kg + (long)(kt + (J) + 2) * 32 * C);                      \
      v##ST_LD.load(vg + (long)(kt + (J) + 2) * 32 * C);                      \
    }                                                                         \
    tile_compute(kt + (J), ldsK + (LBUF) * 32 * C, ldsVt + (LBUF) * C * 32);  \
    if (kt + (J) + 1 < nkt) {                                                 \
      k##ST_WR.write(ldsK + (WBUF) * 32 * C);                                 \
      v##ST_WR.write_tr(ldsVt + (WBUF) * C * 32);                             \
    }                                                                         \
    __syncthreads();                                                          \
  } while (0)

  for (int kt = 0; kt < nkt; kt += 4) {
    // invariant at loop top: buf0 = tile kt (ready); stO holds tile kt+1
    FWD_D2_STEP(0, E, O, 0, 1);  // issue kt+2 (E); compute kt; write kt+1
    FWD_D2_STEP(1, O, E, 1, 2);  // issue kt+3 (O); compute kt+1; write kt+2
    FWD_D2_STEP(2, E, O, 2, 3);
    FWD_D2_STEP(3, O, E, 3, 0);  // write kt+4 into buf0: invariant holds
  }
#undef FWD_D2_STEP

  // epilogue: normalize, bounce through LDS, wide stores
  const float rec = 1.f / lsum;
  float rrow[16];
#pragma unroll
  for (int r = 0; r < 16; ++r) rrow[r] = shfl32(rec, mfma_d_row(lane, r));
  if (lane < 32) lse[bh * T + myq] = m * scale + __logf(lsum);
  float* ob = obuf + w * 32 * 32;
  u16* og = o + (bh * T + qw0) * C;
#pragma unroll
  for (int cb = 0; cb < NCB; ++cb) {
#pragma unroll
    for (int r = 0; r < 16; ++r) {
      const int row = mfma_d_row(lane, r);
      *(float*)((char*)ob + row * 128 + (((lane & 31) * 4) ^ ((row & 7) << 4))) =
          oacc[cb][r] * rrow[r];
    }
    __builtin_amdgcn_s_waitcnt(0);
    const int row = lane & 31;
    const int c16 = 16 * (lane >> 5);
    float tmp[16];
#pragma unroll
    for (int i = 0; i < 4; ++i) {
      f32x4 t = *(const f32x4*)((char*)ob + row * 128 + (((c16 + 4 * i) * 4) ^ ((row & 7) << 4)));
      tmp[4 * i] = t[0]; tmp[4 * i + 1] = t[1]; tmp[4 * i + 2] = t[2]; tmp[4 * i + 3] = t[3];
    }
    u16x8 out0, out1;
#pragma unroll
    for (int j = 0; j < 8; ++j) { out0[j] = f2b(tmp[j]); out1[j] = f2b(tmp[8 + j]); }
    *(u16x8*)(og + (long)row * C + 32 * cb + c16) = out0;
    *(u16x8*)(og + (long)row * C + 32 * cb + c16 + 8) = out1;
    __builtin_amdgcn_s_waitcnt(0);
  }
}

// ===========================================================================
// Forward, PAIRED variant: two 32-row KV tiles per barrier round with ONE
// merged online-softmax rescale — halves barrier rounds and O-rescale
// passes vs the single-tile loop. nkt is always even (multiple of NW).
// ===========================================================================
template <int C, int NW>
__global__ __launch_bounds__(NW * 64, 2) void attn_fwd2_kernel(
    const u16* __restrict__ q, const u16* __restrict__ k,
    const u16* __restrict__ v, u16* __restrict__ o, float* __restrict__ lse,
    int B, int H, int T) {
  constexpr int NCB = C / 32;
  constexpr int NCH = C / 16;
  const float scale = rsqrtf((float)C);
  const long bh = blockIdx.x % ((long)B * H);
  const int qb = blockIdx.x / (B * H);
  const int q0 = qb * (NW * 32);
  const int lane = lane_id();
  const int w = wave_id();
  const int qw0 = q0 + 32 * w;
  const int myq = qw0 + (lane & 31);

  extern __shared__ __attribute__((aligned(16))) char smem[];
  u16* ldsK = (u16*)smem;                       // [2][2][32*C]
  u16* ldsVt = (u16*)(smem + 2 * 64 * C * 2);   // [2][2][C*32]
  float* obuf = (float*)smem;                   // epilogue reuse

  const u16* qg = q + (bh * T) * C;
  const u16* kg = k + (bh * T) * C;
  const u16* vg = v + (bh * T) * C;

  bf16x8_t qf[NCH];
  {
    const u16* qrow = qg + (long)myq * C;
#pragma unroll
    for (int ch = 0; ch < NCH; ++ch)
      qf[ch] = *(const bf16x8_t*)(qrow + 16 * ch + 8 * (lane >> 5));
  }
  f32x16 oacc[NCB];
#pragma unroll
  for (int cb = 0; cb < NCB; ++cb) oacc[cb] = (f32x16)(0.f);
  float m = -1e30f, lsum = 0.f;

  const int nrounds = (q0 + NW * 32) / 64;  // nkt/2, nkt always even
  // prologue: stage round 0 (two tiles) synchronously
  stage_rm<C, NW * 64>(kg, ldsK);
  stage_rm<C, NW * 64>(kg + 32 * C, ldsK + 32 * C);
  stage_tr<C, NW * 64>(vg, ldsVt);
  stage_tr<C, NW * 64>(vg + 32 * C, ldsVt + C * 32);
  __syncthreads();

  RmStage<C, NW * 64> kstA, kstB;
  TrStage<C, NW * 64> vstA, vstB;
  for (int rd = 0; rd < nrounds; ++rd) {
    const int buf = rd & 1;
    const bool pre = rd + 1 < nrounds;
    if (pre) {
      const long nb = (long)(rd + 1) * 64;
      kstA.load(kg + nb * C);
      kstB.load(kg + (nb + 32) * C);
      vstA.load(vg + nb * C);
      vstB.load(vg + (nb + 32) * C);
    }
    const int k0 = rd * 64;
    if (k0 <= qw0 + 31) {  // at least sub-tile A needed by this wave
      const u16* kbuf = ldsK + buf * 64 * C;
      const u16* vbuf = ldsVt + buf * 64 * C;  // (C*32 per sub-tile) x 2
      const bool needB = (k0 + 32) <= qw0 + 31;
      // S for both sub-tiles (independent MFMA chains)
      f32x16 s0 = (f32x16)(0.f), s1 = (f32x16)(0.f);
#pragma unroll
      for (int ch = 0; ch < NCH; ++ch) {
        bf16x8_t a0 = read_rm_frag<C>(kbuf, lane & 31, 16 * ch * 2 + 16 * (lane >> 5));
        s0 = mfma_32x32x16_bf16(a0, qf[ch], s0);
      }
      if (needB) {
#pragma unroll
        for (int ch = 0; ch < NCH; ++ch) {
          bf16x8_t a1 = read_rm_frag<C>(kbuf + 32 * C, lane & 31,
                                        16 * ch * 2 + 16 * (lane >> 5));
          s1 = mfma_32x32x16_bf16(a1, qf[ch], s1);
        }
      }
      // merged mask + stats over up to 64 k
      float sv0[16], sv1[16];
      float mt = -1e30f;
#pragma unroll
      for (int r = 0; r < 16; ++r) {
        sv0[r] = (k0 + mfma_d_row(lane, r) > myq) ? -1e30f : s0[r];
        mt = fmaxf(mt, sv0[r]);
        sv1[r] = (!needB || k0 + 32 + mfma_d_row(lane, r) > myq) ? -1e30f : s1[r];
        mt = fmaxf(mt, sv1[r]);
      }
      mt = fmaxf(mt, __shfl_xor(mt, 32));
      const float mn = fmaxf(m, mt);
      const float alpha = __expf((m - mn) * scale);
      const bool need_rescale = !__all(mt <= m);
      m = mn;
      float p0[16], p1[16], psum = 0.f;
#pragma unroll
      for (int r = 0; r < 16; ++r) {
        p0[r] = __expf((sv0[r] - mn) * scale);
        p1[r] = __expf((sv1[r] - mn) * scale);
        psum += p0[r] + p1[r];
      }
      psum += __shfl_xor(psum, 32);
      lsum = lsum * alpha + psum;
      if (need_rescale) {  // ONE rescale per 64 k
        float arow[16];
#pragma unroll
        for (int r = 0; r < 16; ++r) arow[r] = shfl32(alpha, mfma_d_row(lane, r));
#pragma unroll
        for (int cb = 0; cb < NCB; ++cb)
#pragma unroll
          for (int r = 0; r < 16; ++r) oacc[cb][r] *= arow[r];
      }
      // PV for both sub-tiles
      bf16x8_t pA0 = dlayout_to_afrag(p0);
      bf16x8_t pA1 = dlayout_to_afrag(p0 + 8);
#pragma unroll
      for (int cb = 0; cb < NCB; ++cb) {
        bf16x8_t b0 = read_tr_frag(vbuf, 32 * cb + (lane & 31), 16 * (lane >> 5));
        bf16x8_t b1 = read_tr_frag(vbuf, 32 * cb + (lane & 31), 32 + 16 * (lane >> 5));
        oacc[cb] = mfma_32x32x16_bf16(pA0, b0, oacc[cb]);
        oacc[cb] = mfma_32x32x16_bf16(pA1, b1, oacc[cb]);
      }
      if (needB) {
        bf16x8_t pB0 = dlayout_to_afrag(p1);
        bf16x8_t pB1 = dlayout_to_afrag(p1 + 8);
        const u16* vbufB = vbuf + C * 32;
#pragma unroll
        for (int cb = 0; cb < NCB; ++cb) {
          bf16x8_t b0 = read_tr_frag(vbufB, 32 * cb + (lane & 31), 16 * (lane >> 5));
          bf16x8_t b1 = read_tr_frag(vbufB, 32 * cb + (lane & 31), 32 + 16 * (lane >> 5));
          oacc[cb] = mfma_32x32x16_bf16(pB0, b0, oacc[cb]);
          oacc[cb] = mfma_32x32x16_bf16(pB1, b1, oacc[cb]);
        }
      }
    }
    if (pre) {
      u16* kd = ldsK + (1 - buf) * 64 * C;
      u16* vd = ldsVt + (1 - buf) * 64 * C;
      kstA.write(kd);
      kstB.write(kd + 32 * C);
      vstA.write_tr(vd);
      vstB.write_tr(vd + C * 32);
    }
    __syncthreads();
  }

  // epilogue: identical to the single-tile kernel
  const float rec = 1.f / lsum;
  float rrow[16];
#pragma unroll
  for (int r = 0; r < 16; ++r) rrow[r] = shfl32(rec, mfma_d_row(lane, r));
  if (lane < 32) lse[bh * T + myq] = m * scale + __logf(lsum);
  float* ob = obuf + w * 32 * 32;
  u16* og = o + (bh * T + qw0) * C;
#pragma unroll
  for (int cb = 0; cb < NCB; ++cb) {
#pragma unroll
    for (int r = 0; r < 16; ++r) {
      const int row = mfma_d_row(lane, r);
      *(float*)((char*)ob + row * 128 + (((lane & 31) * 4) ^ ((row & 7) << 4))) =
          oacc[cb][r] * rrow[r];
    }
    __builtin_amdgcn_s_waitcnt(0);
    const int row = lane & 31;
    const int c16 = 16 * (lane >> 5);
    float tmp[16];
#pragma unroll
    for (int i = 0; i < 4; ++i) {
      f32x4 t = *(const f32x4*)((char*)ob + row * 128 + (((c16 + 4 * i) * 4) ^ ((row & 7) << 4)));
      tmp[4 * i] = t[0]; tmp[4 * i + 1] = t[1]; tmp[4 * i + 2] = t[2]; tmp[4 * i + 3] = t[3];
    }
    u16x8 out0, out1;
#pragma unroll
    for (int j = 0; j < 8; ++j) { out0[j] = f2b(tmp[j]); out1[j] = f2b(tmp[8 + j]); }
    *(u16x8*)(og + (long)row * C + 32 * cb + c16) = out0;
    *(u16x8*)(og + (long)row * C + 32 * cb + c16 + 8) = out1;
    __builtin_amdgcn_s_waitcnt(0);
  }
}

// ===========================================================================
// delta = rowsum(dO * O) — one wave per row (prologue of backward)
// ===========================================================================
__global__ void attn_delta_kernel(const u16* __restrict__ dO,
                                  const u16* __restrict__ O,
                                  float* __restrict__ delta, long N, int C) {
  const int lane = lane_id();
  const long row0 = (long)blockIdx.x * (blockDim.x / WAVE) + wave_id();
  const long rstep = (long)gridDim.x * (blockDim.x / WAVE);
  for (long row = row0; row < N; row += rstep) {
    float acc = 0.f;
    for (int i = lane * 8; i < C; i += WAVE * 8) {
      u16x8 a = *(const u16x8*)(dO + row * C + i);
      u16x8 b = *(const u16x8*)(O + row * C + i);
#pragma unroll
      for (int j = 0; j < 8; ++j) acc += b2f(a[j]) * b2f(b[j]);
    }
    acc = group_sum<WAVE>(acc);
    if (lane == 0) delta[row] = acc;
  }
}

// ===========================================================================
// Backward, kernel A (dK/dV): one WG = NW*32 k rows (32/wave); iterates q
// tiles >= its diagonal with DOUBLE-BUFFERED staging (the tile qt+1 is
// staged while qt is computed, hiding the HBM load latency that a
// stage-barrier-compute structure exposes). dK/dV accumulate in registers.
// S and dS are recomputed from Q,K,LSE (standard flash recompute).
// ===========================================================================
template <int C, int NW, int ABLATE = 0, int MINW = 2>
__global__ __launch_bounds__(NW * 64, MINW) void attn_bwd_dkv_kernel(
    const u16* __restrict__ dO, const u16* __restrict__ q,
    const u16* __restrict__ k, const u16* __restrict__ v,
    const float* __restrict__ lse, const float* __restrict__ delta,
    u16* __restrict__ dk, u16* __restrict__ dv, int B, int H, int T) {
  constexpr int NCB = C / 32;
  constexpr int NCH = C / 16;
  constexpr int TILE = 4 * 32 * C;  // u16 elems per buffer set (Q,Qt,dO,dOt)
  const float scale = rsqrtf((float)C);
  const long bh = blockIdx.x % ((long)B * H);
  const int kb = blockIdx.x / (B * H);
  const int lane = lane_id();
  const int w = wave_id();
  const int kw0 = kb * (NW * 32) + 32 * w;
  const int myk = kw0 + (lane & 31);

  extern __shared__ __attribute__((aligned(16))) char smem[];
  u16* base = (u16*)smem;                       // [2][TILE]
  float* ldsLse = (float*)(base + 2 * TILE);    // [2][32]
  float* ldsDelta = ldsLse + 64;                // [2][32]

  const u16* qg = q + (bh * T) * C;
  const u16* kg = k + (bh * T) * C;
  const u16* vg = v + (bh * T) * C;
  const u16* dog = dO + (bh * T) * C;
  const u16* krow = kg + (long)myk * C;
  const u16* vrow = vg + (long)myk * C;

  // K row fragments resident in registers (the L2 re-read variant was
  // measured -20% on dkv C=128 in a full-step profile despite removing
  // the 2-VGPR spill — the per-tile global loads add latency the spill
  // never cost); re-read only under the MINW=3 squeeze experiment.
  constexpr bool KF_RES = (MINW <= 2);
  bf16x8_t kf[KF_RES ? NCH : 1];
  if (KF_RES) {
#pragma unroll
    for (int ch = 0; ch < NCH; ++ch)
      kf[ch] = *(const bf16x8_t*)(krow + 16 * ch + 8 * (lane >> 5));
  }

  f32x16 dvacc[NCB], dkacc[NCB];
#pragma unroll
  for (int cb = 0; cb < NCB; ++cb) { dvacc[cb] = (f32x16)(0.f); dkacc[cb] = (f32x16)(0.f); }

  const int qt0 = kb * NW;  // diagonal q tile
  const int nqt = T / 32;

  auto stage_set = [&](int qt, int buf) {
    const long qbase = (long)qt * 32;
    u16* bq = base + buf * TILE;
    stage_rm<C, NW * 64>(qg + qbase * C, bq);
    stage_tr<C, NW * 64>(qg + qbase * C, bq + 32 * C);
    stage_rm<C, NW * 64>(dog + qbase * C, bq + 2 * 32 * C);
    stage_tr<C, NW * 64>(dog + qbase * C, bq + 3 * 32 * C);
    if (threadIdx.x < 32) {
      ldsLse[buf * 32 + threadIdx.x] = lse[bh * T + qbase + threadIdx.x];
      ldsDelta[buf * 32 + threadIdx.x] = delta[bh * T + qbase + threadIdx.x];
    }
  };

  stage_set(qt0, 0);
  __syncthreads();
  TrStage<C, NW * 64> qst, dost;  // ONE load feeds both rm and tr images
  float lse_r = 0.f, del_r = 0.f;
  for (int qt = qt0; qt < nqt; ++qt) {
    const int buf = (qt - qt0) & 1;
    const bool pre = qt + 1 < nqt;
    if (ABLATE != 2 && pre) {  // T14 split: loads early
      const long nb = (long)(qt + 1) * 32;
      qst.load(qg + nb * C);
      dost.load(dog + nb * C);
      if (threadIdx.x < 32) {
        lse_r = lse[bh * T + nb + threadIdx.x];
        del_r = delta[bh * T + nb + threadIdx.x];
      }
    }
    const int qbase = qt * 32;
    if (ABLATE != 1 && qbase + 31 >= kw0) {  // not fully masked for this wave
      const u16* ldsQ = base + buf * TILE;
      const u16* ldsQt = ldsQ + 32 * C;
      const u16* ldsDO = ldsQt + C * 32;
      const u16* ldsDOt = ldsDO + 32 * C;
      const float* lseb = ldsLse + buf * 32;
      const float* deltab = ldsDelta + buf * 32;
      // S = Q x K^T (D rows = q regs, cols = k lanes)
      f32x16 s = (f32x16)(0.f);
#pragma unroll
      for (int ch = 0; ch < NCH; ++ch) {
        bf16x8_t a = read_rm_frag<C>(ldsQ, lane & 31, 16 * ch * 2 + 16 * (lane >> 5));
        bf16x8_t kfr = KF_RES ? kf[ch]
            : *(const bf16x8_t*)(krow + 16 * ch + 8 * (lane >> 5));
        s = mfma_32x32x16_bf16(a, kfr, s);
      }
      float p[16];
#pragma unroll
      for (int r = 0; r < 16; ++r) {
        const int qrow = qbase + mfma_d_row(lane, r);
        if (ABLATE == 3)
          p[r] = s[r] * scale - lseb[mfma_d_row(lane, r)];  // no exp/mask
        else
          p[r] = (myk > qrow) ? 0.f
               : __expf(s[r] * scale - lseb[mfma_d_row(lane, r)]);
      }
      // dV += P^T x dO
      bf16x8_t pf0, pf1;
      if (ABLATE == 4) {  // skip the permlane pack, keep p live
        asm volatile("" :: "v"(p[0]), "v"(p[8]));
        pf0 = kf[0]; pf1 = kf[0];
      } else {
        pf0 = dlayout_to_afrag(p);
        pf1 = dlayout_to_afrag(p + 8);
      }
#pragma unroll
      for (int cb = 0; cb < NCB; ++cb) {
        bf16x8_t b0, b1;
        if (ABLATE == 5) { b0 = kf[0]; b1 = kf[0]; }  // skip LDS frag reads
        else {
          b0 = read_tr_frag(ldsDOt, 32 * cb + (lane & 31), 16 * (lane >> 5));
          b1 = read_tr_frag(ldsDOt, 32 * cb + (lane & 31), 32 + 16 * (lane >> 5));
        }
        dvacc[cb] = mfma_32x32x16_bf16(pf0, b0, dvacc[cb]);
        dvacc[cb] = mfma_32x32x16_bf16(pf1, b1, dvacc[cb]);
      }
      // dP = dO x V^T; V frags from global (L2-resident)
      f32x16 dp = (f32x16)(0.f);
#pragma unroll
      for (int ch = 0; ch < NCH; ++ch) {
        bf16x8_t a = read_rm_frag<C>(ldsDO, lane & 31, 16 * ch * 2 + 16 * (lane >> 5));
        bf16x8_t vf = *(const bf16x8_t*)(vrow + 16 * ch + 8 * (lane >> 5));
        dp = mfma_32x32x16_bf16(a, vf, dp);
      }
      // dS = P * (dP - delta[q]) * scale; dK += dS^T x Q
      float ds[16];
#pragma unroll
      for (int r = 0; r < 16; ++r)
        ds[r] = p[r] * (dp[r] - deltab[mfma_d_row(lane, r)]) * scale;
      bf16x8_t df0, df1;
      if (ABLATE == 4) {
        asm volatile("" :: "v"(ds[0]), "v"(ds[8]));
        df0 = kf[0]; df1 = kf[0];
      } else {
        df0 = dlayout_to_afrag(ds);
        df1 = dlayout_to_afrag(ds + 8);
      }
#pragma unroll
      for (int cb = 0; cb < NCB; ++cb) {
        bf16x8_t b0, b1;
        if (ABLATE == 5) { b0 = kf[0]; b1 = kf[0]; }
        else {
          b0 = read_tr_frag(ldsQt, 32 * cb + (lane & 31), 16 * (lane >> 5));
          b1 = read_tr_frag(ldsQt, 32 * cb + (lane & 31), 32 + 16 * (lane >> 5));
        }
        dkacc[cb] = mfma_32x32x16_bf16(df0, b0, dkacc[cb]);
        dkacc[cb] = mfma_32x32x16_bf16(df1, b1, dkacc[cb]);
      }
    }
    if (ABLATE != 2 && pre) {  // writes late
      u16* bq = base + (buf ^ 1) * TILE;
      qst.write_rm(bq);
      qst.write_tr(bq + 32 * C);
      dost.write_rm(bq + 2 * 32 * C);
      dost.write_tr(bq + 3 * 32 * C);
      if (threadIdx.x < 32) {
        ldsLse[(buf ^ 1) * 32 + threadIdx.x] = lse_r;
        ldsDelta[(buf ^ 1) * 32 + threadIdx.x] = del_r;
      }
    }
    __syncthreads();
  }

  // epilogue: LDS bounce -> wide bf16 stores (reuses the staging region)
  float* ob = (float*)smem + w * 32 * 32;
  u16* dkg = dk + (bh * T + kw0) * C;
  u16* dvg = dv + (bh * T + kw0) * C;
#pragma unroll
  for (int which = 0; which < 2; ++which) {
    f32x16* acc = which == 0 ? dkacc : dvacc;
    u16* out = which == 0 ? dkg : dvg;
#pragma unroll
    for (int cb = 0; cb < NCB; ++cb) {
#pragma unroll
      for (int r = 0; r < 16; ++r) {
        const int row = mfma_d_row(lane, r);
        *(float*)((char*)ob + row * 128 + (((lane & 31) * 4) ^ ((row & 7) << 4))) =
            acc[cb][r];
      }
      __builtin_amdgcn_s_waitcnt(0);
      const int row = lane & 31;
      const int c16 = 16 * (lane >> 5);
      float tmp[16];
#pragma unroll
      for (int i = 0; i < 4; ++i) {
        f32x4 t = *(const f32x4*)((char*)ob + row * 128 +
                                  (((c16 + 4 * i) * 4) ^ ((row & 7) << 4)));
        tmp[4 * i] = t[0]; tmp[4 * i + 1] = t[1];
        tmp[4 * i + 2] = t[2]; tmp[4 * i + 3] = t[3];
      }
      u16x8 o0, o1;
#pragma unroll
      for (int j = 0; j < 8; ++j) { o0[j] = f2b(tmp[j]); o1[j] = f2b(tmp[8 + j]); }
      *(u16x8*)(out + (long)row * C + 32 * cb + c16) = o0;
      *(u16x8*)(out + (long)row * C + 32 * cb + c16 + 8) = o1;
      __builtin_amdgcn_s_waitcnt(0);
    }
  }
}

// ===========================================================================
// Backward, kernel B (dQ): SPW mirrored q strips per wave (work balance,
// see forward); iterates k tiles up to the max diagonal, DOUBLE-BUFFERED.
// dQ accumulates in registers — no atomics.
// ===========================================================================
template <int C, int NW, int SPW, int MINW>
__global__ __launch_bounds__(NW * 64, MINW) void attn_bwd_dq_kernel(
    const u16* __restrict__ dO, const u16* __restrict__ q,
    const u16* __restrict__ k, const u16* __restrict__ v,
    const float* __restrict__ lse, const float* __restrict__ delta,
    u16* __restrict__ dq, int B, int H, int T) {
  constexpr int NCB = C / 32;
  constexpr int NCH = C / 16;
  constexpr int NSTRIP = NW * SPW;
  constexpr int TILE = 3 * 32 * C;  // u16 elems per buffer set (K, V, Kt)
  const float scale = rsqrtf((float)C);
  const long bh = blockIdx.x % ((long)B * H);
  const int qb = blockIdx.x / (B * H);
  const int q0 = qb * (NSTRIP * 32);
  const int lane = lane_id();
  const int w = wave_id();
  int strip[SPW];
  strip[0] = w;
  if (SPW == 2) strip[1] = NSTRIP - 1 - w;

  extern __shared__ __attribute__((aligned(16))) char smem[];
  u16* base = (u16*)smem;               // [2][TILE] = K rm | V rm | Kt
  u16* ldsDS = base + 2 * TILE;         // per wave + w*32*32 (bf16)

  const u16* qg = q + (bh * T) * C;
  const u16* kg = k + (bh * T) * C;
  const u16* vg = v + (bh * T) * C;
  const u16* dog = dO + (bh * T) * C;

  bf16x8_t qf[NCH];  // registers only for the last (busiest) strip
  f32x16 dqacc[SPW][NCB];
  float mylse[SPW], mydelta[SPW];
#pragma unroll
  for (int sp = 0; sp < SPW; ++sp) {
    const long myq = q0 + 32 * strip[sp] + (lane & 31);
    if (sp == SPW - 1) {
      const u16* qrow = qg + myq * C;
#pragma unroll
      for (int ch = 0; ch < NCH; ++ch)
        qf[ch] = *(const bf16x8_t*)(qrow + 16 * ch + 8 * (lane >> 5));
    }
    mylse[sp] = lse[bh * T + myq];
    mydelta[sp] = delta[bh * T + myq];
#pragma unroll
    for (int cb = 0; cb < NCB; ++cb) dqacc[sp][cb] = (f32x16)(0.f);
  }

  const int nkt = (q0 + NSTRIP * 32) / 32;
  auto stage_set = [&](int kt, int buf) {
    const long k0 = (long)kt * 32;
    u16* bk = base + buf * TILE;
    stage_rm<C, NW * 64>(kg + k0 * C, bk);
    stage_rm<C, NW * 64>(vg + k0 * C, bk + 32 * C);
    stage_tr<C, NW * 64>(kg + k0 * C, bk + 2 * 32 * C);
  };
  stage_set(0, 0);
  __syncthreads();
  TrStage<C, NW * 64> kst;  // ONE load feeds K rm + Kt images
  RmStage<C, NW * 64> vst;
  for (int kt = 0; kt < nkt; ++kt) {
    const int buf = kt & 1;
    const bool pre = kt + 1 < nkt;
    if (pre) {
      kst.load(kg + (long)(kt + 1) * 32 * C);
      vst.load(vg + (long)(kt + 1) * 32 * C);
    }
    const int k0 = kt * 32;
    const u16* ldsK = base + buf * TILE;
    const u16* ldsV = ldsK + 32 * C;
    const u16* ldsKt = ldsV + 32 * C;
#pragma unroll
    for (int sp = 0; sp < SPW; ++sp) {
      const int qw0 = q0 + 32 * strip[sp];
      if (k0 > qw0 + 31) continue;
      const u16* qrow = qg + (long)(qw0 + (lane & 31)) * C;
      const u16* dorow = dog + (long)(qw0 + (lane & 31)) * C;
      // S = Q x K^T and dP = dO x V^T in one pass.
      f32x16 s = (f32x16)(0.f);
      f32x16 dp = (f32x16)(0.f);
#pragma unroll
      for (int ch = 0; ch < NCH; ++ch) {
        bf16x8_t kfrag = read_rm_frag<C>(ldsK, lane & 31, 16 * ch * 2 + 16 * (lane >> 5));
        bf16x8_t qfr = (sp == SPW - 1) ? qf[ch]
            : *(const bf16x8_t*)(qrow + 16 * ch + 8 * (lane >> 5));
        s = mfma_32x32x16_bf16(qfr, kfrag, s);
        bf16x8_t vfrag = read_rm_frag<C>(ldsV, lane & 31, 16 * ch * 2 + 16 * (lane >> 5));
        bf16x8_t dof = *(const bf16x8_t*)(dorow + 16 * ch + 8 * (lane >> 5));
        dp = mfma_32x32x16_bf16(dof, vfrag, dp);
      }
      // rows q are reg-mapped; cols k = lane&31. lse/delta per q via shfl.
      float ds[16];
#pragma unroll
      for (int r = 0; r < 16; ++r) {
        const int qrow = qw0 + mfma_d_row(lane, r);
        const float l_r = shfl32(mylse[sp], mfma_d_row(lane, r));
        const float d_r = shfl32(mydelta[sp], mfma_d_row(lane, r));
        const int kcol = k0 + (lane & 31);
        float pv = (kcol > qrow) ? 0.f : __expf(s[r] * scale - l_r);
        ds[r] = pv * (dp[r] - d_r) * scale;
      }
      // transpose dS through per-wave LDS -> A-frags A[q = lane&31][k]
      u16* dsl = ldsDS + w * 32 * 32;
#pragma unroll
      for (int r = 0; r < 16; ++r) {
        const int row = mfma_d_row(lane, r);  // q-local
        *(u16*)((char*)(dsl + row * 32) + swz_tr(row, (lane & 31) * 2)) = f2b(ds[r]);
      }
      __builtin_amdgcn_s_waitcnt(0);
#pragma unroll
      for (int kc = 0; kc < 2; ++kc) {
        bf16x8_t a = *(const bf16x8_t*)((const char*)(dsl + (lane & 31) * 32) +
                                        swz_tr(lane & 31, (16 * kc + 8 * (lane >> 5)) * 2));
#pragma unroll
        for (int cb = 0; cb < NCB; ++cb) {
          bf16x8_t b = read_tr_frag(ldsKt, 32 * cb + (lane & 31),
                                    (16 * kc + 8 * (lane >> 5)) * 2);
          dqacc[sp][cb] = mfma_32x32x16_bf16(a, b, dqacc[sp][cb]);
        }
      }
      __builtin_amdgcn_s_waitcnt(0);  // dsl reads done before next overwrite
    }
    if (pre) {
      u16* bk = base + (buf ^ 1) * TILE;
      kst.write_rm(bk);
      vst.write(bk + 32 * C);
      kst.write_tr(bk + 2 * 32 * C);
    }
    __syncthreads();
  }

  // epilogue: LDS bounce -> wide stores (per strip)
#pragma unroll
  for (int sp = 0; sp < SPW; ++sp) {
    const int qw0 = q0 + 32 * strip[sp];
    float* ob = (float*)smem + w * 32 * 32;
    u16* dqg = dq + (bh * T + qw0) * C;
#pragma unroll
    for (int cb = 0; cb < NCB; ++cb) {
#pragma unroll
      for (int r = 0; r < 16; ++r) {
        const int row = mfma_d_row(lane, r);
        *(float*)((char*)ob + row * 128 + (((lane & 31) * 4) ^ ((row & 7) << 4))) =
            dqacc[sp][cb][r];
      }
      __builtin_amdgcn_s_waitcnt(0);
      const int row = lane & 31;
      const int c16 = 16 * (lane >> 5);
      float tmp[16];
#pragma unroll
      for (int i = 0; i < 4; ++i) {
        f32x4 t = *(const f32x4*)((char*)ob + row * 128 +
                                  (((c16 + 4 * i) * 4) ^ ((row & 7) << 4)));
        tmp[4 * i] = t[0]; tmp[4 * i + 1] = t[1];
        tmp[4 * i + 2] = t[2]; tmp[4 * i + 3] = t[3];
      }
      u16x8 o0, o1;
#pragma unroll
      for (int j = 0; j < 8; ++j) { o0[j] = f2b(tmp[j]); o1[j] = f2b(tmp[8 + j]); }
      *(u16x8*)(dqg + (long)row * C + 32 * cb + c16) = o0;
      *(u16x8*)(dqg + (long)row * C + 32 * cb + c16 + 8) = o1;
      __builtin_amdgcn_s_waitcnt(0);
    }
  }
}
