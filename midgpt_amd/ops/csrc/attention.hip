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
// Geometry (fwd): one workgroup = 4 waves = 128 q rows (32/wave);
// KV tiles of 32 rows double-buffered in LDS; grid = B*H*(T/128).
// Geometry (bwd): one workgroup = 4 waves = 128 k rows; iterates q tiles;
// dK/dV accumulate in registers, dQ via fp32 atomics (v1; split-q kernel
// is the planned v2).
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
// Cooperative staging helpers (256 threads)
// ---------------------------------------------------------------------------
// Row-major 32 x C tile from global (row stride C) into swizzled LDS.
template <int C, int NT>
DEVINL void stage_rm(const u16* __restrict__ g, u16* lds) {
#pragma unroll
  for (int idx = threadIdx.x * 8; idx < 32 * C; idx += NT * 8) {
    const int row = idx / C, col = idx % C;
    u16x8 val = *(const u16x8*)(g + row * C + col);
    *(u16x8*)((char*)lds + row * C * 2 + swz_rm<C>(row, col * 2)) = val;
  }
}
// Transposed: global 32 x C (row stride C) -> LDS [C][32] swizzled.
// Each thread loads TWO consecutive kv/q rows (u16x8 each) and writes
// 8 ds_write_b32 pairs — consecutive kv are contiguous in the transposed
// row, halving the LDS write instruction count vs scalar b16 stores.
template <int C, int NT>
DEVINL void stage_tr(const u16* __restrict__ g, u16* lds) {
#pragma unroll
  for (int idx = threadIdx.x * 16; idx < 32 * C; idx += NT * 16) {
    const int pair = idx / (2 * C);          // kv pair index (rows 2p, 2p+1)
    const int col = (idx / 2) % C;           // c base (8 wide)
    const int row = 2 * pair;
    u16x8 v0 = *(const u16x8*)(g + (long)row * C + col);
    u16x8 v1 = *(const u16x8*)(g + (long)(row + 1) * C + col);
#pragma unroll
    for (int j = 0; j < 8; ++j) {
      const int r = col + j;  // LDS row = c
      u16x2 pairv; pairv.x = v0[j]; pairv.y = v1[j];
      *(u16x2*)((char*)lds + r * 64 + swz_tr(r, row * 2)) = pairv;
    }
  }
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
// Forward
// ===========================================================================
template <int C, int NW>
__global__ __launch_bounds__(NW * 64, 2) void attn_fwd_kernel(const u16* __restrict__ q,
                                const u16* __restrict__ k,
                                const u16* __restrict__ v,
                                u16* __restrict__ o, float* __restrict__ lse,
                                int B, int H, int T) {
  constexpr int NCB = C / 32;   // 32-col c-blocks
  constexpr int NCH = C / 16;   // 16-deep mfma chunks
  const float scale = rsqrtf((float)C);
  // qb-outermost grid order: all q-blocks of one (b,h) land on the same
  // XCD (b%8 dispatch) for K/V L2 reuse (guide T1).
  const long bh = blockIdx.x % ((long)B * H);
  const int qb = blockIdx.x / (B * H);
  const int q0 = qb * (NW * 32);
  const int lane = lane_id();
  const int w = wave_id();
  const int qw0 = q0 + 32 * w;          // this wave's first q row
  const int myq = qw0 + (lane & 31);    // this lane's q row

  extern __shared__ __attribute__((aligned(16))) char smem[];
  u16* ldsK = (u16*)smem;                       // [2][32*C]
  u16* ldsVt = (u16*)(smem + 2 * 32 * C * 2);   // [2][C*32]
  float* obuf = (float*)smem;                   // epilogue reuse: [4][32*32]

  const u16* qg = q + (bh * T) * C;
  const u16* kg = k + (bh * T) * C;
  const u16* vg = v + (bh * T) * C;

  // Q B-fragments straight from global to registers.
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

  const int nkt = (q0 + NW * 32) / 32;
  stage_rm<C, NW * 64>(kg, ldsK);
  stage_tr<C, NW * 64>(vg, ldsVt);
  __syncthreads();

  for (int kt = 0; kt < nkt; ++kt) {
    const int buf = kt & 1;
    if (kt + 1 < nkt) {
      stage_rm<C, NW * 64>(kg + (long)(kt + 1) * 32 * C, ldsK + (1 - buf) * 32 * C);
      stage_tr<C, NW * 64>(vg + (long)(kt + 1) * 32 * C, ldsVt + (1 - buf) * C * 32);
    }
    const int k0 = kt * 32;
    if (k0 <= qw0 + 31) {  // wave-uniform: tile not fully masked for this wave
      // S = K x Q^T  (swapped: D rows = k, cols = q)
      f32x16 s = (f32x16)(0.f);
      const u16* kb = ldsK + buf * 32 * C;
#pragma unroll
      for (int ch = 0; ch < NCH; ++ch) {
        bf16x8_t a = read_rm_frag<C>(kb, lane & 31, 16 * ch * 2 + 16 * (lane >> 5));
        s = mfma_32x32x16_bf16(a, qf[ch], s);
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
      const float mn = fmaxf(m, mt);
      const float alpha = __expf((m - mn) * scale);
      m = mn;
      float p[16], psum = 0.f;
#pragma unroll
      for (int r = 0; r < 16; ++r) {
        p[r] = __expf((sv[r] - mn) * scale);
        psum += p[r];
      }
      psum += __shfl_xor(psum, 32);
      lsum = lsum * alpha + psum;
      // rescale O by alpha[q_of_reg]
      float arow[16];
#pragma unroll
      for (int r = 0; r < 16; ++r) arow[r] = shfl32(alpha, mfma_d_row(lane, r));
#pragma unroll
      for (int cb = 0; cb < NCB; ++cb)
#pragma unroll
        for (int r = 0; r < 16; ++r) oacc[cb][r] *= arow[r];
      // P -> bf16 A-fragments; PV
      bf16x8_t pf0 = dlayout_to_afrag(p);
      bf16x8_t pf1 = dlayout_to_afrag(p + 8);
      const u16* vb = ldsVt + buf * C * 32;
#pragma unroll
      for (int cb = 0; cb < NCB; ++cb) {
        bf16x8_t b0 = read_tr_frag(vb, 32 * cb + (lane & 31), 16 * (lane >> 5));
        bf16x8_t b1 = read_tr_frag(vb, 32 * cb + (lane & 31), 32 + 16 * (lane >> 5));
        oacc[cb] = mfma_32x32x16_bf16(pf0, b0, oacc[cb]);
        oacc[cb] = mfma_32x32x16_bf16(pf1, b1, oacc[cb]);
      }
    }
    __syncthreads();
  }

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
template <int C, int NW>
__global__ __launch_bounds__(NW * 64, 2) void attn_bwd_dkv_kernel(
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

  // wave-owned K row fragments; V re-read per tile (register budget)
  bf16x8_t kf[NCH];
#pragma unroll
  for (int ch = 0; ch < NCH; ++ch)
    kf[ch] = *(const bf16x8_t*)(krow + 16 * ch + 8 * (lane >> 5));

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
  for (int qt = qt0; qt < nqt; ++qt) {
    const int buf = (qt - qt0) & 1;
    if (qt + 1 < nqt) stage_set(qt + 1, buf ^ 1);
    const int qbase = qt * 32;
    if (qbase + 31 >= kw0) {  // not fully masked for this wave
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
        s = mfma_32x32x16_bf16(a, kf[ch], s);
      }
      float p[16];
#pragma unroll
      for (int r = 0; r < 16; ++r) {
        const int qrow = qbase + mfma_d_row(lane, r);
        p[r] = (myk > qrow) ? 0.f
             : __expf(s[r] * scale - lseb[mfma_d_row(lane, r)]);
      }
      // dV += P^T x dO
      bf16x8_t pf0 = dlayout_to_afrag(p);
      bf16x8_t pf1 = dlayout_to_afrag(p + 8);
#pragma unroll
      for (int cb = 0; cb < NCB; ++cb) {
        bf16x8_t b0 = read_tr_frag(ldsDOt, 32 * cb + (lane & 31), 16 * (lane >> 5));
        bf16x8_t b1 = read_tr_frag(ldsDOt, 32 * cb + (lane & 31), 32 + 16 * (lane >> 5));
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
      bf16x8_t df0 = dlayout_to_afrag(ds);
      bf16x8_t df1 = dlayout_to_afrag(ds + 8);
#pragma unroll
      for (int cb = 0; cb < NCB; ++cb) {
        bf16x8_t b0 = read_tr_frag(ldsQt, 32 * cb + (lane & 31), 16 * (lane >> 5));
        bf16x8_t b1 = read_tr_frag(ldsQt, 32 * cb + (lane & 31), 32 + 16 * (lane >> 5));
        dkacc[cb] = mfma_32x32x16_bf16(df0, b0, dkacc[cb]);
        dkacc[cb] = mfma_32x32x16_bf16(df1, b1, dkacc[cb]);
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
// Backward, kernel B (dQ): one WG = NW*32 q rows; iterates k tiles up to
// its diagonal, DOUBLE-BUFFERED. dQ accumulates in registers — no atomics.
// ===========================================================================
template <int C, int NW>
__global__ __launch_bounds__(NW * 64, 2) void attn_bwd_dq_kernel(
    const u16* __restrict__ dO, const u16* __restrict__ q,
    const u16* __restrict__ k, const u16* __restrict__ v,
    const float* __restrict__ lse, const float* __restrict__ delta,
    u16* __restrict__ dq, int B, int H, int T) {
  constexpr int NCB = C / 32;
  constexpr int NCH = C / 16;
  constexpr int TILE = 3 * 32 * C;  // u16 elems per buffer set (K, V, Kt)
  const float scale = rsqrtf((float)C);
  const long bh = blockIdx.x % ((long)B * H);
  const int qb = blockIdx.x / (B * H);
  const int q0 = qb * (NW * 32);
  const int lane = lane_id();
  const int w = wave_id();
  const int qw0 = q0 + 32 * w;
  const int myq = qw0 + (lane & 31);

  extern __shared__ __attribute__((aligned(16))) char smem[];
  u16* base = (u16*)smem;               // [2][TILE] = K rm | V rm | Kt
  u16* ldsDS = base + 2 * TILE;         // per wave + w*32*32 (bf16)

  const u16* qg = q + (bh * T) * C;
  const u16* kg = k + (bh * T) * C;
  const u16* vg = v + (bh * T) * C;
  const u16* dog = dO + (bh * T) * C;

  // per-lane Q A-fragments; dO fragments re-read per tile (register budget)
  bf16x8_t qf[NCH];
  const u16* dorow = dog + (long)myq * C;
  {
    const u16* qrow = qg + (long)myq * C;
#pragma unroll
    for (int ch = 0; ch < NCH; ++ch)
      qf[ch] = *(const bf16x8_t*)(qrow + 16 * ch + 8 * (lane >> 5));
  }
  const float mylse = lse[bh * T + myq];
  const float mydelta = delta[bh * T + myq];

  f32x16 dqacc[NCB];
#pragma unroll
  for (int cb = 0; cb < NCB; ++cb) dqacc[cb] = (f32x16)(0.f);

  const int nkt = (q0 + NW * 32) / 32;
  auto stage_set = [&](int kt, int buf) {
    const long k0 = (long)kt * 32;
    u16* bk = base + buf * TILE;
    stage_rm<C, NW * 64>(kg + k0 * C, bk);
    stage_rm<C, NW * 64>(vg + k0 * C, bk + 32 * C);
    stage_tr<C, NW * 64>(kg + k0 * C, bk + 2 * 32 * C);
  };
  stage_set(0, 0);
  __syncthreads();
  for (int kt = 0; kt < nkt; ++kt) {
    const int buf = kt & 1;
    if (kt + 1 < nkt) stage_set(kt + 1, buf ^ 1);
    const int k0 = kt * 32;
    if (k0 <= qw0 + 31) {
      const u16* ldsK = base + buf * TILE;
      const u16* ldsV = ldsK + 32 * C;
      const u16* ldsKt = ldsV + 32 * C;
      // S = Q x K^T and dP = dO x V^T in one pass.
      // A = per-lane Q/dO fragments (A[q=lane&31][c]), B = K/V rows from LDS
      // (B[c][k=lane&31]); D rows = q (reg-mapped), cols = k.
      f32x16 s = (f32x16)(0.f);
      f32x16 dp = (f32x16)(0.f);
#pragma unroll
      for (int ch = 0; ch < NCH; ++ch) {
        bf16x8_t kfrag = read_rm_frag<C>(ldsK, lane & 31, 16 * ch * 2 + 16 * (lane >> 5));
        s = mfma_32x32x16_bf16(qf[ch], kfrag, s);
        bf16x8_t vfrag = read_rm_frag<C>(ldsV, lane & 31, 16 * ch * 2 + 16 * (lane >> 5));
        bf16x8_t dof = *(const bf16x8_t*)(dorow + 16 * ch + 8 * (lane >> 5));
        dp = mfma_32x32x16_bf16(dof, vfrag, dp);
      }
      // rows q are reg-mapped; cols k = lane&31. lse/delta per q via shfl.
      float ds[16];
#pragma unroll
      for (int r = 0; r < 16; ++r) {
        const int qrow = qw0 + mfma_d_row(lane, r);
        const float l_r = shfl32(mylse, mfma_d_row(lane, r));
        const float d_r = shfl32(mydelta, mfma_d_row(lane, r));
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
          dqacc[cb] = mfma_32x32x16_bf16(a, b, dqacc[cb]);
        }
      }
      __builtin_amdgcn_s_waitcnt(0);  // dsl reads done before next overwrite
    }
    __syncthreads();
  }

  // epilogue: LDS bounce -> wide stores
  float* ob = (float*)smem + w * 32 * 32;
  u16* dqg = dq + (bh * T + qw0) * C;
#pragma unroll
  for (int cb = 0; cb < NCB; ++cb) {
#pragma unroll
    for (int r = 0; r < 16; ++r) {
      const int row = mfma_d_row(lane, r);
      *(float*)((char*)ob + row * 128 + (((lane & 31) * 4) ^ ((row & 7) << 4))) =
          dqacc[cb][r];
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
