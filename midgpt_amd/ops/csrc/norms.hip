// RMSNorm (plan K5) and fused QK-LayerNorm + RoPE over the packed QKV
// projection (plans K3+K4). Memory-bound rowwise kernels: one wave per row,
// vectorized bf16x8 loads (guide G13), fp32 internal math.
//
// Numerics contract: reference src/layers.py:60-75 (RMSNorm),
// src/model.py:52-53,64-69 (QK-LN eps 1e-6 + GPT-J interleaved RoPE).
#include "common.h"

// ============================================================
// RMSNorm forward: x (N, D) -> y (N, D), invrms (N,) fp32.
// Weightless in the model (block norms / ln_f); weight supported.
// One wave per row; 4 waves (256 thr) per block; grid-stride over rows.
// ============================================================
template <typename T>
DEVINL float load_as_f32(const T* p, long i);
template <> DEVINL float load_as_f32<u16>(const u16* p, long i) { return b2f(p[i]); }
template <> DEVINL float load_as_f32<float>(const float* p, long i) { return p[i]; }
template <typename T>
DEVINL void store_from_f32(T* p, long i, float v);
template <> DEVINL void store_from_f32<u16>(u16* p, long i, float v) { p[i] = f2b(v); }
template <> DEVINL void store_from_f32<float>(float* p, long i, float v) { p[i] = v; }

DEVINL void load8(const u16* p, float* out) {
  u16x8 v = *(const u16x8*)p;
#pragma unroll
  for (int j = 0; j < 8; ++j) out[j] = b2f(v[j]);
}
DEVINL void store8(u16* p, const float* in) {
  u16x8 v;
#pragma unroll
  for (int j = 0; j < 8; ++j) v[j] = f2b(in[j]);
  *(u16x8*)p = v;
}

// bf16, D % 8 == 0 (all model dims). Vectorized u16x8 loads (guide G13).
__global__ void rmsnorm_fwd_bf16(const u16* __restrict__ x, const float* __restrict__ w,
                                 u16* __restrict__ y, float* __restrict__ invrms,
                                 long N, int D, float eps) {
  const int lane = lane_id();
  const long row0 = (long)blockIdx.x * (blockDim.x / WAVE) + wave_id();
  const long rstep = (long)gridDim.x * (blockDim.x / WAVE);
  for (long row = row0; row < N; row += rstep) {
    const u16* xr = x + row * (long)D;
    float ss = 0.f;
    float buf[8];
    for (int i = lane * 8; i < D; i += WAVE * 8) {
      load8(xr + i, buf);
#pragma unroll
      for (int j = 0; j < 8; ++j) ss += buf[j] * buf[j];
    }
    ss = group_sum<WAVE>(ss);
    float r = rsqrtf(ss / D + eps);
    if (lane == 0) invrms[row] = r;
    u16* yr = y + row * (long)D;
    for (int i = lane * 8; i < D; i += WAVE * 8) {
      load8(xr + i, buf);
#pragma unroll
      for (int j = 0; j < 8; ++j) {
        buf[j] *= r;
        if (w) buf[j] *= w[i + j];
      }
      store8(yr + i, buf);
    }
  }
}

template <typename T, int VEC>
__global__ void rmsnorm_fwd_kernel(const T* __restrict__ x, const float* __restrict__ w,
                                   T* __restrict__ y, float* __restrict__ invrms,
                                   long N, int D, float eps) {
  const int lane = lane_id();
  const long row0 = (long)blockIdx.x * (blockDim.x / WAVE) + wave_id();
  const long rstep = (long)gridDim.x * (blockDim.x / WAVE);
  for (long row = row0; row < N; row += rstep) {
    const T* xr = x + row * D;
    float ss = 0.f;
    for (int i = lane * VEC; i < D; i += WAVE * VEC) {
#pragma unroll
      for (int j = 0; j < VEC; ++j) {
        float v = load_as_f32<T>(xr, i + j);
        ss += v * v;
      }
    }
    ss = group_sum<WAVE>(ss);
    float r = rsqrtf(ss / D + eps);
    if (lane == 0) invrms[row] = r;
    T* yr = y + row * D;
    for (int i = lane * VEC; i < D; i += WAVE * VEC) {
#pragma unroll
      for (int j = 0; j < VEC; ++j) {
        float v = load_as_f32<T>(xr, i + j) * r;
        if (w) v *= w[i + j];
        store_from_f32<T>(yr, i + j, v);
      }
    }
  }
}

// bf16 backward, weightless fast path (the model's norms carry no weight).
__global__ void rmsnorm_bwd_bf16(const u16* __restrict__ dy, const u16* __restrict__ x,
                                 const float* __restrict__ invrms,
                                 u16* __restrict__ dx, long N, int D) {
  const int lane = lane_id();
  const long row0 = (long)blockIdx.x * (blockDim.x / WAVE) + wave_id();
  const long rstep = (long)gridDim.x * (blockDim.x / WAVE);
  for (long row = row0; row < N; row += rstep) {
    const u16* dyr = dy + row * (long)D;
    const u16* xr = x + row * (long)D;
    const float r = invrms[row];
    float dot = 0.f;
    float gb[8], xb[8];
    for (int i = lane * 8; i < D; i += WAVE * 8) {
      load8(dyr + i, gb);
      load8(xr + i, xb);
#pragma unroll
      for (int j = 0; j < 8; ++j) dot += gb[j] * xb[j];
    }
    dot = group_sum<WAVE>(dot) / D;
    const float r3dot = r * r * r * dot;
    u16* dxr = dx + row * (long)D;
    for (int i = lane * 8; i < D; i += WAVE * 8) {
      load8(dyr + i, gb);
      load8(xr + i, xb);
#pragma unroll
      for (int j = 0; j < 8; ++j) gb[j] = r * gb[j] - xb[j] * r3dot;
      store8(dxr + i, gb);
    }
  }
}

// ============================================================
// Fused QK-LayerNorm + RoPE over packed QKV (K3+K4).
// qkv: (B, T, 3, H, C) bf16 -> q,k (B,H,T,C) LN'd + RoPE'd, v transposed.
// stats: (B,H,T,2) fp32 {mean, invstd} for q and k.
// Vectorized: C/8 lanes per row, u16x8 loads (8 elems = 4 interleaved RoPE
// pairs per lane, so the GPT-J rotation stays lane-local); a wave covers
// 64/(C/8) rows; reductions via width-(C/8) shuffles.
// ============================================================
__global__ void qkv_prep_fwd_kernel(const u16* __restrict__ qkv,
                                    const float* __restrict__ qw,
                                    const float* __restrict__ kw,
                                    const float* __restrict__ sin_t,
                                    const float* __restrict__ cos_t,
                                    u16* __restrict__ q, u16* __restrict__ k,
                                    u16* __restrict__ v,
                                    float* __restrict__ qstats,
                                    float* __restrict__ kstats,
                                    int B, int T, int H, int C, float eps) {
  const int LPR = C / 8;                 // lanes per row (8 or 16)
  const int RPW = WAVE / LPR;            // rows per wave
  const int sub = lane_id() % LPR;       // lane's slot within its row
  const int rsub = lane_id() / LPR;      // which row of the wave
  const unsigned nrows = (unsigned)(B * T * H * 3);
  const unsigned row0 = ((unsigned)blockIdx.x * (blockDim.x / WAVE) + wave_id()) * RPW + rsub;
  const unsigned rstep = (unsigned)gridDim.x * (blockDim.x / WAVE) * RPW;
  for (unsigned row = row0; row < nrows; row += rstep) {
    const unsigned h = row % (unsigned)H;
    const unsigned rest = row / (unsigned)H;
    const unsigned role = rest % 3u;
    const unsigned bt = rest / 3u;           // = b*T + t
    const unsigned t = bt % (unsigned)T;
    const unsigned b = bt / (unsigned)T;
    const u16* src = qkv + (((long)bt * 3 + role) * H + h) * C + sub * 8;
    const long out_off = ((((long)b * H + h) * T + t) * C) + sub * 8;
    u16x8 raw = *(const u16x8*)src;
    if (role == 2) {  // V: straight transpose copy
      *(u16x8*)(v + out_off) = raw;
      continue;
    }
    const float* w = role == 0 ? qw : kw;
    u16* dst = role == 0 ? q : k;
    float* stats = role == 0 ? qstats : kstats;
    float x[8];
#pragma unroll
    for (int j = 0; j < 8; ++j) x[j] = b2f(raw[j]);
    float s = 0.f;
#pragma unroll
    for (int j = 0; j < 8; ++j) s += x[j];
    // width-LPR reduction within the row's lane group
    for (int o = LPR / 2; o > 0; o >>= 1) s += __shfl_xor(s, o, LPR);
    const float mu = s / C;
    float ss = 0.f;
#pragma unroll
    for (int j = 0; j < 8; ++j) { x[j] -= mu; ss += x[j] * x[j]; }
    for (int o = LPR / 2; o > 0; o >>= 1) ss += __shfl_xor(ss, o, LPR);
    const float invstd = rsqrtf(ss / C + eps);
    if (sub == 0) {
      stats[2 * (((long)b * H + h) * T + t)] = mu;
      stats[2 * (((long)b * H + h) * T + t) + 1] = invstd;
    }
    // normalize + weight + RoPE (pairs p = 4*sub + 0..3)
    const float* srow = sin_t + t * (C / 2) + sub * 4;
    const float* crow = cos_t + t * (C / 2) + sub * 4;
    f32x4 sn = *(const f32x4*)srow;
    f32x4 cs = *(const f32x4*)crow;
    u16x8 out;
#pragma unroll
    for (int p = 0; p < 4; ++p) {
      float n0 = x[2 * p] * invstd * w[sub * 8 + 2 * p];
      float n1 = x[2 * p + 1] * invstd * w[sub * 8 + 2 * p + 1];
      out[2 * p] = f2b(n0 * cs[p] - n1 * sn[p]);
      out[2 * p + 1] = f2b(n1 * cs[p] + n0 * sn[p]);
    }
    *(u16x8*)(dst + out_off) = out;
  }
}

// Backward: dq,dk,dv (B,H,T,C) -> dqkv (B,T,3,H,C); dqw,dkw via partials.
__global__ void qkv_prep_bwd_kernel(const u16* __restrict__ dq,
                                    const u16* __restrict__ dk,
                                    const u16* __restrict__ dv,
                                    const u16* __restrict__ qkv,
                                    const float* __restrict__ qw,
                                    const float* __restrict__ kw,
                                    const float* __restrict__ sin_t,
                                    const float* __restrict__ cos_t,
                                    const float* __restrict__ qstats,
                                    const float* __restrict__ kstats,
                                    u16* __restrict__ dqkv,
                                    float* __restrict__ dqw_partial,
                                    float* __restrict__ dkw_partial,
                                    int B, int T, int H, int C) {
  const int LPR = C / 8;
  const int RPW = WAVE / LPR;
  const int sub = lane_id() % LPR;
  const int rsub = lane_id() / LPR;
  const unsigned nrows = (unsigned)(B * T * H * 3);
  const unsigned row0 = ((unsigned)blockIdx.x * (blockDim.x / WAVE) + wave_id()) * RPW + rsub;
  const unsigned rstep = (unsigned)gridDim.x * (blockDim.x / WAVE) * RPW;
  extern __shared__ __attribute__((aligned(16))) char smem_raw[];
  float* smem = (float*)smem_raw;  // [2][C] partial dqw / dkw
  for (int i = threadIdx.x; i < 2 * C; i += blockDim.x) smem[i] = 0.f;
  __syncthreads();
  // per-thread dw accumulators (8 q-slots + 8 k-slots at this lane's sub
  // offset) -> ONE atomic pass at the end instead of per-row atomics
  float dwq_acc[8] = {0}, dwk_acc[8] = {0};
  for (unsigned row = row0; row < nrows; row += rstep) {
    const unsigned h = row % (unsigned)H;
    const unsigned rest = row / (unsigned)H;
    const unsigned role = rest % 3u;
    const unsigned bt = rest / 3u;           // = b*T + t
    const unsigned t = bt % (unsigned)T;
    const unsigned b = bt / (unsigned)T;
    u16* dst = dqkv + (((long)bt * 3 + role) * H + h) * C + sub * 8;
    const long in_off = ((((long)b * H + h) * T + t) * C) + sub * 8;
    if (role == 2) {
      *(u16x8*)dst = *(const u16x8*)(dv + in_off);
      continue;
    }
    const float* w = role == 0 ? qw : kw;
    const u16* dyp = role == 0 ? dq : dk;
    const float* stats = role == 0 ? qstats : kstats;
    float* dw_acc = role == 0 ? dwq_acc : dwk_acc;
    const float mu = stats[2 * (((long)b * H + h) * T + t)];
    const float invstd = stats[2 * (((long)b * H + h) * T + t) + 1];
    u16x8 rawdy = *(const u16x8*)(dyp + in_off);
    u16x8 rawx = *(const u16x8*)(qkv + (((long)bt * 3 + role) * H + h) * C + sub * 8);
    const float* srow = sin_t + t * (C / 2) + sub * 4;
    const float* crow = cos_t + t * (C / 2) + sub * 4;
    f32x4 sn = *(const f32x4*)srow;
    f32x4 cs = *(const f32x4*)crow;
    float dn[8], xh[8], g[8];
#pragma unroll
    for (int p = 0; p < 4; ++p) {
      // inverse RoPE rotation: dn = dy*cos + rot^T(dy*sin)
      float d0 = b2f(rawdy[2 * p]), d1 = b2f(rawdy[2 * p + 1]);
      dn[2 * p] = d0 * cs[p] + d1 * sn[p];
      dn[2 * p + 1] = d1 * cs[p] - d0 * sn[p];
    }
    float sum_g = 0.f, sum_gx = 0.f;
#pragma unroll
    for (int j = 0; j < 8; ++j) {
      xh[j] = (b2f(rawx[j]) - mu) * invstd;
      g[j] = dn[j] * w[sub * 8 + j];
      sum_g += g[j];
      sum_gx += g[j] * xh[j];
    }
    for (int o = LPR / 2; o > 0; o >>= 1) {
      sum_g += __shfl_xor(sum_g, o, LPR);
      sum_gx += __shfl_xor(sum_gx, o, LPR);
    }
    const float mean_g = sum_g / C, mean_gx = sum_gx / C;
    u16x8 out;
#pragma unroll
    for (int j = 0; j < 8; ++j) {
      out[j] = f2b(invstd * (g[j] - mean_g - xh[j] * mean_gx));
      dw_acc[j] += dn[j] * xh[j];
    }
    *(u16x8*)dst = out;
  }
#pragma unroll
  for (int j = 0; j < 8; ++j) {
    atomicAdd(&smem[sub * 8 + j], dwq_acc[j]);
    atomicAdd(&smem[C + sub * 8 + j], dwk_acc[j]);
  }
  __syncthreads();
  for (int i = threadIdx.x; i < C; i += blockDim.x) {
    dqw_partial[(long)blockIdx.x * C + i] = smem[i];
    dkw_partial[(long)blockIdx.x * C + i] = smem[C + i];
  }
}
