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

// dx = r*g - x * r^3 * mean(x*g), g = dy*w
template <typename T, int VEC>
__global__ void rmsnorm_bwd_kernel(const T* __restrict__ dy, const T* __restrict__ x,
                                   const float* __restrict__ w,
                                   const float* __restrict__ invrms,
                                   T* __restrict__ dx, float* __restrict__ dw_partial,
                                   long N, int D) {
  const int lane = lane_id();
  const long row0 = (long)blockIdx.x * (blockDim.x / WAVE) + wave_id();
  const long rstep = (long)gridDim.x * (blockDim.x / WAVE);
  extern __shared__ __attribute__((aligned(16))) char smem_raw[];
  float* smem = (float*)smem_raw;  // dw partials: one (D,) slab per block
  if (dw_partial) {
    for (int i = threadIdx.x; i < D; i += blockDim.x) smem[i] = 0.f;
    __syncthreads();
  }
  for (long row = row0; row < N; row += rstep) {
    const T* dyr = dy + row * D;
    const T* xr = x + row * D;
    const float r = invrms[row];
    float dot = 0.f;
    for (int i = lane * VEC; i < D; i += WAVE * VEC) {
#pragma unroll
      for (int j = 0; j < VEC; ++j) {
        float g = load_as_f32<T>(dyr, i + j);
        if (w) g *= w[i + j];
        dot += load_as_f32<T>(xr, i + j) * g;
      }
    }
    dot = group_sum<WAVE>(dot) / D;
    T* dxr = dx + row * D;
    for (int i = lane * VEC; i < D; i += WAVE * VEC) {
#pragma unroll
      for (int j = 0; j < VEC; ++j) {
        float g = load_as_f32<T>(dyr, i + j);
        float xv = load_as_f32<T>(xr, i + j);
        float gw = w ? g * w[i + j] : g;
        store_from_f32<T>(dxr, i + j, r * gw - xv * r * r * r * dot);
        if (dw_partial) atomicAdd(&smem[i + j], g * xv * r);
      }
    }
  }
  if (dw_partial) {
    __syncthreads();
    for (int i = threadIdx.x; i < D; i += blockDim.x)
      dw_partial[(long)blockIdx.x * D + i] = smem[i];
  }
}

// ============================================================
// Fused QK-LayerNorm + RoPE over packed QKV (K3+K4).
// qkv: (B, T, 3, H, C) bf16 -> q,k (B,H,T,C) LN'd + RoPE'd, v transposed.
// stats: (B,H,T,2) fp32 {mean, invstd} for q and k.
// One wave per (b,t,h,role) row; lane i < C/2 owns the interleaved RoPE
// pair (2i, 2i+1) so the rotation is lane-local.
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
  const int lane = lane_id();
  const long nrows = (long)B * T * H * 3;
  const long row0 = (long)blockIdx.x * (blockDim.x / WAVE) + wave_id();
  const long rstep = (long)gridDim.x * (blockDim.x / WAVE);
  for (long row = row0; row < nrows; row += rstep) {
    // row index order: (b, t, role, h) — matches qkv memory layout
    const int h = row % H;
    const int role = (row / H) % 3;
    const long t = (row / ((long)3 * H)) % T;
    const long b = row / ((long)3 * H * T);
    const u16* src = qkv + ((((b * T + t) * 3 + role) * H + h) * C);
    // output (B,H,T,C)
    const long out_off = (((b * H + h) * T + t) * C);
    if (role == 2) {  // V: straight transpose copy, 2 bf16/lane (C<=128)
      for (int i = lane * 2; i + 1 < C; i += WAVE * 2) {
        *(u16x2*)(v + out_off + i) = *(const u16x2*)(src + i);
      }
      if (C & 1) { if (lane == 0) v[out_off + C - 1] = src[C - 1]; }
      continue;
    }
    // Q or K: LayerNorm over C (weight, no bias) then RoPE.
    const float* w = role == 0 ? qw : kw;
    u16* dst = role == 0 ? q : k;
    float* stats = role == 0 ? qstats : kstats;
    const int P = C / 2;  // pairs
    float x0 = 0.f, x1 = 0.f;
    if (lane < P) {
      x0 = b2f(src[2 * lane]);
      x1 = b2f(src[2 * lane + 1]);
    }
    float s = group_sum<WAVE>(x0 + x1);
    float mu = s / C;
    float d0 = lane < P ? x0 - mu : 0.f, d1 = lane < P ? x1 - mu : 0.f;
    float ss = group_sum<WAVE>(d0 * d0 + d1 * d1);
    float invstd = rsqrtf(ss / C + eps);
    if (lane == 0) {
      stats[2 * ((b * H + h) * T + t)] = mu;
      stats[2 * ((b * H + h) * T + t) + 1] = invstd;
    }
    if (lane < P) {
      float n0 = d0 * invstd * w[2 * lane];
      float n1 = d1 * invstd * w[2 * lane + 1];
      float sn = sin_t[t * P + lane];
      float cs = cos_t[t * P + lane];
      // out = x*cos + rotate_every_two(x)*sin ; rot([a,b]) = [-b, a]
      float o0 = n0 * cs - n1 * sn;
      float o1 = n1 * cs + n0 * sn;
      u16x2 o; o.x = f2b(o0); o.y = f2b(o1);
      *(u16x2*)(dst + out_off + 2 * lane) = o;
    }
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
  const int lane = lane_id();
  const long nrows = (long)B * T * H * 3;
  const long row0 = (long)blockIdx.x * (blockDim.x / WAVE) + wave_id();
  const long rstep = (long)gridDim.x * (blockDim.x / WAVE);
  extern __shared__ __attribute__((aligned(16))) char smem_raw[];
  float* smem = (float*)smem_raw;  // [2][C] partial dqw / dkw
  for (int i = threadIdx.x; i < 2 * C; i += blockDim.x) smem[i] = 0.f;
  __syncthreads();
  for (long row = row0; row < nrows; row += rstep) {
    const int h = row % H;
    const int role = (row / H) % 3;
    const long t = (row / ((long)3 * H)) % T;
    const long b = row / ((long)3 * H * T);
    u16* dst = dqkv + ((((b * T + t) * 3 + role) * H + h) * C);
    const long in_off = (((b * H + h) * T + t) * C);
    if (role == 2) {
      for (int i = lane * 2; i + 1 < C; i += WAVE * 2)
        *(u16x2*)(dst + i) = *(const u16x2*)(dv + in_off + i);
      continue;
    }
    const float* w = role == 0 ? qw : kw;
    const u16* dyp = role == 0 ? dq : dk;
    const float* stats = role == 0 ? qstats : kstats;
    float* dwp = role == 0 ? smem : smem + C;
    const int P = C / 2;
    float g0 = 0.f, g1 = 0.f, xh0 = 0.f, xh1 = 0.f, dn0 = 0.f, dn1 = 0.f;
    const float mu = stats[2 * ((b * H + h) * T + t)];
    const float invstd = stats[2 * ((b * H + h) * T + t) + 1];
    if (lane < P) {
      // inverse RoPE rotation: dn = dy*cos + rot^T(dy*sin)
      float d0 = b2f(dyp[in_off + 2 * lane]);
      float d1 = b2f(dyp[in_off + 2 * lane + 1]);
      float sn = sin_t[t * P + lane];
      float cs = cos_t[t * P + lane];
      dn0 = d0 * cs + d1 * sn;
      dn1 = d1 * cs - d0 * sn;
      const u16* src = qkv + ((((b * T + t) * 3 + role) * H + h) * C);
      xh0 = (b2f(src[2 * lane]) - mu) * invstd;
      xh1 = (b2f(src[2 * lane + 1]) - mu) * invstd;
      g0 = dn0 * w[2 * lane];
      g1 = dn1 * w[2 * lane + 1];
    }
    float mean_g = group_sum<WAVE>(g0 + g1) / C;
    float mean_gx = group_sum<WAVE>(g0 * xh0 + g1 * xh1) / C;
    if (lane < P) {
      float dx0 = invstd * (g0 - mean_g - xh0 * mean_gx);
      float dx1 = invstd * (g1 - mean_g - xh1 * mean_gx);
      u16x2 o; o.x = f2b(dx0); o.y = f2b(dx1);
      *(u16x2*)(dst + 2 * lane) = o;
      atomicAdd(&dwp[2 * lane], dn0 * xh0);
      atomicAdd(&dwp[2 * lane + 1], dn1 * xh1);
    }
  }
  __syncthreads();
  for (int i = threadIdx.x; i < C; i += blockDim.x) {
    dqw_partial[(long)blockIdx.x * C + i] = smem[i];
    dkw_partial[(long)blockIdx.x * C + i] = smem[C + i];
  }
}
