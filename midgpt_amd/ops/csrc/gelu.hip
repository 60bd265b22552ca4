// GELU (tanh approximation) forward/backward (plan K7 — reference
// jax.nn.gelu in src/model.py:30; numerics match torch's
// gelu(approximate="tanh")). Memory-bound elementwise: u16x8 vectorized
// grid-stride (guide G13), fp32 internal math, tanh via exp2 (one
// hardware transcendental per element: tanh(z) = 1 - 2/(exp2(2z*log2e)+1)).
#include "common.h"

#define GELU_C0 0.7978845608028654f   // sqrt(2/pi)
#define GELU_C1 0.044715f
#define LOG2E_G 1.4426950408889634f

DEVINL float tanh_fast(float z) {
  // bounded input (|inner| grows ~x^3): clamp to avoid exp overflow
  z = fminf(fmaxf(z, -15.f), 15.f);
  return 1.f - 2.f / (exp2f(2.f * LOG2E_G * z) + 1.f);
}

DEVINL float gelu_f(float x) {
  const float inner = GELU_C0 * (x + GELU_C1 * x * x * x);
  return 0.5f * x * (1.f + tanh_fast(inner));
}

DEVINL float dgelu_f(float x) {
  const float x2 = x * x;
  const float inner = GELU_C0 * (x + GELU_C1 * x * x2);
  const float t = tanh_fast(inner);
  const float dinner = GELU_C0 * (1.f + 3.f * GELU_C1 * x2);
  return 0.5f * (1.f + t) + 0.5f * x * (1.f - t * t) * dinner;
}

// NT: nontemporal loads/stores (1 GB streams have no reuse — keep them
// out of L2); UNROLL: independent chunks per iteration for load-latency
// overlap within a wave.
template <int NT>
__global__ void gelu_fwd_bf16(const u16* __restrict__ x, u16* __restrict__ y,
                              long n) {
  long i = ((long)blockIdx.x * blockDim.x + threadIdx.x) * 8;
  const long step = (long)gridDim.x * blockDim.x * 8;
  for (; i + 7 < n; i += step) {
    u16x8 v = NT ? __builtin_nontemporal_load((const u16x8*)(x + i))
                 : *(const u16x8*)(x + i);
    u16x8 o;
#pragma unroll
    for (int j = 0; j < 8; ++j) o[j] = f2b(gelu_f(b2f(v[j])));
    if (NT) __builtin_nontemporal_store(o, (u16x8*)(y + i));
    else *(u16x8*)(y + i) = o;
  }
}

template <int NT>
__global__ void gelu_bwd_bf16(const u16* __restrict__ dy,
                              const u16* __restrict__ x,
                              u16* __restrict__ dx, long n) {
  long i = ((long)blockIdx.x * blockDim.x + threadIdx.x) * 8;
  const long step = (long)gridDim.x * blockDim.x * 8;
  for (; i + 7 < n; i += step) {
    u16x8 g = NT ? __builtin_nontemporal_load((const u16x8*)(dy + i))
                 : *(const u16x8*)(dy + i);
    u16x8 v = NT ? __builtin_nontemporal_load((const u16x8*)(x + i))
                 : *(const u16x8*)(x + i);
    u16x8 o;
#pragma unroll
    for (int j = 0; j < 8; ++j)
      o[j] = f2b(b2f(g[j]) * dgelu_f(b2f(v[j])));
    if (NT) __builtin_nontemporal_store(o, (u16x8*)(dx + i));
    else *(u16x8*)(dx + i) = o;
  }
}
