// Common device helpers for midgpt_amd CDNA4 (gfx950) kernels.
// Wave size is 64 on CDNA; hard-coded per the MI355X programming guide.
#pragma once
#include <hip/hip_runtime.h>
#include <cstdint>

#define WAVE 64
#define DEVINL __device__ __forceinline__

using u16 = unsigned short;
using u16x2 = __attribute__((ext_vector_type(2))) unsigned short;
using u16x4 = __attribute__((ext_vector_type(4))) unsigned short;
using u16x8 = __attribute__((ext_vector_type(8))) unsigned short;
using f32x2 = __attribute__((ext_vector_type(2))) float;
using f32x4 = __attribute__((ext_vector_type(4))) float;
using f32x8 = __attribute__((ext_vector_type(8))) float;
using f32x16 = __attribute__((ext_vector_type(16))) float;
using bf16x4 = __attribute__((ext_vector_type(4))) __bf16;
using bf16x8 = __attribute__((ext_vector_type(8))) __bf16;

// ---- bf16 <-> f32 (bit-level; bf16 stored as u16) ----
DEVINL float b2f(u16 h) {
  union { unsigned u; float f; } c;
  c.u = ((unsigned)h) << 16;
  return c.f;
}
DEVINL u16 f2b(float f) {  // round-to-nearest-even
  union { float f; unsigned u; } c;
  c.f = f;
  unsigned u = c.u;
  unsigned rounded = u + 0x7fffu + ((u >> 16) & 1u);
  if ((u & 0x7fffffffu) > 0x7f800000u) return (u16)((u >> 16) | 0x0040u);  // nan
  return (u16)(rounded >> 16);
}

// ---- wave-level reductions over a lane group of width W (power of 2) ----
template <int W> DEVINL float group_sum(float x) {
#pragma unroll
  for (int o = W / 2; o > 0; o >>= 1) x += __shfl_xor(x, o, W);
  return x;
}
template <int W> DEVINL float group_max(float x) {
#pragma unroll
  for (int o = W / 2; o > 0; o >>= 1) x = fmaxf(x, __shfl_xor(x, o, W));
  return x;
}

DEVINL int lane_id() { return threadIdx.x & (WAVE - 1); }
DEVINL int wave_id() { return threadIdx.x / WAVE; }

#define HIP_CHECK_LAUNCH()                                              \
  do {                                                                  \
    hipError_t e_ = hipGetLastError();                                  \
    TORCH_CHECK(e_ == hipSuccess, "HIP launch failed: ",                \
                hipGetErrorString(e_));                                 \
  } while (0)
