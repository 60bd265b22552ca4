// Embedding backward: scatter-add of dY rows into dW (plan K8).
// Counterpart of the reference's Embedding bwd (reference src/layers.py:13-34,
// jnp.take fwd -> scatter-add bwd synthesized by XLA).
//
// Design: fp32 accumulator image of dW in HBM, one 256-thread block per
// group of rows (grid-stride), per-element atomicAdd. Token repeats at
// GPT batch sizes are modest (N ~ 64K rows over V ~ 50K vocab), so L2
// atomic contention is low; the fp32 image keeps accumulation exact
// regardless of index collisions, then one vectorized pass converts to
// bf16. dY rows are read as u16x8 (guide G13: vectorize bf16 loads).
#include "common.h"

__global__ void embed_bwd_scatter_kernel(const u16* __restrict__ dy,
                                         const long* __restrict__ idx,
                                         float* __restrict__ dw32,
                                         long N, int D) {
  // thread t of the block covers elements [8t, 8t+8) of each row
  const int t = threadIdx.x;
  for (long row = blockIdx.x; row < N; row += gridDim.x) {
    const long v = idx[row];
    const u16* src = dy + row * D;
    float* dst = dw32 + v * D;
    for (int i = t * 8; i + 7 < D; i += blockDim.x * 8) {
      u16x8 h = *(const u16x8*)(src + i);
#pragma unroll
      for (int j = 0; j < 8; ++j) atomicAdd(dst + i + j, b2f(h[j]));
    }
  }
}

// dw32 (V*D fp32) -> out bf16, grid-stride vectorized
__global__ void f32_to_bf16_kernel(const float* __restrict__ src,
                                   u16* __restrict__ dst, long n) {
  long i = ((long)blockIdx.x * blockDim.x + threadIdx.x) * 4;
  const long step = (long)gridDim.x * blockDim.x * 4;
  for (; i + 3 < n; i += step) {
    f32x4 v = *(const f32x4*)(src + i);
    u16x4 o;
#pragma unroll
    for (int j = 0; j < 4; ++j) o[j] = f2b(v[j]);
    *(u16x4*)(dst + i) = o;
  }
}


// ---------------------------------------------------------------------
// Sort-based variant (default): host sorts the token ids; one WAVE per
// run of equal tokens accumulates the permuted dY rows in fp32
// REGISTERS and writes the bf16 dW row once — no atomics, no fp32
// image, no conversion pass (the atomic variant above stays as the
// MIDGPT_EMBED_ATOMIC=1 fallback). Lane l owns elements
// [l*c, l*c+c) of the row (c = D/64 <= 32 per pass; larger D loops).
// ---------------------------------------------------------------------
__global__ void embed_bwd_sorted_kernel(const u16* __restrict__ dy,
                                        const long* __restrict__ sorted,
                                        const long* __restrict__ perm,
                                        u16* __restrict__ dw,
                                        long N, int D, int d0) {
  const long p = (long)blockIdx.x * (blockDim.x / WAVE) + wave_id();
  if (p >= N) return;
  const long tok = sorted[p];
  if (p > 0 && sorted[p - 1] == tok) return;  // not a run start
  const int lane = lane_id();
  const int dpass = min(D - d0, 2048);
  const int c = dpass / WAVE;                 // fp32 per lane this pass
  float acc[32];
#pragma unroll
  for (int j = 0; j < 32; ++j) acc[j] = 0.f;
  for (long q = p; q < N && sorted[q] == tok; ++q) {
    const u16* row = dy + perm[q] * (long)D + d0 + lane * c;
    int j = 0;
    for (; j + 7 < c; j += 8) {
      u16x8 v = *(const u16x8*)(row + j);
#pragma unroll
      for (int t = 0; t < 8; ++t) acc[j + t] += b2f(v[t]);
    }
    for (; j < c; ++j) acc[j] += b2f(row[j]);  // c %% 8 tail (e.g. D=768)
  }
  u16* out = dw + tok * (long)D + d0 + lane * c;
  int j = 0;
  for (; j + 7 < c; j += 8) {
    u16x8 o;
#pragma unroll
    for (int t = 0; t < 8; ++t) o[t] = f2b(acc[j + t]);
    *(u16x8*)(out + j) = o;
  }
  for (; j < c; ++j) out[j] = f2b(acc[j]);
}
