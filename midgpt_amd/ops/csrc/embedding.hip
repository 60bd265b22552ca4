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
