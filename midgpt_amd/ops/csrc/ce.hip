// Fused softmax cross-entropy with integer labels (plan K9).
// Reference numerics: fp32 logits -> logsumexp - logit[y], mean over rows
// (reference src/train.py:76-77). The fp32 softmax over V is never
// materialized: forward does one online (max, sumexp) pass; backward does
// one pass writing bf16 dlogits. V = 50304 (bf16 row = ~100 KB): one
// 256-thread block per row, vectorized u16x8 loads, block reduce in LDS.
#include "common.h"

// online-max merge of (m, s): s' holds sum(exp(x - m))
DEVINL void online_merge(float& m, float& s, float m2, float s2) {
  float mn = fmaxf(m, m2);
  s = s * __expf(m - mn) + s2 * __expf(m2 - mn);
  m = mn;
}

__global__ void ce_fwd_kernel(const u16* __restrict__ logits,
                              const long* __restrict__ targets,
                              float* __restrict__ lse,
                              float* __restrict__ loss_sum,
                              long N, int V) {
  const int tid = threadIdx.x;
  const int lane = lane_id();
  const int wid = wave_id();
  const int nwaves = blockDim.x / WAVE;
  __shared__ float sm[32], ssum[32];
  for (long row = blockIdx.x; row < N; row += gridDim.x) {
    const u16* lr = logits + row * V;
    float m = -1e30f, s = 0.f;
    int i = tid * 8;
    const int step = blockDim.x * 8;
    for (; i + 7 < V; i += step) {
      u16x8 v = *(const u16x8*)(lr + i);
      float mx = -1e30f;
      float x[8];
#pragma unroll
      for (int j = 0; j < 8; ++j) { x[j] = b2f(v[j]); mx = fmaxf(mx, x[j]); }
      float ls = 0.f;
#pragma unroll
      for (int j = 0; j < 8; ++j) ls += __expf(x[j] - mx);
      online_merge(m, s, mx, ls);
    }
    for (; i < V; ++i) online_merge(m, s, b2f(lr[i]), 1.f);
    // wave reduce (m, s)
#pragma unroll
    for (int o = WAVE / 2; o > 0; o >>= 1)
      online_merge(m, s, __shfl_xor(m, o), __shfl_xor(s, o));
    if (lane == 0) { sm[wid] = m; ssum[wid] = s; }
    __syncthreads();
    if (tid == 0) {
      float M = sm[0], S = ssum[0];
      for (int w = 1; w < nwaves; ++w) online_merge(M, S, sm[w], ssum[w]);
      float l = M + __logf(S);
      lse[row] = l;
      atomicAdd(loss_sum, l - b2f(lr[targets[row]]));
    }
    __syncthreads();
  }
}

// dlogits[i,j] = (exp(l_ij - lse_i) - [j == y_i]) * gscale
__global__ void ce_bwd_kernel(const u16* __restrict__ logits,
                              const long* __restrict__ targets,
                              const float* __restrict__ lse,
                              const float* __restrict__ gscale,  // dloss / N
                              u16* __restrict__ dlogits,
                              long N, int V) {
  const int tid = threadIdx.x;
  const float gs = gscale[0];
  for (long row = blockIdx.x; row < N; row += gridDim.x) {
    const u16* lr = logits + row * V;
    u16* dr = dlogits + row * V;
    const float l = lse[row];
    const long y = targets[row];
    int i = tid * 8;
    const int step = blockDim.x * 8;
    for (; i + 7 < V; i += step) {
      u16x8 v = *(const u16x8*)(lr + i);
      u16x8 o;
#pragma unroll
      for (int j = 0; j < 8; ++j) {
        float p = __expf(b2f(v[j]) - l);
        if (i + j == y) p -= 1.f;
        o[j] = f2b(p * gs);
      }
      *(u16x8*)(dr + i) = o;
    }
    for (; i < V; ++i) {
      float p = __expf(b2f(lr[i]) - l);
      if (i == y) p -= 1.f;
      dr[i] = f2b(p * gs);
    }
  }
}
