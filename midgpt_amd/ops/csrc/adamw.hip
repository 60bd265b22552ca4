// Fused AdamW with independent weight decay + on-device global-norm clip
// (plans K10/K13). Operates on the engine's FLAT fp32 buffers (master, grad
// accumulator, m, v) and writes the bf16 working-weight image in the same
// pass — the fp32->bf16 cast (reference src/train.py:83) costs no extra
// memory pass.
//
// Optimizer semantics = the reference optax chain (src/train.py:153-159):
//   g   <- grad * grad_scale * min(1, clip_norm / ||g*grad_scale||)
//   m   <- b1*m + (1-b1)*g ;  v <- b2*v + (1-b2)*g^2
//   upd <- mhat/(sqrt(vhat)+eps) + (wd/lr_peak)*theta
//   theta <- theta - lr_t * upd
// The clip coefficient is computed ON DEVICE from the all-reduced squared
// norm (sq_sum) so the step path never syncs to host.
#include "common.h"

__global__ void adamw_kernel(float* __restrict__ master,
                             const float* __restrict__ grad,
                             float* __restrict__ m, float* __restrict__ v,
                             u16* __restrict__ out_bf16, int write_bf16,
                             const float* __restrict__ sq_sum,
                             float lr, float b1, float b2, float eps,
                             float wd, float grad_scale, float clip_norm,
                             float bc1, float bc2, long n) {
  const float gnorm = sqrtf(sq_sum[0]) * grad_scale + 1e-12f;
  const float coef = grad_scale * fminf(1.f, clip_norm / gnorm);
  const long i0 = ((long)blockIdx.x * blockDim.x + threadIdx.x) * 4;
  const long stride = (long)gridDim.x * blockDim.x * 4;
  for (long i = i0; i + 3 < n; i += stride) {
    f32x4 g = *(const f32x4*)(grad + i);
    f32x4 mm = *(const f32x4*)(m + i);
    f32x4 vv = *(const f32x4*)(v + i);
    f32x4 th = *(const f32x4*)(master + i);
    u16x4 ob;
#pragma unroll
    for (int j = 0; j < 4; ++j) {
      float gg = g[j] * coef;
      mm[j] = b1 * mm[j] + (1.f - b1) * gg;
      vv[j] = b2 * vv[j] + (1.f - b2) * gg * gg;
      float mhat = mm[j] * bc1;
      float vhat = vv[j] * bc2;
      float upd = mhat / (sqrtf(vhat) + eps) + wd * th[j];
      th[j] -= lr * upd;
      ob[j] = f2b(th[j]);
    }
    *(f32x4*)(m + i) = mm;
    *(f32x4*)(v + i) = vv;
    *(f32x4*)(master + i) = th;
    if (write_bf16) *(u16x4*)(out_bf16 + i) = ob;
  }
  // tail (n % 4): handled by the first few threads of block 0
  if (blockIdx.x == 0 && threadIdx.x < (n & 3)) {
    const long i = (n & ~3L) + threadIdx.x;
    float gg = grad[i] * coef;
    m[i] = b1 * m[i] + (1.f - b1) * gg;
    v[i] = b2 * v[i] + (1.f - b2) * gg * gg;
    float upd = (m[i] * bc1) / (sqrtf(v[i] * bc2) + eps) + wd * master[i];
    master[i] -= lr * upd;
    if (write_bf16) out_bf16[i] = f2b(master[i]);
  }
}
