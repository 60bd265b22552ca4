// MFMA layout probes: verify on hardware the fragment-layout assumptions in
// mfma.h (tests/test_gpu_mfma.py). Asymmetric inputs catch transposes
// (guide section 3 "Always A=I-check with ASYMMETRIC B").
#include "common.h"
#include "mfma.h"

// D = A(32x16) @ B(16x32) with fragments loaded per mfma.h's layout maps.
__global__ void probe_mfma_kernel(const float* __restrict__ A,
                                  const float* __restrict__ B,
                                  float* __restrict__ D) {
  const int lane = threadIdx.x;  // 64 threads
  bf16x8_t a, b;
#pragma unroll
  for (int j = 0; j < 8; ++j) {
    // A[row = lane&31][k = 8*(lane>>5) + j]
    a[j] = (__bf16)A[(lane & 31) * 16 + 8 * (lane >> 5) + j];
    // B[k = 8*(lane>>5) + j][col = lane&31]
    b[j] = (__bf16)B[(8 * (lane >> 5) + j) * 32 + (lane & 31)];
  }
  f32x16 d = (f32x16)(0.f);
  d = mfma_32x32x16_bf16(a, b, d);
#pragma unroll
  for (int r = 0; r < 16; ++r)
    D[mfma_d_row(lane, r) * 32 + (lane & 31)] = d[r];
}

// D = M^T @ B where M (32x32) enters via D-layout registers and is packed to
// A-fragments by dlayout_to_afrag (the fwd P->PV path). Expected: M^T @ B.
__global__ void probe_pack_kernel(const float* __restrict__ M,
                                  const float* __restrict__ B,
                                  float* __restrict__ D) {
  const int lane = threadIdx.x;
  float m[16];
#pragma unroll
  for (int r = 0; r < 16; ++r)
    m[r] = M[mfma_d_row(lane, r) * 32 + (lane & 31)];
  bf16x8_t a0 = dlayout_to_afrag(m);
  bf16x8_t a1 = dlayout_to_afrag(m + 8);
  bf16x8_t b0, b1;
#pragma unroll
  for (int j = 0; j < 8; ++j) {
    b0[j] = (__bf16)B[(8 * (lane >> 5) + j) * 32 + (lane & 31)];
    b1[j] = (__bf16)B[(16 + 8 * (lane >> 5) + j) * 32 + (lane & 31)];
  }
  f32x16 d = (f32x16)(0.f);
  d = mfma_32x32x16_bf16(a0, b0, d);
  d = mfma_32x32x16_bf16(a1, b1, d);
#pragma unroll
  for (int r = 0; r < 16; ++r)
    D[mfma_d_row(lane, r) * 32 + (lane & 31)] = d[r];
}
