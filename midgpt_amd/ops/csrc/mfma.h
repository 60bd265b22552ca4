// MFMA fragment-layout helpers for v_mfma_f32_32x32x16_bf16 (gfx950).
//
// Layouts (verified on hardware by the mfma_probe test, tests/test_gpu_mfma.py):
//   D/C: lane l, reg r  ->  row = (r&3) + 8*(r>>2) + 4*(l>>5), col = l&31
//   A  : lane l, j      ->  A[row = l&31][k = 8*(l>>5) + j],  j = 0..7
//   B  : lane l, j      ->  B[k = 8*(l>>5) + j][col = l&31]
// (guide cdna_hip_programming.md section 3; A/B extrapolated from the CDNA
//  32x32x8 pattern with K doubled — the probe test guards this assumption.)
#pragma once
#include "common.h"

using bf16x8_t = __attribute__((ext_vector_type(8))) __bf16;

DEVINL f32x16 mfma_32x32x16_bf16(bf16x8_t a, bf16x8_t b, f32x16 c) {
  return __builtin_amdgcn_mfma_f32_32x32x16_bf16(a, b, c, 0, 0, 0);
}

DEVINL int mfma_d_row(int lane, int r) {
  return (r & 3) + 8 * (r >> 2) + 4 * (lane >> 5);
}

// pack two f32 into one dword of two bf16 (lo = first)
DEVINL unsigned pack_bf16(float lo, float hi) {
  return ((unsigned)f2b(hi) << 16) | (unsigned)f2b(lo);
}

// Convert 8 f32 D-layout registers (regs rb..rb+7 of a 32x32 accumulator,
// holding rows {0..3, 8..11} + 4*(lane>>5) of some logical axis) into ONE
// bf16x8 A-fragment covering rows 0..15 of that axis (k-chunk), using
// v_permlane32_swap half exchanges (guide T12/T21 pattern).
DEVINL bf16x8_t dlayout_to_afrag(const float* s /*8 vals*/) {
  unsigned x0 = pack_bf16(s[0], s[1]);
  unsigned x1 = pack_bf16(s[2], s[3]);
  unsigned x2 = pack_bf16(s[4], s[5]);
  unsigned x3 = pack_bf16(s[6], s[7]);
  {
    auto r = __builtin_amdgcn_permlane32_swap(x0, x2, false, false);
    x0 = r[0]; x2 = r[1];
  }
  {
    auto r = __builtin_amdgcn_permlane32_swap(x1, x3, false, false);
    x1 = r[0]; x3 = r[1];
  }
  union { unsigned u[4]; bf16x8_t v; } out;
  out.u[0] = x0; out.u[1] = x1; out.u[2] = x2; out.u[3] = x3;
  return out.v;
}
