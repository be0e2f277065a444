// On-device verification of the v_mfma_f32_32x32x16_bf16 fragment maps the
// prefill attention kernel assumes (§3 of the CDNA guide warns the C/D map is
// non-obvious and that transposed-output bugs pass symmetric-input tests —
// the python test drives this with ASYMMETRIC A and B).
//
// Assumed maps (lane l, 64 lanes):
//   A[i][k]: i = l&31, k = (l>>5)*8 + j          (j = 0..7)
//   B[k][j]: j = l&31, k = (l>>5)*8 + jj
//   D[i][j]: j = l&31, i = (r&3) + 8*(r>>2) + 4*(l>>5)   (r = 0..15)

#include "common.h"

typedef __bf16 bf16x8_mfma __attribute__((ext_vector_type(8)));

extern "C" __global__ void __launch_bounds__(64) mfma_selftest_kernel(
    float* __restrict__ D,          // [32, 32] row-major
    const bf16_t* __restrict__ A,   // [32, 16] row-major
    const bf16_t* __restrict__ B) { // [16, 32] row-major
  const int lane = threadIdx.x;
  const int half = lane >> 5;
  const int col = lane & 31;

  bf16x8_mfma a_frag, b_frag;
#pragma unroll
  for (int j = 0; j < 8; ++j) {
    const int k = half * 8 + j;
    a_frag[j] = *reinterpret_cast<const __bf16*>(&A[col * 16 + k]);   // A[i=col][k]
    b_frag[j] = *reinterpret_cast<const __bf16*>(&B[k * 32 + col]);   // B[k][j=col]
  }
  f32x16 acc = {};
  acc = __builtin_amdgcn_mfma_f32_32x32x16_bf16(a_frag, b_frag, acc, 0, 0, 0);
#pragma unroll
  for (int r = 0; r < 16; ++r) {
    const int row = (r & 3) + 8 * (r >> 2) + 4 * half;
    D[row * 32 + col] = acc[r];
  }
}
