// Diagnostic probe for the gfx950 v_mfma_f32_16x16x32_bf16 A/B fragment
// lane->element mappings (the C/D map is documented: col=lane&15,
// row=(lane>>4)*4+reg; the A/B maps are ISA-doc territory — this probe
// identifies them empirically so the round-2 attention kernel starts from a
// verified layout).
//
// One wave computes D = A[16x32] @ B[32x16] with a CANDIDATE lane map; the
// host compares against a torch reference per candidate.
#include "common.h"

typedef short bf16x8 __attribute__((ext_vector_type(8)));
typedef float f32x4_p __attribute__((ext_vector_type(4)));

// candidate k-index for element j (0..7) of lane l
__device__ __forceinline__ int probe_k(int cand, int l, int j) {
  switch (cand) {
    case 0: return (l >> 4) * 8 + j;            // contiguous 8 per lane
    case 1: return (l >> 4) + 4 * j;            // stride-4 interleave
    case 2: return (l >> 4) * 4 + (j & 3) + (j >> 2) * 16;  // two 4-blocks
    default: return j;
  }
}

__global__ void mfma_probe_kernel(const __hip_bfloat16* __restrict__ A,
                                  const __hip_bfloat16* __restrict__ B,
                                  float* __restrict__ D, int cand) {
  int l = threadIdx.x;
  if (l >= 64) return;
  bf16x8 a, b;
  int m = l & 15;
#pragma unroll
  for (int j = 0; j < 8; ++j) {
    int k = probe_k(cand, l, j);
    a[j] = reinterpret_cast<const short*>(A)[m * 32 + k];   // A[m][k]
    b[j] = reinterpret_cast<const short*>(B)[k * 16 + m];   // B[k][n=m]
  }
  f32x4_p c = {0.f, 0.f, 0.f, 0.f};
  c = __builtin_amdgcn_mfma_f32_16x16x32_bf16(a, b, c, 0, 0, 0);
#pragma unroll
  for (int r = 0; r < 4; ++r) {
    int row = (l >> 4) * 4 + r;
    D[row * 16 + (l & 15)] = c[r];
  }
}
