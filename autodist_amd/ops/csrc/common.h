// Common helpers for the gfx950 (CDNA4) kernels.
//
// All sync-path kernels here are HBM3E-bandwidth-bound elementwise ops:
// the design rules (cdna_hip_programming guide) are 64-wide wavefronts,
// 256-thread blocks, >>256 workgroups to fill 8 XCDs, and 16 B/lane
// vectorized access (float4 / 8x bf16) for peak coalescing.
#pragma once
#include <hip/hip_runtime.h>
#include <hip/hip_bf16.h>

#define WAVE_SIZE 64
#define BLOCK_THREADS 256

static inline int grid_for(long n, int per_thread = 4) {
  // enough workgroups to fill 256 CUs across 8 XCDs, capped for small n
  long blocks = (n + (long)BLOCK_THREADS * per_thread - 1) /
                ((long)BLOCK_THREADS * per_thread);
  if (blocks > 65535L * 8) blocks = 65535L * 8;
  if (blocks < 1) blocks = 1;
  return (int)blocks;
}

#define HIP_CHECK(cmd)                                                   \
  do {                                                                   \
    hipError_t e = (cmd);                                                \
    if (e != hipSuccess) {                                               \
      throw std::runtime_error(std::string("HIP error: ") +              \
                               hipGetErrorString(e));                    \
    }                                                                    \
  } while (0)

// ---- 16-lane cross-lane reductions via DPP row_ror (VALU-only) ----
// The 16-lane "row" of DPP matches the MFMA C-column group exactly; four
// rotate-accumulate steps give every lane the full reduction without any
// ds_bpermute traffic (shfl_xor lowers to LDS-pipe ops which contended
// with fragment reads/writes — measured 20:1 VALU:MFMA before this).
__device__ __forceinline__ float dpp16_sum(float v) {
  int x;
  x = __builtin_amdgcn_update_dpp(0, __builtin_bit_cast(int, v),
                                  0x121, 0xF, 0xF, true);  // row_ror:1
  v += __builtin_bit_cast(float, x);
  x = __builtin_amdgcn_update_dpp(0, __builtin_bit_cast(int, v),
                                  0x122, 0xF, 0xF, true);  // row_ror:2
  v += __builtin_bit_cast(float, x);
  x = __builtin_amdgcn_update_dpp(0, __builtin_bit_cast(int, v),
                                  0x124, 0xF, 0xF, true);  // row_ror:4
  v += __builtin_bit_cast(float, x);
  x = __builtin_amdgcn_update_dpp(0, __builtin_bit_cast(int, v),
                                  0x128, 0xF, 0xF, true);  // row_ror:8
  v += __builtin_bit_cast(float, x);
  return v;
}

__device__ __forceinline__ float dpp16_max(float v) {
  int x;
  x = __builtin_amdgcn_update_dpp(0, __builtin_bit_cast(int, v),
                                  0x121, 0xF, 0xF, true);
  v = fmaxf(v, __builtin_bit_cast(float, x));
  x = __builtin_amdgcn_update_dpp(0, __builtin_bit_cast(int, v),
                                  0x122, 0xF, 0xF, true);
  v = fmaxf(v, __builtin_bit_cast(float, x));
  x = __builtin_amdgcn_update_dpp(0, __builtin_bit_cast(int, v),
                                  0x124, 0xF, 0xF, true);
  v = fmaxf(v, __builtin_bit_cast(float, x));
  x = __builtin_amdgcn_update_dpp(0, __builtin_bit_cast(int, v),
                                  0x128, 0xF, 0xF, true);
  v = fmaxf(v, __builtin_bit_cast(float, x));
  return v;
}

#define ATTN_LOG2E 1.4426950408889634f
