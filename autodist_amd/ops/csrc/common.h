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
