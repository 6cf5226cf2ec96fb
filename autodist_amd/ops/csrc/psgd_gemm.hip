// PowerSGD factor GEMMs on gfx950 matrix cores (MFMA) + fused
// decompress/error-feedback kernel.
//
// The compressor's hot ops (parallel/powersgd.py):
//   P = M @ Q        (tall-skinny: [n x s] @ [s x 16])      -> psgd_mq
//   Qn = M^T @ P     ([s x n] @ [n x 16], M stored row-major) -> psgd_mtp
//   hat = P @ Q^T * scale ; err = M - hat ; flat = hat       -> psgd_decompress_ef
//
// MFMA: v_mfma_f32_16x16x4_f32 (exact f32 at the f32 vector rate; guide §3)
// with the canonical staging: coalesced global -> LDS tiles (+1 padding
// kills the 16-lane bank conflicts on the strided A reads), per-wave 16x16
// C tiles, split-K across blocks with fp32 atomics into the small C.
//
// Contract (enforced by the python wrapper): n % 64 == 0, s % 64 == 0,
// factor width fixed at 16 columns (rank <= 16, zero-padded).
#include "common.h"

typedef float f32x4 __attribute__((ext_vector_type(4)));

#define PSGD_R 16   // padded factor width
#define PSGD_BK 64  // K-chunk staged per iteration

// ---------------------------------------------------------------- M @ Q
// grid: (n/64, splitk); block 256 = 4 waves, wave w owns rows w*16..w*16+16.
// K (= s) is sliced across gridDim.y in units of PSGD_BK.
__global__ void psgd_mq_kernel(const float* __restrict__ M,
                               const float* __restrict__ Q,
                               float* __restrict__ C, long n, long s) {
  __shared__ float Mt[64][PSGD_BK + 1];
  __shared__ float Qt[PSGD_BK][PSGD_R + 1];
  int tid = threadIdx.x;
  int wave = tid / WAVE_SIZE;
  int lane = tid % WAVE_SIZE;
  long row0 = (long)blockIdx.x * 64;
  // K-slice for this blockIdx.y
  long chunks = s / PSGD_BK;
  long per = (chunks + gridDim.y - 1) / gridDim.y;
  long c0 = (long)blockIdx.y * per;
  long c1 = min(c0 + per, chunks);
  f32x4 acc = {0.f, 0.f, 0.f, 0.f};
  int a_m = lane & 15, a_k = lane >> 4;      // A/B fragment coords (ISA map)
  for (long c = c0; c < c1; ++c) {
    long k0 = c * PSGD_BK;
    // stage M[row0+0..64)[k0..k0+64): thread t -> row t/4, 16 floats
    {
      int r = tid >> 2, q = tid & 3;
      const float* src = M + (row0 + r) * s + k0 + q * 16;
#pragma unroll
      for (int u = 0; u < 4; ++u) {
        *reinterpret_cast<float4*>(&Mt[r][q * 16 + u * 4]) =
            *reinterpret_cast<const float4*>(src + u * 4);
      }
    }
    // stage Q[k0..k0+64)[0..16): thread t<256 -> 4 floats
    {
      int r = tid >> 2, q = tid & 3;
      *reinterpret_cast<float4*>(&Qt[r][q * 4]) =
          *reinterpret_cast<const float4*>(Q + (k0 + r) * PSGD_R + q * 4);
    }
    __syncthreads();
#pragma unroll
    for (int kk = 0; kk < PSGD_BK; kk += 4) {
      float a = Mt[wave * 16 + a_m][kk + a_k];
      float b = Qt[kk + a_k][a_m];
      acc = __builtin_amdgcn_mfma_f32_16x16x4f32(a, b, acc, 0, 0, 0);
    }
    __syncthreads();
  }
  // C/D map: col = lane&15, row = (lane>>4)*4 + reg
  long crow0 = row0 + wave * 16 + (lane >> 4) * 4;
  int col = lane & 15;
#pragma unroll
  for (int r = 0; r < 4; ++r) {
    atomicAdd(C + (crow0 + r) * PSGD_R + col, acc[r]);
  }
}

// ---------------------------------------------------------------- M^T @ P
// C[s x 16] = M^T @ P. grid: (s/64, splitk over n); block 256 = 4 waves,
// wave w owns output rows (= M columns) wk = w*16..w*16+16.
__global__ void psgd_mtp_kernel(const float* __restrict__ M,
                                const float* __restrict__ P,
                                float* __restrict__ C, long n, long s) {
  __shared__ float Mt[64][PSGD_BK + 1];   // [i][k] slab
  __shared__ float Pt[64][PSGD_R + 1];
  int tid = threadIdx.x;
  int wave = tid / WAVE_SIZE;
  int lane = tid % WAVE_SIZE;
  long kcol0 = (long)blockIdx.x * 64;     // this block's M-column window
  long chunks = n / PSGD_BK;
  long per = (chunks + gridDim.y - 1) / gridDim.y;
  long c0 = (long)blockIdx.y * per;
  long c1 = min(c0 + per, chunks);
  f32x4 acc = {0.f, 0.f, 0.f, 0.f};
  int a_m = lane & 15, a_k = lane >> 4;
  for (long c = c0; c < c1; ++c) {
    long i0 = c * PSGD_BK;
    {
      int r = tid >> 2, q = tid & 3;
      const float* src = M + (i0 + r) * s + kcol0 + q * 16;
#pragma unroll
      for (int u = 0; u < 4; ++u) {
        *reinterpret_cast<float4*>(&Mt[r][q * 16 + u * 4]) =
            *reinterpret_cast<const float4*>(src + u * 4);
      }
    }
    {
      int r = tid >> 2, q = tid & 3;
      *reinterpret_cast<float4*>(&Pt[r][q * 4]) =
          *reinterpret_cast<const float4*>(P + (i0 + r) * PSGD_R + q * 4);
    }
    __syncthreads();
#pragma unroll
    for (int ii = 0; ii < PSGD_BK; ii += 4) {
      // A'[m = output row = M column][k-dim = i]
      float a = Mt[ii + a_k][wave * 16 + a_m];
      float b = Pt[ii + a_k][a_m];
      acc = __builtin_amdgcn_mfma_f32_16x16x4f32(a, b, acc, 0, 0, 0);
    }
    __syncthreads();
  }
  long crow0 = kcol0 + wave * 16 + (lane >> 4) * 4;
  int col = lane & 15;
#pragma unroll
  for (int r = 0; r < 4; ++r) {
    atomicAdd(C + (crow0 + r) * PSGD_R + col, acc[r]);
  }
}

// ------------------------------------------- fused decompress + error fb
// For e < numel (e = i*s + j): hat = dot(Ppad[i], Qpad[j]) * scale;
// err[e] = m_local[e] - hat; flat[e] = hat. P/Q rows are 64 B (L2-resident).
__global__ void psgd_decompress_ef_kernel(float* __restrict__ flat,
                                          float* __restrict__ err,
                                          const float* __restrict__ m_local,
                                          const float* __restrict__ P,
                                          const float* __restrict__ Q,
                                          long numel, long s, float scale) {
  long stride = (long)gridDim.x * blockDim.x;
  for (long e = (long)blockIdx.x * blockDim.x + threadIdx.x; e < numel;
       e += stride) {
    long i = e / s, j = e - i * s;
    const float4* pr = reinterpret_cast<const float4*>(P + i * PSGD_R);
    const float4* qr = reinterpret_cast<const float4*>(Q + j * PSGD_R);
    float hat = 0.f;
#pragma unroll
    for (int u = 0; u < 4; ++u) {
      float4 a = pr[u], b = qr[u];
      hat += a.x * b.x + a.y * b.y + a.z * b.z + a.w * b.w;
    }
    hat *= scale;
    err[e] = m_local[e] - hat;
    flat[e] = hat;
  }
}

// padded M build: out[e] = e < numel ? flat[e] + err[e] : 0
__global__ void psgd_add_err_pad_kernel(const float* __restrict__ flat,
                                        const float* __restrict__ err,
                                        float* __restrict__ out, long numel,
                                        long total) {
  long stride = (long)gridDim.x * blockDim.x;
  for (long e = (long)blockIdx.x * blockDim.x + threadIdx.x; e < total;
       e += stride) {
    out[e] = e < numel ? flat[e] + err[e] : 0.f;
  }
}
