// Fused attention FORWARD on gfx950 matrix cores.
//
// O = dropout(softmax(Q K^T / sqrt(D) + mask)) V for [B, H, S, D] bf16,
// D = 64, S % 32 == 0. mask is an optional ADDITIVE key-padding mask
// [B, 1, 1, S] (the BERT attention_mask form); dropout uses a counter-based
// hash of (seed, bh, q, k) so the backward regenerates the identical mask
// without materializing S x S state (flash-attention-style).
//
// v2 structure (4 waves x 16-query rows = 64-query tile per block):
//   * K/V 32-key tiles are staged in LDS ONCE per block and consumed by all
//     4 waves -> 4x less L2/HBM K/V traffic than the round-1 one-wave
//     kernel (which re-read K/V per 16-row tile and was 0.5-0.9x SDPA).
//   * QK^T and P@V on v_mfma_f32_16x16x32_bf16. A/B fragments use the
//     contiguous-8 k-map; per the measured probe (tests/test_mfma_probe.py)
//     any SELF-CONSISTENT A/B k-map is valid, and P is routed through a
//     per-wave LDS buffer (C-layout write, A-layout read).
//   * online softmax in fp32: per-row running max m / sum l, row reductions
//     via 4-step shfl_xor over the 16-lane C-column group. l accumulates
//     the UNdropped probabilities (torch semantics: dropout after softmax);
//     the 1/(1-p) rescale is folded into the epilogue.
#include "common.h"

typedef short bf16x8_t __attribute__((ext_vector_type(8)));
typedef float f32x4_t __attribute__((ext_vector_type(4)));

#define ATTN_D 64
#define KPAD 8  // LDS row padding (shorts) to stagger banks

__device__ __forceinline__ float row_reduce_max(float v, int width16) {
  // max across the 16-lane group (lanes sharing l>>4)
#pragma unroll
  for (int off = 1; off < 16; off <<= 1) {
    v = fmaxf(v, __shfl_xor(v, off, 64));
  }
  return v;
}

__device__ __forceinline__ float row_reduce_sum(float v) {
#pragma unroll
  for (int off = 1; off < 16; off <<= 1) {
    v += __shfl_xor(v, off, 64);
  }
  return v;
}

// counter-based dropout hash: uniform in [0, 2^32); keep iff >= p*2^32.
// Pure function of (seed, bh, q, k) so forward and both backward kernels
// regenerate the identical mask.
__device__ __forceinline__ unsigned int drop_hash(unsigned int seed,
                                                  unsigned int bh,
                                                  unsigned int q,
                                                  unsigned int k) {
  unsigned int x = seed ^ (bh * 0x9E3779B9u) ^ (q * 0x85EBCA6Bu)
                   ^ (k * 0xC2B2AE35u);
  x ^= x >> 16; x *= 0x7FEB352Du;
  x ^= x >> 15; x *= 0x846CA68Bu;
  x ^= x >> 16;
  return x;
}

// grid (ceil(S/64), B*H), 256 threads (4 waves). mask may be nullptr.
__global__ void attn_fwd_kernel(const __hip_bfloat16* __restrict__ Q,
                                const __hip_bfloat16* __restrict__ K,
                                const __hip_bfloat16* __restrict__ V,
                                __hip_bfloat16* __restrict__ O,
                                const float* __restrict__ mask, long S,
                                long H, float scale, float p_drop,
                                unsigned int seed) {
  __shared__ short Ks[32][ATTN_D + KPAD];
  __shared__ short VsT[ATTN_D][32 + KPAD];  // transposed: B-frag reads are
                                            // row-contiguous (1 ds_read_b128)
  __shared__ float P[4][16][32 + 1];
  int t = threadIdx.x;
  int w = t >> 6;        // wave 0..3
  int l = t & 63;        // lane
  long bh = blockIdx.y;
  long q0 = (long)blockIdx.x * 64 + w * 16;  // this wave's 16 q rows
  const short* q_p = reinterpret_cast<const short*>(Q) + bh * S * ATTN_D;
  const short* k_p = reinterpret_cast<const short*>(K) + bh * S * ATTN_D;
  const short* v_p = reinterpret_cast<const short*>(V) + bh * S * ATTN_D;
  short* o_p = reinterpret_cast<short*>(O) + bh * S * ATTN_D;
  const float* m_p = mask ? mask + (bh / H) * S : nullptr;

  int am = l & 15;       // A-fragment m index / C column index
  int kg = l >> 4;       // k-group (0..3)
  const unsigned int thresh =
      (unsigned int)fminf(p_drop * 4294967296.0f, 4294967040.0f);
  const bool do_drop = p_drop > 0.0f;

  // Q fragments: q row (q0+am) clamped for partial tiles; d = c*32+kg*8+j
  long qrow_a = q0 + am < S ? q0 + am : S - 1;
  bf16x8_t qf[2];
#pragma unroll
  for (int c = 0; c < 2; ++c) {
#pragma unroll
    for (int j = 0; j < 8; ++j) {
      qf[c][j] = q_p[qrow_a * ATTN_D + c * 32 + kg * 8 + j];
    }
  }

  float m_run[4], l_run[4];
  f32x4_t o_acc[4];  // d-tiles of 16 cols each
#pragma unroll
  for (int r = 0; r < 4; ++r) {
    m_run[r] = -1e30f;
    l_run[r] = 0.f;
  }
#pragma unroll
  for (int dt = 0; dt < 4; ++dt) o_acc[dt] = {0.f, 0.f, 0.f, 0.f};

  // cooperative K/V stage indices: 256 threads x bf16x8 = one 32x64 tile
  int srow = t >> 3, scol = (t & 7) * 8;

  for (long kt = 0; kt < S; kt += 32) {
    __syncthreads();  // previous tile's consumers done
    *reinterpret_cast<bf16x8_t*>(&Ks[srow][scol]) =
        *reinterpret_cast<const bf16x8_t*>(&k_p[(kt + srow) * ATTN_D + scol]);
    {
      bf16x8_t vrow = *reinterpret_cast<const bf16x8_t*>(
          &v_p[(kt + srow) * ATTN_D + scol]);
#pragma unroll
      for (int j = 0; j < 8; ++j) VsT[scol + j][srow] = vrow[j];
    }
    __syncthreads();

    // ---- S tile = Q[16] x K[32]^T : two 16x16 C tiles (key halves)
    f32x4_t s_acc[2] = {{0.f, 0.f, 0.f, 0.f}, {0.f, 0.f, 0.f, 0.f}};
#pragma unroll
    for (int h = 0; h < 2; ++h) {
#pragma unroll
      for (int c = 0; c < 2; ++c) {
        bf16x8_t kf = *reinterpret_cast<const bf16x8_t*>(
            &Ks[h * 16 + am][c * 32 + kg * 8]);
        s_acc[h] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(qf[c], kf,
                                                           s_acc[h], 0, 0, 0);
      }
    }
    // additive key mask (same value for every q row / r)
    float mv0 = 0.f, mv1 = 0.f;
    if (m_p) {
      mv0 = m_p[kt + am];
      mv1 = m_p[kt + 16 + am];
    }
    // scale + online softmax bookkeeping (row q = kg*4 + r)
    float p0[4], p1[4], alpha[4];
#pragma unroll
    for (int r = 0; r < 4; ++r) {
      float s0 = s_acc[0][r] * scale + mv0;
      float s1 = s_acc[1][r] * scale + mv1;
      float tmax = row_reduce_max(fmaxf(s0, s1), 16);
      float m_new = fmaxf(m_run[r], tmax);
      alpha[r] = __expf(m_run[r] - m_new);
      p0[r] = __expf(s0 - m_new);
      p1[r] = __expf(s1 - m_new);
      float rsum = row_reduce_sum(p0[r] + p1[r]);
      l_run[r] = l_run[r] * alpha[r] + rsum;  // UNdropped sum
      m_run[r] = m_new;
      if (do_drop) {
        unsigned int qrow = (unsigned int)(q0 + kg * 4 + r);
        if (drop_hash(seed, (unsigned int)bh, qrow,
                      (unsigned int)(kt + am)) < thresh)
          p0[r] = 0.f;
        if (drop_hash(seed, (unsigned int)bh, qrow,
                      (unsigned int)(kt + 16 + am)) < thresh)
          p1[r] = 0.f;
      }
    }
#pragma unroll
    for (int dt = 0; dt < 4; ++dt) {
#pragma unroll
      for (int r = 0; r < 4; ++r) o_acc[dt][r] *= alpha[r];
    }
    // ---- stage (dropped) P through per-wave LDS: C-write, A-read
#pragma unroll
    for (int r = 0; r < 4; ++r) {
      P[w][kg * 4 + r][am] = p0[r];
      P[w][kg * 4 + r][16 + am] = p1[r];
    }
    __syncthreads();
    bf16x8_t pf;
#pragma unroll
    for (int j = 0; j < 8; ++j) {
      __hip_bfloat16 pb = __float2bfloat16(P[w][am][kg * 8 + j]);
      pf[j] = reinterpret_cast<short&>(pb);
    }
    // ---- O += P @ V : one mfma per 16-col d tile (k = 32 keys)
#pragma unroll
    for (int dt = 0; dt < 4; ++dt) {
      bf16x8_t vf = *reinterpret_cast<const bf16x8_t*>(
          &VsT[dt * 16 + am][kg * 8]);
      o_acc[dt] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(pf, vf,
                                                          o_acc[dt], 0, 0, 0);
    }
  }
  // ---- epilogue: normalize (+ dropout keep-rescale) + store
  float rkeep = do_drop ? 1.0f / (1.0f - p_drop) : 1.0f;
#pragma unroll
  for (int dt = 0; dt < 4; ++dt) {
#pragma unroll
    for (int r = 0; r < 4; ++r) {
      long qrow = q0 + kg * 4 + r;
      if (qrow >= S) continue;
      float val = o_acc[dt][r] / l_run[r] * rkeep;
      __hip_bfloat16 ob = __float2bfloat16(val);
      o_p[qrow * ATTN_D + dt * 16 + am] = reinterpret_cast<short&>(ob);
    }
  }
}

// debug/testing: materialize the dropout keep-mask (1=kept) the kernels
// derive from (seed, bh, q, k) — lets tests build an exact torch reference.
__global__ void attn_dropmask_kernel(unsigned char* __restrict__ out, long S,
                                     float p_drop, unsigned int seed) {
  long bh = blockIdx.y;
  long q = blockIdx.x;
  unsigned int thresh =
      (unsigned int)fminf(p_drop * 4294967296.0f, 4294967040.0f);
  for (long k = threadIdx.x; k < S; k += blockDim.x) {
    out[(bh * S + q) * S + k] =
        drop_hash(seed, (unsigned int)bh, (unsigned int)q,
                  (unsigned int)k) >= thresh;
  }
}
