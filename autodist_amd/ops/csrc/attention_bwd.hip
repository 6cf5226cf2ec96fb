// MFMA attention BACKWARD (GPU-validated vs fp32 autograd, round 2).
//
// Flash-attention-2-style backward for the forward in attention.hip
// (bf16, D=64, S%32==0; optional additive key mask [B,1,1,S] and
// hash-counter dropout regenerated from (seed, bh, q, k) — identical to
// the forward's mask, no S x S state):
//
//   delta_i = rowsum(dO_i * O_i)
//   dV = P^T dO          dP = dO V^T
//   dS = P o (dP - delta)            (row-wise subtract)
//   dQ = dS K * scale    dK = dS^T Q * scale
//
// Two kernels, no atomics:
//   K1 (per q-tile):   recompute per-row m/l (pass A), write M/L/delta to
//                      global, then dQ (pass B).
//   K2 (per key-tile): mirror structure with swapped roles — S' = K Q^T
//                      gives C tiles [key][q]; per-column softmax stats are
//                      read from the M/L arrays K1 wrote; accumulates dV and
//                      dK.
// Every MFMA fragment pattern below reuses the GPU-verified forward
// patterns (QK^T loader, LDS C-layout->A-layout staging for P, V-style
// B-fragments); see tests/test_mfma_probe.py for the layout contract.
#include "common.h"

typedef short bwd_bf16x8 __attribute__((ext_vector_type(8)));
typedef float bwd_f32x4 __attribute__((ext_vector_type(4)));

#define ATTN_BD 64

__device__ __forceinline__ unsigned int bwd_drop_hash(unsigned int seed,
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

// 16-lane reductions via DPP (common.h) — no LDS-pipe traffic
#define bwd_red_max dpp16_max
#define bwd_red_sum dpp16_sum

// load an A/B fragment row-block: elem j from src[(row)*64 + c*32 + kg*8+j]
__device__ __forceinline__ bwd_bf16x8 frag_rowmajor(const short* src,
                                                    long row, int c, int kg) {
  bwd_bf16x8 f;
#pragma unroll
  for (int j = 0; j < 8; ++j) f[j] = src[row * ATTN_BD + c * 32 + kg * 8 + j];
  return f;
}

// ---------------------------------------------------------------- K1
// grid (S/16, B*H), 64 threads. Writes M, L, delta [B*H, S] fp32 and dQ.
__global__ void attn_bwd_q_kernel(const __hip_bfloat16* __restrict__ Q,
                                  const __hip_bfloat16* __restrict__ K,
                                  const __hip_bfloat16* __restrict__ V,
                                  const __hip_bfloat16* __restrict__ O,
                                  const __hip_bfloat16* __restrict__ dO,
                                  __hip_bfloat16* __restrict__ dQ,
                                  float* __restrict__ Mbuf,
                                  float* __restrict__ Lbuf,
                                  float* __restrict__ Dbuf,
                                  const float* __restrict__ mask, long S,
                                  long H, float scale, float p_drop,
                                  unsigned int seed) {
  __shared__ float PS[16][32 + 1];
  int l = threadIdx.x;
  long bh = blockIdx.y;
  long q0 = (long)blockIdx.x * 16;
  const short* q_p = reinterpret_cast<const short*>(Q) + bh * S * ATTN_BD;
  const short* k_p = reinterpret_cast<const short*>(K) + bh * S * ATTN_BD;
  const short* v_p = reinterpret_cast<const short*>(V) + bh * S * ATTN_BD;
  const short* o_p = reinterpret_cast<const short*>(O) + bh * S * ATTN_BD;
  const short* do_p = reinterpret_cast<const short*>(dO) + bh * S * ATTN_BD;
  short* dq_p = reinterpret_cast<short*>(dQ) + bh * S * ATTN_BD;
  int am = l & 15, kg = l >> 4;
  const float* m_p = mask ? mask + (bh / H) * S : nullptr;
  const unsigned int thresh =
      (unsigned int)fminf(p_drop * 4294967296.0f, 4294967040.0f);
  const bool do_drop = p_drop > 0.0f;
  const float rkeep = do_drop ? 1.0f / (1.0f - p_drop) : 1.0f;
  const float scale2 = scale * ATTN_LOG2E;  // Mbuf/Lbuf stats are in the
                                            // exp2 (log2) domain

  bwd_bf16x8 qf[2], dof[2];
#pragma unroll
  for (int c = 0; c < 2; ++c) {
    qf[c] = frag_rowmajor(q_p, q0 + am, c, kg);
    dof[c] = frag_rowmajor(do_p, q0 + am, c, kg);
  }

  // ---- pass A: softmax stats m, l per row
  float m_run[4], l_run[4];
#pragma unroll
  for (int r = 0; r < 4; ++r) { m_run[r] = -1e30f; l_run[r] = 0.f; }
  for (long kt = 0; kt < S; kt += 32) {
    bwd_f32x4 s_acc[2] = {{0, 0, 0, 0}, {0, 0, 0, 0}};
#pragma unroll
    for (int h = 0; h < 2; ++h) {
#pragma unroll
      for (int c = 0; c < 2; ++c) {
        bwd_bf16x8 kf = frag_rowmajor(k_p, kt + h * 16 + am, c, kg);
        s_acc[h] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(qf[c], kf,
                                                           s_acc[h], 0, 0, 0);
      }
    }
    float mv0 = m_p ? m_p[kt + am] * ATTN_LOG2E : 0.f;
    float mv1 = m_p ? m_p[kt + 16 + am] * ATTN_LOG2E : 0.f;
#pragma unroll
    for (int r = 0; r < 4; ++r) {
      float s0 = s_acc[0][r] * scale2 + mv0;
      float s1 = s_acc[1][r] * scale2 + mv1;
      float tmax = bwd_red_max(fmaxf(s0, s1));
      float m_new = fmaxf(m_run[r], tmax);
      float alpha = exp2f(m_run[r] - m_new);
      float rsum = bwd_red_sum(exp2f(s0 - m_new) + exp2f(s1 - m_new));
      l_run[r] = l_run[r] * alpha + rsum;
      m_run[r] = m_new;
    }
  }
  // ---- delta = rowsum(dO * O); write stats (lane am==0 writes per row)
  float delta[4];
#pragma unroll
  for (int r = 0; r < 4; ++r) {
    long row = q0 + kg * 4 + r;
    float part = 0.f;
    // each of the 16 lanes in the group sums 4 d-columns: d = am*4..am*4+3
#pragma unroll
    for (int u = 0; u < 4; ++u) {
      int d = am * 4 + u;
      float ov = __bfloat162float(reinterpret_cast<const __hip_bfloat16*>(
          o_p)[row * ATTN_BD + d]);
      float dv = __bfloat162float(reinterpret_cast<const __hip_bfloat16*>(
          do_p)[row * ATTN_BD + d]);
      part += ov * dv;
    }
    delta[r] = bwd_red_sum(part);
    if (am == 0) {  // one lane per (kg, r) row writes the stats
      Mbuf[bh * S + row] = m_run[r];
      Lbuf[bh * S + row] = l_run[r];
      Dbuf[bh * S + row] = delta[r];
    }
  }
  // ---- pass B: dQ accumulation
  bwd_f32x4 dq_acc[4];
#pragma unroll
  for (int dt = 0; dt < 4; ++dt) dq_acc[dt] = {0, 0, 0, 0};
  for (long kt = 0; kt < S; kt += 32) {
    // S tile and dP tile (dP = dO V^T: same shape as QK^T with Q->dO, K->V)
    bwd_f32x4 s_acc[2] = {{0, 0, 0, 0}, {0, 0, 0, 0}};
    bwd_f32x4 dp_acc[2] = {{0, 0, 0, 0}, {0, 0, 0, 0}};
#pragma unroll
    for (int h = 0; h < 2; ++h) {
#pragma unroll
      for (int c = 0; c < 2; ++c) {
        bwd_bf16x8 kf = frag_rowmajor(k_p, kt + h * 16 + am, c, kg);
        bwd_bf16x8 vf = frag_rowmajor(v_p, kt + h * 16 + am, c, kg);
        s_acc[h] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(qf[c], kf,
                                                           s_acc[h], 0, 0, 0);
        dp_acc[h] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(dof[c], vf,
                                                            dp_acc[h], 0, 0, 0);
      }
    }
    // dS = P * (keep*dP/(1-p) - delta) * scale (fold dQ's trailing scale)
#pragma unroll
    for (int h = 0; h < 2; ++h) {
      float mv = m_p ? m_p[kt + h * 16 + am] * ATTN_LOG2E : 0.f;
#pragma unroll
      for (int r = 0; r < 4; ++r) {
        float p = exp2f(s_acc[h][r] * scale2 + mv - m_run[r]) / l_run[r];
        float dp = dp_acc[h][r];
        if (do_drop) {
          unsigned int keep = bwd_drop_hash(
              seed, (unsigned int)bh, (unsigned int)(q0 + kg * 4 + r),
              (unsigned int)(kt + h * 16 + am)) >= thresh;
          dp = keep ? dp * rkeep : 0.f;
        }
        float ds = p * (dp - delta[r]) * scale;
        PS[kg * 4 + r][h * 16 + am] = ds;
      }
    }
    __syncthreads();
    bwd_bf16x8 dsf;
#pragma unroll
    for (int j = 0; j < 8; ++j) {
      __hip_bfloat16 b = __float2bfloat16(PS[am][kg * 8 + j]);
      dsf[j] = reinterpret_cast<short&>(b);
    }
    // dQ += dS K : m=q, n=d(16/tile), k=key(32)
#pragma unroll
    for (int dt = 0; dt < 4; ++dt) {
      bwd_bf16x8 kf;
#pragma unroll
      for (int j = 0; j < 8; ++j) {
        kf[j] = k_p[(kt + kg * 8 + j) * ATTN_BD + dt * 16 + am];
      }
      dq_acc[dt] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(dsf, kf,
                                                           dq_acc[dt], 0, 0, 0);
    }
    __syncthreads();
  }
#pragma unroll
  for (int dt = 0; dt < 4; ++dt) {
#pragma unroll
    for (int r = 0; r < 4; ++r) {
      long row = q0 + kg * 4 + r;
      __hip_bfloat16 b = __float2bfloat16(dq_acc[dt][r]);
      dq_p[row * ATTN_BD + dt * 16 + am] = reinterpret_cast<short&>(b);
    }
  }
}

// ---------------------------------------------------------------- K2
// grid (S/16, B*H): per KEY tile, accumulate dK and dV over all q.
__global__ void attn_bwd_kv_kernel(const __hip_bfloat16* __restrict__ Q,
                                   const __hip_bfloat16* __restrict__ K,
                                   const __hip_bfloat16* __restrict__ V,
                                   const __hip_bfloat16* __restrict__ dO,
                                   __hip_bfloat16* __restrict__ dK,
                                   __hip_bfloat16* __restrict__ dV,
                                   const float* __restrict__ Mbuf,
                                   const float* __restrict__ Lbuf,
                                   const float* __restrict__ Dbuf,
                                   const float* __restrict__ mask, long S,
                                   long H, float scale, float p_drop,
                                   unsigned int seed) {
  __shared__ float PS[16][32 + 1];   // P' or dS' tile [key][q-chunk]
  __shared__ float PS2[16][32 + 1];
  int l = threadIdx.x;
  long bh = blockIdx.y;
  long k0 = (long)blockIdx.x * 16;   // this block's 16 keys
  const short* q_p = reinterpret_cast<const short*>(Q) + bh * S * ATTN_BD;
  const short* k_p = reinterpret_cast<const short*>(K) + bh * S * ATTN_BD;
  const short* v_p = reinterpret_cast<const short*>(V) + bh * S * ATTN_BD;
  const short* do_p = reinterpret_cast<const short*>(dO) + bh * S * ATTN_BD;
  short* dk_p = reinterpret_cast<short*>(dK) + bh * S * ATTN_BD;
  short* dv_p = reinterpret_cast<short*>(dV) + bh * S * ATTN_BD;
  int am = l & 15, kg = l >> 4;
  const float* m_p = mask ? mask + (bh / H) * S : nullptr;
  const unsigned int thresh =
      (unsigned int)fminf(p_drop * 4294967296.0f, 4294967040.0f);
  const bool do_drop = p_drop > 0.0f;
  const float rkeep = do_drop ? 1.0f / (1.0f - p_drop) : 1.0f;
  const float scale2 = scale * ATTN_LOG2E;  // Mbuf/Lbuf stats are in the
                                            // exp2 (log2) domain

  bwd_bf16x8 kf[2], vf[2];
#pragma unroll
  for (int c = 0; c < 2; ++c) {
    kf[c] = frag_rowmajor(k_p, k0 + am, c, kg);
    vf[c] = frag_rowmajor(v_p, k0 + am, c, kg);
  }
  bwd_f32x4 dk_acc[4], dv_acc[4];
#pragma unroll
  for (int dt = 0; dt < 4; ++dt) {
    dk_acc[dt] = {0, 0, 0, 0};
    dv_acc[dt] = {0, 0, 0, 0};
  }
  for (long qt = 0; qt < S; qt += 32) {
    // S' = K Q^T and dP' = V dO^T : C tiles [key][q] (two q halves)
    bwd_f32x4 s_acc[2] = {{0, 0, 0, 0}, {0, 0, 0, 0}};
    bwd_f32x4 dp_acc[2] = {{0, 0, 0, 0}, {0, 0, 0, 0}};
#pragma unroll
    for (int h = 0; h < 2; ++h) {
#pragma unroll
      for (int c = 0; c < 2; ++c) {
        bwd_bf16x8 qf = frag_rowmajor(q_p, qt + h * 16 + am, c, kg);
        bwd_bf16x8 dof = frag_rowmajor(do_p, qt + h * 16 + am, c, kg);
        s_acc[h] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(kf[c], qf,
                                                           s_acc[h], 0, 0, 0);
        dp_acc[h] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(vf[c], dof,
                                                            dp_acc[h], 0, 0, 0);
      }
    }
    // per-column (q) stats; column index = h*16 + am
#pragma unroll
    for (int h = 0; h < 2; ++h) {
      long qrow = qt + h * 16 + am;
      float m_q = Mbuf[bh * S + qrow];
      float l_q = Lbuf[bh * S + qrow];
      float d_q = Dbuf[bh * S + qrow];
#pragma unroll
      for (int r = 0; r < 4; ++r) {
        float mv = m_p ? m_p[k0 + kg * 4 + r] * ATTN_LOG2E : 0.f;  // key mask
        float p = exp2f(s_acc[h][r] * scale2 + mv - m_q) / l_q;
        float pd = p, dp = dp_acc[h][r];
        if (do_drop) {
          unsigned int keep = bwd_drop_hash(
              seed, (unsigned int)bh, (unsigned int)qrow,
              (unsigned int)(k0 + kg * 4 + r)) >= thresh;
          pd = keep ? p * rkeep : 0.f;   // dropped P' for dV
          dp = keep ? dp * rkeep : 0.f;  // dropped dP' for dK
        }
        PS[kg * 4 + r][h * 16 + am] = pd;                       // P'
        PS2[kg * 4 + r][h * 16 + am] = p * (dp - d_q) * scale;
      }
    }
    __syncthreads();
    bwd_bf16x8 pf, dsf;
#pragma unroll
    for (int j = 0; j < 8; ++j) {
      __hip_bfloat16 b1 = __float2bfloat16(PS[am][kg * 8 + j]);
      __hip_bfloat16 b2 = __float2bfloat16(PS2[am][kg * 8 + j]);
      pf[j] = reinterpret_cast<short&>(b1);
      dsf[j] = reinterpret_cast<short&>(b2);
    }
    // dV += P' dO : m=key, n=d, k=q ; dK += dS' Q : m=key, n=d, k=q
#pragma unroll
    for (int dt = 0; dt < 4; ++dt) {
      bwd_bf16x8 dof, qf;
#pragma unroll
      for (int j = 0; j < 8; ++j) {
        long qq = qt + kg * 8 + j;
        dof[j] = do_p[qq * ATTN_BD + dt * 16 + am];
        qf[j] = q_p[qq * ATTN_BD + dt * 16 + am];
      }
      dv_acc[dt] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(pf, dof,
                                                           dv_acc[dt], 0, 0, 0);
      dk_acc[dt] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(dsf, qf,
                                                           dk_acc[dt], 0, 0, 0);
    }
    __syncthreads();
  }
#pragma unroll
  for (int dt = 0; dt < 4; ++dt) {
#pragma unroll
    for (int r = 0; r < 4; ++r) {
      long row = k0 + kg * 4 + r;
      __hip_bfloat16 bk = __float2bfloat16(dk_acc[dt][r]);
      __hip_bfloat16 bv = __float2bfloat16(dv_acc[dt][r]);
      dk_p[row * ATTN_BD + dt * 16 + am] = reinterpret_cast<short&>(bk);
      dv_p[row * ATTN_BD + dt * 16 + am] = reinterpret_cast<short&>(bv);
    }
  }
}
