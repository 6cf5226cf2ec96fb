// MFMA attention BACKWARD (GPU-validated vs fp32 autograd, round 2).
//
// Flash-attention-2-style backward for the forward in attention.hip
// (bf16, D=64, S%32==0; optional additive key mask [B,1,1,S] and
// hash-counter dropout regenerated from (seed, bh, q, k) — identical to
// the forward's mask, no S x S state):
//
//   delta_i = rowsum(dO_i * O_i)
//   dV = D(P)^T dO       dP = mask/(1-p) o (dO V^T)
//   dS = P o (dP - delta)            (row-wise subtract)
//   dQ = dS K * scale    dK = dS^T Q * scale
//
// Two kernels, no atomics, both 4-wave (256-thread) blocks that stage the
// streamed operand tiles in LDS ONCE per block (the round-1 one-wave
// version re-read K/V from L2 per 16 rows and was K/V-bandwidth-bound —
// same diagnosis as the forward, fixed the same way):
//   K1 (per 64-query block): pass A recomputes per-row m/l (K tiles via
//      LDS), writes M/L/delta, then pass B accumulates dQ (K, V, and
//      K-transposed tiles via LDS).
//   K2 (per 64-key block): mirror structure with swapped roles — S' =
//      K Q^T gives C tiles [key][q]; per-column softmax stats are read
//      from the M/L arrays K1 wrote; accumulates dV and dK (Q, dO and
//      their transposes via LDS).
// Grids are 1-D with (b,h) as the fast dimension: blocks that re-read
// the same tensors land on the same XCD's L2 (see attention.hip).
// M/L stats are in the exp2 (log2) domain; reductions are DPP row_ror
// (common.h). Every MFMA fragment pattern reuses the GPU-verified
// forward patterns; see tests/test_mfma_probe.py for the layout contract.
#include "common.h"

typedef short bwd_bf16x8 __attribute__((ext_vector_type(8)));
typedef float bwd_f32x4 __attribute__((ext_vector_type(4)));

#define ATTN_BD 64
#define BKPAD 8

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

// load an A/B fragment row-block from LDS rows [64][D+BKPAD]
template <int W>
__device__ __forceinline__ bwd_bf16x8 lds_frag(const short (*buf)[W],
                                               int row, int col) {
  return *reinterpret_cast<const bwd_bf16x8*>(&buf[row][col]);
}

// ---------------------------------------------------------------- K1
// grid 1-D (ceil(S/64) * B*H), 256 threads (4 waves x 16 q rows).
// Writes M, L (exp2-domain), delta [B*H, S] fp32 and dQ.
template <int D>
__global__ void
__launch_bounds__(256, D == 64 ? 2 : 1)
attn_bwd_q_kernel(const __hip_bfloat16* __restrict__ Q,
                  const __hip_bfloat16* __restrict__ K,
                  const __hip_bfloat16* __restrict__ V,
                  const __hip_bfloat16* __restrict__ O,
                  const __hip_bfloat16* __restrict__ dO,
                  __hip_bfloat16* __restrict__ dQ,
                  float* __restrict__ Mbuf, float* __restrict__ Lbuf,
                  float* __restrict__ Dbuf,
                  const float* __restrict__ mask, long S, long H, long NBH,
                  long q_sb, long q_sh, long q_ss, long k_sb, long k_sh,
                  long k_ss, long v_sb, long v_sh, long v_ss,
                  float scale, float p_drop, unsigned int seed) {
  __shared__ short Ks[64][D + BKPAD];
  __shared__ short Vs[64][D + BKPAD];
  __shared__ short KsT[D][64 + BKPAD];
  __shared__ short DSw[4][16][64 + BKPAD];  // per-wave bf16 dS staging
  int t = threadIdx.x;
  int w = t >> 6, l = t & 63;
  long bh = (long)blockIdx.x % NBH;
  long q0 = ((long)blockIdx.x / NBH) * 64 + w * 16;
  long bb = bh / H, hh = bh % H;
  const short* q_p = reinterpret_cast<const short*>(Q) + bb * q_sb
                     + hh * q_sh;
  const short* k_p = reinterpret_cast<const short*>(K) + bb * k_sb
                     + hh * k_sh;
  const short* v_p = reinterpret_cast<const short*>(V) + bb * v_sb
                     + hh * v_sh;
  const short* o_p = reinterpret_cast<const short*>(O) + bh * S * D;
  const short* do_p = reinterpret_cast<const short*>(dO) + bh * S * D;
  short* dq_p = reinterpret_cast<short*>(dQ) + bh * S * D;
  int am = l & 15, kg = l >> 4;
  constexpr int NC = D / 32;
  constexpr int ND = D / 16;
  const float* m_p = mask ? mask + (bh / H) * S : nullptr;
  const unsigned int thresh =
      (unsigned int)fminf(p_drop * 4294967296.0f, 4294967040.0f);
  const bool do_drop = p_drop > 0.0f;
  const float rkeep = do_drop ? 1.0f / (1.0f - p_drop) : 1.0f;
  const float scale2 = scale * ATTN_LOG2E;  // exp2-domain logit scale

  int qrow_a = q0 + am < S ? (int)q0 + am : (int)S - 1;
  bwd_bf16x8 qf[NC], dof[NC];
#pragma unroll
  for (int c = 0; c < NC; ++c) {
#pragma unroll
    for (int j = 0; j < 8; ++j) {
      qf[c][j] = q_p[qrow_a * q_ss + c * 32 + kg * 8 + j];
      dof[c][j] = do_p[qrow_a * D + c * 32 + kg * 8 + j];
    }
  }

  int srow = t >> 2, scol = (t & 3) * (D / 4);
  auto stage_k = [&](long kt) {
    int krow = kt + srow < S ? (int)kt + srow : (int)S - 1;
#pragma unroll
    for (int half = 0; half < NC; ++half) {
      *reinterpret_cast<bwd_bf16x8*>(&Ks[srow][scol + half * 8]) =
          *reinterpret_cast<const bwd_bf16x8*>(
              &k_p[krow * k_ss + scol + half * 8]);
    }
  };
  auto stage_kvt = [&](long kt) {
    int krow = kt + srow < S ? (int)kt + srow : (int)S - 1;
#pragma unroll
    for (int half = 0; half < NC; ++half) {
      bwd_bf16x8 kv = *reinterpret_cast<const bwd_bf16x8*>(
          &k_p[krow * k_ss + scol + half * 8]);
      *reinterpret_cast<bwd_bf16x8*>(&Ks[srow][scol + half * 8]) = kv;
#pragma unroll
      for (int j = 0; j < 8; ++j) KsT[scol + half * 8 + j][srow] = kv[j];
      *reinterpret_cast<bwd_bf16x8*>(&Vs[srow][scol + half * 8]) =
          *reinterpret_cast<const bwd_bf16x8*>(
              &v_p[krow * v_ss + scol + half * 8]);
    }
  };

  // ---- pass A: softmax stats m, l per row (exp2 domain)
  float m_run[4], l_run[4];
#pragma unroll
  for (int r = 0; r < 4; ++r) { m_run[r] = -1e30f; l_run[r] = 0.f; }
  for (long kt = 0; kt < S; kt += 64) {
    __syncthreads();
    stage_k(kt);
    __syncthreads();
    bwd_f32x4 s_acc[4] = {{0, 0, 0, 0}, {0, 0, 0, 0},
                          {0, 0, 0, 0}, {0, 0, 0, 0}};
#pragma unroll
    for (int h = 0; h < 4; ++h) {
#pragma unroll
      for (int c = 0; c < NC; ++c) {
        bwd_bf16x8 kf = lds_frag(Ks, h * 16 + am, c * 32 + kg * 8);
        s_acc[h] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(qf[c], kf,
                                                           s_acc[h], 0, 0, 0);
      }
    }
    float mv2[4], oob[4];
#pragma unroll
    for (int h = 0; h < 4; ++h) {
      int key = (int)kt + h * 16 + am;
      mv2[h] = (m_p && key < (int)S) ? m_p[key] * ATTN_LOG2E : 0.f;
      oob[h] = key < (int)S ? 0.f : -1e30f;
    }
#pragma unroll
    for (int r = 0; r < 4; ++r) {
      float sv[4];
#pragma unroll
      for (int h = 0; h < 4; ++h)
        sv[h] = s_acc[h][r] * scale2 + mv2[h] + oob[h];
      float tmax = dpp16_max(fmaxf(fmaxf(sv[0], sv[1]),
                                   fmaxf(sv[2], sv[3])));
      float m_new = fmaxf(m_run[r], tmax);
      float alpha = exp2f(m_run[r] - m_new);
      float rsum = dpp16_sum(exp2f(sv[0] - m_new) + exp2f(sv[1] - m_new) +
                             exp2f(sv[2] - m_new) + exp2f(sv[3] - m_new));
      l_run[r] = l_run[r] * alpha + rsum;
      m_run[r] = m_new;
    }
  }
  // ---- delta = rowsum(dO * O); write stats (lane am==0 writes per row)
  float delta[4];
#pragma unroll
  for (int r = 0; r < 4; ++r) {
    long row = q0 + kg * 4 + r;
    long rowc = row < S ? row : S - 1;
    float part = 0.f;
#pragma unroll
    for (int u = 0; u < D / 16; ++u) {  // D/16 cols per lane
      int d = am * (D / 16) + u;
      float ov = __bfloat162float(reinterpret_cast<const __hip_bfloat16*>(
          o_p)[rowc * D + d]);
      float dv = __bfloat162float(reinterpret_cast<const __hip_bfloat16*>(
          do_p)[rowc * D + d]);
      part += ov * dv;
    }
    delta[r] = dpp16_sum(part);
    if (am == 0 && row < S) {  // one lane per (kg, r) row writes the stats
      Mbuf[bh * S + row] = m_run[r];
      Lbuf[bh * S + row] = l_run[r];
      Dbuf[bh * S + row] = delta[r];
    }
  }
  // ---- pass B: dQ accumulation
  bwd_f32x4 dq_acc[ND];
#pragma unroll
  for (int dt = 0; dt < ND; ++dt) dq_acc[dt] = {0, 0, 0, 0};
  for (long kt = 0; kt < S; kt += 64) {
    __syncthreads();
    stage_kvt(kt);
    __syncthreads();
    // S tile and dP tile (dP = dO V^T: same shape as QK^T with Q->dO, K->V)
    bwd_f32x4 s_acc[4] = {{0, 0, 0, 0}, {0, 0, 0, 0},
                          {0, 0, 0, 0}, {0, 0, 0, 0}};
    bwd_f32x4 dp_acc[4] = {{0, 0, 0, 0}, {0, 0, 0, 0},
                           {0, 0, 0, 0}, {0, 0, 0, 0}};
#pragma unroll
    for (int h = 0; h < 4; ++h) {
#pragma unroll
      for (int c = 0; c < NC; ++c) {
        bwd_bf16x8 kf = lds_frag(Ks, h * 16 + am, c * 32 + kg * 8);
        bwd_bf16x8 vf = lds_frag(Vs, h * 16 + am, c * 32 + kg * 8);
        s_acc[h] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(qf[c], kf,
                                                           s_acc[h], 0, 0, 0);
        dp_acc[h] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(dof[c], vf,
                                                            dp_acc[h], 0, 0, 0);
      }
    }
    // dS = P * (keep*dP/(1-p) - delta) * scale (fold dQ's trailing scale)
#pragma unroll
    for (int h = 0; h < 4; ++h) {
      int key = (int)kt + h * 16 + am;
      float mv2 = (m_p && key < (int)S) ? m_p[key] * ATTN_LOG2E : 0.f;
      float oob = key < (int)S ? 0.f : -1e30f;
#pragma unroll
      for (int r = 0; r < 4; ++r) {
        float p = exp2f(s_acc[h][r] * scale2 + mv2 + oob - m_run[r])
                  / l_run[r];
        float dp = dp_acc[h][r];
        if (do_drop) {
          unsigned int keep = bwd_drop_hash(
              seed, (unsigned int)bh, (unsigned int)(q0 + kg * 4 + r),
              (unsigned int)key) >= thresh;
          dp = keep ? dp * rkeep : 0.f;
        }
        float ds = p * (dp - delta[r]) * scale;
        __hip_bfloat16 b = __float2bfloat16(ds);
        DSw[w][kg * 4 + r][h * 16 + am] = reinterpret_cast<short&>(b);
      }
    }
    wave_lds_fence();
    bwd_bf16x8 dsf[2];
#pragma unroll
    for (int kc = 0; kc < 2; ++kc)
      dsf[kc] = *reinterpret_cast<const bwd_bf16x8*>(
          &DSw[w][am][kc * 32 + kg * 8]);
    // dQ += dS K : m=q, n=d(16/tile), k=keys(64 in 2 chunks)
#pragma unroll
    for (int dt = 0; dt < ND; ++dt) {
#pragma unroll
      for (int kc = 0; kc < 2; ++kc) {
        bwd_bf16x8 kfT = *reinterpret_cast<const bwd_bf16x8*>(
            &KsT[dt * 16 + am][kc * 32 + kg * 8]);
        dq_acc[dt] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
            dsf[kc], kfT, dq_acc[dt], 0, 0, 0);
      }
    }
    wave_lds_fence();
  }
#pragma unroll
  for (int dt = 0; dt < ND; ++dt) {
#pragma unroll
    for (int r = 0; r < 4; ++r) {
      long row = q0 + kg * 4 + r;
      if (row >= S) continue;
      __hip_bfloat16 b = __float2bfloat16(dq_acc[dt][r]);
      dq_p[row * D + dt * 16 + am] = reinterpret_cast<short&>(b);
    }
  }
}

// ---------------------------------------------------------------- K2
// grid 1-D (ceil(S/64) * B*H), 256 threads (4 waves x 16 keys): per
// 64-KEY block, accumulate dK and dV over all q (Q/dO tiles via LDS).
template <int D>
__global__ void
__launch_bounds__(256, D == 64 ? 2 : 1)
attn_bwd_kv_kernel(const __hip_bfloat16* __restrict__ Q,
                   const __hip_bfloat16* __restrict__ K,
                   const __hip_bfloat16* __restrict__ V,
                   const __hip_bfloat16* __restrict__ dO,
                   __hip_bfloat16* __restrict__ dK,
                   __hip_bfloat16* __restrict__ dV,
                   const float* __restrict__ Mbuf,
                   const float* __restrict__ Lbuf,
                   const float* __restrict__ Dbuf,
                   const float* __restrict__ mask, long S, long H, long NBH,
                   long q_sb, long q_sh, long q_ss, long k_sb, long k_sh,
                   long k_ss, long v_sb, long v_sh, long v_ss,
                   float scale, float p_drop, unsigned int seed) {
  __shared__ short Qs[64][D + BKPAD];
  __shared__ short dOs[64][D + BKPAD];
  __shared__ short QsT[D][64 + BKPAD];
  __shared__ short dOsT[D][64 + BKPAD];
  __shared__ short Pw[4][16][64 + BKPAD];   // P' staging (bf16)
  __shared__ short DSw[4][16][64 + BKPAD];  // dS' staging (bf16)
  int t = threadIdx.x;
  int w = t >> 6, l = t & 63;
  long bh = (long)blockIdx.x % NBH;
  long k0 = ((long)blockIdx.x / NBH) * 64 + w * 16;  // this wave's 16 keys
  long bb = bh / H, hh = bh % H;
  const short* q_p = reinterpret_cast<const short*>(Q) + bb * q_sb
                     + hh * q_sh;
  const short* k_p = reinterpret_cast<const short*>(K) + bb * k_sb
                     + hh * k_sh;
  const short* v_p = reinterpret_cast<const short*>(V) + bb * v_sb
                     + hh * v_sh;
  const short* do_p = reinterpret_cast<const short*>(dO) + bh * S * D;
  short* dk_p = reinterpret_cast<short*>(dK) + bh * S * D;
  short* dv_p = reinterpret_cast<short*>(dV) + bh * S * D;
  int am = l & 15, kg = l >> 4;
  constexpr int NC = D / 32;
  constexpr int ND = D / 16;
  const float* m_p = mask ? mask + (bh / H) * S : nullptr;
  const unsigned int thresh =
      (unsigned int)fminf(p_drop * 4294967296.0f, 4294967040.0f);
  const bool do_drop = p_drop > 0.0f;
  const float rkeep = do_drop ? 1.0f / (1.0f - p_drop) : 1.0f;
  const float scale2 = scale * ATTN_LOG2E;

  int krow_a = k0 + am < S ? (int)k0 + am : (int)S - 1;
  bwd_bf16x8 kf[NC], vf[NC];
#pragma unroll
  for (int c = 0; c < NC; ++c) {
#pragma unroll
    for (int j = 0; j < 8; ++j) {
      kf[c][j] = k_p[krow_a * k_ss + c * 32 + kg * 8 + j];
      vf[c][j] = v_p[krow_a * v_ss + c * 32 + kg * 8 + j];
    }
  }
  // additive mask value of this lane's OWN key rows (kg*4+r)
  float mvk2[4];
#pragma unroll
  for (int r = 0; r < 4; ++r) {
    long key = k0 + kg * 4 + r;
    mvk2[r] = (m_p && key < S) ? m_p[key] * ATTN_LOG2E : 0.f;
  }

  bwd_f32x4 dk_acc[ND], dv_acc[ND];
#pragma unroll
  for (int dt = 0; dt < ND; ++dt) {
    dk_acc[dt] = {0, 0, 0, 0};
    dv_acc[dt] = {0, 0, 0, 0};
  }

  int srow = t >> 2, scol = (t & 3) * (D / 4);
  auto stage_qdo = [&](long qt) {
    int qrow = qt + srow < S ? (int)qt + srow : (int)S - 1;
#pragma unroll
    for (int half = 0; half < NC; ++half) {
      bwd_bf16x8 qv = *reinterpret_cast<const bwd_bf16x8*>(
          &q_p[qrow * q_ss + scol + half * 8]);
      *reinterpret_cast<bwd_bf16x8*>(&Qs[srow][scol + half * 8]) = qv;
#pragma unroll
      for (int j = 0; j < 8; ++j) QsT[scol + half * 8 + j][srow] = qv[j];
      bwd_bf16x8 dv8 = *reinterpret_cast<const bwd_bf16x8*>(
          &do_p[qrow * D + scol + half * 8]);
      *reinterpret_cast<bwd_bf16x8*>(&dOs[srow][scol + half * 8]) = dv8;
#pragma unroll
      for (int j = 0; j < 8; ++j) dOsT[scol + half * 8 + j][srow] = dv8[j];
    }
  };

  for (long qt = 0; qt < S; qt += 64) {
    __syncthreads();
    stage_qdo(qt);
    __syncthreads();
    // S' = K Q^T and dP' = V dO^T : C tiles [key][q] (four q quarters)
    bwd_f32x4 s_acc[4] = {{0, 0, 0, 0}, {0, 0, 0, 0},
                          {0, 0, 0, 0}, {0, 0, 0, 0}};
    bwd_f32x4 dp_acc[4] = {{0, 0, 0, 0}, {0, 0, 0, 0},
                           {0, 0, 0, 0}, {0, 0, 0, 0}};
#pragma unroll
    for (int h = 0; h < 4; ++h) {
#pragma unroll
      for (int c = 0; c < NC; ++c) {
        bwd_bf16x8 qfr = lds_frag(Qs, h * 16 + am, c * 32 + kg * 8);
        bwd_bf16x8 dofr = lds_frag(dOs, h * 16 + am, c * 32 + kg * 8);
        s_acc[h] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(kf[c], qfr,
                                                           s_acc[h], 0, 0, 0);
        dp_acc[h] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(vf[c], dofr,
                                                            dp_acc[h], 0, 0, 0);
      }
    }
    // per-column (q) stats; column index = h*16 + am
#pragma unroll
    for (int h = 0; h < 4; ++h) {
      int qrow = (int)qt + h * 16 + am;
      int qrc = qrow < (int)S ? qrow : (int)S - 1;
      float m_q = Mbuf[bh * S + qrc];
      float l_q = Lbuf[bh * S + qrc];
      float d_q = Dbuf[bh * S + qrc];
      float oob = qrow < (int)S ? 0.f : -1e30f;  // oob q contributes 0
#pragma unroll
      for (int r = 0; r < 4; ++r) {
        float p = exp2f(s_acc[h][r] * scale2 + mvk2[r] + oob - m_q) / l_q;
        float pd = p, dp = dp_acc[h][r];
        if (do_drop) {
          unsigned int keep = bwd_drop_hash(
              seed, (unsigned int)bh, (unsigned int)qrow,
              (unsigned int)(k0 + kg * 4 + r)) >= thresh;
          pd = keep ? p * rkeep : 0.f;   // dropped P' for dV
          dp = keep ? dp * rkeep : 0.f;  // dropped dP' for dK
        }
        float ds = p * (dp - d_q) * scale;
        __hip_bfloat16 b1 = __float2bfloat16(pd);
        __hip_bfloat16 b2 = __float2bfloat16(ds);
        Pw[w][kg * 4 + r][h * 16 + am] = reinterpret_cast<short&>(b1);
        DSw[w][kg * 4 + r][h * 16 + am] = reinterpret_cast<short&>(b2);
      }
    }
    wave_lds_fence();
    bwd_bf16x8 pf[2], dsf[2];
#pragma unroll
    for (int kc = 0; kc < 2; ++kc) {
      pf[kc] = *reinterpret_cast<const bwd_bf16x8*>(
          &Pw[w][am][kc * 32 + kg * 8]);
      dsf[kc] = *reinterpret_cast<const bwd_bf16x8*>(
          &DSw[w][am][kc * 32 + kg * 8]);
    }
    // dV += P' dO : m=key, n=d, k=q ; dK += dS' Q : m=key, n=d, k=q
#pragma unroll
    for (int dt = 0; dt < ND; ++dt) {
#pragma unroll
      for (int kc = 0; kc < 2; ++kc) {
        bwd_bf16x8 dofT = *reinterpret_cast<const bwd_bf16x8*>(
            &dOsT[dt * 16 + am][kc * 32 + kg * 8]);
        bwd_bf16x8 qfT = *reinterpret_cast<const bwd_bf16x8*>(
            &QsT[dt * 16 + am][kc * 32 + kg * 8]);
        dv_acc[dt] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
            pf[kc], dofT, dv_acc[dt], 0, 0, 0);
        dk_acc[dt] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
            dsf[kc], qfT, dk_acc[dt], 0, 0, 0);
      }
    }
    wave_lds_fence();
  }
#pragma unroll
  for (int dt = 0; dt < ND; ++dt) {
#pragma unroll
    for (int r = 0; r < 4; ++r) {
      long row = k0 + kg * 4 + r;
      if (row >= S) continue;
      __hip_bfloat16 bk = __float2bfloat16(dk_acc[dt][r]);
      __hip_bfloat16 bv = __float2bfloat16(dv_acc[dt][r]);
      dk_p[row * D + dt * 16 + am] = reinterpret_cast<short&>(bk);
      dv_p[row * D + dt * 16 + am] = reinterpret_cast<short&>(bv);
    }
  }
}
