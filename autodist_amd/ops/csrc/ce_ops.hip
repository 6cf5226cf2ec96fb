// Fused cross-entropy over bf16 logits (gfx950) — the LM1B 793k-vocab
// loss. torch's F.cross_entropy runs log_softmax fwd+bwd and SAVES the
// [N, V] log-probabilities (4 GB at the benchmark shape: extra write +
// re-read). These kernels keep only per-row (max, log-sum-exp) fp32
// stats: forward is ONE streaming read of the logits (online softmax in
// the exp2 domain), backward one read + one write producing dlogits.
// No ignore_index (the LM1B loss does not use one).
#include "common.h"

// one block per row; 256 threads stride the row (coalesced)
__global__ void
__launch_bounds__(256)
ce_fwd_kernel(const __hip_bfloat16* __restrict__ logits,
              const long* __restrict__ targets,
              float* __restrict__ loss_rows, float* __restrict__ m_out,
              float* __restrict__ l2s_out, long N, long H) {
  __shared__ float lm[256], ls[256];
  long row = blockIdx.x;
  if (row >= N) return;
  const __hip_bfloat16* lr = logits + row * H;
  // online (max, sum-exp2) per thread
  float m = -1e30f, s = 0.f;
  for (long i = threadIdx.x; i < H; i += 256) {
    float v = __bfloat162float(lr[i]) * ATTN_LOG2E;
    if (v > m) {
      s = s * exp2f(m - v) + 1.f;
      m = v;
    } else {
      s += exp2f(v - m);
    }
  }
  lm[threadIdx.x] = m;
  ls[threadIdx.x] = s;
  __syncthreads();
  // tree-combine the 256 (m, s) pairs
  for (int off = 128; off > 0; off >>= 1) {
    if (threadIdx.x < off) {
      float m2 = lm[threadIdx.x + off], s2 = ls[threadIdx.x + off];
      float m1 = lm[threadIdx.x], s1 = ls[threadIdx.x];
      float M = fmaxf(m1, m2);
      lm[threadIdx.x] = M;
      ls[threadIdx.x] = s1 * exp2f(m1 - M) + s2 * exp2f(m2 - M);
    }
    __syncthreads();
  }
  if (threadIdx.x == 0) {
    float M = lm[0];
    float l2s = log2f(ls[0]);
    float tgt = __bfloat162float(lr[targets[row]]);
    m_out[row] = M;
    l2s_out[row] = l2s;
    loss_rows[row] = (M + l2s) / ATTN_LOG2E - tgt;
  }
}

// dlogits[row][j] = (softmax - onehot) * dloss_rows[row]
__global__ void
__launch_bounds__(256)
ce_bwd_kernel(const __hip_bfloat16* __restrict__ logits,
              const long* __restrict__ targets,
              const float* __restrict__ m_in,
              const float* __restrict__ l2s_in,
              const float* __restrict__ dloss_rows,
              __hip_bfloat16* __restrict__ dlogits, long N, long H) {
  long row = blockIdx.x;
  if (row >= N) return;
  const __hip_bfloat16* lr = logits + row * H;
  __hip_bfloat16* dr = dlogits + row * H;
  float shift = m_in[row] + l2s_in[row];  // exp2-domain log-normalizer
  float g = dloss_rows[row];
  long t = targets[row];
  for (long i = threadIdx.x; i < H; i += 256) {
    float p = exp2f(__bfloat162float(lr[i]) * ATTN_LOG2E - shift);
    float d = (p - (i == t ? 1.f : 0.f)) * g;
    dr[i] = __float2bfloat16(d);
  }
}
