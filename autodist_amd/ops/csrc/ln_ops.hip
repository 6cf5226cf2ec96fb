// Fused LayerNorm (+ residual add) for bf16 transformer streams, gfx950.
//
// torch's LN on a bf16 autocast stream runs fp32 kernels bracketed by
// bfloat16<->float32 copy kernels (measured: the cast/elementwise cluster
// around LN was ~13% of the BERT-base step). These kernels read/write
// bf16 directly with fp32 accumulation, and fold the residual add
// y = LN(a + b) into the same pass (saving the separate add kernel AND
// its extra HBM round trip). Saved for backward: the summed input u
// (bf16, doubles as the residual stream), mean and rstd (fp32 per row).
//
// Shapes: x [N, H] rows normalized over H; H <= 256 * LN_MAX_PER_THREAD
// and H % 2 == 0 (bf16 pairs). One 256-thread block per row; row sums
// via DPP 16-lane reduction + 2 cross-group shuffles + LDS across the 4
// waves.
#include "common.h"

#define LN_MAX_PER_THREAD 16  // H up to 4096

typedef short ln_bf16x2 __attribute__((ext_vector_type(2)));

__device__ __forceinline__ float ln_wave_sum(float v) {
  v = dpp16_sum(v);
  v += __shfl_xor(v, 16, 64);
  v += __shfl_xor(v, 32, 64);
  return v;
}

// block-wide sum of two values at once (saves one barrier round)
__device__ __forceinline__ void ln_block_sum2(float& a, float& b,
                                              float* lds /* >= 8 */) {
  int w = threadIdx.x >> 6;
  a = ln_wave_sum(a);
  b = ln_wave_sum(b);
  if ((threadIdx.x & 63) == 0) {
    lds[w] = a;
    lds[4 + w] = b;
  }
  __syncthreads();
  a = lds[0] + lds[1] + lds[2] + lds[3];
  b = lds[4] + lds[5] + lds[6] + lds[7];
  __syncthreads();
}

// paired (bf16x2) loads/stores: H must be even (the module guards)
__global__ void
__launch_bounds__(256)
ln_fwd_kernel(const __hip_bfloat16* __restrict__ x,
              const __hip_bfloat16* __restrict__ res,  // nullable
              const float* __restrict__ gamma, const float* __restrict__ beta,
              __hip_bfloat16* __restrict__ y,
              __hip_bfloat16* __restrict__ u_out,      // nullable
              float* __restrict__ mean_out, float* __restrict__ rstd_out,
              long N, int H, float eps) {
  __shared__ float lds[8];
  long row = blockIdx.x;
  if (row >= N) return;
  const __hip_bfloat162* xr =
      reinterpret_cast<const __hip_bfloat162*>(x + row * H);
  const __hip_bfloat162* rr = res
      ? reinterpret_cast<const __hip_bfloat162*>(res + row * H) : nullptr;
  int H2 = H >> 1;
  float2 v[LN_MAX_PER_THREAD / 2];
  int nper = (H2 + 255) >> 8;
  float acc = 0.f, acc2 = 0.f;
  for (int k = 0; k < nper; ++k) {
    int i = threadIdx.x + (k << 8);
    float2 u = {0.f, 0.f};
    if (i < H2) {
      u = __bfloat1622float2(xr[i]);
      if (rr) {
        float2 r2 = __bfloat1622float2(rr[i]);
        u.x += r2.x;
        u.y += r2.y;
      }
    }
    v[k] = u;
    acc += u.x + u.y;
    acc2 += u.x * u.x + u.y * u.y;
  }
  ln_block_sum2(acc, acc2, lds);
  float mean = acc / H;
  float var = fmaxf(acc2 / H - mean * mean, 0.f);
  float rstd = rsqrtf(var + eps);
  if (threadIdx.x == 0) {
    mean_out[row] = mean;
    rstd_out[row] = rstd;
  }
  __hip_bfloat162* yr = reinterpret_cast<__hip_bfloat162*>(y + row * H);
  __hip_bfloat162* ur = u_out
      ? reinterpret_cast<__hip_bfloat162*>(u_out + row * H) : nullptr;
  const float2* g2 = reinterpret_cast<const float2*>(gamma);
  const float2* b2 = reinterpret_cast<const float2*>(beta);
  for (int k = 0; k < nper; ++k) {
    int i = threadIdx.x + (k << 8);
    if (i < H2) {
      float2 gg = g2[i], bb = b2[i];
      float2 o = {(v[k].x - mean) * rstd * gg.x + bb.x,
                  (v[k].y - mean) * rstd * gg.y + bb.y};
      yr[i] = __float22bfloat162_rn(o);
      if (ur) ur[i] = __float22bfloat162_rn(v[k]);
    }
  }
}

// dx = rstd * (g - mean(g) - xhat * mean(g*xhat)), g = dy*gamma (fp32)
__global__ void
__launch_bounds__(256)
ln_bwd_dx_kernel(const __hip_bfloat16* __restrict__ dy,
                 const __hip_bfloat16* __restrict__ u,
                 const float* __restrict__ gamma,
                 const float* __restrict__ mean_in,
                 const float* __restrict__ rstd_in,
                 __hip_bfloat16* __restrict__ dx, long N, int H) {
  __shared__ float lds[8];
  long row = blockIdx.x;
  if (row >= N) return;
  const __hip_bfloat162* dyr =
      reinterpret_cast<const __hip_bfloat162*>(dy + row * H);
  const __hip_bfloat162* ur =
      reinterpret_cast<const __hip_bfloat162*>(u + row * H);
  const float2* g2 = reinterpret_cast<const float2*>(gamma);
  float mean = mean_in[row], rstd = rstd_in[row];
  int H2 = H >> 1;
  float2 g[LN_MAX_PER_THREAD / 2], xh[LN_MAX_PER_THREAD / 2];
  int nper = (H2 + 255) >> 8;
  float c1 = 0.f, c2 = 0.f;
  for (int k = 0; k < nper; ++k) {
    int i = threadIdx.x + (k << 8);
    float2 gv = {0.f, 0.f}, xv = {0.f, 0.f};
    if (i < H2) {
      float2 d2 = __bfloat1622float2(dyr[i]);
      float2 u2 = __bfloat1622float2(ur[i]);
      float2 gg = g2[i];
      gv = {d2.x * gg.x, d2.y * gg.y};
      xv = {(u2.x - mean) * rstd, (u2.y - mean) * rstd};
    }
    g[k] = gv;
    xh[k] = xv;
    c1 += gv.x + gv.y;
    c2 += gv.x * xv.x + gv.y * xv.y;
  }
  ln_block_sum2(c1, c2, lds);
  c1 /= H;
  c2 /= H;
  __hip_bfloat162* dxr = reinterpret_cast<__hip_bfloat162*>(dx + row * H);
  for (int k = 0; k < nper; ++k) {
    int i = threadIdx.x + (k << 8);
    if (i < H2) {
      float2 o = {rstd * (g[k].x - c1 - xh[k].x * c2),
                  rstd * (g[k].y - c1 - xh[k].y * c2)};
      dxr[i] = __float22bfloat162_rn(o);
    }
  }
}

// per-chunk partial dgamma/dbeta: grid (H/64, nchunks); 256 threads =
// 64 cols x 4 row-lanes; partials [nchunks, 2, H] fp32 summed by caller.
__global__ void
__launch_bounds__(256)
ln_bwd_gb_kernel(const __hip_bfloat16* __restrict__ dy,
                 const __hip_bfloat16* __restrict__ u,
                 const float* __restrict__ mean_in,
                 const float* __restrict__ rstd_in,
                 float* __restrict__ partials, long N, int H,
                 long rows_per_chunk) {
  __shared__ float pg[4][64];
  __shared__ float pb[4][64];
  int col = (blockIdx.x << 6) + (threadIdx.x & 63);
  int rl = threadIdx.x >> 6;  // row-lane 0..3
  long r0 = (long)blockIdx.y * rows_per_chunk;
  long r1 = min(r0 + rows_per_chunk, N);
  float sg = 0.f, sb = 0.f;
  if (col < H) {
    for (long r = r0 + rl; r < r1; r += 4) {
      float d = __bfloat162float(dy[r * H + col]);
      float xh = (__bfloat162float(u[r * H + col]) - mean_in[r]) * rstd_in[r];
      sg += d * xh;
      sb += d;
    }
  }
  pg[rl][threadIdx.x & 63] = sg;
  pb[rl][threadIdx.x & 63] = sb;
  __syncthreads();
  if (rl == 0 && col < H) {
    int c = threadIdx.x & 63;
    float tg = pg[0][c] + pg[1][c] + pg[2][c] + pg[3][c];
    float tb = pb[0][c] + pb[1][c] + pb[2][c] + pb[3][c];
    partials[((long)blockIdx.y * 2) * H + col] = tg;
    partials[((long)blockIdx.y * 2 + 1) * H + col] = tb;
  }
}

// column sum of a bf16 [N, H] matrix -> fp32 partials [nchunks, H]
// (bias gradients: torch's generic reduce_kernel<BFloat16> ran at
// ~0.35 TB/s on these shapes; this streams rows coalesced per column
// block and lets the caller sum the small partial matrix).
__global__ void
__launch_bounds__(256)
col_sum_kernel(const __hip_bfloat16* __restrict__ x,
               float* __restrict__ partials, long N, long H,
               long rows_per_chunk) {
  __shared__ float ps[4][64];
  long col = ((long)blockIdx.x << 6) + (threadIdx.x & 63);
  int rl = threadIdx.x >> 6;  // row-lane 0..3
  long r0 = (long)blockIdx.y * rows_per_chunk;
  long r1 = min(r0 + rows_per_chunk, N);
  float s = 0.f;
  if (col < H) {
    for (long r = r0 + rl; r < r1; r += 4) {
      s += __bfloat162float(x[r * H + col]);
    }
  }
  ps[rl][threadIdx.x & 63] = s;
  __syncthreads();
  if (rl == 0 && col < H) {
    int c = threadIdx.x & 63;
    partials[(long)blockIdx.y * H + col] =
        ps[0][c] + ps[1][c] + ps[2][c] + ps[3][c];
  }
}
