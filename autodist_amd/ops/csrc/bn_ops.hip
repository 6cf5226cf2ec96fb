// Fused NHWC BatchNorm kernels for gfx950 — the hot elementwise path of the
// ResNet benchmark (BASELINE headline). Replaces MIOpen's unfused
// BatchNorm + separate residual-add + ReLU (+ fp32 autocast conversions)
// with bf16-native fused kernels:
//
//   fwd:  reduce(sum,sumsq) -> finalize(scale/shift, running stats)
//         -> apply: y = relu(x*scale + shift [+ residual])
//   bwd:  reduce(sum_dz, sum_dz*xhat; dz = relu-masked dy)
//         -> finalize(k1..k3, dweight, dbias)
//         -> apply: dx = k1*(dz - k2 - xhat*k3)  [+ dres = dz]
//
// Layout: channels_last => x is M x C row-major (M = N*H*W), C contiguous.
// Each thread owns a FIXED 8-channel group (bf16x8 = 16 B loads), so
// per-channel coefficients live in registers, not LDS lookups. Reductions:
// in-register accumulate over rows -> LDS cross-row reduce -> one fp32
// atomicAdd per channel per block. fp32 math throughout.
#include "common.h"

#define BN_VEC 8  // channels per thread (8 x bf16 = 16B; C must be mult of 8)

// vector load/store helpers: 8 channels
__device__ __forceinline__ void load8(const __hip_bfloat16* p, float* v) {
  // 16B load
  const __hip_bfloat162* p2 = reinterpret_cast<const __hip_bfloat162*>(p);
  float2 a = __bfloat1622float2(p2[0]);
  float2 b = __bfloat1622float2(p2[1]);
  float2 c = __bfloat1622float2(p2[2]);
  float2 d = __bfloat1622float2(p2[3]);
  v[0] = a.x; v[1] = a.y; v[2] = b.x; v[3] = b.y;
  v[4] = c.x; v[5] = c.y; v[6] = d.x; v[7] = d.y;
}
__device__ __forceinline__ void store8(__hip_bfloat16* p, const float* v) {
  __hip_bfloat162* p2 = reinterpret_cast<__hip_bfloat162*>(p);
  p2[0] = __float22bfloat162_rn({v[0], v[1]});
  p2[1] = __float22bfloat162_rn({v[2], v[3]});
  p2[2] = __float22bfloat162_rn({v[4], v[5]});
  p2[3] = __float22bfloat162_rn({v[6], v[7]});
}
__device__ __forceinline__ void load8(const float* p, float* v) {
  float4 a = *reinterpret_cast<const float4*>(p);
  float4 b = *reinterpret_cast<const float4*>(p + 4);
  v[0] = a.x; v[1] = a.y; v[2] = a.z; v[3] = a.w;
  v[4] = b.x; v[5] = b.y; v[6] = b.z; v[7] = b.w;
}
__device__ __forceinline__ void store8(float* p, const float* v) {
  *reinterpret_cast<float4*>(p) = make_float4(v[0], v[1], v[2], v[3]);
  *reinterpret_cast<float4*>(p + 4) = make_float4(v[4], v[5], v[6], v[7]);
}

// ---------------------------------------------------------------- fwd reduce
// Two-stage, atomic-free (deterministic): each block writes one partial row
// partial[blockIdx.x][c]; the finalize kernel sums the gridM rows. (A global
// fp32 atomicAdd design serialized ~gridM-way per channel address — ~200 us
// on small tensors.)
template <typename T>
__global__ void bn_fwd_reduce_kernel(const T* __restrict__ x,
                                     float* __restrict__ partial_sum,
                                     float* __restrict__ partial_sq, long M,
                                     int C) {
  int tx_count = min(C / BN_VEC, (int)blockDim.x);
  int tx = threadIdx.x % tx_count;
  int ty = threadIdx.x / tx_count;
  int rows_per_blk = blockDim.x / tx_count;
  int c0 = (blockIdx.y * tx_count + tx) * BN_VEC;
  if (c0 >= C) return;
  float s[BN_VEC] = {0}, q[BN_VEC] = {0};
  for (long m = (long)blockIdx.x * rows_per_blk + ty; m < M;
       m += (long)gridDim.x * rows_per_blk) {
    float v[BN_VEC];
    load8(x + m * C + c0, v);
#pragma unroll
    for (int k = 0; k < BN_VEC; ++k) { s[k] += v[k]; q[k] += v[k] * v[k]; }
  }
  __shared__ float ls[BLOCK_THREADS * BN_VEC];
  __shared__ float lq[BLOCK_THREADS * BN_VEC];
#pragma unroll
  for (int k = 0; k < BN_VEC; ++k) {
    ls[threadIdx.x * BN_VEC + k] = s[k];
    lq[threadIdx.x * BN_VEC + k] = q[k];
  }
  __syncthreads();
  if (ty == 0) {
    for (int r = 1; r < rows_per_blk; ++r) {
      int o = (r * tx_count + tx) * BN_VEC;
#pragma unroll
      for (int k = 0; k < BN_VEC; ++k) { s[k] += ls[o + k]; q[k] += lq[o + k]; }
    }
    store8(partial_sum + (long)blockIdx.x * C + c0, s);
    store8(partial_sq + (long)blockIdx.x * C + c0, q);
  }
}

// ------------------------------------------------------------ fwd finalize
// scale = w*rstd; shift = b - mean*scale; running stats updated in place.
// block = 64 channels x 4 row-lanes: lanes split the grid_m partial rows so
// small-C layers still read the partial matrix with thousands of threads.
#define FIN_CH 64
#define FIN_LANES 4
__global__ void bn_fwd_finalize_kernel(const float* __restrict__ partial_sum,
                                       const float* __restrict__ partial_sq,
                                       int grid_m,
                                       const float* __restrict__ weight,
                                       const float* __restrict__ bias,
                                       float* __restrict__ running_mean,
                                       float* __restrict__ running_var,
                                       float* __restrict__ save_mean,
                                       float* __restrict__ save_rstd,
                                       float* __restrict__ scale,
                                       float* __restrict__ shift, long M,
                                       int C, float eps, float momentum) {
  int tx = threadIdx.x % FIN_CH;
  int ty = threadIdx.x / FIN_CH;
  int c = blockIdx.x * FIN_CH + tx;
  if (c >= C) return;
  float s = 0.f, q = 0.f;
  for (int r = ty; r < grid_m; r += FIN_LANES) {
    s += partial_sum[(long)r * C + c];
    q += partial_sq[(long)r * C + c];
  }
  __shared__ float ls[FIN_LANES * FIN_CH], lq[FIN_LANES * FIN_CH];
  ls[ty * FIN_CH + tx] = s;
  lq[ty * FIN_CH + tx] = q;
  __syncthreads();
  if (ty != 0) return;
#pragma unroll
  for (int r = 1; r < FIN_LANES; ++r) {
    s += ls[r * FIN_CH + tx];
    q += lq[r * FIN_CH + tx];
  }
  float mean = s / (float)M;
  float var = fmaxf(q / (float)M - mean * mean, 0.f);
  float rstd = rsqrtf(var + eps);
  float sc = weight[c] * rstd;
  save_mean[c] = mean;
  save_rstd[c] = rstd;
  scale[c] = sc;
  shift[c] = bias[c] - mean * sc;
  if (momentum > 0.f) {
    float unbiased = M > 1 ? var * (float)M / (float)(M - 1) : var;
    running_mean[c] = (1.f - momentum) * running_mean[c] + momentum * mean;
    running_var[c] = (1.f - momentum) * running_var[c] + momentum * unbiased;
  }
}

// ------------------------------------------------------------- fwd apply
template <typename T, bool RELU, bool ADD>
__global__ void bn_fwd_apply_kernel(const T* __restrict__ x,
                                    const T* __restrict__ res,
                                    T* __restrict__ y,
                                    const float* __restrict__ scale,
                                    const float* __restrict__ shift, long M,
                                    int C) {
  int tx_count = min(C / BN_VEC, (int)blockDim.x);
  int tx = threadIdx.x % tx_count;
  int ty = threadIdx.x / tx_count;
  int rows_per_blk = blockDim.x / tx_count;
  int c0 = (blockIdx.y * tx_count + tx) * BN_VEC;
  if (c0 >= C) return;
  float sc[BN_VEC], sh[BN_VEC];
  load8(scale + c0, sc);
  load8(shift + c0, sh);
  for (long m = (long)blockIdx.x * rows_per_blk + ty; m < M;
       m += (long)gridDim.x * rows_per_blk) {
    float v[BN_VEC];
    load8(x + m * C + c0, v);
#pragma unroll
    for (int k = 0; k < BN_VEC; ++k) v[k] = v[k] * sc[k] + sh[k];
    if (ADD) {
      float r[BN_VEC];
      load8(res + m * C + c0, r);
#pragma unroll
      for (int k = 0; k < BN_VEC; ++k) v[k] += r[k];
    }
    if (RELU) {
#pragma unroll
      for (int k = 0; k < BN_VEC; ++k) v[k] = fmaxf(v[k], 0.f);
    }
    store8(y + m * C + c0, v);
  }
}

// ------------------------------------------------------------- bwd reduce
// dz = RELU ? (y>0 ? dy : 0) : dy ; accumulate sum_dz, sum_dz*xhat.
template <typename T, bool RELU>
__global__ void bn_bwd_reduce_kernel(const T* __restrict__ x,
                                     const T* __restrict__ dy,
                                     const T* __restrict__ y,
                                     const float* __restrict__ save_mean,
                                     const float* __restrict__ save_rstd,
                                     float* __restrict__ partial_dz,
                                     float* __restrict__ partial_dzxh, long M,
                                     int C) {
  int tx_count = min(C / BN_VEC, (int)blockDim.x);
  int tx = threadIdx.x % tx_count;
  int ty = threadIdx.x / tx_count;
  int rows_per_blk = blockDim.x / tx_count;
  int c0 = (blockIdx.y * tx_count + tx) * BN_VEC;
  if (c0 >= C) return;
  float mean[BN_VEC], rstd[BN_VEC];
  load8(save_mean + c0, mean);
  load8(save_rstd + c0, rstd);
  float s[BN_VEC] = {0}, t[BN_VEC] = {0};
  for (long m = (long)blockIdx.x * rows_per_blk + ty; m < M;
       m += (long)gridDim.x * rows_per_blk) {
    float xv[BN_VEC], dv[BN_VEC];
    load8(x + m * C + c0, xv);
    load8(dy + m * C + c0, dv);
    if (RELU) {
      float yv[BN_VEC];
      load8(y + m * C + c0, yv);
#pragma unroll
      for (int k = 0; k < BN_VEC; ++k) dv[k] = yv[k] > 0.f ? dv[k] : 0.f;
    }
#pragma unroll
    for (int k = 0; k < BN_VEC; ++k) {
      s[k] += dv[k];
      t[k] += dv[k] * (xv[k] - mean[k]) * rstd[k];
    }
  }
  __shared__ float ls[BLOCK_THREADS * BN_VEC];
  __shared__ float lt[BLOCK_THREADS * BN_VEC];
#pragma unroll
  for (int k = 0; k < BN_VEC; ++k) {
    ls[threadIdx.x * BN_VEC + k] = s[k];
    lt[threadIdx.x * BN_VEC + k] = t[k];
  }
  __syncthreads();
  if (ty == 0) {
    for (int r = 1; r < rows_per_blk; ++r) {
      int o = (r * tx_count + tx) * BN_VEC;
#pragma unroll
      for (int k = 0; k < BN_VEC; ++k) { s[k] += ls[o + k]; t[k] += lt[o + k]; }
    }
    store8(partial_dz + (long)blockIdx.x * C + c0, s);
    store8(partial_dzxh + (long)blockIdx.x * C + c0, t);
  }
}

// ------------------------------------------------------------ bwd finalize
// k1 = w*rstd ; k2 = sum_dz/M ; k3 = sum_dzxh/M ; dweight = sum_dzxh ;
// dbias = sum_dz
__global__ void bn_bwd_finalize_kernel(const float* __restrict__ partial_dz,
                                       const float* __restrict__ partial_dzxh,
                                       int grid_m,
                                       const float* __restrict__ weight,
                                       const float* __restrict__ save_rstd,
                                       float* __restrict__ k1,
                                       float* __restrict__ k2,
                                       float* __restrict__ k3,
                                       float* __restrict__ dweight,
                                       float* __restrict__ dbias, long M,
                                       int C) {
  int tx = threadIdx.x % FIN_CH;
  int ty = threadIdx.x / FIN_CH;
  int c = blockIdx.x * FIN_CH + tx;
  if (c >= C) return;
  float sdz = 0.f, sdzxh = 0.f;
  for (int r = ty; r < grid_m; r += FIN_LANES) {
    sdz += partial_dz[(long)r * C + c];
    sdzxh += partial_dzxh[(long)r * C + c];
  }
  __shared__ float ls[FIN_LANES * FIN_CH], lq[FIN_LANES * FIN_CH];
  ls[ty * FIN_CH + tx] = sdz;
  lq[ty * FIN_CH + tx] = sdzxh;
  __syncthreads();
  if (ty != 0) return;
#pragma unroll
  for (int r = 1; r < FIN_LANES; ++r) {
    sdz += ls[r * FIN_CH + tx];
    sdzxh += lq[r * FIN_CH + tx];
  }
  k1[c] = weight[c] * save_rstd[c];
  k2[c] = sdz / (float)M;
  k3[c] = sdzxh / (float)M;
  dweight[c] = sdzxh;
  dbias[c] = sdz;
}

// -------------------------------------------------------------- bwd apply
// dx = k1*(dz - k2 - xhat*k3); dres = dz (if ADD)
template <typename T, bool RELU, bool ADD>
__global__ void bn_bwd_apply_kernel(const T* __restrict__ x,
                                    const T* __restrict__ dy,
                                    const T* __restrict__ y,
                                    const float* __restrict__ save_mean,
                                    const float* __restrict__ save_rstd,
                                    const float* __restrict__ k1,
                                    const float* __restrict__ k2,
                                    const float* __restrict__ k3,
                                    T* __restrict__ dx, T* __restrict__ dres,
                                    long M, int C) {
  int tx_count = min(C / BN_VEC, (int)blockDim.x);
  int tx = threadIdx.x % tx_count;
  int ty = threadIdx.x / tx_count;
  int rows_per_blk = blockDim.x / tx_count;
  int c0 = (blockIdx.y * tx_count + tx) * BN_VEC;
  if (c0 >= C) return;
  float mean[BN_VEC], rstd[BN_VEC], a1[BN_VEC], a2[BN_VEC], a3[BN_VEC];
  load8(save_mean + c0, mean);
  load8(save_rstd + c0, rstd);
  load8(k1 + c0, a1);
  load8(k2 + c0, a2);
  load8(k3 + c0, a3);
  for (long m = (long)blockIdx.x * rows_per_blk + ty; m < M;
       m += (long)gridDim.x * rows_per_blk) {
    float xv[BN_VEC], dv[BN_VEC];
    load8(x + m * C + c0, xv);
    load8(dy + m * C + c0, dv);
    if (RELU) {
      float yv[BN_VEC];
      load8(y + m * C + c0, yv);
#pragma unroll
      for (int k = 0; k < BN_VEC; ++k) dv[k] = yv[k] > 0.f ? dv[k] : 0.f;
    }
    if (ADD) store8(dres + m * C + c0, dv);
    float o[BN_VEC];
#pragma unroll
    for (int k = 0; k < BN_VEC; ++k) {
      float xhat = (xv[k] - mean[k]) * rstd[k];
      o[k] = a1[k] * (dv[k] - a2[k] - xhat * a3[k]);
    }
    store8(dx + m * C + c0, o);
  }
}
