// Fused optimizer-apply kernels over flat bucket buffers — the MI355X
// equivalents of TF's ResourceApplyGradientDescent / ResourceApplyKerasMomentum /
// ResourceApplyAdam ops (reference table: autodist/kernel/common/op_info.py:24-68).
//
// One launch updates an ENTIRE gradient bucket (param/grad/state are flat,
// contiguous views — see parallel/buckets.py), so the optimizer step costs
// exactly the read+write HBM traffic of its operands. Kernels are elementwise
// and HBM3E-bound: float4 (16 B/lane) access, 256-thread blocks (4 waves),
// grid-stride loop sized to fill all 8 XCDs.
#include "common.h"
#include <cstdint>

// ---------------------------------------------------------------- SGD
// d = g (+ wd*p); buf = mom*buf + (1-damp)*d (or = d on first step);
// d = nesterov ? d + mom*buf : buf; p -= lr*d
template <bool HAS_MOMENTUM>
__global__ void fused_sgd_kernel(float* __restrict__ p,
                                 const float* __restrict__ g,
                                 float* __restrict__ buf, long n, float lr,
                                 float momentum, float dampening,
                                 float weight_decay, int nesterov,
                                 int first_step, int maximize) {
  long stride = (long)gridDim.x * blockDim.x * 4;
  for (long base = ((long)blockIdx.x * blockDim.x + threadIdx.x) * 4; base < n;
       base += stride) {
    if (base + 3 < n) {
      float4 pv = *reinterpret_cast<float4*>(p + base);
      float4 gv = *reinterpret_cast<const float4*>(g + base);
      float d[4] = {gv.x, gv.y, gv.z, gv.w};
      float pr[4] = {pv.x, pv.y, pv.z, pv.w};
#pragma unroll
      for (int k = 0; k < 4; ++k) {
        if (maximize) d[k] = -d[k];
        d[k] += weight_decay * pr[k];
      }
      if (HAS_MOMENTUM) {
        float4 bv = *reinterpret_cast<float4*>(buf + base);
        float br[4] = {bv.x, bv.y, bv.z, bv.w};
#pragma unroll
        for (int k = 0; k < 4; ++k) {
          br[k] = first_step ? d[k] : momentum * br[k] + (1.f - dampening) * d[k];
          d[k] = nesterov ? d[k] + momentum * br[k] : br[k];
        }
        *reinterpret_cast<float4*>(buf + base) =
            make_float4(br[0], br[1], br[2], br[3]);
      }
#pragma unroll
      for (int k = 0; k < 4; ++k) pr[k] -= lr * d[k];
      *reinterpret_cast<float4*>(p + base) =
          make_float4(pr[0], pr[1], pr[2], pr[3]);
    } else {
      for (long i = base; i < n; ++i) {
        float d = maximize ? -g[i] : g[i];
        d += weight_decay * p[i];
        if (HAS_MOMENTUM) {
          float b = first_step ? d : momentum * buf[i] + (1.f - dampening) * d;
          buf[i] = b;
          d = nesterov ? d + momentum * b : b;
        }
        p[i] -= lr * d;
      }
    }
  }
}

// ---------------------------------------------------------------- Adam/AdamW
// adamw: p *= (1 - lr*wd) else g += wd*p
// m = b1*m + (1-b1)*g ; v = b2*v + (1-b2)*g^2
// p -= lr/bc1 * m / (sqrt(v)/sqrt(bc2) + eps)
__global__ void fused_adam_kernel(float* __restrict__ p,
                                  const float* __restrict__ g,
                                  float* __restrict__ m, float* __restrict__ v,
                                  long n, float lr, float beta1, float beta2,
                                  float eps, float weight_decay, int adamw,
                                  float bc1, float sqrt_bc2, int maximize) {
  long stride = (long)gridDim.x * blockDim.x * 4;
  float step_size = lr / bc1;
  float wd_mul = 1.f - lr * weight_decay;
  for (long base = ((long)blockIdx.x * blockDim.x + threadIdx.x) * 4; base < n;
       base += stride) {
    long lim = base + 4 <= n ? 4 : n - base;
    if (lim == 4) {
      float4 pv = *reinterpret_cast<float4*>(p + base);
      float4 gv = *reinterpret_cast<const float4*>(g + base);
      float4 mv = *reinterpret_cast<float4*>(m + base);
      float4 vv = *reinterpret_cast<float4*>(v + base);
      float pr[4] = {pv.x, pv.y, pv.z, pv.w};
      float gr[4] = {gv.x, gv.y, gv.z, gv.w};
      float mr[4] = {mv.x, mv.y, mv.z, mv.w};
      float vr[4] = {vv.x, vv.y, vv.z, vv.w};
#pragma unroll
      for (int k = 0; k < 4; ++k) {
        if (maximize) gr[k] = -gr[k];
        if (adamw) pr[k] *= wd_mul; else gr[k] += weight_decay * pr[k];
        mr[k] = beta1 * mr[k] + (1.f - beta1) * gr[k];
        vr[k] = beta2 * vr[k] + (1.f - beta2) * gr[k] * gr[k];
        pr[k] -= step_size * mr[k] / (sqrtf(vr[k]) / sqrt_bc2 + eps);
      }
      *reinterpret_cast<float4*>(p + base) = make_float4(pr[0], pr[1], pr[2], pr[3]);
      *reinterpret_cast<float4*>(m + base) = make_float4(mr[0], mr[1], mr[2], mr[3]);
      *reinterpret_cast<float4*>(v + base) = make_float4(vr[0], vr[1], vr[2], vr[3]);
    } else {
      for (long i = base; i < base + lim; ++i) {
        float gr = maximize ? -g[i] : g[i];
        float pr = p[i];
        if (adamw) pr *= wd_mul; else gr += weight_decay * pr;
        float mr = beta1 * m[i] + (1.f - beta1) * gr;
        float vr = beta2 * v[i] + (1.f - beta2) * gr * gr;
        p[i] = pr - step_size * mr / (sqrtf(vr) / sqrt_bc2 + eps);
        m[i] = mr;
        v[i] = vr;
      }
    }
  }
}

// ------------------------------------------------- compressor cast kernels
// scale+cast fp32 -> bf16 (wire), 8 bf16 out per lane iteration.
__global__ void scale_cast_bf16_kernel(const float* __restrict__ in,
                                       __hip_bfloat16* __restrict__ out,
                                       long n, float scale) {
  long stride = (long)gridDim.x * blockDim.x * 4;
  for (long base = ((long)blockIdx.x * blockDim.x + threadIdx.x) * 4; base < n;
       base += stride) {
    if (base + 3 < n) {
      float4 v = *reinterpret_cast<const float4*>(in + base);
      __hip_bfloat162 lo = __float22bfloat162_rn({v.x * scale, v.y * scale});
      __hip_bfloat162 hi = __float22bfloat162_rn({v.z * scale, v.w * scale});
      *reinterpret_cast<__hip_bfloat162*>(out + base) = lo;
      *reinterpret_cast<__hip_bfloat162*>(out + base + 2) = hi;
    } else {
      for (long i = base; i < n; ++i) out[i] = __float2bfloat16(in[i] * scale);
    }
  }
}

// cast bf16 wire back to fp32
__global__ void cast_back_f32_kernel(const __hip_bfloat16* __restrict__ in,
                                     float* __restrict__ out, long n) {
  long stride = (long)gridDim.x * blockDim.x * 4;
  for (long base = ((long)blockIdx.x * blockDim.x + threadIdx.x) * 4; base < n;
       base += stride) {
    if (base + 3 < n) {
      __hip_bfloat162 lo = *reinterpret_cast<const __hip_bfloat162*>(in + base);
      __hip_bfloat162 hi = *reinterpret_cast<const __hip_bfloat162*>(in + base + 2);
      float2 l = __bfloat1622float2(lo);
      float2 h = __bfloat1622float2(hi);
      *reinterpret_cast<float4*>(out + base) = make_float4(l.x, l.y, h.x, h.y);
    } else {
      for (long i = base; i < n; ++i) out[i] = __bfloat162float(in[i]);
    }
  }
}

// fused error-feedback compress: flat += err; wire = bf16(flat*scale);
// err = flat - fp32(wire)/scale   (one pass, reference compressor.py:204-205)
__global__ void ef_compress_kernel(float* __restrict__ flat,
                                   float* __restrict__ err,
                                   __hip_bfloat16* __restrict__ wire, long n,
                                   float scale) {
  long stride = (long)gridDim.x * blockDim.x;
  for (long i = (long)blockIdx.x * blockDim.x + threadIdx.x; i < n;
       i += stride) {
    float f = flat[i] + err[i];
    __hip_bfloat16 w = __float2bfloat16(f * scale);
    wire[i] = w;
    err[i] = f - __bfloat162float(w) / scale;
    flat[i] = f;
  }
}

// --------------------------------------------- sparse segment accumulate
// scatter-add rows: out[idx[r]] += vals[r] for row-sparse gradients
// (the SparseConditionalAccumulator dedup-sum, reference
// ps_synchronizer.py:498-535). One thread per (row, col) element.
__global__ void scatter_add_rows_kernel(float* __restrict__ out,
                                        const int64_t* __restrict__ idx,
                                        const float* __restrict__ vals,
                                        long nnz, long dim) {
  long total = nnz * dim;
  long stride = (long)gridDim.x * blockDim.x;
  for (long i = (long)blockIdx.x * blockDim.x + threadIdx.x; i < total;
       i += stride) {
    long r = i / dim, c = i % dim;
    atomicAdd(out + idx[r] * dim + c, vals[i]);
  }
}

// rows gather: out[r] = src[idx[r]] (embedding shard gather)
__global__ void gather_rows_kernel(const float* __restrict__ src,
                                   const int64_t* __restrict__ idx,
                                   float* __restrict__ out, long nrows,
                                   long dim) {
  long total = nrows * dim;
  long stride = (long)gridDim.x * blockDim.x;
  for (long i = (long)blockIdx.x * blockDim.x + threadIdx.x; i < total;
       i += stride) {
    long r = i / dim, c = i % dim;
    out[i] = src[idx[r] * dim + c];
  }
}

// ---------------------------------------------------------------- Adagrad
// g' = (max? -g : g) + wd*p ; sum += g'^2 ; p -= clr * g'/(sqrt(sum)+eps)
// (clr = lr / (1 + (step-1)*lr_decay), host-computed)
__global__ void fused_adagrad_kernel(float* __restrict__ p,
                                     const float* __restrict__ g,
                                     float* __restrict__ sum, long n,
                                     float clr, float eps,
                                     float weight_decay, int maximize) {
  long stride = (long)gridDim.x * blockDim.x * 4;
  for (long base = ((long)blockIdx.x * blockDim.x + threadIdx.x) * 4;
       base < n; base += stride) {
    if (base + 3 < n) {
      float4 pv = *reinterpret_cast<float4*>(p + base);
      float4 gv = *reinterpret_cast<const float4*>(g + base);
      float4 sv = *reinterpret_cast<float4*>(sum + base);
      float pr[4] = {pv.x, pv.y, pv.z, pv.w};
      float gr[4] = {gv.x, gv.y, gv.z, gv.w};
      float sr[4] = {sv.x, sv.y, sv.z, sv.w};
#pragma unroll
      for (int k = 0; k < 4; ++k) {
        float d = maximize ? -gr[k] : gr[k];
        d += weight_decay * pr[k];
        sr[k] += d * d;
        pr[k] -= clr * d / (sqrtf(sr[k]) + eps);
      }
      *reinterpret_cast<float4*>(sum + base) =
          make_float4(sr[0], sr[1], sr[2], sr[3]);
      *reinterpret_cast<float4*>(p + base) =
          make_float4(pr[0], pr[1], pr[2], pr[3]);
    } else {
      for (long i = base; i < n; ++i) {
        float d = maximize ? -g[i] : g[i];
        d += weight_decay * p[i];
        sum[i] += d * d;
        p[i] -= clr * d / (sqrtf(sum[i]) + eps);
      }
    }
  }
}

// ---------------------------------------------------------------- RMSprop
// g' = (max? -g : g) + wd*p ; sq = a*sq + (1-a)g'^2 ;
// centered: ga = ga + (1-a)(g'-ga), denom = sqrt(sq - ga^2)+eps
//           else denom = sqrt(sq)+eps
// momentum: buf = mom*buf + g'/denom, p -= lr*buf ; else p -= lr*g'/denom
__global__ void fused_rmsprop_kernel(float* __restrict__ p,
                                     const float* __restrict__ g,
                                     float* __restrict__ sq,
                                     float* __restrict__ ga,   // nullable
                                     float* __restrict__ buf,  // nullable
                                     long n, float lr, float alpha,
                                     float eps, float weight_decay,
                                     float momentum, int maximize) {
  long stride = (long)gridDim.x * blockDim.x;
  for (long i = (long)blockIdx.x * blockDim.x + threadIdx.x; i < n;
       i += stride) {
    float d = maximize ? -g[i] : g[i];
    d += weight_decay * p[i];
    float s = alpha * sq[i] + (1.f - alpha) * d * d;
    sq[i] = s;
    float denom;
    if (ga != nullptr) {
      float a = ga[i] + (1.f - alpha) * (d - ga[i]);
      ga[i] = a;
      denom = sqrtf(s - a * a) + eps;
    } else {
      denom = sqrtf(s) + eps;
    }
    if (buf != nullptr) {
      float b = momentum * buf[i] + d / denom;
      buf[i] = b;
      p[i] -= lr * b;
    } else {
      p[i] -= lr * d / denom;
    }
  }
}
