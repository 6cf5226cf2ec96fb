// Python bindings for the gfx950 HIP ops (torch extension).
// Launches on the caller's current HIP stream so the engine's comm-stream /
// compute-stream ordering (parallel/engine.py) applies to these kernels too.
#include <torch/extension.h>
#include <ATen/cuda/CUDAContext.h>

#include "multi_tensor.hip"
#include "bn_ops.hip"
#include "psgd_gemm.hip"
#include "mfma_probe.hip"
#include "attention.hip"
#include "attention_bwd.hip"
#include "ln_ops.hip"
#include "ce_ops.hip"

namespace {

inline hipStream_t cur_stream() {
  return at::cuda::getCurrentCUDAStream().stream();
}

// ---------------------------------------------------------------- fused BN
struct BNGeom {
  long M;
  int C;
  int tx_count, rows_per_blk, grid_c;
};

BNGeom bn_geom(const torch::Tensor& x) {
  TORCH_CHECK(x.dim() == 4, "BN expects NCHW-logical channels_last tensor");
  int C = (int)x.size(1);
  TORCH_CHECK(C % BN_VEC == 0, "C must be a multiple of 8, got ", C);
  TORCH_CHECK(x.is_contiguous(at::MemoryFormat::ChannelsLast),
              "x must be channels_last");
  BNGeom g;
  g.M = x.size(0) * x.size(2) * x.size(3);
  g.C = C;
  g.tx_count = std::min(C / BN_VEC, BLOCK_THREADS);
  g.rows_per_blk = BLOCK_THREADS / g.tx_count;
  g.grid_c = (C / BN_VEC + g.tx_count - 1) / g.tx_count;
  return g;
}

inline int bn_grid_m(const BNGeom& g, long target_blocks) {
  long gm = (g.M + g.rows_per_blk - 1) / g.rows_per_blk;
  long cap = std::max<long>(target_blocks / std::max(g.grid_c, 1), 1);
  return (int)std::min(gm, cap);
}

// max partial rows for the two-stage BN reductions (2 blocks per CU).
// NOTE: raising this to 2048 (8 blocks/CU) was MEASURED SLOWER on every
// ResNet shape (tools/bn_microbench.py, 2026-09-14): the reduce stage is
// not occupancy-bound and the larger partial matrix taxes the finalize.
#define BN_GM_MAX 512

inline int bn_reduce_gm(const BNGeom& g) {
  long rows = (g.M + g.rows_per_blk - 1) / g.rows_per_blk;
  long gm = (rows + 31) / 32;
  if (gm < 64) gm = 64;
  if (gm > BN_GM_MAX) gm = BN_GM_MAX;
  if (gm > rows) gm = rows;
  return (int)gm;
}

std::vector<torch::Tensor> bn_fwd_train(torch::Tensor x, torch::Tensor weight,
                                        torch::Tensor bias,
                                        torch::Tensor running_mean,
                                        torch::Tensor running_var,
                                        c10::optional<torch::Tensor> res,
                                        double eps, double momentum,
                                        bool relu,
                                        c10::optional<torch::Tensor> ws) {
  // ws: persistent per-module workspace [2*BN_GM_MAX + 4, C] fp32:
  //   rows [0, GM)               partial sums (per reduce block)
  //   rows [BN_GM_MAX, BN_GM_MAX+GM) partial sumsq
  //   rows 2*BN_GM_MAX + {0,1,2,3}: save_mean, save_rstd, scale, shift
  auto g = bn_geom(x);
  auto fopt = weight.options().dtype(torch::kFloat32);
  torch::Tensor w6 = ws.has_value() ? *ws
      : torch::empty({2 * BN_GM_MAX + 4, (long)g.C}, fopt);
  TORCH_CHECK(w6.size(0) >= 2 * BN_GM_MAX + 4 && w6.size(1) == g.C &&
              w6.is_contiguous());
  float* wp = w6.data_ptr<float>();
  float* partial_sum = wp;
  float* partial_sq = wp + (long)BN_GM_MAX * g.C;
  auto save_mean = w6[2 * BN_GM_MAX + 0];
  auto save_rstd = w6[2 * BN_GM_MAX + 1];
  auto scale = w6[2 * BN_GM_MAX + 2];
  auto shift = w6[2 * BN_GM_MAX + 3];
  auto y = torch::empty_like(x);
  dim3 block(BLOCK_THREADS);
  int gm = bn_reduce_gm(g);
  dim3 grid_r(gm, g.grid_c);
  dim3 grid_a(bn_grid_m(g, 4096), g.grid_c);
  auto st = cur_stream();

#define BN_FWD_T(T, GET)                                                      \
  {                                                                           \
    hipLaunchKernelGGL((bn_fwd_reduce_kernel<T>), grid_r, block, 0, st,       \
                       GET(x), partial_sum, partial_sq, g.M, g.C);            \
    hipLaunchKernelGGL(bn_fwd_finalize_kernel,                                \
                       dim3((g.C + FIN_CH - 1) / FIN_CH),                     \
                       dim3(FIN_CH * FIN_LANES), 0, st, partial_sum,          \
                       partial_sq, gm,                                        \
                       weight.data_ptr<float>(),                              \
                       bias.data_ptr<float>(), running_mean.data_ptr<float>(),\
                       running_var.data_ptr<float>(),                         \
                       save_mean.data_ptr<float>(),                           \
                       save_rstd.data_ptr<float>(), scale.data_ptr<float>(),  \
                       shift.data_ptr<float>(), g.M, g.C, (float)eps,         \
                       (float)momentum);                                      \
    if (res.has_value()) {                                                    \
      if (relu)                                                               \
        hipLaunchKernelGGL((bn_fwd_apply_kernel<T, true, true>), grid_a,      \
                           block, 0, st, GET(x), GET(*res), GET(y),           \
                           scale.data_ptr<float>(), shift.data_ptr<float>(),  \
                           g.M, g.C);                                         \
      else                                                                    \
        hipLaunchKernelGGL((bn_fwd_apply_kernel<T, false, true>), grid_a,     \
                           block, 0, st, GET(x), GET(*res), GET(y),           \
                           scale.data_ptr<float>(), shift.data_ptr<float>(),  \
                           g.M, g.C);                                         \
    } else {                                                                  \
      if (relu)                                                               \
        hipLaunchKernelGGL((bn_fwd_apply_kernel<T, true, false>), grid_a,     \
                           block, 0, st, GET(x), (const T*)nullptr, GET(y),   \
                           scale.data_ptr<float>(), shift.data_ptr<float>(),  \
                           g.M, g.C);                                         \
      else                                                                    \
        hipLaunchKernelGGL((bn_fwd_apply_kernel<T, false, false>), grid_a,    \
                           block, 0, st, GET(x), (const T*)nullptr, GET(y),   \
                           scale.data_ptr<float>(), shift.data_ptr<float>(),  \
                           g.M, g.C);                                         \
    }                                                                         \
  }

#define GET_BF16(t) reinterpret_cast<__hip_bfloat16*>((t).data_ptr())
#define GET_F32(t) (t).data_ptr<float>()
  if (x.scalar_type() == torch::kBFloat16) {
    BN_FWD_T(__hip_bfloat16, GET_BF16)
  } else {
    TORCH_CHECK(x.scalar_type() == torch::kFloat32);
    BN_FWD_T(float, GET_F32)
  }
  return {y, save_mean, save_rstd};
}

std::vector<torch::Tensor> bn_bwd(torch::Tensor x, torch::Tensor dy,
                                  torch::Tensor y, torch::Tensor save_mean,
                                  torch::Tensor save_rstd,
                                  torch::Tensor weight, bool relu,
                                  bool need_dres,
                                  c10::optional<torch::Tensor> ws) {
  // ws rows: [0,GM) partial_dz, [BN_GM_MAX, BN_GM_MAX+GM) partial_dzxh,
  //          2*BN_GM_MAX + {0,1,2}: k1, k2, k3.
  auto g = bn_geom(x);
  auto fopt = weight.options().dtype(torch::kFloat32);
  torch::Tensor w5 = ws.has_value() ? *ws
      : torch::empty({2 * BN_GM_MAX + 3, (long)g.C}, fopt);
  TORCH_CHECK(w5.size(0) >= 2 * BN_GM_MAX + 3 && w5.size(1) == g.C &&
              w5.is_contiguous());
  float* wp = w5.data_ptr<float>();
  float* partial_dz = wp;
  float* partial_dzxh = wp + (long)BN_GM_MAX * g.C;
  auto k1 = w5[2 * BN_GM_MAX + 0];
  auto k2 = w5[2 * BN_GM_MAX + 1];
  auto k3 = w5[2 * BN_GM_MAX + 2];
  // fresh allocations: returned to autograd as parameter gradients
  auto dweight = torch::empty({g.C}, fopt);
  auto dbias = torch::empty({g.C}, fopt);
  auto dx = torch::empty_like(x);
  torch::Tensor dres;
  if (need_dres) dres = torch::empty_like(x);
  dim3 block(BLOCK_THREADS);
  int gm = bn_reduce_gm(g);
  dim3 grid_r(gm, g.grid_c);
  dim3 grid_a(bn_grid_m(g, 4096), g.grid_c);
  auto st = cur_stream();

#define BN_BWD_T(T, GET)                                                      \
  {                                                                           \
    if (relu)                                                                 \
      hipLaunchKernelGGL((bn_bwd_reduce_kernel<T, true>), grid_r, block, 0,   \
                         st, GET(x), GET(dy), GET(y),                         \
                         save_mean.data_ptr<float>(),                         \
                         save_rstd.data_ptr<float>(),                         \
                         partial_dz, partial_dzxh, g.M, g.C);                 \
    else                                                                      \
      hipLaunchKernelGGL((bn_bwd_reduce_kernel<T, false>), grid_r, block, 0,  \
                         st, GET(x), GET(dy), GET(y),                         \
                         save_mean.data_ptr<float>(),                         \
                         save_rstd.data_ptr<float>(),                         \
                         partial_dz, partial_dzxh, g.M, g.C);                 \
    hipLaunchKernelGGL(bn_bwd_finalize_kernel,                                \
                       dim3((g.C + FIN_CH - 1) / FIN_CH),                     \
                       dim3(FIN_CH * FIN_LANES), 0, st, partial_dz,           \
                       partial_dzxh, gm,                                      \
                       weight.data_ptr<float>(),                              \
                       save_rstd.data_ptr<float>(), k1.data_ptr<float>(),     \
                       k2.data_ptr<float>(), k3.data_ptr<float>(),            \
                       dweight.data_ptr<float>(), dbias.data_ptr<float>(),    \
                       g.M, g.C);                                             \
    T* dres_p = need_dres ? GET(dres) : (T*)nullptr;                          \
    if (relu) {                                                               \
      if (need_dres)                                                          \
        hipLaunchKernelGGL((bn_bwd_apply_kernel<T, true, true>), grid_a,      \
                           block, 0, st, GET(x), GET(dy), GET(y),             \
                           save_mean.data_ptr<float>(),                       \
                           save_rstd.data_ptr<float>(), k1.data_ptr<float>(), \
                           k2.data_ptr<float>(), k3.data_ptr<float>(),        \
                           GET(dx), dres_p, g.M, g.C);                        \
      else                                                                    \
        hipLaunchKernelGGL((bn_bwd_apply_kernel<T, true, false>), grid_a,     \
                           block, 0, st, GET(x), GET(dy), GET(y),             \
                           save_mean.data_ptr<float>(),                       \
                           save_rstd.data_ptr<float>(), k1.data_ptr<float>(), \
                           k2.data_ptr<float>(), k3.data_ptr<float>(),        \
                           GET(dx), dres_p, g.M, g.C);                        \
    } else {                                                                  \
      if (need_dres)                                                          \
        hipLaunchKernelGGL((bn_bwd_apply_kernel<T, false, true>), grid_a,     \
                           block, 0, st, GET(x), GET(dy), GET(y),             \
                           save_mean.data_ptr<float>(),                       \
                           save_rstd.data_ptr<float>(), k1.data_ptr<float>(), \
                           k2.data_ptr<float>(), k3.data_ptr<float>(),        \
                           GET(dx), dres_p, g.M, g.C);                        \
      else                                                                    \
        hipLaunchKernelGGL((bn_bwd_apply_kernel<T, false, false>), grid_a,    \
                           block, 0, st, GET(x), GET(dy), GET(y),             \
                           save_mean.data_ptr<float>(),                       \
                           save_rstd.data_ptr<float>(), k1.data_ptr<float>(), \
                           k2.data_ptr<float>(), k3.data_ptr<float>(),        \
                           GET(dx), dres_p, g.M, g.C);                        \
    }                                                                         \
  }
  if (x.scalar_type() == torch::kBFloat16) {
    BN_BWD_T(__hip_bfloat16, GET_BF16)
  } else {
    TORCH_CHECK(x.scalar_type() == torch::kFloat32);
    BN_BWD_T(float, GET_F32)
  }
  std::vector<torch::Tensor> out = {dx, dweight, dbias};
  if (need_dres) out.push_back(dres);
  return out;
}

void check_f32_flat(const torch::Tensor& t, const char* name) {
  TORCH_CHECK(t.is_cuda(), name, " must be on GPU");
  TORCH_CHECK(t.is_contiguous(), name, " must be contiguous");
  TORCH_CHECK(t.scalar_type() == torch::kFloat32, name, " must be fp32");
}

void fused_sgd(torch::Tensor p, torch::Tensor g,
               c10::optional<torch::Tensor> buf, double lr, double momentum,
               double dampening, double weight_decay, bool nesterov,
               bool first_step, bool maximize) {
  check_f32_flat(p, "param");
  check_f32_flat(g, "grad");
  long n = p.numel();
  dim3 grid(grid_for(n)), block(BLOCK_THREADS);
  if (buf.has_value()) {
    check_f32_flat(*buf, "momentum_buffer");
    hipLaunchKernelGGL(fused_sgd_kernel<true>, grid, block, 0, cur_stream(),
                       p.data_ptr<float>(), g.data_ptr<float>(),
                       buf->data_ptr<float>(), n, (float)lr, (float)momentum,
                       (float)dampening, (float)weight_decay, nesterov,
                       first_step, maximize);
  } else {
    hipLaunchKernelGGL(fused_sgd_kernel<false>, grid, block, 0, cur_stream(),
                       p.data_ptr<float>(), g.data_ptr<float>(), nullptr, n,
                       (float)lr, (float)momentum, (float)dampening,
                       (float)weight_decay, nesterov, first_step, maximize);
  }
}

void fused_adam(torch::Tensor p, torch::Tensor g, torch::Tensor m,
                torch::Tensor v, double lr, double beta1, double beta2,
                double eps, double weight_decay, bool adamw, double bc1,
                double sqrt_bc2, bool maximize) {
  check_f32_flat(p, "param");
  check_f32_flat(g, "grad");
  check_f32_flat(m, "exp_avg");
  check_f32_flat(v, "exp_avg_sq");
  long n = p.numel();
  hipLaunchKernelGGL(fused_adam_kernel, dim3(grid_for(n)),
                     dim3(BLOCK_THREADS), 0, cur_stream(),
                     p.data_ptr<float>(), g.data_ptr<float>(),
                     m.data_ptr<float>(), v.data_ptr<float>(), n, (float)lr,
                     (float)beta1, (float)beta2, (float)eps,
                     (float)weight_decay, adamw, (float)bc1, (float)sqrt_bc2,
                     maximize);
}

void fused_adagrad(torch::Tensor p, torch::Tensor g, torch::Tensor sum,
                   double clr, double eps, double weight_decay,
                   bool maximize) {
  check_f32_flat(p, "param");
  check_f32_flat(g, "grad");
  check_f32_flat(sum, "sum");
  long n = p.numel();
  hipLaunchKernelGGL(fused_adagrad_kernel, dim3(grid_for(n)),
                     dim3(BLOCK_THREADS), 0, cur_stream(),
                     p.data_ptr<float>(), g.data_ptr<float>(),
                     sum.data_ptr<float>(), n, (float)clr, (float)eps,
                     (float)weight_decay, maximize);
}

void fused_rmsprop(torch::Tensor p, torch::Tensor g, torch::Tensor sq,
                   c10::optional<torch::Tensor> ga,
                   c10::optional<torch::Tensor> buf, double lr, double alpha,
                   double eps, double weight_decay, double momentum,
                   bool maximize) {
  check_f32_flat(p, "param");
  check_f32_flat(g, "grad");
  check_f32_flat(sq, "square_avg");
  long n = p.numel();
  float* gap = (ga.has_value() && ga->defined()) ? ga->data_ptr<float>()
                                                 : nullptr;
  float* bufp = (buf.has_value() && buf->defined()) ? buf->data_ptr<float>()
                                                    : nullptr;
  hipLaunchKernelGGL(fused_rmsprop_kernel, dim3(grid_for(n, 1)),
                     dim3(BLOCK_THREADS), 0, cur_stream(),
                     p.data_ptr<float>(), g.data_ptr<float>(),
                     sq.data_ptr<float>(), gap, bufp, n, (float)lr,
                     (float)alpha, (float)eps, (float)weight_decay,
                     (float)momentum, maximize);
}

void scale_cast_bf16(torch::Tensor in, torch::Tensor out, double scale) {
  check_f32_flat(in, "in");
  TORCH_CHECK(out.scalar_type() == torch::kBFloat16 && out.is_contiguous());
  long n = in.numel();
  hipLaunchKernelGGL(scale_cast_bf16_kernel, dim3(grid_for(n)),
                     dim3(BLOCK_THREADS), 0, cur_stream(),
                     in.data_ptr<float>(),
                     reinterpret_cast<__hip_bfloat16*>(out.data_ptr()), n,
                     (float)scale);
}

void cast_back_f32(torch::Tensor in, torch::Tensor out) {
  TORCH_CHECK(in.scalar_type() == torch::kBFloat16 && in.is_contiguous());
  check_f32_flat(out, "out");
  long n = in.numel();
  hipLaunchKernelGGL(cast_back_f32_kernel, dim3(grid_for(n)),
                     dim3(BLOCK_THREADS), 0, cur_stream(),
                     reinterpret_cast<const __hip_bfloat16*>(in.data_ptr()),
                     out.data_ptr<float>(), n);
}

void ef_compress(torch::Tensor flat, torch::Tensor err, torch::Tensor wire,
                 double scale) {
  check_f32_flat(flat, "flat");
  check_f32_flat(err, "err");
  TORCH_CHECK(wire.scalar_type() == torch::kBFloat16);
  long n = flat.numel();
  hipLaunchKernelGGL(ef_compress_kernel, dim3(grid_for(n, 1)),
                     dim3(BLOCK_THREADS), 0, cur_stream(),
                     flat.data_ptr<float>(), err.data_ptr<float>(),
                     reinterpret_cast<__hip_bfloat16*>(wire.data_ptr()), n,
                     (float)scale);
}

std::tuple<torch::Tensor, torch::Tensor> segment_coalesce(
    torch::Tensor indices, torch::Tensor values) {
  TORCH_CHECK(indices.is_cuda() && values.is_cuda());
  TORCH_CHECK(indices.scalar_type() == torch::kInt64);
  auto vals = values.contiguous().to(torch::kFloat32);
  auto [uniq, inverse] = at::_unique(indices, /*sorted=*/true,
                                     /*return_inverse=*/true);
  long dim = values.numel() / std::max<long>(values.size(0), 1);
  auto out_sizes = values.sizes().vec();
  out_sizes[0] = uniq.size(0);
  auto out = torch::zeros(out_sizes, vals.options());
  long nnz = indices.numel();
  if (nnz > 0) {
    hipLaunchKernelGGL(scatter_add_rows_kernel,
                       dim3(grid_for(nnz * dim, 1)), dim3(BLOCK_THREADS), 0,
                       cur_stream(), out.data_ptr<float>(),
                       inverse.contiguous().data_ptr<int64_t>(),
                       vals.data_ptr<float>(), nnz, dim);
  }
  return {uniq, out.to(values.scalar_type())};
}

void scatter_add_rows(torch::Tensor out, torch::Tensor idx,
                      torch::Tensor vals) {
  check_f32_flat(out, "out");
  TORCH_CHECK(idx.scalar_type() == torch::kInt64);
  long nnz = idx.numel();
  long dim = vals.numel() / std::max<long>(nnz, 1);
  if (nnz == 0) return;
  hipLaunchKernelGGL(scatter_add_rows_kernel, dim3(grid_for(nnz * dim, 1)),
                     dim3(BLOCK_THREADS), 0, cur_stream(),
                     out.data_ptr<float>(), idx.contiguous().data_ptr<int64_t>(),
                     vals.contiguous().data_ptr<float>(), nnz, dim);
}

torch::Tensor gather_rows(torch::Tensor src, torch::Tensor idx) {
  check_f32_flat(src, "src");
  TORCH_CHECK(idx.scalar_type() == torch::kInt64);
  long nrows = idx.numel();
  long dim = src.size(1);
  auto out = torch::empty({nrows, dim}, src.options());
  if (nrows > 0) {
    hipLaunchKernelGGL(gather_rows_kernel, dim3(grid_for(nrows * dim, 1)),
                       dim3(BLOCK_THREADS), 0, cur_stream(),
                       src.data_ptr<float>(), idx.contiguous().data_ptr<int64_t>(),
                       out.data_ptr<float>(), nrows, dim);
  }
  return out;
}

// ------------------------------------------------------------- PowerSGD
torch::Tensor psgd_mq(torch::Tensor M2d, torch::Tensor Qp) {
  long n = M2d.size(0), s = M2d.size(1);
  TORCH_CHECK(n % 64 == 0 && s % 64 == 0, "psgd_mq needs 64-multiples");
  TORCH_CHECK(Qp.size(0) == s && Qp.size(1) == PSGD_R);
  auto C = torch::zeros({n, (long)PSGD_R},
                        M2d.options().dtype(torch::kFloat32));
  long gridx = n / 64;
  long chunks = s / PSGD_BK;
  long splitk = std::min(std::max<long>((512 + gridx - 1) / gridx, 1), chunks);
  hipLaunchKernelGGL(psgd_mq_kernel, dim3(gridx, splitk), dim3(256), 0,
                     cur_stream(), M2d.data_ptr<float>(),
                     Qp.contiguous().data_ptr<float>(), C.data_ptr<float>(),
                     n, s);
  return C;
}

torch::Tensor psgd_mtp(torch::Tensor M2d, torch::Tensor Pp) {
  long n = M2d.size(0), s = M2d.size(1);
  TORCH_CHECK(n % 64 == 0 && s % 64 == 0, "psgd_mtp needs 64-multiples");
  TORCH_CHECK(Pp.size(0) == n && Pp.size(1) == PSGD_R);
  auto C = torch::zeros({s, (long)PSGD_R},
                        M2d.options().dtype(torch::kFloat32));
  long gridx = s / 64;
  long chunks = n / PSGD_BK;
  long splitk = std::min(std::max<long>((512 + gridx - 1) / gridx, 1), chunks);
  hipLaunchKernelGGL(psgd_mtp_kernel, dim3(gridx, splitk), dim3(256), 0,
                     cur_stream(), M2d.data_ptr<float>(),
                     Pp.contiguous().data_ptr<float>(), C.data_ptr<float>(),
                     n, s);
  return C;
}

void psgd_decompress_ef(torch::Tensor flat, torch::Tensor err,
                        torch::Tensor m_local, torch::Tensor Pp,
                        torch::Tensor Qp, double scale) {
  long numel = flat.numel();
  long s = m_local.size(1);
  hipLaunchKernelGGL(psgd_decompress_ef_kernel, dim3(grid_for(numel, 1)),
                     dim3(BLOCK_THREADS), 0, cur_stream(),
                     flat.data_ptr<float>(), err.data_ptr<float>(),
                     m_local.data_ptr<float>(), Pp.data_ptr<float>(),
                     Qp.data_ptr<float>(), numel, s, (float)scale);
}

// Q/K/V may be views of a fused qkv projection: last dim must be
// contiguous, other dims carried as element strides into the kernels.
static void attn_check_strided(const torch::Tensor& t, long D,
                               const char* name) {
  TORCH_CHECK(t.dim() == 4 && t.size(3) == D && t.stride(3) == 1,
              name, ": expected [B,H,S,D] with contiguous head dim");
  TORCH_CHECK(t.scalar_type() == torch::kBFloat16);
}

static const float* attn_mask_ptr(const c10::optional<torch::Tensor>& mask,
                                  long B, long S) {
  if (!mask.has_value() || !mask->defined()) return nullptr;
  TORCH_CHECK(mask->scalar_type() == torch::kFloat32 && mask->is_contiguous(),
              "attention mask must be contiguous fp32");
  TORCH_CHECK(mask->numel() == B * S,
              "attention mask must be an additive [B,1,1,S] key mask");
  return mask->data_ptr<float>();
}

torch::Tensor attn_fwd(torch::Tensor q, torch::Tensor k, torch::Tensor v,
                       double scale,
                       c10::optional<torch::Tensor> mask = c10::nullopt,
                       double p_drop = 0.0, int64_t seed = 0) {
  long D = q.size(3);
  TORCH_CHECK(q.dim() == 4 && (D == 64 || D == 128),
              "attn_fwd expects [B,H,S,64|128]");
  attn_check_strided(q, D, "q");
  attn_check_strided(k, D, "k");
  attn_check_strided(v, D, "v");
  long B = q.size(0), H = q.size(1), S = q.size(2);
  TORCH_CHECK(S % 32 == 0 && S >= 32, "S must be a multiple of 32");
  TORCH_CHECK(k.sizes() == q.sizes() && v.sizes() == q.sizes());
  auto o = torch::empty({B, H, S, D}, q.options());
  // RB=2 (128-row blocks) halves K/V re-read traffic for long sequences;
  // RB=1 keeps more blocks in flight for short ones. 1-D grid with bh as
  // the fast dimension = XCD-locality swizzle (see kernel comment).
  long nbh = B * H;
  auto launch = [&](auto kern, long rows) {
    long nqb = (S + rows - 1) / rows;
    hipLaunchKernelGGL(kern, dim3(nqb * nbh), dim3(256), 0, cur_stream(),
                       reinterpret_cast<__hip_bfloat16*>(q.data_ptr()),
                       reinterpret_cast<__hip_bfloat16*>(k.data_ptr()),
                       reinterpret_cast<__hip_bfloat16*>(v.data_ptr()),
                       reinterpret_cast<__hip_bfloat16*>(o.data_ptr()),
                       attn_mask_ptr(mask, B, S), S, H, nbh,
                       q.stride(0), q.stride(1), q.stride(2),
                       k.stride(0), k.stride(1), k.stride(2),
                       v.stride(0), v.stride(1), v.stride(2),
                       (float)scale, (float)p_drop,
                       (unsigned int)(uint64_t)seed);
  };
  // RB=3 (192-row blocks) for S>=512 was MEASURED SLOWER (0.034 -> 0.048
  // ms at S=512): occupancy 3 -> 2 waves/SIMD plus partial-tile waste
  // outweigh the traffic saving. RB=2 is the sweet spot.
  if (q.size(3) == 64) {
    if (S >= 256) launch(attn_fwd_kernel<2, 64>, 128);
    else launch(attn_fwd_kernel<1, 64>, 64);
  } else {
    if (S >= 256) launch(attn_fwd_kernel<2, 128>, 128);
    else launch(attn_fwd_kernel<1, 128>, 64);
  }
  return o;
}

torch::Tensor attn_dropmask(long B, long H, long S, double p_drop,
                            int64_t seed, torch::Device dev) {
  auto out = torch::empty({B, H, S, S},
                          torch::dtype(torch::kUInt8).device(dev));
  hipLaunchKernelGGL(attn_dropmask_kernel, dim3(S, B * H), dim3(256), 0,
                     cur_stream(), out.data_ptr<unsigned char>(), S,
                     (float)p_drop, (unsigned int)(uint64_t)seed);
  return out;
}

std::vector<torch::Tensor> attn_bwd(torch::Tensor q, torch::Tensor k,
                                    torch::Tensor v, torch::Tensor o,
                                    torch::Tensor dout, double scale,
                                    c10::optional<torch::Tensor> mask
                                        = c10::nullopt,
                                    double p_drop = 0.0, int64_t seed = 0) {
  long D = q.size(3);
  TORCH_CHECK(q.dim() == 4 && (D == 64 || D == 128));
  attn_check_strided(q, D, "q");
  attn_check_strided(k, D, "k");
  attn_check_strided(v, D, "v");
  TORCH_CHECK(o.is_contiguous() && dout.is_contiguous());
  long B = q.size(0), H = q.size(1), S = q.size(2);
  TORCH_CHECK(S % 32 == 0 && S >= 32);
  auto dq = torch::empty({B, H, S, D}, q.options());
  auto dk = torch::empty({B, H, S, D}, q.options());
  auto dv = torch::empty({B, H, S, D}, q.options());
  auto fopt = q.options().dtype(torch::kFloat32);
  auto Mbuf = torch::empty({B * H * S}, fopt);
  auto Lbuf = torch::empty({B * H * S}, fopt);
  auto Dbuf = torch::empty({B * H * S}, fopt);
  const float* mp = attn_mask_ptr(mask, B, S);
  unsigned int sd = (unsigned int)(uint64_t)seed;
  // 1-D grids, (b,h) fast for XCD L2 locality (see attention.hip)
  long nbh = B * H;
  dim3 grid(((S + 63) / 64) * nbh);
  auto st = cur_stream();
#define BF16P(t) reinterpret_cast<__hip_bfloat16*>((t).data_ptr())
  if (q.size(3) == 64) {
    hipLaunchKernelGGL(attn_bwd_q_kernel<64>, grid, dim3(256), 0, st,
                       BF16P(q), BF16P(k), BF16P(v), BF16P(o), BF16P(dout),
                       BF16P(dq), Mbuf.data_ptr<float>(),
                       Lbuf.data_ptr<float>(), Dbuf.data_ptr<float>(), mp,
                       S, H, nbh, q.stride(0), q.stride(1), q.stride(2),
                       k.stride(0), k.stride(1), k.stride(2), v.stride(0),
                       v.stride(1), v.stride(2), (float)scale,
                       (float)p_drop, sd);
    hipLaunchKernelGGL(attn_bwd_kv_kernel<64>, grid, dim3(256), 0, st,
                       BF16P(q), BF16P(k), BF16P(v), BF16P(dout), BF16P(dk),
                       BF16P(dv), Mbuf.data_ptr<float>(),
                       Lbuf.data_ptr<float>(), Dbuf.data_ptr<float>(), mp,
                       S, H, nbh, q.stride(0), q.stride(1), q.stride(2),
                       k.stride(0), k.stride(1), k.stride(2), v.stride(0),
                       v.stride(1), v.stride(2), (float)scale,
                       (float)p_drop, sd);
  } else {
    hipLaunchKernelGGL(attn_bwd_q_kernel<128>, grid, dim3(256), 0, st,
                       BF16P(q), BF16P(k), BF16P(v), BF16P(o), BF16P(dout),
                       BF16P(dq), Mbuf.data_ptr<float>(),
                       Lbuf.data_ptr<float>(), Dbuf.data_ptr<float>(), mp,
                       S, H, nbh, q.stride(0), q.stride(1), q.stride(2),
                       k.stride(0), k.stride(1), k.stride(2), v.stride(0),
                       v.stride(1), v.stride(2), (float)scale,
                       (float)p_drop, sd);
    hipLaunchKernelGGL(attn_bwd_kv_kernel<128>, grid, dim3(256), 0, st,
                       BF16P(q), BF16P(k), BF16P(v), BF16P(dout), BF16P(dk),
                       BF16P(dv), Mbuf.data_ptr<float>(),
                       Lbuf.data_ptr<float>(), Dbuf.data_ptr<float>(), mp,
                       S, H, nbh, q.stride(0), q.stride(1), q.stride(2),
                       k.stride(0), k.stride(1), k.stride(2), v.stride(0),
                       v.stride(1), v.stride(2), (float)scale,
                       (float)p_drop, sd);
  }
#undef BF16P
  return {dq, dk, dv};
}

std::vector<torch::Tensor> ln_fwd(torch::Tensor x,
                                  c10::optional<torch::Tensor> res,
                                  torch::Tensor gamma, torch::Tensor beta,
                                  double eps) {
  TORCH_CHECK(x.scalar_type() == torch::kBFloat16 && x.is_contiguous());
  long H = x.size(-1);
  long N = x.numel() / H;
  TORCH_CHECK(H <= 4096 && H % 2 == 0,
              "ln_fwd supports even H <= 4096");
  TORCH_CHECK(gamma.scalar_type() == torch::kFloat32 && gamma.is_contiguous());
  auto y = torch::empty_like(x);
  bool has_res = res.has_value() && res->defined();
  torch::Tensor u = has_res ? torch::empty_like(x) : x;
  auto fopt = x.options().dtype(torch::kFloat32);
  auto mean = torch::empty({N}, fopt);
  auto rstd = torch::empty({N}, fopt);
  const __hip_bfloat16* rp = has_res
      ? reinterpret_cast<__hip_bfloat16*>(res->data_ptr()) : nullptr;
  __hip_bfloat16* up = has_res
      ? reinterpret_cast<__hip_bfloat16*>(u.data_ptr()) : nullptr;
  hipLaunchKernelGGL(ln_fwd_kernel, dim3(N), dim3(256), 0, cur_stream(),
                     reinterpret_cast<__hip_bfloat16*>(x.data_ptr()), rp,
                     gamma.data_ptr<float>(), beta.data_ptr<float>(),
                     reinterpret_cast<__hip_bfloat16*>(y.data_ptr()), up,
                     mean.data_ptr<float>(), rstd.data_ptr<float>(), N,
                     (int)H, (float)eps);
  return {y, u, mean, rstd};
}

std::vector<torch::Tensor> ln_bwd(torch::Tensor dy, torch::Tensor u,
                                  torch::Tensor gamma, torch::Tensor mean,
                                  torch::Tensor rstd) {
  TORCH_CHECK(dy.scalar_type() == torch::kBFloat16 && dy.is_contiguous());
  long H = dy.size(-1);
  long N = dy.numel() / H;
  auto dx = torch::empty_like(dy);
  hipLaunchKernelGGL(ln_bwd_dx_kernel, dim3(N), dim3(256), 0, cur_stream(),
                     reinterpret_cast<__hip_bfloat16*>(dy.data_ptr()),
                     reinterpret_cast<__hip_bfloat16*>(u.data_ptr()),
                     gamma.data_ptr<float>(), mean.data_ptr<float>(),
                     rstd.data_ptr<float>(),
                     reinterpret_cast<__hip_bfloat16*>(dx.data_ptr()), N,
                     (int)H);
  // chunked dgamma/dbeta partials, reduced on the torch side
  long rows_per_chunk = 256;
  long nchunks = (N + rows_per_chunk - 1) / rows_per_chunk;
  auto partials = torch::empty({nchunks, 2, H},
                               dy.options().dtype(torch::kFloat32));
  hipLaunchKernelGGL(ln_bwd_gb_kernel, dim3((H + 63) / 64, nchunks),
                     dim3(256), 0, cur_stream(),
                     reinterpret_cast<__hip_bfloat16*>(dy.data_ptr()),
                     reinterpret_cast<__hip_bfloat16*>(u.data_ptr()),
                     mean.data_ptr<float>(), rstd.data_ptr<float>(),
                     partials.data_ptr<float>(), N, (int)H, rows_per_chunk);
  auto gb = partials.sum(0);
  return {dx, gb[0], gb[1]};
}

torch::Tensor col_sum(torch::Tensor x) {
  TORCH_CHECK(x.dim() == 2 && x.scalar_type() == torch::kBFloat16 &&
              x.is_contiguous());
  long N = x.size(0), H = x.size(1);
  // chunk rows so the grid fills the 8 XCDs even for narrow H
  long rows_per_chunk = 512;
  long nchunks = (N + rows_per_chunk - 1) / rows_per_chunk;
  if (nchunks > 256) { nchunks = 256;
    rows_per_chunk = (N + nchunks - 1) / nchunks; }
  auto partials = torch::empty({nchunks, H},
                               x.options().dtype(torch::kFloat32));
  hipLaunchKernelGGL(col_sum_kernel, dim3((H + 63) / 64, nchunks),
                     dim3(256), 0, cur_stream(),
                     reinterpret_cast<__hip_bfloat16*>(x.data_ptr()),
                     partials.data_ptr<float>(), N, H, rows_per_chunk);
  return partials.sum(0);
}

std::vector<torch::Tensor> ce_fwd(torch::Tensor logits,
                                  torch::Tensor targets) {
  TORCH_CHECK(logits.dim() == 2 && logits.scalar_type() == torch::kBFloat16
              && logits.is_contiguous());
  TORCH_CHECK(targets.scalar_type() == torch::kLong);
  long N = logits.size(0), H = logits.size(1);
  auto fopt = logits.options().dtype(torch::kFloat32);
  auto loss_rows = torch::empty({N}, fopt);
  auto m = torch::empty({N}, fopt);
  auto l2s = torch::empty({N}, fopt);
  hipLaunchKernelGGL(ce_fwd_kernel, dim3(N), dim3(256), 0, cur_stream(),
                     reinterpret_cast<__hip_bfloat16*>(logits.data_ptr()),
                     targets.data_ptr<long>(), loss_rows.data_ptr<float>(),
                     m.data_ptr<float>(), l2s.data_ptr<float>(), N, H);
  return {loss_rows, m, l2s};
}

torch::Tensor ce_bwd(torch::Tensor logits, torch::Tensor targets,
                     torch::Tensor m, torch::Tensor l2s,
                     torch::Tensor dloss_rows) {
  long N = logits.size(0), H = logits.size(1);
  auto dlogits = torch::empty_like(logits);
  hipLaunchKernelGGL(ce_bwd_kernel, dim3(N), dim3(256), 0, cur_stream(),
                     reinterpret_cast<__hip_bfloat16*>(logits.data_ptr()),
                     targets.data_ptr<long>(), m.data_ptr<float>(),
                     l2s.data_ptr<float>(),
                     dloss_rows.contiguous().data_ptr<float>(),
                     reinterpret_cast<__hip_bfloat16*>(dlogits.data_ptr()),
                     N, H);
  return dlogits;
}

torch::Tensor mfma_probe(torch::Tensor A, torch::Tensor B, long cand) {
  TORCH_CHECK(A.scalar_type() == torch::kBFloat16 && A.numel() == 16 * 32);
  TORCH_CHECK(B.scalar_type() == torch::kBFloat16 && B.numel() == 32 * 16);
  auto D = torch::zeros({16, 16}, A.options().dtype(torch::kFloat32));
  hipLaunchKernelGGL(mfma_probe_kernel, dim3(1), dim3(64), 0, cur_stream(),
                     reinterpret_cast<__hip_bfloat16*>(A.data_ptr()),
                     reinterpret_cast<__hip_bfloat16*>(B.data_ptr()),
                     D.data_ptr<float>(), (int)cand);
  return D;
}

void psgd_add_err_pad(torch::Tensor flat, torch::Tensor err,
                      torch::Tensor out) {
  long numel = flat.numel();
  long total = out.numel();
  hipLaunchKernelGGL(psgd_add_err_pad_kernel, dim3(grid_for(total, 1)),
                     dim3(BLOCK_THREADS), 0, cur_stream(),
                     flat.data_ptr<float>(), err.data_ptr<float>(),
                     out.data_ptr<float>(), numel, total);
}

}  // namespace

PYBIND11_MODULE(TORCH_EXTENSION_NAME, m) {
  m.def("fused_sgd", &fused_sgd, "fused flat SGD update (gfx950)");
  m.def("fused_adam", &fused_adam, "fused flat Adam/AdamW update (gfx950)");
  m.def("fused_adagrad", &fused_adagrad,
        "fused flat Adagrad update (gfx950)");
  m.def("fused_rmsprop", &fused_rmsprop,
        "fused flat RMSprop update (gfx950, centered/momentum variants)");
  m.def("scale_cast_bf16", &scale_cast_bf16, "scale+cast fp32->bf16");
  m.def("cast_back_f32", &cast_back_f32, "cast bf16->fp32");
  m.def("ef_compress", &ef_compress, "fused error-feedback bf16 compress");
  m.def("segment_coalesce", &segment_coalesce,
        "dedup-sum row-sparse gradient");
  m.def("scatter_add_rows", &scatter_add_rows, "rowwise scatter-add");
  m.def("gather_rows", &gather_rows, "rowwise gather");
  m.def("bn_fwd_train", &bn_fwd_train,
        "fused NHWC batchnorm fwd (+add+relu), returns y/save_mean/save_rstd");
  m.def("bn_bwd", &bn_bwd,
        "fused NHWC batchnorm bwd (+relu-mask+dres), returns "
        "dx/dweight/dbias[/dres]");
  m.def("psgd_mq", &psgd_mq, "PowerSGD P = M @ Q (MFMA f32 16x16x4)");
  m.def("psgd_mtp", &psgd_mtp, "PowerSGD Qn = M^T @ P (MFMA f32 16x16x4)");
  m.def("psgd_decompress_ef", &psgd_decompress_ef,
        "fused hat = P Q^T * scale; err = M - hat; flat = hat");
  m.def("psgd_add_err_pad", &psgd_add_err_pad,
        "padded M = flat + err (zero tail)");
  m.def("mfma_probe", &mfma_probe,
        "diagnostic: v_mfma_f32_16x16x32_bf16 A/B layout probe");
  m.def("ce_fwd", &ce_fwd,
        "fused online-softmax cross-entropy fwd -> loss_rows, m, l2s");
  m.def("ce_bwd", &ce_bwd, "fused cross-entropy bwd -> dlogits (bf16)");
  m.def("col_sum", &col_sum, "bf16 [N,H] column sum -> fp32 [H]");
  m.def("ln_fwd", &ln_fwd,
        "fused bf16 LayerNorm(+residual) forward -> y, u, mean, rstd");
  m.def("ln_bwd", &ln_bwd, "fused bf16 LayerNorm backward -> dx, dgamma, "
        "dbeta");
  m.def("attn_dropmask", &attn_dropmask,
        "materialize the hash dropout keep-mask (testing)");
  m.def("attn_fwd", &attn_fwd,
        "fused MFMA attention forward (bf16, D=64; optional additive key "
        "mask + hash dropout) — 4-wave LDS-tiled",
        py::arg("q"), py::arg("k"), py::arg("v"), py::arg("scale"),
        py::arg("mask") = py::none(), py::arg("p_drop") = 0.0,
        py::arg("seed") = 0);
  m.def("attn_bwd", &attn_bwd,
        "fused MFMA attention backward (GPU-validated) -> dq, dk, dv",
        py::arg("q"), py::arg("k"), py::arg("v"), py::arg("o"),
        py::arg("dout"), py::arg("scale"), py::arg("mask") = py::none(),
        py::arg("p_drop") = 0.0, py::arg("seed") = 0);
}
