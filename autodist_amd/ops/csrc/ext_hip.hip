#include "hip/hip_runtime.h"
// Python bindings for the gfx950 HIP ops (torch extension).
// Launches on the caller's current HIP stream so the engine's comm-stream /
// compute-stream ordering (parallel/engine.py) applies to these kernels too.
#include <torch/extension.h>
#include <ATen/hip/HIPContext.h>

#include "multi_tensor.hip"

namespace {

inline hipStream_t cur_stream() {
  return at::hip::getCurrentHIPStreamMasqueradingAsCUDA().stream();
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

}  // namespace

PYBIND11_MODULE(TORCH_EXTENSION_NAME, m) {
  m.def("fused_sgd", &fused_sgd, "fused flat SGD update (gfx950)");
  m.def("fused_adam", &fused_adam, "fused flat Adam/AdamW update (gfx950)");
  m.def("scale_cast_bf16", &scale_cast_bf16, "scale+cast fp32->bf16");
  m.def("cast_back_f32", &cast_back_f32, "cast bf16->fp32");
  m.def("ef_compress", &ef_compress, "fused error-feedback bf16 compress");
  m.def("segment_coalesce", &segment_coalesce,
        "dedup-sum row-sparse gradient");
  m.def("scatter_add_rows", &scatter_add_rows, "rowwise scatter-add");
  m.def("gather_rows", &gather_rows, "rowwise gather");
}
