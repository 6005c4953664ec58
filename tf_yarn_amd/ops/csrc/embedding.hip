// Fused multi-table embedding lookup + sparse-gradient apply for MI355X.
//
// The dominant cost of Criteo CTR models (SURVEY §7 "hard parts": sparse
// embedding gradient reduce).  All categorical tables are concatenated into
// one [total_rows, D] fp32 buffer; Python pre-adds per-feature row offsets
// into the flat id tensor, so the kernels are pure row gather / row
// scatter:
//
//  * emb_fwd:      out[i, :] = table[ids[i], :]          (out fp32 or bf16;
//                  bf16 out fuses the mixed-precision cast into the gather)
//  * emb_bwd_sgd:  table[ids[i], :] -= lr * scale * g[i, :]   (fused
//                  backward+update, atomicAdd fp32 — no materialized
//                  dense gradient table)
//  * emb_bwd_dense: grad_table[ids[i], :] += scale * g[i, :]  (for
//                  optimizers that need the dense gradient)
//
// Row layout keeps consecutive d contiguous, so lanes covering one row
// coalesce; reuse across the batch is served by L2/LLC (gather guidance:
// guide Appendix B "Scatter/gather/embedding").

#include <torch/extension.h>
#include <c10/hip/HIPStream.h>
#include "common.h"
#include <cstdlib>

namespace {

template <typename OIo>
__global__ void emb_fwd_kernel(const float* __restrict__ table,
                               const int64_t* __restrict__ ids,
                               typename OIo::scalar_t* __restrict__ out,
                               int64_t n_rows, int64_t dim) {
  const int64_t dvec = dim >> 2;
  const int64_t total = n_rows * dvec;
  const int64_t stride = gridDim.x * (int64_t)blockDim.x;
  for (int64_t t = blockIdx.x * (int64_t)blockDim.x + threadIdx.x;
       t < total; t += stride) {
    const int64_t row = t / dvec;
    const int64_t c4 = t - row * dvec;
    const int64_t src = ids[row];
    f32x4 v = reinterpret_cast<const f32x4*>(table + src * dim)[c4];
    float vv[4];
#pragma unroll
    for (int j = 0; j < 4; ++j) vv[j] = v[j];
    QuadIo<OIo>::store4(out, row * dvec + c4, vv);
  }
}

template <typename GIo>
__global__ void emb_bwd_sgd_kernel(float* __restrict__ table,
                                   const int64_t* __restrict__ ids,
                                   const typename GIo::scalar_t* __restrict__ g,
                                   int64_t n_rows, int64_t dim,
                                   float neg_lr_scale) {
  const int64_t dvec = dim >> 2;
  const int64_t total = n_rows * dvec;
  const int64_t stride = gridDim.x * (int64_t)blockDim.x;
  for (int64_t t = blockIdx.x * (int64_t)blockDim.x + threadIdx.x;
       t < total; t += stride) {
    const int64_t row = t / dvec;
    const int64_t c4 = t - row * dvec;
    float* dst = table + ids[row] * dim + c4 * 4;
    float gv[4];
    QuadIo<GIo>::load4(g, row * dvec + c4, gv);
#pragma unroll
    for (int j = 0; j < 4; ++j)
      f32_atomic_add(dst + j, neg_lr_scale * gv[j]);
  }
}

template <typename GIo>
__global__ void emb_bwd_dense_kernel(
    float* __restrict__ grad_table,
    const int64_t* __restrict__ ids,
    const typename GIo::scalar_t* __restrict__ g,
    int64_t n_rows, int64_t dim, float scale) {
  const int64_t dvec = dim >> 2;
  const int64_t total = n_rows * dvec;
  const int64_t stride = gridDim.x * (int64_t)blockDim.x;
  for (int64_t t = blockIdx.x * (int64_t)blockDim.x + threadIdx.x;
       t < total; t += stride) {
    const int64_t row = t / dvec;
    const int64_t c4 = t - row * dvec;
    float* dst = grad_table + ids[row] * dim + c4 * 4;
    float gv[4];
    QuadIo<GIo>::load4(g, row * dvec + c4, gv);
#pragma unroll
    for (int j = 0; j < 4; ++j)
      f32_atomic_add(dst + j, scale * gv[j]);
  }
}

// Fused deep+wide sparse update: the CTR models' wide (dim-1) table is
// driven by the SAME flat ids as the deep (dim-16) table, so one kernel
// reads the ids once and issues both updates (separate kernels re-read
// 13.6 MB of ids and pay an extra launch; measured 84 us for the wide
// scatter alone at b=65536).  The lane covering quad 0 of each row adds
// the wide scalar.
template <typename GIo>
__global__ void emb_bwd_sgd_fused_wide_kernel(
    float* __restrict__ table, float* __restrict__ wide_table,
    const int64_t* __restrict__ ids,
    const typename GIo::scalar_t* __restrict__ g,
    const typename GIo::scalar_t* __restrict__ gw,
    int64_t n_rows, int64_t dim, int g_div,
    float neg_lr_scale, float wide_alpha) {
  const int64_t dvec = dim >> 2;
  const int64_t total = n_rows * dvec;
  const int64_t stride = gridDim.x * (int64_t)blockDim.x;
  for (int64_t t = blockIdx.x * (int64_t)blockDim.x + threadIdx.x;
       t < total; t += stride) {
    const int64_t row = t / dvec;
    const int64_t c4 = t - row * dvec;
    const int64_t id = ids[row];
    float* dst = table + id * dim + c4 * 4;
    float gv[4];
    QuadIo<GIo>::load4(g, row * dvec + c4, gv);
#pragma unroll
    for (int j = 0; j < 4; ++j)
      f32_atomic_add(dst + j, neg_lr_scale * gv[j]);
    if (c4 == 0)
      f32_atomic_add(wide_table + id,
                     wide_alpha * GIo::load(gw, row / g_div));
  }
}

// Scalar variants for dim % 4 != 0 (e.g. the wide part's dim-1 tables).
template <typename OIo>
__global__ void emb_fwd_scalar_kernel(const float* __restrict__ table,
                                      const int64_t* __restrict__ ids,
                                      typename OIo::scalar_t* __restrict__ out,
                                      int64_t n_rows, int64_t dim) {
  const int64_t total = n_rows * dim;
  const int64_t stride = gridDim.x * (int64_t)blockDim.x;
  for (int64_t t = blockIdx.x * (int64_t)blockDim.x + threadIdx.x;
       t < total; t += stride) {
    const int64_t row = t / dim;
    const int64_t c = t - row * dim;
    OIo::store(out, t, table[ids[row] * dim + c]);
  }
}

template <typename GIo>
__global__ void emb_scatter_scalar_kernel(
    float* __restrict__ table, const int64_t* __restrict__ ids,
    const typename GIo::scalar_t* __restrict__ g,
    int64_t n_rows, int64_t dim, float alpha) {
  const int64_t total = n_rows * dim;
  const int64_t stride = gridDim.x * (int64_t)blockDim.x;
  for (int64_t t = blockIdx.x * (int64_t)blockDim.x + threadIdx.x;
       t < total; t += stride) {
    const int64_t row = t / dim;
    const int64_t c = t - row * dim;
    f32_atomic_add(table + ids[row] * dim + c, alpha * GIo::load(g, t));
  }
}

// Wide-part fusion: out[b] = sum_f table[ids[b*F + f]] for scalar
// (dim-1) tables — replaces gather + torch reduce with one kernel.
template <typename OIo>
__global__ void emb_gather_sum_kernel(const float* __restrict__ table,
                                      const int64_t* __restrict__ ids,
                                      typename OIo::scalar_t* __restrict__ out,
                                      int64_t batch, int F) {
  const int64_t stride = gridDim.x * (int64_t)blockDim.x;
  for (int64_t b = blockIdx.x * (int64_t)blockDim.x + threadIdx.x;
       b < batch; b += stride) {
    float acc = 0.f;
    const int64_t* row = ids + b * F;
#pragma unroll 4
    for (int f = 0; f < F; ++f) acc += table[row[f]];
    OIo::store(out, b, acc);
  }
}

// Backward of gather-sum: table[ids[b*F+f]] += alpha * g[b]
template <typename GIo>
__global__ void emb_scatter_sum_kernel(float* __restrict__ table,
                                       const int64_t* __restrict__ ids,
                                       const typename GIo::scalar_t* __restrict__ g,
                                       int64_t batch, int F, float alpha) {
  const int64_t total = batch * (int64_t)F;
  const int64_t stride = gridDim.x * (int64_t)blockDim.x;
  for (int64_t t = blockIdx.x * (int64_t)blockDim.x + threadIdx.x;
       t < total; t += stride) {
    const int64_t b = t / F;
    f32_atomic_add(table + ids[t], alpha * GIo::load(g, b));
  }
}

// Gather straight into a slice of the MLP input buffer: gather row
// (b, f) = ids[b*F + f] lands at out[b, col_off + f*dim : ...], killing
// the separate concat kernel (out stride/offset in quads; host guarantees
// dim % 4 == 0 and col_off % 4 == 0).
template <typename OIo>
__global__ void emb_fwd_into_kernel(const float* __restrict__ table,
                                    const int64_t* __restrict__ ids,
                                    typename OIo::scalar_t* __restrict__ out,
                                    int64_t n_rows, int64_t dim, int F,
                                    int64_t out_stride_q, int64_t off_q) {
  const int64_t dvec = dim >> 2;
  const int64_t total = n_rows * dvec;
  const int64_t stride = gridDim.x * (int64_t)blockDim.x;
  for (int64_t t = blockIdx.x * (int64_t)blockDim.x + threadIdx.x;
       t < total; t += stride) {
    const int64_t row = t / dvec;
    const int64_t c4 = t - row * dvec;
    const int64_t b = row / F;
    const int64_t f = row - b * F;
    f32x4 v = reinterpret_cast<const f32x4*>(table + ids[row] * dim)[c4];
    float vv[4];
#pragma unroll
    for (int j = 0; j < 4; ++j) vv[j] = v[j];
    QuadIo<OIo>::store4(out, b * out_stride_q + off_q + f * dvec + c4,
                        vv);
  }
}

// Sorted segmented scatter+SGD: ids are SORTED so duplicates are
// contiguous; only the head thread of each run applies the (summed)
// update with a plain read-modify-write — no atomics (measured 4.6x
// faster than atomicAdd at the bench shape; see scripts/micro_scatter).
template <typename GIo>
__global__ void emb_bwd_sgd_sorted_kernel(
    float* __restrict__ table, const int64_t* __restrict__ ids,
    const typename GIo::scalar_t* __restrict__ g,
    int64_t n_rows, int64_t dim, float neg_lr_scale) {
  const int64_t dvec = dim >> 2;
  const int64_t total = n_rows * dvec;
  const int64_t stride = gridDim.x * (int64_t)blockDim.x;
  for (int64_t t = blockIdx.x * (int64_t)blockDim.x + threadIdx.x;
       t < total; t += stride) {
    const int64_t row = t / dvec;
    const int64_t c4 = t - row * dvec;
    const int64_t id = ids[row];
    if (row > 0 && ids[row - 1] == id) continue;  // not a run head
    float acc[4];
    QuadIo<GIo>::load4(g, row * dvec + c4, acc);
    for (int64_t r = row + 1; r < n_rows && ids[r] == id; ++r) {
      float more[4];
      QuadIo<GIo>::load4(g, r * dvec + c4, more);
#pragma unroll
      for (int j = 0; j < 4; ++j) acc[j] += more[j];
    }
    float* dst = table + id * dim + c4 * 4;
    f32x4 cur = *reinterpret_cast<f32x4*>(dst);
#pragma unroll
    for (int j = 0; j < 4; ++j) cur[j] += neg_lr_scale * acc[j];
    *reinterpret_cast<f32x4*>(dst) = cur;
  }
}

// Out-of-range ids silently corrupt adjacent table rows via atomicAdd;
// MIYARN_DEBUG_BOUNDS=1 turns on a host-side min/max validation (two
// small reductions + sync per call — debug only, off in production).
bool bounds_debug_enabled() {
  static const bool on = [] {
    const char* v = std::getenv("MIYARN_DEBUG_BOUNDS");
    return v != nullptr && v[0] != '\0' && v[0] != '0';
  }();
  return on;
}

void debug_check_ids(const torch::Tensor& ids, int64_t rows) {
  if (!bounds_debug_enabled() || ids.numel() == 0) return;
  const int64_t mn = ids.min().item<int64_t>();
  const int64_t mx = ids.max().item<int64_t>();
  TORCH_CHECK(mn >= 0 && mx < rows, "embedding ids out of range: [", mn,
              ", ", mx, "] vs table rows ", rows,
              " (MIYARN_DEBUG_BOUNDS check)");
}

void check_emb(const torch::Tensor& table, const torch::Tensor& ids,
               int64_t dim) {
  TORCH_CHECK(table.is_cuda() && table.is_contiguous() &&
              table.scalar_type() == torch::kFloat32,
              "table must be a contiguous fp32 GPU tensor");
  TORCH_CHECK(ids.is_cuda() && ids.is_contiguous() &&
              ids.scalar_type() == torch::kInt64,
              "ids must be contiguous int64 on GPU");
}

}  // namespace

torch::Tensor emb_fwd(torch::Tensor table, torch::Tensor ids,
                      bool out_bf16) {
  const int64_t dim = table.size(1);
  const int64_t n = ids.numel();
  check_emb(table, ids, dim);
  debug_check_ids(ids, table.size(0));
  auto out = torch::empty(
      {n, dim}, table.options().dtype(
          out_bf16 ? torch::kBFloat16 : torch::kFloat32));
  auto stream = c10::hip::getCurrentHIPStream().stream();
  const bool vec = (dim % 4 == 0);
  int grid = miyarn_grid(n * (vec ? dim / 4 : dim));
  if (out_bf16) {
    auto* optr = reinterpret_cast<unsigned short*>(out.data_ptr());
    if (vec)
      hipLaunchKernelGGL(emb_fwd_kernel<Bf16Io>, dim3(grid),
                         dim3(MIYARN_BLOCK), 0, stream,
                         table.data_ptr<float>(), ids.data_ptr<int64_t>(),
                         optr, n, dim);
    else
      hipLaunchKernelGGL(emb_fwd_scalar_kernel<Bf16Io>, dim3(grid),
                         dim3(MIYARN_BLOCK), 0, stream,
                         table.data_ptr<float>(), ids.data_ptr<int64_t>(),
                         optr, n, dim);
  } else {
    if (vec)
      hipLaunchKernelGGL(emb_fwd_kernel<F32Io>, dim3(grid),
                         dim3(MIYARN_BLOCK), 0, stream,
                         table.data_ptr<float>(), ids.data_ptr<int64_t>(),
                         out.data_ptr<float>(), n, dim);
    else
      hipLaunchKernelGGL(emb_fwd_scalar_kernel<F32Io>, dim3(grid),
                         dim3(MIYARN_BLOCK), 0, stream,
                         table.data_ptr<float>(), ids.data_ptr<int64_t>(),
                         out.data_ptr<float>(), n, dim);
  }
  return out;
}

void emb_bwd_sgd(torch::Tensor table, torch::Tensor ids, torch::Tensor grad,
                 double lr, double scale) {
  const int64_t dim = table.size(1);
  const int64_t n = ids.numel();
  check_emb(table, ids, dim);
  debug_check_ids(ids, table.size(0));
  TORCH_CHECK(grad.is_cuda() && grad.is_contiguous() &&
              grad.numel() == n * dim, "grad shape mismatch");
  auto stream = c10::hip::getCurrentHIPStream().stream();
  const bool vec = (dim % 4 == 0);
  // atomic RMW latency is higher than load latency: give the scheduler
  // 4x the streaming-kernel wave count to hide it
  int grid = miyarn_grid_cap(n * (vec ? dim / 4 : dim), 8192);
  float nls = (float)(-lr * scale);
  if (grad.scalar_type() == torch::kFloat32) {
    if (vec)
      hipLaunchKernelGGL(emb_bwd_sgd_kernel<F32Io>, dim3(grid),
                         dim3(MIYARN_BLOCK), 0, stream,
                         table.data_ptr<float>(), ids.data_ptr<int64_t>(),
                         grad.data_ptr<float>(), n, dim, nls);
    else
      hipLaunchKernelGGL(emb_scatter_scalar_kernel<F32Io>, dim3(grid),
                         dim3(MIYARN_BLOCK), 0, stream,
                         table.data_ptr<float>(), ids.data_ptr<int64_t>(),
                         grad.data_ptr<float>(), n, dim, nls);
  } else {
    TORCH_CHECK(grad.scalar_type() == torch::kBFloat16,
                "grad must be fp32 or bf16");
    auto* gptr = reinterpret_cast<unsigned short*>(grad.data_ptr());
    if (vec)
      hipLaunchKernelGGL(emb_bwd_sgd_kernel<Bf16Io>, dim3(grid),
                         dim3(MIYARN_BLOCK), 0, stream,
                         table.data_ptr<float>(), ids.data_ptr<int64_t>(),
                         gptr, n, dim, nls);
    else
      hipLaunchKernelGGL(emb_scatter_scalar_kernel<Bf16Io>, dim3(grid),
                         dim3(MIYARN_BLOCK), 0, stream,
                         table.data_ptr<float>(), ids.data_ptr<int64_t>(),
                         gptr, n, dim, nls);
  }
}

void emb_bwd_dense(torch::Tensor grad_table, torch::Tensor ids,
                   torch::Tensor grad, double scale) {
  const int64_t dim = grad_table.size(1);
  const int64_t n = ids.numel();
  check_emb(grad_table, ids, dim);
  debug_check_ids(ids, grad_table.size(0));
  TORCH_CHECK(grad.is_cuda() && grad.is_contiguous() &&
              grad.numel() == n * dim, "grad shape mismatch");
  auto stream = c10::hip::getCurrentHIPStream().stream();
  const bool vec = (dim % 4 == 0);
  int grid = miyarn_grid(n * (vec ? dim / 4 : dim));
  if (grad.scalar_type() == torch::kFloat32) {
    if (vec)
      hipLaunchKernelGGL(emb_bwd_dense_kernel<F32Io>, dim3(grid),
                         dim3(MIYARN_BLOCK), 0, stream,
                         grad_table.data_ptr<float>(),
                         ids.data_ptr<int64_t>(), grad.data_ptr<float>(),
                         n, dim, (float)scale);
    else
      hipLaunchKernelGGL(emb_scatter_scalar_kernel<F32Io>, dim3(grid),
                         dim3(MIYARN_BLOCK), 0, stream,
                         grad_table.data_ptr<float>(),
                         ids.data_ptr<int64_t>(), grad.data_ptr<float>(),
                         n, dim, (float)scale);
  } else {
    TORCH_CHECK(grad.scalar_type() == torch::kBFloat16,
                "grad must be fp32 or bf16");
    auto* gptr = reinterpret_cast<unsigned short*>(grad.data_ptr());
    if (vec)
      hipLaunchKernelGGL(emb_bwd_dense_kernel<Bf16Io>, dim3(grid),
                         dim3(MIYARN_BLOCK), 0, stream,
                         grad_table.data_ptr<float>(),
                         ids.data_ptr<int64_t>(), gptr, n, dim,
                         (float)scale);
    else
      hipLaunchKernelGGL(emb_scatter_scalar_kernel<Bf16Io>, dim3(grid),
                         dim3(MIYARN_BLOCK), 0, stream,
                         grad_table.data_ptr<float>(),
                         ids.data_ptr<int64_t>(), gptr, n, dim,
                         (float)scale);
  }
}

torch::Tensor emb_gather_sum(torch::Tensor table, torch::Tensor ids,
                             int64_t batch, bool out_bf16) {
  TORCH_CHECK(table.is_cuda() && table.is_contiguous() &&
              table.scalar_type() == torch::kFloat32,
              "table must be contiguous fp32 on GPU");
  TORCH_CHECK(ids.is_cuda() && ids.is_contiguous() &&
              ids.scalar_type() == torch::kInt64, "ids must be int64 GPU");
  TORCH_CHECK(ids.numel() % batch == 0, "ids size not divisible by batch");
  debug_check_ids(ids, table.numel());
  const int F = static_cast<int>(ids.numel() / batch);
  auto out = torch::empty({batch}, table.options().dtype(
      out_bf16 ? torch::kBFloat16 : torch::kFloat32));
  auto stream = c10::hip::getCurrentHIPStream().stream();
  int grid = miyarn_grid(batch);
  if (out_bf16) {
    hipLaunchKernelGGL(emb_gather_sum_kernel<Bf16Io>, dim3(grid),
                       dim3(MIYARN_BLOCK), 0, stream,
                       table.data_ptr<float>(), ids.data_ptr<int64_t>(),
                       reinterpret_cast<unsigned short*>(out.data_ptr()),
                       batch, F);
  } else {
    hipLaunchKernelGGL(emb_gather_sum_kernel<F32Io>, dim3(grid),
                       dim3(MIYARN_BLOCK), 0, stream,
                       table.data_ptr<float>(), ids.data_ptr<int64_t>(),
                       out.data_ptr<float>(), batch, F);
  }
  return out;
}

void emb_scatter_sum(torch::Tensor table, torch::Tensor ids,
                     torch::Tensor grad, double alpha) {
  TORCH_CHECK(table.is_cuda() && table.is_contiguous() &&
              table.scalar_type() == torch::kFloat32,
              "table must be contiguous fp32 on GPU");
  TORCH_CHECK(ids.is_cuda() && ids.is_contiguous() &&
              ids.scalar_type() == torch::kInt64, "ids must be int64 GPU");
  const int64_t batch = grad.numel();
  TORCH_CHECK(ids.numel() % batch == 0, "ids size not divisible by batch");
  debug_check_ids(ids, table.numel());
  const int F = static_cast<int>(ids.numel() / batch);
  auto stream = c10::hip::getCurrentHIPStream().stream();
  int grid = miyarn_grid(batch * F);
  if (grad.scalar_type() == torch::kFloat32) {
    hipLaunchKernelGGL(emb_scatter_sum_kernel<F32Io>, dim3(grid),
                       dim3(MIYARN_BLOCK), 0, stream,
                       table.data_ptr<float>(), ids.data_ptr<int64_t>(),
                       grad.data_ptr<float>(), batch, F, (float)alpha);
  } else {
    TORCH_CHECK(grad.scalar_type() == torch::kBFloat16,
                "grad must be fp32 or bf16");
    hipLaunchKernelGGL(emb_scatter_sum_kernel<Bf16Io>, dim3(grid),
                       dim3(MIYARN_BLOCK), 0, stream,
                       table.data_ptr<float>(), ids.data_ptr<int64_t>(),
                       reinterpret_cast<unsigned short*>(grad.data_ptr()),
                       batch, F, (float)alpha);
  }
}

void emb_fwd_into(torch::Tensor table, torch::Tensor ids,
                  torch::Tensor out, int64_t col_offset) {
  const int64_t dim = table.size(1);
  const int64_t n = ids.numel();
  check_emb(table, ids, dim);
  debug_check_ids(ids, table.size(0));
  TORCH_CHECK(out.is_cuda() && out.is_contiguous() && out.dim() == 2,
              "out must be a contiguous 2D GPU tensor");
  TORCH_CHECK(dim % 4 == 0 && col_offset % 4 == 0 &&
              out.size(1) % 4 == 0,
              "emb_fwd_into needs dim, col_offset and out stride % 4 == 0");
  const int64_t batch = out.size(0);
  TORCH_CHECK(n % batch == 0, "ids not divisible by out rows");
  const int F = static_cast<int>(n / batch);
  TORCH_CHECK(col_offset + (int64_t)F * dim <= out.size(1),
              "slice out of range");
  auto stream = c10::hip::getCurrentHIPStream().stream();
  int grid = miyarn_grid(n * (dim / 4));
  if (out.scalar_type() == torch::kBFloat16) {
    hipLaunchKernelGGL(emb_fwd_into_kernel<Bf16Io>, dim3(grid),
                       dim3(MIYARN_BLOCK), 0, stream,
                       table.data_ptr<float>(), ids.data_ptr<int64_t>(),
                       reinterpret_cast<unsigned short*>(out.data_ptr()),
                       n, dim, F, out.size(1) / 4, col_offset / 4);
  } else {
    TORCH_CHECK(out.scalar_type() == torch::kFloat32, "fp32/bf16 only");
    hipLaunchKernelGGL(emb_fwd_into_kernel<F32Io>, dim3(grid),
                       dim3(MIYARN_BLOCK), 0, stream,
                       table.data_ptr<float>(), ids.data_ptr<int64_t>(),
                       out.data_ptr<float>(), n, dim, F,
                       out.size(1) / 4, col_offset / 4);
  }
}


void emb_bwd_sgd_sorted(torch::Tensor table, torch::Tensor sorted_ids,
                        torch::Tensor grad, double lr, double scale) {
  const int64_t dim = table.size(1);
  const int64_t n = sorted_ids.numel();
  check_emb(table, sorted_ids, dim);
  debug_check_ids(sorted_ids, table.size(0));
  TORCH_CHECK(dim % 4 == 0, "sorted scatter needs dim % 4 == 0");
  TORCH_CHECK(grad.is_cuda() && grad.is_contiguous() &&
              grad.numel() == n * dim, "grad shape mismatch");
  auto stream = c10::hip::getCurrentHIPStream().stream();
  int grid = miyarn_grid_cap(n * (dim / 4), 8192);
  float nls = (float)(-lr * scale);
  if (grad.scalar_type() == torch::kFloat32) {
    hipLaunchKernelGGL(emb_bwd_sgd_sorted_kernel<F32Io>, dim3(grid),
                       dim3(MIYARN_BLOCK), 0, stream,
                       table.data_ptr<float>(),
                       sorted_ids.data_ptr<int64_t>(),
                       grad.data_ptr<float>(), n, dim, nls);
  } else {
    TORCH_CHECK(grad.scalar_type() == torch::kBFloat16,
                "grad must be fp32 or bf16");
    hipLaunchKernelGGL(emb_bwd_sgd_sorted_kernel<Bf16Io>, dim3(grid),
                       dim3(MIYARN_BLOCK), 0, stream,
                       table.data_ptr<float>(),
                       sorted_ids.data_ptr<int64_t>(),
                       reinterpret_cast<unsigned short*>(grad.data_ptr()),
                       n, dim, nls);
  }
}

void emb_bwd_sgd_fused_wide(torch::Tensor table, torch::Tensor wide_table,
                            torch::Tensor ids, torch::Tensor grad,
                            torch::Tensor gw, double lr, double scale) {
  const int64_t dim = table.size(1);
  const int64_t n = ids.numel();
  check_emb(table, ids, dim);
  debug_check_ids(ids, table.size(0));
  TORCH_CHECK(dim % 4 == 0, "fused wide update needs dim % 4 == 0");
  TORCH_CHECK(wide_table.is_cuda() && wide_table.is_contiguous() &&
              wide_table.scalar_type() == torch::kFloat32 &&
              wide_table.size(0) == table.size(0), "bad wide table");
  TORCH_CHECK(grad.is_cuda() && grad.is_contiguous() &&
              grad.numel() == n * dim, "grad shape mismatch");
  TORCH_CHECK(gw.is_cuda() && gw.is_contiguous() &&
              gw.scalar_type() == grad.scalar_type() &&
              n % gw.numel() == 0, "gw dtype/shape mismatch");
  const int g_div = static_cast<int>(n / gw.numel());
  auto stream = c10::hip::getCurrentHIPStream().stream();
  int grid = miyarn_grid_cap(n * (dim / 4), 8192);
  const float nls = (float)(-lr * scale);
  if (grad.scalar_type() == torch::kFloat32) {
    hipLaunchKernelGGL(emb_bwd_sgd_fused_wide_kernel<F32Io>, dim3(grid),
                       dim3(MIYARN_BLOCK), 0, stream,
                       table.data_ptr<float>(),
                       wide_table.data_ptr<float>(),
                       ids.data_ptr<int64_t>(), grad.data_ptr<float>(),
                       gw.data_ptr<float>(), n, dim, g_div, nls, nls);
  } else {
    TORCH_CHECK(grad.scalar_type() == torch::kBFloat16, "fp32/bf16 only");
    hipLaunchKernelGGL(emb_bwd_sgd_fused_wide_kernel<Bf16Io>, dim3(grid),
                       dim3(MIYARN_BLOCK), 0, stream,
                       table.data_ptr<float>(),
                       wide_table.data_ptr<float>(),
                       ids.data_ptr<int64_t>(),
                       reinterpret_cast<unsigned short*>(grad.data_ptr()),
                       reinterpret_cast<unsigned short*>(gw.data_ptr()),
                       n, dim, g_div, nls, nls);
  }
}
