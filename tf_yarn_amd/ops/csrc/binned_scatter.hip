// Radix-binned sparse scatter+SGD for MI355X — the round-2 replacement
// for the atomic-bound embedding backward (docs/Kernels.md "Round-2
// kernel designs" #1; executable algorithm spec:
// tests/test_binned_scatter_spec.py).
//
// Round-1 floor: 27M random fp32 atomicAdds/step over a 1.7 GB table ran
// at ~805 GB/s effective (340 us emb_bwd_sgd + 90 us emb_scatter_sum at
// b=65536).  Design:
//
//  Pass A (binned_permutation):
//    A1 histogram ids by table region (row >> region_bits).  Counters
//       are padded to ONE PER 64 B CACHE LINE: v2's 16-way privatized
//       dense counters still serialized at L2 line granularity
//       (measured 65 us for 1.7M int adds; ~134 atomics landed on each
//       line).
//    A2 single-block exclusive scan -> bin_starts + reservation
//       cursors (G=1 after padding, so the scan walks n_bins once),
//    A3 slot reservation; order[slot] packs (row << 31 | update index)
//       so pass B never touches the ids array again (the v2 applies
//       paid a random 8 B ids[j] load per update).
//    No sort: slot order within a bin is irrelevant (sum is
//    commutative).
//
//  Pass B (one workgroup per bin, grid-stride over bins):
//    Each bin's region belongs to EXACTLY ONE workgroup, so the final
//    table update needs no global atomics at all:
//      1. dedup/accumulate the bin's updates in an LDS hash keyed by
//         row id (LDS atomicCAS insert + LDS float atomicAdd, value
//         stride padded to 17 — the natural *16 stride is a 16-way
//         bank conflict);
//      2. barrier; write each occupied hash slot back with a plain
//         vectorized read-modify-write, resetting the slot lazily for
//         the block's next bin.
//    Bins whose update count exceeds 3/4 of the hash capacity (tiny
//    tables, heavy skew) fall back to direct global atomics — still
//    region-grouped, so they keep the L2 locality.
//
// The deep (dim=16) and wide (dim=1) tables of the CTR models use the
// SAME flat ids, so the host computes one permutation and feeds both
// apply kernels (tf_yarn_amd/models/sharded_embedding.py).

#include <torch/extension.h>
#include <c10/hip/HIPStream.h>
#include "common.h"

namespace {

constexpr int kPad = 16;          // int32 counters, one per 64 B line
constexpr int kScanBlock = 1024;
constexpr int kHashDeep = 1024;   // slots; 1024*(4 + 17*4) B = 72 KB LDS
constexpr int kValStride = 17;    // bank-conflict-free value stride
constexpr int kHashScalar = 2048; // slots; 2048*(4+4) B = 16 KB LDS
constexpr int kProbeMax = 64;
constexpr int kJBits = 31;
constexpr int64_t kJMask = (int64_t{1} << kJBits) - 1;

__global__ void bin_count_kernel(const int64_t* __restrict__ ids,
                                 int64_t n, int32_t* __restrict__ counts,
                                 int region_bits) {
  const int64_t stride = gridDim.x * (int64_t)blockDim.x;
  for (int64_t i = blockIdx.x * (int64_t)blockDim.x + threadIdx.x;
       i < n; i += stride)
    atomicAdd(&counts[(ids[i] >> region_bits) * kPad], 1);
}

// Single-block exclusive scan over the line-padded counters ->
// starts[n_bins+1] and the (padded) reservation cursors.
__global__ void bin_scan_kernel(const int32_t* __restrict__ counts,
                                int32_t* __restrict__ starts,
                                int32_t* __restrict__ cursor, int n_bins) {
  __shared__ int32_t chunk_tot[kScanBlock];
  const int per = (n_bins + kScanBlock - 1) / kScanBlock;
  const int b0 = threadIdx.x * per;
  const int b1 = min(n_bins, b0 + per);
  int32_t local = 0;
  for (int b = b0; b < b1; ++b) local += counts[(int64_t)b * kPad];
  chunk_tot[threadIdx.x] = local;
  __syncthreads();
  for (int off = 1; off < kScanBlock; off <<= 1) {
    int32_t t = (threadIdx.x >= off) ? chunk_tot[threadIdx.x - off] : 0;
    __syncthreads();
    chunk_tot[threadIdx.x] += t;
    __syncthreads();
  }
  int32_t run = (threadIdx.x == 0) ? 0 : chunk_tot[threadIdx.x - 1];
  if (threadIdx.x == 0) starts[0] = 0;
  for (int b = b0; b < b1; ++b) {
    cursor[(int64_t)b * kPad] = run;
    run += counts[(int64_t)b * kPad];
    starts[b + 1] = run;
  }
}

__global__ void bin_slot_kernel(const int64_t* __restrict__ ids,
                                int64_t n, int32_t* __restrict__ cursor,
                                int64_t* __restrict__ order,
                                int region_bits) {
  const int64_t stride = gridDim.x * (int64_t)blockDim.x;
  for (int64_t i = blockIdx.x * (int64_t)blockDim.x + threadIdx.x;
       i < n; i += stride) {
    const int64_t row = ids[i];
    const int slot = atomicAdd(&cursor[(row >> region_bits) * kPad], 1);
    order[slot] = (row << kJBits) | i;
  }
}

__device__ __forceinline__ int hash_slot(int32_t key, int capacity) {
  return static_cast<int>((static_cast<uint32_t>(key) * 2654435761u)
                          & (capacity - 1));
}

// Find-or-insert via LDS atomicCAS; the probe sequence is deterministic,
// so every thread that gives up on a key gives up consistently (their
// updates all take the global-atomic fallback — no mixed RMW/atomic race
// on one row).
__device__ __forceinline__ int hash_find_or_insert(int32_t* h_id,
                                                   int32_t key,
                                                   int capacity) {
  int s = hash_slot(key, capacity);
  for (int p = 0; p < kProbeMax; ++p) {
    const int32_t old = atomicCAS(&h_id[s], -1, key);
    if (old == -1 || old == key) return s;
    s = (s + 1) & (capacity - 1);
  }
  return -1;
}

// Pass B, deep tables (dim compile-time, 16 for the CTR models).
template <typename GIo, int DIM>
__global__ void binned_apply_deep_kernel(
    float* __restrict__ table,
    const typename GIo::scalar_t* __restrict__ g,
    const int64_t* __restrict__ order,
    const int32_t* __restrict__ starts, int n_bins, float neg_lr_scale) {
  __shared__ int32_t h_id[kHashDeep];
  __shared__ float h_val[kHashDeep * kValStride];
  constexpr int DVEC = DIM / 4;
  // Init the hash ONCE; each bin's writeback resets only the slots it
  // occupied.
  for (int s = threadIdx.x; s < kHashDeep; s += blockDim.x) {
    h_id[s] = -1;
#pragma unroll
    for (int d = 0; d < DIM; ++d) h_val[s * kValStride + d] = 0.f;
  }
  __syncthreads();
  for (int bin = blockIdx.x; bin < n_bins; bin += gridDim.x) {
    const int start = starts[bin], end = starts[bin + 1];
    const int count = end - start;
    if (count == 0) continue;
    const bool use_hash = count <= (3 * kHashDeep) / 4;
    for (int k = start + threadIdx.x; k < end; k += blockDim.x) {
      const int64_t packed = order[k];
      const int64_t j = packed & kJMask;
      const int64_t row = packed >> kJBits;
      float gv[DIM];
#pragma unroll
      for (int q = 0; q < DVEC; ++q) {
        float quad[4];
        QuadIo<GIo>::load4(g, j * DVEC + q, quad);
#pragma unroll
        for (int x = 0; x < 4; ++x) gv[q * 4 + x] = quad[x];
      }
      int slot = -1;
      if (use_hash)
        slot = hash_find_or_insert(h_id, static_cast<int32_t>(row),
                                   kHashDeep);
      if (slot >= 0) {
#pragma unroll
        for (int d = 0; d < DIM; ++d)
          atomicAdd(&h_val[slot * kValStride + d], gv[d]);
      } else {
        float* dst = table + row * DIM;
#pragma unroll
        for (int d = 0; d < DIM; ++d)
          f32_atomic_add(dst + d, neg_lr_scale * gv[d]);
      }
    }
    if (use_hash) {
      __syncthreads();
      for (int s = threadIdx.x; s < kHashDeep; s += blockDim.x) {
        const int32_t row = h_id[s];
        if (row < 0) continue;
        float* dst = table + (int64_t)row * DIM;
#pragma unroll
        for (int q = 0; q < DVEC; ++q) {
          f32x4 cur = reinterpret_cast<f32x4*>(dst)[q];
#pragma unroll
          for (int x = 0; x < 4; ++x) {
            cur[x] += neg_lr_scale * h_val[s * kValStride + q * 4 + x];
            h_val[s * kValStride + q * 4 + x] = 0.f;  // lazy reset
          }
          reinterpret_cast<f32x4*>(dst)[q] = cur;
        }
        h_id[s] = -1;
      }
      __syncthreads();
    }
  }
}

// Pass B, scalar (dim-1 wide) tables.  grad for update j is g[j / g_div]
// (g_div = features-per-output of the fused gather-sum).
template <typename GIo>
__global__ void binned_apply_scalar_kernel(
    float* __restrict__ table,
    const typename GIo::scalar_t* __restrict__ g,
    const int64_t* __restrict__ order,
    const int32_t* __restrict__ starts, int n_bins, int g_div,
    float alpha) {
  __shared__ int32_t h_id[kHashScalar];
  __shared__ float h_val[kHashScalar];
  for (int s = threadIdx.x; s < kHashScalar; s += blockDim.x) {
    h_id[s] = -1;
    h_val[s] = 0.f;
  }
  __syncthreads();
  for (int bin = blockIdx.x; bin < n_bins; bin += gridDim.x) {
    const int start = starts[bin], end = starts[bin + 1];
    const int count = end - start;
    if (count == 0) continue;
    const bool use_hash = count <= (3 * kHashScalar) / 4;
    for (int k = start + threadIdx.x; k < end; k += blockDim.x) {
      const int64_t packed = order[k];
      const int64_t j = packed & kJMask;
      const int64_t row = packed >> kJBits;
      const float gv = GIo::load(g, j / g_div);
      int slot = -1;
      if (use_hash)
        slot = hash_find_or_insert(h_id, static_cast<int32_t>(row),
                                   kHashScalar);
      if (slot >= 0)
        atomicAdd(&h_val[slot], gv);
      else
        f32_atomic_add(table + row, alpha * gv);
    }
    if (use_hash) {
      __syncthreads();
      for (int s = threadIdx.x; s < kHashScalar; s += blockDim.x) {
        const int32_t row = h_id[s];
        if (row >= 0) {
          table[row] += alpha * h_val[s];
          h_val[s] = 0.f;  // lazy reset
          h_id[s] = -1;
        }
      }
      __syncthreads();
    }
  }
}

}  // namespace

std::vector<torch::Tensor> binned_permutation(torch::Tensor ids,
                                              int64_t n_rows,
                                              int64_t region_bits) {
  TORCH_CHECK(ids.is_cuda() && ids.is_contiguous() &&
              ids.scalar_type() == torch::kInt64,
              "ids must be contiguous int64 on GPU");
  TORCH_CHECK(region_bits >= 1 && region_bits <= 26, "bad region_bits");
  TORCH_CHECK(n_rows < (int64_t{1} << 31), "table too large for int32 rows");
  const int64_t n = ids.numel();
  TORCH_CHECK(n < (int64_t{1} << kJBits), "too many updates to pack");
  const int64_t n_bins64 =
      (n_rows + (int64_t{1} << region_bits) - 1) >> region_bits;
  TORCH_CHECK(n_bins64 < (1 << 24), "too many bins; raise region_bits");
  const int n_bins = static_cast<int>(n_bins64);
  auto opts = ids.options().dtype(torch::kInt32);
  auto counts = torch::zeros({n_bins * kPad}, opts);
  auto starts = torch::empty({n_bins + 1}, opts);
  auto cursor = torch::empty({n_bins * kPad}, opts);
  auto order = torch::empty({n}, ids.options());
  auto stream = c10::hip::getCurrentHIPStream().stream();
  const int grid = miyarn_grid(n);
  hipLaunchKernelGGL(bin_count_kernel, dim3(grid), dim3(MIYARN_BLOCK), 0,
                     stream, ids.data_ptr<int64_t>(), n,
                     counts.data_ptr<int32_t>(),
                     static_cast<int>(region_bits));
  hipLaunchKernelGGL(bin_scan_kernel, dim3(1), dim3(kScanBlock), 0,
                     stream, counts.data_ptr<int32_t>(),
                     starts.data_ptr<int32_t>(),
                     cursor.data_ptr<int32_t>(), n_bins);
  hipLaunchKernelGGL(bin_slot_kernel, dim3(grid), dim3(MIYARN_BLOCK), 0,
                     stream, ids.data_ptr<int64_t>(), n,
                     cursor.data_ptr<int32_t>(),
                     order.data_ptr<int64_t>(),
                     static_cast<int>(region_bits));
  return {order, starts};
}

void emb_bwd_sgd_binned(torch::Tensor table, torch::Tensor ids,
                        torch::Tensor grad, double lr, double scale,
                        torch::Tensor order, torch::Tensor starts) {
  const int64_t dim = table.size(1);
  const int64_t n = ids.numel();
  TORCH_CHECK(table.is_cuda() && table.is_contiguous() &&
              table.scalar_type() == torch::kFloat32, "bad table");
  TORCH_CHECK(dim == 16, "binned deep apply supports dim=16 (got ", dim,
              "); use emb_bwd_sgd");
  TORCH_CHECK(grad.is_cuda() && grad.is_contiguous() &&
              grad.numel() == n * dim, "grad shape mismatch");
  TORCH_CHECK(order.scalar_type() == torch::kInt64 &&
              starts.scalar_type() == torch::kInt32, "bad perm dtypes");
  const int n_bins = static_cast<int>(starts.numel() - 1);
  auto stream = c10::hip::getCurrentHIPStream().stream();
  const int grid = std::min(n_bins, 8192);
  const float nls = static_cast<float>(-lr * scale);
  if (grad.scalar_type() == torch::kFloat32) {
    hipLaunchKernelGGL((binned_apply_deep_kernel<F32Io, 16>), dim3(grid),
                       dim3(MIYARN_BLOCK), 0, stream,
                       table.data_ptr<float>(),
                       grad.data_ptr<float>(), order.data_ptr<int64_t>(),
                       starts.data_ptr<int32_t>(), n_bins, nls);
  } else {
    TORCH_CHECK(grad.scalar_type() == torch::kBFloat16, "fp32/bf16 only");
    hipLaunchKernelGGL((binned_apply_deep_kernel<Bf16Io, 16>), dim3(grid),
                       dim3(MIYARN_BLOCK), 0, stream,
                       table.data_ptr<float>(),
                       reinterpret_cast<unsigned short*>(grad.data_ptr()),
                       order.data_ptr<int64_t>(),
                       starts.data_ptr<int32_t>(), n_bins, nls);
  }
}

void emb_scatter_sum_binned(torch::Tensor table, torch::Tensor ids,
                            torch::Tensor grad, int64_t g_div,
                            double alpha, torch::Tensor order,
                            torch::Tensor starts) {
  const int64_t n = ids.numel();
  TORCH_CHECK(table.is_cuda() && table.is_contiguous() &&
              table.scalar_type() == torch::kFloat32, "bad table");
  TORCH_CHECK(grad.is_cuda() && grad.is_contiguous() &&
              grad.numel() * g_div == n, "grad/g_div mismatch");
  TORCH_CHECK(order.scalar_type() == torch::kInt64 &&
              starts.scalar_type() == torch::kInt32, "bad perm dtypes");
  const int n_bins = static_cast<int>(starts.numel() - 1);
  auto stream = c10::hip::getCurrentHIPStream().stream();
  const int grid = std::min(n_bins, 8192);
  if (grad.scalar_type() == torch::kFloat32) {
    hipLaunchKernelGGL(binned_apply_scalar_kernel<F32Io>, dim3(grid),
                       dim3(MIYARN_BLOCK), 0, stream,
                       table.data_ptr<float>(),
                       grad.data_ptr<float>(), order.data_ptr<int64_t>(),
                       starts.data_ptr<int32_t>(), n_bins,
                       static_cast<int>(g_div),
                       static_cast<float>(alpha));
  } else {
    TORCH_CHECK(grad.scalar_type() == torch::kBFloat16, "fp32/bf16 only");
    hipLaunchKernelGGL(binned_apply_scalar_kernel<Bf16Io>, dim3(grid),
                       dim3(MIYARN_BLOCK), 0, stream,
                       table.data_ptr<float>(),
                       reinterpret_cast<unsigned short*>(grad.data_ptr()),
                       order.data_ptr<int64_t>(),
                       starts.data_ptr<int32_t>(), n_bins,
                       static_cast<int>(g_div),
                       static_cast<float>(alpha));
  }
}
