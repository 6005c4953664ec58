// Elementwise fused kernels: bias+ReLU (fwd/bwd) and bucket dtype convert.
//
// The MLP hot path of the wide-and-deep model: GEMMs go through
// rocBLAS/hipBLASLt (library GEMMs), while the epilogue (bias+activation)
// and the reducer's mixed-precision bucket casts are fused here so
// activations make one HBM round trip instead of three
// (guide Appendix B: element-wise — vectorize bf16 as packed shorts,
// grid-stride, grid capped).

#include <torch/extension.h>
#include <c10/hip/HIPStream.h>
#include "common.h"

namespace {

// y = relu(x + bias), mask output for backward packed in sign of y (y==0).
template <typename Io>
__global__ void bias_relu_fwd_kernel(
    const typename Io::scalar_t* __restrict__ x,
    const typename Io::scalar_t* __restrict__ bias,
    typename Io::scalar_t* __restrict__ y,
    int64_t rows, int64_t cols) {
  const int64_t total = rows * cols;
  const int64_t stride = gridDim.x * (int64_t)blockDim.x;
  for (int64_t i = blockIdx.x * (int64_t)blockDim.x + threadIdx.x;
       i < total; i += stride) {
    const int64_t c = i % cols;
    float v = Io::load(x, i) + Io::load(bias, c);
    Io::store(y, i, v > 0.f ? v : 0.f);
  }
}

// Vectorized variant (cols % 4 == 0): one quad per iteration, 8-16 B/lane.
template <typename Io>
__global__ void bias_relu_fwd_vec_kernel(
    const typename Io::scalar_t* __restrict__ x,
    const typename Io::scalar_t* __restrict__ bias,
    typename Io::scalar_t* __restrict__ y,
    int64_t total_quads, int64_t qcols) {
  const int64_t stride = gridDim.x * (int64_t)blockDim.x;
  for (int64_t q = blockIdx.x * (int64_t)blockDim.x + threadIdx.x;
       q < total_quads; q += stride) {
    const int64_t cq = q % qcols;
    float v[4], b[4];
    QuadIo<Io>::load4(x, q, v);
    QuadIo<Io>::load4(bias, cq, b);
#pragma unroll
    for (int j = 0; j < 4; ++j) {
      v[j] += b[j];
      v[j] = v[j] > 0.f ? v[j] : 0.f;
    }
    QuadIo<Io>::store4(y, q, v);
  }
}

// bf16 oct variant (cols % 8 == 0): b128 loads/stores, 16 B/lane.
__global__ void bias_relu_fwd_oct_kernel(
    const unsigned short* __restrict__ x,
    const unsigned short* __restrict__ bias,
    unsigned short* __restrict__ y,
    int64_t total_octs, int64_t ocols) {
  const int64_t stride = gridDim.x * (int64_t)blockDim.x;
  for (int64_t q = blockIdx.x * (int64_t)blockDim.x + threadIdx.x;
       q < total_octs; q += stride) {
    const int64_t cq = q % ocols;
    bf16x8 xv = reinterpret_cast<const bf16x8*>(x)[q];
    bf16x8 bv = reinterpret_cast<const bf16x8*>(bias)[cq];
    bf16x8 out;
#pragma unroll
    for (int j = 0; j < 8; ++j) {
      float v = bf16_to_f32(xv[j]) + bf16_to_f32(bv[j]);
      out[j] = f32_to_bf16(v > 0.f ? v : 0.f);
    }
    reinterpret_cast<bf16x8*>(y)[q] = out;
  }
}

// Vectorized ReLU backward: dx = dy * (y > 0), quads.
template <typename Io>
__global__ void bias_relu_bwd_vec_kernel(
    const typename Io::scalar_t* __restrict__ dy,
    const typename Io::scalar_t* __restrict__ y,
    typename Io::scalar_t* __restrict__ dx,
    int64_t total_quads) {
  const int64_t stride = gridDim.x * (int64_t)blockDim.x;
  for (int64_t q = blockIdx.x * (int64_t)blockDim.x + threadIdx.x;
       q < total_quads; q += stride) {
    float g[4], yy[4];
    QuadIo<Io>::load4(dy, q, g);
    QuadIo<Io>::load4(y, q, yy);
#pragma unroll
    for (int j = 0; j < 4; ++j) g[j] = yy[j] > 0.f ? g[j] : 0.f;
    QuadIo<Io>::store4(dx, q, g);
  }
}

// Fused dx + per-block dbias PARTIALS (atomic-free): block b owns rows
// [b*DB_ROWS :: grid*DB_ROWS]; thread partials in registers; at the end
// each block writes its own row of dbias_part[grid][cols], which the host
// reduces with one tiny torch sum (2048 x cols).
#define DB_MAX_K 8  // max column-quads per thread (cols <= 4*block*K)
#define DB_ROWS 4   // rows in flight per thread (ILP for HBM latency)

template <typename Io>
__global__ void bias_relu_bwd_dbpart_kernel(
    const typename Io::scalar_t* __restrict__ dy,
    const typename Io::scalar_t* __restrict__ y,
    typename Io::scalar_t* __restrict__ dx,
    float* __restrict__ dbias_part,
    int64_t rows, int64_t cols) {
  const int64_t quads = cols >> 2;
  float acc[DB_MAX_K][4];
#pragma unroll
  for (int k = 0; k < DB_MAX_K; ++k)
#pragma unroll
    for (int j = 0; j < 4; ++j) acc[k][j] = 0.f;
  const int64_t row_stride = (int64_t)gridDim.x * DB_ROWS;
  for (int64_t r0 = (int64_t)blockIdx.x * DB_ROWS; r0 < rows;
       r0 += row_stride) {
    const int nr = min((int64_t)DB_ROWS, rows - r0);
    int k = 0;
    for (int64_t q = threadIdx.x; q < quads; q += blockDim.x, ++k) {
      float v[DB_ROWS][4], yy[DB_ROWS][4];
#pragma unroll
      for (int rr = 0; rr < DB_ROWS; ++rr)
        if (rr < nr) QuadIo<Io>::load4(dy, (r0 + rr) * quads + q, v[rr]);
#pragma unroll
      for (int rr = 0; rr < DB_ROWS; ++rr)
        if (rr < nr) QuadIo<Io>::load4(y, (r0 + rr) * quads + q, yy[rr]);
#pragma unroll
      for (int rr = 0; rr < DB_ROWS; ++rr)
        if (rr < nr) {
#pragma unroll
          for (int j = 0; j < 4; ++j) {
            v[rr][j] = yy[rr][j] > 0.f ? v[rr][j] : 0.f;
            acc[k][j] += v[rr][j];
          }
          QuadIo<Io>::store4(dx, (r0 + rr) * quads + q, v[rr]);
        }
    }
  }
  float* part = dbias_part + (int64_t)blockIdx.x * cols;
  int k = 0;
  for (int64_t q = threadIdx.x; q < quads; q += blockDim.x, ++k) {
    f32x4 o;
#pragma unroll
    for (int j = 0; j < 4; ++j) o[j] = acc[k][j];
    reinterpret_cast<f32x4*>(part)[q] = o;
  }
}

// bf16 oct variant of the dbpart kernel: b128 loads/stores (16 B/lane,
// 8 bf16) instead of b64 quads — the quad version streams at ~4 TB/s,
// leaving HBM bandwidth on the table for these 2-read/1-write layers.
// 2D thread map: each thread owns exactly ONE column-oct and the block
// covers blockDim/octs row slots concurrently, so narrow layers (the
// 256-wide MLP tail ran at 1.9 TB/s with a 1D map) still fill the chip.
// Requires octs to divide blockDim; one partial row per (block, slot).
// dx keeps the original dy bits where the mask passes (no round trip).
__global__ void bias_relu_bwd_dbpart8_kernel(
    const unsigned short* __restrict__ dy,
    const unsigned short* __restrict__ y,
    unsigned short* __restrict__ dx,
    float* __restrict__ dbias_part,  // [gridDim.x * rpb][cols]
    int64_t rows, int64_t cols) {
  const int octs = (int)(cols >> 3);
  const int rpb = blockDim.x / octs;       // row slots per block
  const int rid = threadIdx.x / octs;      // my slot
  const int64_t q = threadIdx.x - rid * octs;  // my oct (fixed)
  float acc[8];
#pragma unroll
  for (int j = 0; j < 8; ++j) acc[j] = 0.f;
  const int64_t slot = (int64_t)blockIdx.x * rpb + rid;
  const int64_t row_stride = (int64_t)gridDim.x * rpb * DB_ROWS;
  for (int64_t r0 = slot * DB_ROWS; r0 < rows; r0 += row_stride) {
    const int nr = min((int64_t)DB_ROWS, rows - r0);
    bf16x8 gv[DB_ROWS], yv[DB_ROWS];
#pragma unroll
    for (int rr = 0; rr < DB_ROWS; ++rr)
      if (rr < nr)
        gv[rr] = reinterpret_cast<const bf16x8*>(
            dy)[(r0 + rr) * octs + q];
#pragma unroll
    for (int rr = 0; rr < DB_ROWS; ++rr)
      if (rr < nr)
        yv[rr] = reinterpret_cast<const bf16x8*>(
            y)[(r0 + rr) * octs + q];
#pragma unroll
    for (int rr = 0; rr < DB_ROWS; ++rr)
      if (rr < nr) {
        bf16x8 out;
#pragma unroll
        for (int j = 0; j < 8; ++j) {
          const bool keep = bf16_to_f32(yv[rr][j]) > 0.f;
          out[j] = keep ? gv[rr][j] : (unsigned short)0;
          if (keep) acc[j] += bf16_to_f32(gv[rr][j]);
        }
        reinterpret_cast<bf16x8*>(dx)[(r0 + rr) * octs + q] = out;
      }
  }
  float* part = dbias_part + slot * cols;
  f32x4 lo, hi;
#pragma unroll
  for (int j = 0; j < 4; ++j) { lo[j] = acc[j]; hi[j] = acc[4 + j]; }
  reinterpret_cast<f32x4*>(part)[q * 2] = lo;
  reinterpret_cast<f32x4*>(part)[q * 2 + 1] = hi;
}

// dx = dy * (y > 0)
template <typename Io>
__global__ void bias_relu_bwd_kernel(
    const typename Io::scalar_t* __restrict__ dy,
    const typename Io::scalar_t* __restrict__ y,
    typename Io::scalar_t* __restrict__ dx,
    int64_t total) {
  const int64_t stride = gridDim.x * (int64_t)blockDim.x;
  for (int64_t i = blockIdx.x * (int64_t)blockDim.x + threadIdx.x;
       i < total; i += stride) {
    float g = Io::load(dy, i);
    float yy = Io::load(y, i);
    Io::store(dx, i, yy > 0.f ? g : 0.f);
  }
}

__global__ void bf16_to_f32_kernel(const unsigned short* __restrict__ src,
                                   float* __restrict__ dst, int64_t n,
                                   float scale) {
  const int64_t stride = gridDim.x * (int64_t)blockDim.x;
  const int64_t nvec = n >> 2;
  const int64_t tid = blockIdx.x * (int64_t)blockDim.x + threadIdx.x;
  for (int64_t i = tid; i < nvec; i += stride) {
    bf16x4 v = reinterpret_cast<const bf16x4*>(src)[i];
    f32x4 o;
#pragma unroll
    for (int j = 0; j < 4; ++j) o[j] = bf16_to_f32(v[j]) * scale;
    reinterpret_cast<f32x4*>(dst)[i] = o;
  }
  for (int64_t i = (nvec << 2) + tid; i < n; i += stride)
    dst[i] = bf16_to_f32(src[i]) * scale;
}

__global__ void f32_to_bf16_kernel(const float* __restrict__ src,
                                   unsigned short* __restrict__ dst,
                                   int64_t n, float scale) {
  const int64_t stride = gridDim.x * (int64_t)blockDim.x;
  const int64_t nvec = n >> 2;
  const int64_t tid = blockIdx.x * (int64_t)blockDim.x + threadIdx.x;
  for (int64_t i = tid; i < nvec; i += stride) {
    f32x4 v = reinterpret_cast<const f32x4*>(src)[i];
    bf16x4 o;
#pragma unroll
    for (int j = 0; j < 4; ++j) o[j] = f32_to_bf16(v[j] * scale);
    reinterpret_cast<bf16x4*>(dst)[i] = o;
  }
  for (int64_t i = (nvec << 2) + tid; i < n; i += stride)
    dst[i] = f32_to_bf16(src[i] * scale);
}


// Split-K / column-partial reduction with the output cast fused:
// out[j] (fp32 or bf16) = sum_s part[s][j].  Replaces torch's
// reduce_kernel + the separate .to(bf16) elementwise pass on the wgrad
// and dbias paths (8 reduce + 3 cast kernels, ~125 us/step at the
// bench shape).  Threads own f32x4 quads; s-slices stream coalesced.
template <typename OIo>
__global__ void reduce_splitk_kernel(
    const float* __restrict__ part,
    typename OIo::scalar_t* __restrict__ out, int S, int64_t nm) {
  const int64_t quads = nm >> 2;
  const int64_t stride = gridDim.x * (int64_t)blockDim.x;
  for (int64_t q = blockIdx.x * (int64_t)blockDim.x + threadIdx.x;
       q < quads; q += stride) {
    f32x4 acc = reinterpret_cast<const f32x4*>(part)[q];
    for (int s = 1; s < S; ++s) {
      const f32x4 v =
          reinterpret_cast<const f32x4*>(part + (int64_t)s * nm)[q];
#pragma unroll
      for (int j = 0; j < 4; ++j) acc[j] += v[j];
    }
    float vv[4];
#pragma unroll
    for (int j = 0; j < 4; ++j) vv[j] = acc[j];
    QuadIo<OIo>::store4(out, q, vv);
  }
}

}  // namespace

torch::Tensor bias_relu_fwd(torch::Tensor x, torch::Tensor bias) {
  TORCH_CHECK(x.is_cuda() && x.is_contiguous(), "x must be GPU contiguous");
  TORCH_CHECK(bias.is_cuda() && bias.is_contiguous() &&
              bias.scalar_type() == x.scalar_type(), "bias mismatch");
  const int64_t cols = x.size(-1);
  const int64_t rows = x.numel() / cols;
  TORCH_CHECK(bias.numel() == cols, "bias size mismatch");
  auto y = torch::empty_like(x);
  auto stream = c10::hip::getCurrentHIPStream().stream();
  if (x.scalar_type() == torch::kBFloat16 && cols % 8 == 0) {
    const int64_t tq = rows * (cols / 8);
    int grid = miyarn_grid(tq);
    hipLaunchKernelGGL(bias_relu_fwd_oct_kernel, dim3(grid),
                       dim3(MIYARN_BLOCK), 0, stream,
                       reinterpret_cast<unsigned short*>(x.data_ptr()),
                       reinterpret_cast<unsigned short*>(bias.data_ptr()),
                       reinterpret_cast<unsigned short*>(y.data_ptr()),
                       tq, cols / 8);
    return y;
  }
  const bool vec = (cols % 4 == 0);
  if (vec) {
    const int64_t tq = rows * (cols / 4);
    int grid = miyarn_grid(tq);
    if (x.scalar_type() == torch::kFloat32) {
      hipLaunchKernelGGL(bias_relu_fwd_vec_kernel<F32Io>, dim3(grid),
                         dim3(MIYARN_BLOCK), 0, stream,
                         x.data_ptr<float>(), bias.data_ptr<float>(),
                         y.data_ptr<float>(), tq, cols / 4);
    } else {
      TORCH_CHECK(x.scalar_type() == torch::kBFloat16, "fp32/bf16 only");
      hipLaunchKernelGGL(bias_relu_fwd_vec_kernel<Bf16Io>, dim3(grid),
                         dim3(MIYARN_BLOCK), 0, stream,
                         reinterpret_cast<unsigned short*>(x.data_ptr()),
                         reinterpret_cast<unsigned short*>(bias.data_ptr()),
                         reinterpret_cast<unsigned short*>(y.data_ptr()),
                         tq, cols / 4);
    }
    return y;
  }
  int grid = miyarn_grid(rows * cols);
  if (x.scalar_type() == torch::kFloat32) {
    hipLaunchKernelGGL(bias_relu_fwd_kernel<F32Io>, dim3(grid),
                       dim3(MIYARN_BLOCK), 0, stream, x.data_ptr<float>(),
                       bias.data_ptr<float>(), y.data_ptr<float>(),
                       rows, cols);
  } else {
    TORCH_CHECK(x.scalar_type() == torch::kBFloat16, "fp32/bf16 only");
    hipLaunchKernelGGL(bias_relu_fwd_kernel<Bf16Io>, dim3(grid),
                       dim3(MIYARN_BLOCK), 0, stream,
                       reinterpret_cast<unsigned short*>(x.data_ptr()),
                       reinterpret_cast<unsigned short*>(bias.data_ptr()),
                       reinterpret_cast<unsigned short*>(y.data_ptr()),
                       rows, cols);
  }
  return y;
}

torch::Tensor bias_relu_bwd(torch::Tensor dy, torch::Tensor y) {
  TORCH_CHECK(dy.is_cuda() && dy.is_contiguous() && y.is_contiguous(),
              "dy/y must be GPU contiguous");
  TORCH_CHECK(dy.scalar_type() == y.scalar_type() &&
              dy.numel() == y.numel(), "dy/y mismatch");
  auto dx = torch::empty_like(dy);
  auto stream = c10::hip::getCurrentHIPStream().stream();
  int64_t total = dy.numel();
  if (total % 4 == 0) {
    const int64_t tq = total / 4;
    int grid = miyarn_grid(tq);
    if (dy.scalar_type() == torch::kFloat32) {
      hipLaunchKernelGGL(bias_relu_bwd_vec_kernel<F32Io>, dim3(grid),
                         dim3(MIYARN_BLOCK), 0, stream,
                         dy.data_ptr<float>(), y.data_ptr<float>(),
                         dx.data_ptr<float>(), tq);
    } else {
      TORCH_CHECK(dy.scalar_type() == torch::kBFloat16, "fp32/bf16 only");
      hipLaunchKernelGGL(bias_relu_bwd_vec_kernel<Bf16Io>, dim3(grid),
                         dim3(MIYARN_BLOCK), 0, stream,
                         reinterpret_cast<unsigned short*>(dy.data_ptr()),
                         reinterpret_cast<unsigned short*>(y.data_ptr()),
                         reinterpret_cast<unsigned short*>(dx.data_ptr()),
                         tq);
    }
    return dx;
  }
  int grid = miyarn_grid(total);
  if (dy.scalar_type() == torch::kFloat32) {
    hipLaunchKernelGGL(bias_relu_bwd_kernel<F32Io>, dim3(grid),
                       dim3(MIYARN_BLOCK), 0, stream, dy.data_ptr<float>(),
                       y.data_ptr<float>(), dx.data_ptr<float>(), total);
  } else {
    TORCH_CHECK(dy.scalar_type() == torch::kBFloat16, "fp32/bf16 only");
    hipLaunchKernelGGL(bias_relu_bwd_kernel<Bf16Io>, dim3(grid),
                       dim3(MIYARN_BLOCK), 0, stream,
                       reinterpret_cast<unsigned short*>(dy.data_ptr()),
                       reinterpret_cast<unsigned short*>(y.data_ptr()),
                       reinterpret_cast<unsigned short*>(dx.data_ptr()),
                       total);
  }
  return dx;
}

torch::Tensor reduce_splitk(torch::Tensor part, bool out_bf16);

std::vector<torch::Tensor> bias_relu_bwd_db(torch::Tensor dy,
                                            torch::Tensor y,
                                            bool dbias_bf16) {
  TORCH_CHECK(dy.is_cuda() && dy.is_contiguous() && y.is_contiguous(),
              "dy/y must be GPU contiguous");
  TORCH_CHECK(dy.scalar_type() == y.scalar_type() &&
              dy.numel() == y.numel(), "dy/y mismatch");
  const int64_t cols = dy.size(-1);
  const int64_t rows = dy.numel() / cols;
  TORCH_CHECK(cols % 4 == 0 && cols <= 4 * MIYARN_BLOCK * DB_MAX_K,
              "bias_relu_bwd_db needs cols % 4 == 0 and cols <= ",
              4 * MIYARN_BLOCK * DB_MAX_K);
  auto dx = torch::empty_like(dy);
  auto stream = c10::hip::getCurrentHIPStream().stream();
  // Block size matched to the quad count so no thread idles at small cols.
  const int64_t quads = cols / 4;
  int block = static_cast<int>(
      std::min<int64_t>(MIYARN_BLOCK, ((quads + 63) / 64) * 64));
  int grid = static_cast<int>(std::min<int64_t>(
      (rows + DB_ROWS - 1) / DB_ROWS, MIYARN_MAX_BLOCKS));
  const int64_t octs = cols / 8;
  const bool oct_ok = dy.scalar_type() == torch::kBFloat16 &&
                      cols % 8 == 0 && octs <= MIYARN_BLOCK &&
                      MIYARN_BLOCK % octs == 0;
  // Wide layers measured best with one row slot per block (block = octs);
  // narrow ones need multiple slots to fill the chip (65536x256 ran at
  // 1.9 TB/s single-slot vs 2.2 multi-slot).
  const int block8 = octs >= 128 ? static_cast<int>(octs) : MIYARN_BLOCK;
  const int64_t rpb = oct_ok ? block8 / octs : 1;
  if (oct_ok)
    // Fill the chip: narrow layers (256 cols -> rpb=8) were capped at
    // 256 blocks (1/CU) by a MAX_BLOCKS/rpb partial-row bound and ran
    // at 1.9 TB/s.  Allow up to 4x MAX_BLOCKS partial rows (>= 4
    // blocks/CU for every width) — unbounded partials measured the
    // dbias reduce growing +18 us/step, eating half the streaming win.
    grid = static_cast<int>(std::min<int64_t>(
        (rows + rpb * DB_ROWS - 1) / (rpb * DB_ROWS),
        std::min<int64_t>(MIYARN_MAX_BLOCKS,
                          4 * MIYARN_MAX_BLOCKS / rpb)));
  // Atomic-free column partials: one row per block (per slot for the
  // oct kernel), reduced below.
  auto part = torch::empty({grid * rpb, cols},
                           dy.options().dtype(torch::kFloat32));
  if (oct_ok) {
    hipLaunchKernelGGL(bias_relu_bwd_dbpart8_kernel, dim3(grid),
                       dim3(block8), 0, stream,
                       reinterpret_cast<unsigned short*>(dy.data_ptr()),
                       reinterpret_cast<unsigned short*>(y.data_ptr()),
                       reinterpret_cast<unsigned short*>(dx.data_ptr()),
                       part.data_ptr<float>(), rows, cols);
  } else if (dy.scalar_type() == torch::kFloat32) {
    hipLaunchKernelGGL(bias_relu_bwd_dbpart_kernel<F32Io>, dim3(grid),
                       dim3(block), 0, stream,
                       dy.data_ptr<float>(), y.data_ptr<float>(),
                       dx.data_ptr<float>(), part.data_ptr<float>(),
                       rows, cols);
  } else {
    TORCH_CHECK(dy.scalar_type() == torch::kBFloat16, "fp32/bf16 only");
    hipLaunchKernelGGL(bias_relu_bwd_dbpart_kernel<Bf16Io>, dim3(grid),
                       dim3(block), 0, stream,
                       reinterpret_cast<unsigned short*>(dy.data_ptr()),
                       reinterpret_cast<unsigned short*>(y.data_ptr()),
                       reinterpret_cast<unsigned short*>(dx.data_ptr()),
                       part.data_ptr<float>(), rows, cols);
  }
  torch::Tensor dbias;
  if (cols % 4 == 0)
    dbias = reduce_splitk(part, dbias_bf16);
  else {
    dbias = part.sum(0);
    if (dbias_bf16) dbias = dbias.to(torch::kBFloat16);
  }
  return {dx, dbias};
}

void convert_scaled(torch::Tensor src, torch::Tensor dst, double scale) {
  TORCH_CHECK(src.is_cuda() && src.is_contiguous() &&
              dst.is_cuda() && dst.is_contiguous(), "GPU contiguous only");
  TORCH_CHECK(src.numel() == dst.numel(), "numel mismatch");
  auto stream = c10::hip::getCurrentHIPStream().stream();
  int64_t n = src.numel();
  int grid = miyarn_grid((n + 3) / 4);
  if (src.scalar_type() == torch::kBFloat16 &&
      dst.scalar_type() == torch::kFloat32) {
    hipLaunchKernelGGL(bf16_to_f32_kernel, dim3(grid), dim3(MIYARN_BLOCK), 0,
                       stream,
                       reinterpret_cast<unsigned short*>(src.data_ptr()),
                       dst.data_ptr<float>(), n, (float)scale);
  } else if (src.scalar_type() == torch::kFloat32 &&
             dst.scalar_type() == torch::kBFloat16) {
    hipLaunchKernelGGL(f32_to_bf16_kernel, dim3(grid), dim3(MIYARN_BLOCK), 0,
                       stream, src.data_ptr<float>(),
                       reinterpret_cast<unsigned short*>(dst.data_ptr()),
                       n, (float)scale);
  } else {
    TORCH_CHECK(false, "convert_scaled supports bf16<->fp32 only");
  }
}

namespace {

// dw[m] = sum_b dy[b] * x[b, m]  — the wgrad of a single-logit head as a
// memory-bound column reduction (hipBLASLt runs this M=1-ish GEMM at
// ~190 us for 33 MB; this kernel is a straight stream).  Atomic-free:
// per-block partials + host-side sum, like bias_relu_bwd_dbpart.
template <typename Io>
__global__ void col_reduce_dot_kernel(
    const typename Io::scalar_t* __restrict__ x,
    const typename Io::scalar_t* __restrict__ dy,
    float* __restrict__ part,  // [gridDim.x * slots][cols]
    int64_t rows, int64_t cols) {
  const int64_t quads = cols >> 2;
  // 2D map for narrow inputs: when the quad count divides the block,
  // the block covers blockDim/quads row slots concurrently (a 64-thread
  // 1D map left the chip underfilled on the 256-wide head).
  const int bd = blockDim.x;
  const bool multi = quads <= bd && bd % (int)quads == 0;
  const int tpq = multi ? (int)quads : bd;
  const int slots = multi ? bd / (int)quads : 1;
  const int rid = threadIdx.x / tpq;
  const int ltid = threadIdx.x - rid * tpq;
  float acc[DB_MAX_K][4];
#pragma unroll
  for (int k = 0; k < DB_MAX_K; ++k)
#pragma unroll
    for (int j = 0; j < 4; ++j) acc[k][j] = 0.f;
  const int64_t slot = (int64_t)blockIdx.x * slots + rid;
  const int64_t row_stride = (int64_t)gridDim.x * slots * DB_ROWS;
  for (int64_t r0 = slot * DB_ROWS; r0 < rows; r0 += row_stride) {
    const int nr = min((int64_t)DB_ROWS, rows - r0);
    float w[DB_ROWS];
#pragma unroll
    for (int rr = 0; rr < DB_ROWS; ++rr)
      w[rr] = rr < nr ? Io::load(dy, r0 + rr) : 0.f;
    int k = 0;
    for (int64_t q = ltid; q < quads; q += tpq, ++k) {
      float v[DB_ROWS][4];
#pragma unroll
      for (int rr = 0; rr < DB_ROWS; ++rr)
        if (rr < nr) QuadIo<Io>::load4(x, (r0 + rr) * quads + q, v[rr]);
#pragma unroll
      for (int rr = 0; rr < DB_ROWS; ++rr)
        if (rr < nr)
#pragma unroll
          for (int j = 0; j < 4; ++j) acc[k][j] += w[rr] * v[rr][j];
    }
  }
  float* prow = part + slot * cols;
  int k = 0;
  for (int64_t q = ltid; q < quads; q += tpq, ++k) {
    f32x4 o;
#pragma unroll
    for (int j = 0; j < 4; ++j) o[j] = acc[k][j];
    reinterpret_cast<f32x4*>(prow)[q] = o;
  }
}

}  // namespace

torch::Tensor col_reduce_dot(torch::Tensor x, torch::Tensor dy) {
  TORCH_CHECK(x.is_cuda() && x.is_contiguous() && x.dim() == 2,
              "x must be contiguous 2D on GPU");
  TORCH_CHECK(dy.is_cuda() && dy.is_contiguous() &&
              dy.numel() == x.size(0), "dy must be [rows]");
  TORCH_CHECK(dy.scalar_type() == x.scalar_type(), "dtype mismatch");
  const int64_t rows = x.size(0);
  const int64_t cols = x.size(1);
  TORCH_CHECK(cols % 4 == 0 && cols <= 4 * MIYARN_BLOCK * DB_MAX_K,
              "col_reduce_dot needs cols % 4 == 0 and small cols");
  auto stream = c10::hip::getCurrentHIPStream().stream();
  const int64_t quads = cols / 4;
  const bool multi = quads <= MIYARN_BLOCK && MIYARN_BLOCK % quads == 0;
  int block = multi ? MIYARN_BLOCK
                    : static_cast<int>(std::min<int64_t>(
                          MIYARN_BLOCK, ((quads + 63) / 64) * 64));
  const int64_t slots = multi ? MIYARN_BLOCK / quads : 1;
  int grid = static_cast<int>(std::max<int64_t>(
      1, std::min<int64_t>((rows + slots * DB_ROWS - 1) / (slots * DB_ROWS),
                           MIYARN_MAX_BLOCKS / slots)));
  auto part = torch::empty({grid * slots, cols},
                           x.options().dtype(torch::kFloat32));
  if (x.scalar_type() == torch::kFloat32) {
    hipLaunchKernelGGL(col_reduce_dot_kernel<F32Io>, dim3(grid),
                       dim3(block), 0, stream, x.data_ptr<float>(),
                       dy.data_ptr<float>(), part.data_ptr<float>(),
                       rows, cols);
  } else {
    TORCH_CHECK(x.scalar_type() == torch::kBFloat16, "fp32/bf16 only");
    hipLaunchKernelGGL(col_reduce_dot_kernel<Bf16Io>, dim3(grid),
                       dim3(block), 0, stream,
                       reinterpret_cast<unsigned short*>(x.data_ptr()),
                       reinterpret_cast<unsigned short*>(dy.data_ptr()),
                       part.data_ptr<float>(), rows, cols);
  }
  return cols % 4 == 0 ? reduce_splitk(part, false) : part.sum(0);
}

namespace {

// y[r] = x[r] . w + bias for the single-logit head FORWARD.  hipBLASLt
// runs this M=1 bias-GEMV at ~175 GB/s (192 us measured at 65536x256 in
// the bench profile); this is a plain streaming dot: 16 lanes per row,
// fp32 accumulate, cross-lane shuffle reduce.  w is preloaded into
// registers once (cols <= 512) and reused for every row.
#define RD_MAX_Q 8  // quads per lane -> cols <= 16*4*8 = 512

template <typename Io>
__global__ void row_dot_kernel(
    const typename Io::scalar_t* __restrict__ x,
    const typename Io::scalar_t* __restrict__ w,
    const typename Io::scalar_t* __restrict__ bias,  // 1 elem or null
    typename Io::scalar_t* __restrict__ y,
    int64_t rows, int64_t cols) {
  const int64_t quads = cols >> 2;
  const int lane16 = threadIdx.x & 15;
  const int nq = (int)((quads - lane16 + 15) >> 4);  // quads this lane
  float wreg[RD_MAX_Q][4];
  for (int i = 0; i < nq; ++i)
    QuadIo<Io>::load4(w, lane16 + (int64_t)i * 16, wreg[i]);
  const float bv = bias ? Io::load(bias, 0) : 0.f;

  const int64_t groups = ((int64_t)gridDim.x * blockDim.x) >> 4;
  const int64_t g0 = ((int64_t)blockIdx.x * blockDim.x + threadIdx.x) >> 4;
  for (int64_t r = g0; r < rows; r += groups) {
    float acc = 0.f;
    for (int i = 0; i < nq; ++i) {
      float v[4];
      QuadIo<Io>::load4(x, r * quads + lane16 + (int64_t)i * 16, v);
#pragma unroll
      for (int j = 0; j < 4; ++j) acc += v[j] * wreg[i][j];
    }
#pragma unroll
    for (int off = 8; off > 0; off >>= 1)
      acc += __shfl_xor(acc, off, 16);
    if (lane16 == 0) Io::store(y, r, acc + bv);
  }
}

}  // namespace

torch::Tensor row_dot(torch::Tensor x, torch::Tensor w,
                      c10::optional<torch::Tensor> bias) {
  TORCH_CHECK(x.is_cuda() && x.is_contiguous() && x.dim() == 2,
              "x must be contiguous 2D on GPU");
  TORCH_CHECK(w.is_cuda() && w.is_contiguous() &&
              w.numel() == x.size(1), "w must be [cols]");
  TORCH_CHECK(w.scalar_type() == x.scalar_type(), "dtype mismatch");
  const int64_t rows = x.size(0);
  const int64_t cols = x.size(1);
  TORCH_CHECK(cols % 4 == 0 && cols <= 64 * RD_MAX_Q,
              "row_dot needs cols % 4 == 0 and cols <= 512");
  const void* bias_ptr = nullptr;
  if (bias.has_value()) {
    TORCH_CHECK(bias->is_cuda() && bias->numel() == 1 &&
                bias->scalar_type() == x.scalar_type(),
                "bias must be a 1-element tensor of x dtype");
    bias_ptr = bias->data_ptr();
  }
  auto y = torch::empty({rows}, x.options());
  auto stream = c10::hip::getCurrentHIPStream().stream();
  const int grid = miyarn_grid_cap(rows * 16, MIYARN_MAX_BLOCKS);
  if (x.scalar_type() == torch::kFloat32) {
    hipLaunchKernelGGL(row_dot_kernel<F32Io>, dim3(grid),
                       dim3(MIYARN_BLOCK), 0, stream, x.data_ptr<float>(),
                       w.data_ptr<float>(),
                       static_cast<const float*>(bias_ptr),
                       y.data_ptr<float>(), rows, cols);
  } else {
    TORCH_CHECK(x.scalar_type() == torch::kBFloat16, "fp32/bf16 only");
    hipLaunchKernelGGL(row_dot_kernel<Bf16Io>, dim3(grid),
                       dim3(MIYARN_BLOCK), 0, stream,
                       reinterpret_cast<unsigned short*>(x.data_ptr()),
                       reinterpret_cast<unsigned short*>(w.data_ptr()),
                       static_cast<const unsigned short*>(bias_ptr),
                       reinterpret_cast<unsigned short*>(y.data_ptr()),
                       rows, cols);
  }
  return y;
}

namespace {

// Fused wide-and-deep head + BCEWithLogits loss: one kernel sums the
// three logit parts in fp32 and emits the stable per-example loss plus
// sigmoid(z) for the backward (replaces ~5 small elementwise kernels:
// two adds, a bf16->f32 cast, the BCE forward, and its backward prep).
__global__ void bce_head_fwd_kernel(
    const unsigned short* __restrict__ deep,
    const unsigned short* __restrict__ wide,
    const unsigned short* __restrict__ dhead,
    const float* __restrict__ labels,
    float* __restrict__ loss,
    unsigned short* __restrict__ sig,
    int64_t n) {
  const int64_t stride = gridDim.x * (int64_t)blockDim.x;
  for (int64_t i = blockIdx.x * (int64_t)blockDim.x + threadIdx.x;
       i < n; i += stride) {
    const float z = bf16_to_f32(deep[i]) + bf16_to_f32(wide[i]) +
                    bf16_to_f32(dhead[i]);
    const float yv = labels[i];
    // numerically stable: max(z,0) - z*y + log1p(exp(-|z|))
    loss[i] = fmaxf(z, 0.f) - z * yv + log1pf(expf(-fabsf(z)));
    sig[i] = f32_to_bf16(1.f / (1.f + expf(-z)));
  }
}

__global__ void bce_head_bwd_kernel(
    const unsigned short* __restrict__ sig,
    const float* __restrict__ labels,
    const float* __restrict__ g,  // upstream grad per element
    unsigned short* __restrict__ dl,
    int64_t n) {
  const int64_t stride = gridDim.x * (int64_t)blockDim.x;
  for (int64_t i = blockIdx.x * (int64_t)blockDim.x + threadIdx.x;
       i < n; i += stride)
    dl[i] = f32_to_bf16((bf16_to_f32(sig[i]) - labels[i]) * g[i]);
}

}  // namespace

std::vector<torch::Tensor> bce_head_fwd(torch::Tensor deep,
                                        torch::Tensor wide,
                                        torch::Tensor dhead,
                                        torch::Tensor labels) {
  TORCH_CHECK(deep.is_cuda() && deep.is_contiguous() &&
              deep.scalar_type() == torch::kBFloat16, "deep bf16 GPU");
  const int64_t n = deep.numel();
  for (const auto& t : {wide, dhead}) {
    TORCH_CHECK(t.is_cuda() && t.is_contiguous() && t.numel() == n &&
                t.scalar_type() == torch::kBFloat16, "logit part bf16");
  }
  TORCH_CHECK(labels.is_cuda() && labels.is_contiguous() &&
              labels.numel() == n &&
              labels.scalar_type() == torch::kFloat32, "labels fp32");
  auto loss = torch::empty({n}, deep.options().dtype(torch::kFloat32));
  auto sig = torch::empty({n}, deep.options());
  auto stream = c10::hip::getCurrentHIPStream().stream();
  hipLaunchKernelGGL(bce_head_fwd_kernel, dim3(miyarn_grid(n)),
                     dim3(MIYARN_BLOCK), 0, stream,
                     reinterpret_cast<unsigned short*>(deep.data_ptr()),
                     reinterpret_cast<unsigned short*>(wide.data_ptr()),
                     reinterpret_cast<unsigned short*>(dhead.data_ptr()),
                     labels.data_ptr<float>(), loss.data_ptr<float>(),
                     reinterpret_cast<unsigned short*>(sig.data_ptr()), n);
  return {loss, sig};
}

torch::Tensor bce_head_bwd(torch::Tensor sig, torch::Tensor labels,
                           torch::Tensor g) {
  const int64_t n = sig.numel();
  TORCH_CHECK(sig.is_cuda() && sig.is_contiguous() &&
              sig.scalar_type() == torch::kBFloat16, "sig bf16 GPU");
  TORCH_CHECK(labels.numel() == n && g.numel() == n &&
              g.scalar_type() == torch::kFloat32 && g.is_contiguous(),
              "labels/g mismatch");
  auto dl = torch::empty({n}, sig.options());
  auto stream = c10::hip::getCurrentHIPStream().stream();
  hipLaunchKernelGGL(bce_head_bwd_kernel, dim3(miyarn_grid(n)),
                     dim3(MIYARN_BLOCK), 0, stream,
                     reinterpret_cast<unsigned short*>(sig.data_ptr()),
                     labels.data_ptr<float>(), g.data_ptr<float>(),
                     reinterpret_cast<unsigned short*>(dl.data_ptr()), n);
  return dl;
}

torch::Tensor reduce_splitk(torch::Tensor part, bool out_bf16) {
  TORCH_CHECK(part.is_cuda() && part.is_contiguous() && part.dim() >= 2 &&
              part.scalar_type() == torch::kFloat32, "bad partials");
  const int S = static_cast<int>(part.size(0));
  const int64_t nm = part.numel() / S;
  TORCH_CHECK(nm % 4 == 0, "reduce_splitk needs inner numel %% 4 == 0");
  if (S > 64 || nm < 4 * MIYARN_BLOCK) {
    // The vectorized kernel parallelizes over the inner dim only; deep
    // stacks of narrow partials (the dbias case: S up to 8192, nm 256)
    // would serialize S-long latency chains on a handful of blocks
    // (measured 4x whole-step regression).  torch's tree reduce is the
    // right shape there.
    auto out = part.sum(0);
    return out_bf16 ? out.to(torch::kBFloat16) : out;
  }
  auto sizes = part.sizes().vec();
  sizes.erase(sizes.begin());
  auto out = torch::empty(sizes, part.options().dtype(
      out_bf16 ? torch::kBFloat16 : torch::kFloat32));
  auto stream = c10::hip::getCurrentHIPStream().stream();
  const int grid = miyarn_grid(nm / 4);
  if (out_bf16) {
    hipLaunchKernelGGL(reduce_splitk_kernel<Bf16Io>, dim3(grid),
                       dim3(MIYARN_BLOCK), 0, stream,
                       part.data_ptr<float>(),
                       reinterpret_cast<unsigned short*>(out.data_ptr()),
                       S, nm);
  } else {
    hipLaunchKernelGGL(reduce_splitk_kernel<F32Io>, dim3(grid),
                       dim3(MIYARN_BLOCK), 0, stream,
                       part.data_ptr<float>(), out.data_ptr<float>(),
                       S, nm);
  }
  return out;
}
