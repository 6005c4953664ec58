// Split-K MFMA weight-gradient kernel for MI355X (gfx950).
//
// dW[N, M] = sum_b dy[b, n] * x[b, m]  with dy [B, N], x [B, M] bf16
// row-major and B ~ 65536: the "reduction GEMM" shape hipBLASLt runs at
// 140-380 TF on this chip (measured, scripts/micro_gemm.py) because its
// split-K solutions are weak.  This kernel:
//
//  * tiles the OUTPUT 64x64 per workgroup (4 waves, each a 32x32 quadrant
//    of 2x2 mfma_f32_16x16x32_bf16 fragments),
//  * splits K (the batch) across gridDim.y, each block reducing its
//    contiguous K-chunk and writing an fp32 partial slab
//    part[sk][N][M] (atomic-free, deterministic); the host sums slabs,
//  * stages both operands TRANSPOSED into padded LDS ([n][k] / [m][k],
//    k-stride padded to 40 elems so the 16-lane ds_read_b128 groups hit
//    distinct banks) because BOTH mfma operand fragments want 8
//    consecutive k per lane (A[i][k], B[k][j] with j = lane/16*8 + reg).
//
// Guide refs: §3 fragment layout (16x16x32_bf16: 8 bf16/lane in, 4 fp32
// acc, C/D col=lane&15 row=(lane>>4)*4+reg), §2 LDS banking, G13.

#include <torch/extension.h>
#include <c10/hip/HIPStream.h>
#include "common.h"

namespace {

#define WG_BN 64
#define WG_BM 64
#define WG_BK 64
// LDS k-stride (elements): 136 B rows -> transposed b64 writes and b128
// fragment reads both bank-conflict-free (banking analysis in module doc)
#define BKPAD 68

typedef __attribute__((ext_vector_type(8))) __bf16 bf16x8_v;
typedef __attribute__((ext_vector_type(4))) float f32x4_v;

union U16x4 {
  unsigned short u[4];
  unsigned long long ll;
};

__global__ __launch_bounds__(256)
void wgrad_nt_kernel(const unsigned short* __restrict__ dy,
                     const unsigned short* __restrict__ x,
                     float* __restrict__ part,
                     int64_t B, int N, int M, int64_t chunk) {
  const int tiles_m = M / WG_BM;
  const int tile_n = blockIdx.x / tiles_m;
  const int tile_m = blockIdx.x - tile_n * tiles_m;
  const int n0 = tile_n * WG_BN;
  const int m0 = tile_m * WG_BM;
  const int64_t k_begin = (int64_t)blockIdx.y * chunk;
  const int64_t k_end = min(B, k_begin + chunk);

  __shared__ unsigned short lds[(WG_BN + WG_BM) * BKPAD];
  unsigned short* dyT = lds;                       // [WG_BN][BKPAD]
  unsigned short* xT = lds + WG_BN * BKPAD;        // [WG_BM][BKPAD]

  const int tid = threadIdx.x;
  const int lane = tid & 63;
  const int wave = tid >> 6;
  const int wn = (wave >> 1) * 32;  // wave quadrant in the 64x64 tile
  const int wm = (wave & 1) * 32;

  // staging: threads 0-127 stage dy, 128-255 stage x.  Each thread loads
  // 4 k-rows x 8 columns (4 x 16 B coalesced) and writes 8 packed
  // ds_write_b64 (4 k-values per write) into the transposed image.
  const int st_tile = tid >> 7;          // 0 = dy, 1 = x
  const int st_t = tid & 127;
  const int st_kg = (st_t >> 3) * 4;     // k base: 0,4,...,60
  const int st_c = (st_t & 7) * 8;       // col base: 0,8,...,56
  const unsigned short* st_src = st_tile ? x : dy;
  unsigned short* st_dst = st_tile ? xT : dyT;
  const int st_ld = st_tile ? M : N;
  const int st_o0 = st_tile ? m0 : n0;

  f32x4_v acc[2][2];
#pragma unroll
  for (int i = 0; i < 2; ++i)
#pragma unroll
    for (int j = 0; j < 2; ++j) acc[i][j] = {0.f, 0.f, 0.f, 0.f};

  const int a_row = lane & 15;        // fragment row (n) / col (m)
  const int a_k = (lane >> 4) * 8;    // fragment k base

  for (int64_t k0 = k_begin; k0 < k_end; k0 += WG_BK) {
    // ---- stage transposed (packed b64 writes) ---------------------------
    {
      const unsigned short* src =
          st_src + (k0 + st_kg) * (int64_t)st_ld + st_o0 + st_c;
      bf16x4 r0a = reinterpret_cast<const bf16x4*>(src)[0];
      bf16x4 r0b = reinterpret_cast<const bf16x4*>(src)[1];
      src += st_ld;
      bf16x4 r1a = reinterpret_cast<const bf16x4*>(src)[0];
      bf16x4 r1b = reinterpret_cast<const bf16x4*>(src)[1];
      src += st_ld;
      bf16x4 r2a = reinterpret_cast<const bf16x4*>(src)[0];
      bf16x4 r2b = reinterpret_cast<const bf16x4*>(src)[1];
      src += st_ld;
      bf16x4 r3a = reinterpret_cast<const bf16x4*>(src)[0];
      bf16x4 r3b = reinterpret_cast<const bf16x4*>(src)[1];
#pragma unroll
      for (int j = 0; j < 4; ++j) {
        U16x4 pa{{r0a[j], r1a[j], r2a[j], r3a[j]}};
        *reinterpret_cast<unsigned long long*>(
            st_dst + (st_c + j) * BKPAD + st_kg) = pa.ll;
        U16x4 pb{{r0b[j], r1b[j], r2b[j], r3b[j]}};
        *reinterpret_cast<unsigned long long*>(
            st_dst + (st_c + 4 + j) * BKPAD + st_kg) = pb.ll;
      }
    }
    __syncthreads();
    // ---- MFMA: 2x2 fragments x 2 k-halves -------------------------------
#pragma unroll
    for (int kh = 0; kh < 2; ++kh) {
      bf16x8_v a[2], b[2];
#pragma unroll
      for (int i = 0; i < 2; ++i) {
        a[i] = *reinterpret_cast<const bf16x8_v*>(
            dyT + (wn + i * 16 + a_row) * BKPAD + kh * 32 + a_k);
        b[i] = *reinterpret_cast<const bf16x8_v*>(
            xT + (wm + i * 16 + a_row) * BKPAD + kh * 32 + a_k);
      }
#pragma unroll
      for (int i = 0; i < 2; ++i)
#pragma unroll
        for (int j = 0; j < 2; ++j)
          acc[i][j] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
              a[i], b[j], acc[i][j], 0, 0, 0);
    }
    __syncthreads();
  }

  // ---- epilogue: fp32 partial slab --------------------------------------
  float* out = part + (int64_t)blockIdx.y * N * M;
  const int c_col = lane & 15;
  const int c_row = (lane >> 4) * 4;
#pragma unroll
  for (int i = 0; i < 2; ++i)
#pragma unroll
    for (int j = 0; j < 2; ++j)
#pragma unroll
      for (int r = 0; r < 4; ++r) {
        const int row = n0 + wn + i * 16 + c_row + r;
        const int col = m0 + wm + j * 16 + c_col;
        out[(int64_t)row * M + col] = acc[i][j][r];
      }
}

}  // namespace

torch::Tensor reduce_splitk(torch::Tensor part, bool out_bf16);

torch::Tensor wgrad_nt(torch::Tensor dy, torch::Tensor x,
                       int64_t splitk, bool out_bf16) {
  TORCH_CHECK(dy.is_cuda() && dy.is_contiguous() && dy.dim() == 2 &&
              dy.scalar_type() == torch::kBFloat16,
              "dy must be [B, N] bf16 contiguous");
  TORCH_CHECK(x.is_cuda() && x.is_contiguous() && x.dim() == 2 &&
              x.scalar_type() == torch::kBFloat16,
              "x must be [B, M] bf16 contiguous");
  const int64_t B = dy.size(0);
  const int N = static_cast<int>(dy.size(1));
  const int M = static_cast<int>(x.size(1));
  TORCH_CHECK(x.size(0) == B, "batch mismatch");
  TORCH_CHECK(N % WG_BN == 0 && M % WG_BM == 0 && B % WG_BK == 0,
              "wgrad_nt needs N, M % 64 == 0 and B % 32 == 0");
  if (splitk <= 0) {
    // enough blocks for ~2 per CU
    const int tiles = (N / WG_BN) * (M / WG_BM);
    splitk = std::max<int64_t>(1, 512 / std::max(1, tiles));
  }
  // chunk must be a multiple of WG_BK covering B
  int64_t chunk = ((B + splitk - 1) / splitk + WG_BK - 1) / WG_BK * WG_BK;
  splitk = (B + chunk - 1) / chunk;
  auto part = torch::empty({splitk, N, M},
                           dy.options().dtype(torch::kFloat32));
  auto stream = c10::hip::getCurrentHIPStream().stream();
  dim3 grid((N / WG_BN) * (M / WG_BM), splitk);
  hipLaunchKernelGGL(wgrad_nt_kernel, grid, dim3(256), 0, stream,
                     reinterpret_cast<unsigned short*>(dy.data_ptr()),
                     reinterpret_cast<unsigned short*>(x.data_ptr()),
                     part.data_ptr<float>(), B, N, M, chunk);
  // fused split-K reduce (+bf16 cast when the consumer is a
  // bf16 parameter) — replaces torch reduce + .to elementwise
  return reduce_splitk(part, out_bf16);
}
