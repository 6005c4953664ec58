// Split-K MFMA weight-gradient kernel, 128x128 output tiles (v3).
//
// v2 (wgrad.hip, 64x64 tiles) measured 118-129 TF: redundant operand
// traffic (dy re-read tiles_m times, x re-read tiles_n times) plus LDS
// write conflicts bound it.  This version:
//  * 128x128 tiles halve the redundant traffic,
//  * LDS images are [n][64k] rows of 128 B with the guide's
//    `byte ^= ((row&7)<<4)` XOR swizzle (T2): packed b64 transposed
//    writes and b128 fragment reads both conflict-light,
//  * column edges are masked (M need not divide 128), so it serves the
//    model's 432-wide first layer directly,
//  * fp32 partial slabs per split-K block, host-side sum (deterministic).
//
// Fragment maps (guide §3, mfma_f32_16x16x32_bf16): per lane
//   A[i = lane%16][k = (lane/16)*8 + j],  B[k][m] same shape,
//   C/D col = lane&15, row = (lane>>4)*4 + reg.

#include <torch/extension.h>
#include <c10/hip/HIPStream.h>
#include "common.h"

namespace {

#define W3_BN 128
#define W3_BM 128
#define W3_BK 64

typedef __attribute__((ext_vector_type(8))) __bf16 bf16x8_v;
typedef __attribute__((ext_vector_type(4))) float f32x4_v;

union U16x4b {
  unsigned short u[4];
  unsigned long long ll;
};

// LDS byte offset of element [row n][col k] in a [128][64] bf16 image,
// XOR-swizzled per the guide (row stride 128 B).
__device__ __forceinline__ int lds_off(int n, int k) {
  int byte = n * 128 + k * 2;
  // combined swizzle (v1.5): (n&7) spreads the b128 READ groups
  // (consecutive rows), (n>>3)&7 spreads the b64 WRITE groups (the
  // plain (n&7) term made those a 16-way conflict), and the (n>>6)
  // bit-7 term splits the write groups across the two 128 B halves of
  // a bank sweep — 4-way -> 2-way write conflicts (the 16 B-granular
  // XOR floor for this staging shape; verified by bank simulation and
  // PMC SQ_LDS_BANK_CONFLICT).  Reads stay at the conflict-free 4
  // (bit 7 is constant across each 16-row read group).
  const int s = ((n ^ (n >> 3)) & 7) | (((n >> 6) & 1) << 3);
  return byte ^ (s << 4);
}

__global__ __launch_bounds__(256)
void wgrad128_kernel(const unsigned short* __restrict__ dy,
                     const unsigned short* __restrict__ x,
                     float* __restrict__ part,
                     int64_t B, int N, int M, int64_t chunk) {
  const int tiles_m = (M + W3_BM - 1) / W3_BM;
  // XCD-aware slab-major mapping (see wgrad256.hip): same-slab tiles
  // co-locate on one XCD and share operand column reads via its L2.
  const int tile_n = blockIdx.y / tiles_m;
  const int tile_m = blockIdx.y - tile_n * tiles_m;
  const int n0 = tile_n * W3_BN;
  const int m0 = tile_m * W3_BM;
  const int64_t k_begin = (int64_t)blockIdx.x * chunk;
  const int64_t k_end = min(B, k_begin + chunk);

  __shared__ __attribute__((aligned(16))) unsigned char lds_raw[2 * 128 * 128];
  unsigned char* dyT = lds_raw;            // [128 n][64 k] swizzled
  unsigned char* xT = lds_raw + 128 * 128;

  const int tid = threadIdx.x;
  const int lane = tid & 63;
  const int wave = tid >> 6;
  const int wn = (wave >> 1) * 64;  // wave quadrant 64x64
  const int wm = (wave & 1) * 64;

  // staging map (per tile): 256 threads cover [64 k][128 cols]:
  // thread: kg = (t>>4)*4 (4 k-rows), cg = (t&15)*8 (8 cols), i.e.
  // 4 x 16B coalesced global loads -> 8 packed b64 transposed LDS writes.
  const int st_kg = (tid >> 4) * 4;
  const int st_c = (tid & 15) * 8;

  f32x4_v acc[4][4];
#pragma unroll
  for (int i = 0; i < 4; ++i)
#pragma unroll
    for (int j = 0; j < 4; ++j) acc[i][j] = {0.f, 0.f, 0.f, 0.f};

  const int a_row = lane & 15;
  const int a_k = (lane >> 4) * 8;
  const bool m_edge = (m0 + W3_BM) > M;
  const bool in0 = !m_edge || (m0 + st_c + 4) <= M;
  const bool in1 = !m_edge || (m0 + st_c + 8) <= M;

  // T14 split (guide G15): issue the NEXT step's global loads before this
  // step's MFMA phase so HBM latency hides under compute; the register
  // tile is written to LDS after the barrier.
  bf16x4 rdy[4][2], rx[4][2];

  auto load_step = [&](int64_t k0) {
#pragma unroll
    for (int rr = 0; rr < 4; ++rr) {
      const unsigned short* src =
          dy + (k0 + st_kg + rr) * (int64_t)N + n0 + st_c;
      rdy[rr][0] = reinterpret_cast<const bf16x4*>(src)[0];
      rdy[rr][1] = reinterpret_cast<const bf16x4*>(src)[1];
      const unsigned short* srcx =
          x + (k0 + st_kg + rr) * (int64_t)M + m0 + st_c;
      // real branches (EXEC-masked loads): an out-of-range lane must not
      // issue the load at all (last row would read past the tensor)
      rx[rr][0] = bf16x4{0, 0, 0, 0};
      rx[rr][1] = bf16x4{0, 0, 0, 0};
      if (in0) rx[rr][0] = reinterpret_cast<const bf16x4*>(srcx)[0];
      if (in1) rx[rr][1] = reinterpret_cast<const bf16x4*>(srcx)[1];
    }
  };

  auto write_step = [&] {
#pragma unroll
    for (int h = 0; h < 2; ++h)
#pragma unroll
      for (int j = 0; j < 4; ++j) {
        U16x4b p{{rdy[0][h][j], rdy[1][h][j], rdy[2][h][j],
                  rdy[3][h][j]}};
        *reinterpret_cast<unsigned long long*>(
            dyT + lds_off(st_c + h * 4 + j, st_kg)) = p.ll;
        U16x4b q{{rx[0][h][j], rx[1][h][j], rx[2][h][j], rx[3][h][j]}};
        *reinterpret_cast<unsigned long long*>(
            xT + lds_off(st_c + h * 4 + j, st_kg)) = q.ll;
      }
  };

  load_step(k_begin);
  for (int64_t k0 = k_begin; k0 < k_end; k0 += W3_BK) {
    write_step();
    __syncthreads();
    if (k0 + W3_BK < k_end)
      load_step(k0 + W3_BK);  // in flight under the MFMA phase
    // ---- MFMA: 4x4 fragments x 2 k-halves -------------------------------
#pragma unroll
    for (int kh = 0; kh < 2; ++kh) {
      bf16x8_v a[4], b[4];
#pragma unroll
      for (int i = 0; i < 4; ++i) {
        a[i] = *reinterpret_cast<const bf16x8_v*>(
            dyT + lds_off(wn + i * 16 + a_row, kh * 32 + a_k));
        b[i] = *reinterpret_cast<const bf16x8_v*>(
            xT + lds_off(wm + i * 16 + a_row, kh * 32 + a_k));
      }
#pragma unroll
      for (int i = 0; i < 4; ++i)
#pragma unroll
        for (int j = 0; j < 4; ++j)
          acc[i][j] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
              a[i], b[j], acc[i][j], 0, 0, 0);
    }
    __syncthreads();
  }

  // ---- epilogue: fp32 partial slab (column-masked) ----------------------
  float* out = part + (int64_t)blockIdx.x * N * M;
  const int c_col = lane & 15;
  const int c_row = (lane >> 4) * 4;
#pragma unroll
  for (int i = 0; i < 4; ++i)
#pragma unroll
    for (int j = 0; j < 4; ++j) {
      const int col = m0 + wm + j * 16 + c_col;
      if (col >= M) continue;
#pragma unroll
      for (int r = 0; r < 4; ++r) {
        const int row = n0 + wn + i * 16 + c_row + r;
        out[(int64_t)row * M + col] = acc[i][j][r];
      }
    }
}

}  // namespace

torch::Tensor reduce_splitk(torch::Tensor part, bool out_bf16);

torch::Tensor wgrad_nt128(torch::Tensor dy, torch::Tensor x,
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
  TORCH_CHECK(N % W3_BN == 0 && M % 8 == 0 && B % W3_BK == 0,
              "wgrad_nt128 needs N % 128 == 0, M % 8 == 0, B % 64 == 0");
  const int tiles = (N / W3_BN) * ((M + W3_BM - 1) / W3_BM);
  if (splitk <= 0)
    // sk sweep (scripts/micro_gemm.py): >32 splits exceed one residency
    // wave of blocks and cliff; 32 is the measured optimum
    splitk = std::min<int64_t>(
        32, std::max<int64_t>(1, 512 / std::max(1, tiles)));
  int64_t chunk = ((B + splitk - 1) / splitk + W3_BK - 1) / W3_BK * W3_BK;
  splitk = (B + chunk - 1) / chunk;
  auto part = torch::empty({splitk, N, M},
                           dy.options().dtype(torch::kFloat32));
  auto stream = c10::hip::getCurrentHIPStream().stream();
  dim3 grid(splitk, tiles);
  hipLaunchKernelGGL(wgrad128_kernel, grid, dim3(256), 0, stream,
                     reinterpret_cast<unsigned short*>(dy.data_ptr()),
                     reinterpret_cast<unsigned short*>(x.data_ptr()),
                     part.data_ptr<float>(), B, N, M, chunk);
  // fused split-K reduce (+bf16 cast when the consumer is a
  // bf16 parameter) — replaces torch reduce + .to elementwise
  return reduce_splitk(part, out_bf16);
}
