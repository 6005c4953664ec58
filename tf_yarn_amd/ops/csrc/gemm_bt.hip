// Forward GEMM C[M, N] = A[M, K] @ B[N, K]^T with optional fused
// bias+ReLU epilogue — the nn.Linear forward shape (x @ W.t()).
//
// Both operands are stored K-MAJOR (row-major with K innermost), which is
// exactly what the mfma_f32_16x16x32_bf16 fragments want (8 consecutive k
// per lane): staging into LDS is a LINEAR copy, no transpose — only the
// fragment reads/writes use the combined XOR swizzle proven in
// wgrad256.hip (same 256x256 tile / 8-wave / 4x8-fragment geometry).
// K is tail-masked (K % 64 != 0 OK, K % 16 == 0 required); B rows are
// masked so N % 256 != 0 works (the 432-wide dgrad shape).
//
// The bias+ReLU epilogue writes bf16 C directly, eliminating the separate
// bias_relu kernel AND the pre-activation round trip through HBM.

#include <torch/extension.h>
#include <c10/hip/HIPStream.h>
#include "common.h"

namespace {

#define G_BM 256  // rows of A per tile
#define G_BN 256  // rows of B (= output cols) per tile
#define G_BK 64

typedef __attribute__((ext_vector_type(8))) __bf16 bf16x8_g;
typedef __attribute__((ext_vector_type(4))) float f32x4_g;
typedef __attribute__((ext_vector_type(8))) unsigned short u16x8_g;

__device__ __forceinline__ int g_lds_off(int r, int k) {
  int byte = r * 128 + k * 2;
  return byte ^ (((r ^ (r >> 3)) & 7) << 4);
}

template <bool RELU>
__global__ __launch_bounds__(512)
void gemm_bt_kernel(const unsigned short* __restrict__ A,
                    const unsigned short* __restrict__ B,
                    const unsigned short* __restrict__ bias,  // may be null
                    unsigned short* __restrict__ C,
                    int64_t M, int N, int K) {
  const int tiles_n = (N + G_BN - 1) / G_BN;
  const int tiles_m = static_cast<int>(M / G_BM);
  int tile_m, tile_n;
  if ((tiles_m & 7) == 0) {
    // XCD-aware grouping (see wgrad256.hip): linear block b lands on
    // XCD b%8, so map the tiles_n column-tiles of each row-block to
    // indices sharing b%8 — they re-read the SAME 256 A rows and now
    // share them through one XCD's L2 instead of 4 different XCDs.
    const int c = blockIdx.x & 7;
    const int q = blockIdx.x >> 3;
    tile_n = q % tiles_n;
    tile_m = c + 8 * (q / tiles_n);
  } else {
    tile_m = blockIdx.x / tiles_n;
    tile_n = blockIdx.x - tile_m * tiles_n;
  }
  const int64_t m0 = (int64_t)tile_m * G_BM;
  const int n0 = tile_n * G_BN;

  __shared__ __attribute__((aligned(16))) unsigned char lds_raw[2 * 256 * 128];
  unsigned char* aT = lds_raw;              // [256 m][64 k] swizzled
  unsigned char* bT = lds_raw + 256 * 128;  // [256 n][64 k]

  const int tid = threadIdx.x;
  const int lane = tid & 63;
  const int wave = tid >> 6;
  const int wm = (wave >> 1) * 64;   // 4 M-quadrants of 64
  const int wn = (wave & 1) * 128;   // 2 N-halves of 128

  // staging: 512 threads cover one [256 rows][64 k] tile as
  // row = t >> 1, k seg = (t & 1) * 32: each thread 32 bf16 (64 B) per
  // operand per step, loaded as two 32 B vectors.
  const int st_r = tid >> 1;
  const int st_k = (tid & 1) * 32;
  const bool b_row_ok = (n0 + st_r) < N;  // B rows masked for ragged N

  f32x4_g acc[4][8];
#pragma unroll
  for (int i = 0; i < 4; ++i)
#pragma unroll
    for (int j = 0; j < 8; ++j) acc[i][j] = {0.f, 0.f, 0.f, 0.f};

  const int a_row = lane & 15;
  const int a_k = (lane >> 4) * 8;

  u16x8_g ra[4], rb[4];  // 4 x 8 bf16 = 32 elements per operand

  auto load_step = [&](int k0) {
    const unsigned short* pa = A + (m0 + st_r) * (int64_t)K + k0 + st_k;
    const unsigned short* pb = B + (int64_t)(n0 + st_r) * K + k0 + st_k;
    if (k0 + st_k + 32 <= K) {  // full 32-element segment in range
#pragma unroll
      for (int c = 0; c < 4; ++c) {
        ra[c] = *reinterpret_cast<const u16x8_g*>(pa + c * 8);
        rb[c] = b_row_ok ? *reinterpret_cast<const u16x8_g*>(pb + c * 8)
                         : u16x8_g{0, 0, 0, 0, 0, 0, 0, 0};
      }
    } else {
      // K tail: valid prefix is a multiple of 16 elements (host checks
      // K % 16 == 0), so whole 8-element vectors are either in or out.
#pragma unroll
      for (int c = 0; c < 4; ++c) {
        const bool in = (k0 + st_k + c * 8 + 8) <= K;
        ra[c] = in ? *reinterpret_cast<const u16x8_g*>(pa + c * 8)
                   : u16x8_g{0, 0, 0, 0, 0, 0, 0, 0};
        rb[c] = (in && b_row_ok)
                    ? *reinterpret_cast<const u16x8_g*>(pb + c * 8)
                    : u16x8_g{0, 0, 0, 0, 0, 0, 0, 0};
      }
    }
  };

  auto write_step = [&] {
#pragma unroll
    for (int c = 0; c < 4; ++c) {
      *reinterpret_cast<u16x8_g*>(aT + g_lds_off(st_r, st_k + c * 8)) =
          ra[c];
      *reinterpret_cast<u16x8_g*>(bT + g_lds_off(st_r, st_k + c * 8)) =
          rb[c];
    }
  };

  load_step(0);
  for (int k0 = 0; k0 < K; k0 += G_BK) {
    write_step();
    __syncthreads();
    if (k0 + G_BK < K) load_step(k0 + G_BK);  // in flight under MFMA
#pragma unroll
    for (int kh = 0; kh < 2; ++kh) {
      bf16x8_g a[4], b[8];
#pragma unroll
      for (int i = 0; i < 4; ++i)
        a[i] = *reinterpret_cast<const bf16x8_g*>(
            aT + g_lds_off(wm + i * 16 + a_row, kh * 32 + a_k));
#pragma unroll
      for (int j = 0; j < 8; ++j)
        b[j] = *reinterpret_cast<const bf16x8_g*>(
            bT + g_lds_off(wn + j * 16 + a_row, kh * 32 + a_k));
#pragma unroll
      for (int i = 0; i < 4; ++i)
#pragma unroll
        for (int j = 0; j < 8; ++j)
          acc[i][j] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
              a[i], b[j], acc[i][j], 0, 0, 0);
    }
    __syncthreads();
  }

  // epilogue: bf16 C with optional bias+ReLU; A-fragments index M (rows),
  // B-fragments index N (cols): D col = lane&15 -> N, row -> M.
  const int c_col = lane & 15;
  const int c_row = (lane >> 4) * 4;
#pragma unroll
  for (int i = 0; i < 4; ++i)
#pragma unroll
    for (int j = 0; j < 8; ++j) {
      const int col = n0 + wn + j * 16 + c_col;
      if (col >= N) continue;
      const float bv = bias ? bf16_to_f32(bias[col]) : 0.f;
#pragma unroll
      for (int r = 0; r < 4; ++r) {
        const int64_t row = m0 + wm + i * 16 + c_row + r;
        float v = acc[i][j][r] + bv;
        if (RELU) v = v > 0.f ? v : 0.f;
        C[row * N + col] = f32_to_bf16(v);
      }
    }
}

}  // namespace

torch::Tensor gemm_bt(torch::Tensor a, torch::Tensor b,
                      c10::optional<torch::Tensor> bias, bool relu) {
  TORCH_CHECK(a.is_cuda() && a.is_contiguous() && a.dim() == 2 &&
              a.scalar_type() == torch::kBFloat16,
              "a must be [M, K] bf16 contiguous");
  TORCH_CHECK(b.is_cuda() && b.is_contiguous() && b.dim() == 2 &&
              b.scalar_type() == torch::kBFloat16,
              "b must be [N, K] bf16 contiguous");
  const int64_t M = a.size(0);
  const int K = static_cast<int>(a.size(1));
  const int N = static_cast<int>(b.size(0));
  TORCH_CHECK(b.size(1) == K, "K mismatch");
  TORCH_CHECK(M % G_BM == 0 && N % 16 == 0 && K % 16 == 0,
              "gemm_bt needs M % 256 == 0, N % 16 == 0, K % 16 == 0");
  const unsigned short* bias_ptr = nullptr;
  if (bias.has_value()) {
    TORCH_CHECK(bias->is_cuda() && bias->is_contiguous() &&
                bias->scalar_type() == torch::kBFloat16 &&
                bias->numel() == N, "bias must be [N] bf16");
    bias_ptr = reinterpret_cast<const unsigned short*>(bias->data_ptr());
  }
  auto c = torch::empty({M, (int64_t)N},
                        a.options().dtype(torch::kBFloat16));
  auto stream = c10::hip::getCurrentHIPStream().stream();
  dim3 grid((M / G_BM) * ((N + G_BN - 1) / G_BN));
  auto* kern = relu ? gemm_bt_kernel<true> : gemm_bt_kernel<false>;
  hipLaunchKernelGGL(kern, grid, dim3(512), 0, stream,
                     reinterpret_cast<const unsigned short*>(a.data_ptr()),
                     reinterpret_cast<const unsigned short*>(b.data_ptr()),
                     bias_ptr,
                     reinterpret_cast<unsigned short*>(c.data_ptr()),
                     M, N, K);
  return c;
}
