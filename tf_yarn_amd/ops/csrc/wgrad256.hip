// Split-K MFMA weight-gradient kernel, 256x256 output tiles (v4).
//
// v3 (wgrad128.hip) is bound by redundant operand traffic: at B=65536 a
// 128x128 tiling re-reads dy tiles_m times and x tiles_n times (~1 GB for
// the 1024x432 layer).  256x256 tiles halve that again (~0.5 GB), the
// regime where the B-sweep says the structure already matches hipBLASLt.
//
// Double-buffering post-mortem (round 2): two LDS buffer pairs with one
// barrier per k-step were tried twice — runtime buffer select (537 TF)
// and a 2x-unrolled constant-offset pipeline (544/580 TF) — and BOTH
// lost to this single-buffered loop (581/655 TF measured).  At 2
// waves/SIMD the two-barrier alternation keeps the MFMA and LDS-staging
// phases of the co-resident waves interleaved; the "saved" barrier was
// not the bottleneck.  Keep single-buffered.
//
// Geometry: 8 waves (512 threads); wave quadrants 4(N) x 2(M), each wave
// 64x128 = 4x8 fragments of mfma_f32_16x16x32_bf16 (128 fp32 acc/lane —
// ~2 waves/SIMD occupancy, 1 block/CU with 64 KB LDS).  Staging, LDS
// swizzle, split-K slabs and edge masking are v3's (see wgrad128.hip).

#include <torch/extension.h>
#include <c10/hip/HIPStream.h>
#include "common.h"

namespace {

#define W4_BN 256
#define W4_BM 256
#define W4_BK 64

typedef __attribute__((ext_vector_type(8))) __bf16 bf16x8_v4;
typedef __attribute__((ext_vector_type(4))) float f32x4_v4;

union U16x4c {
  unsigned short u[4];
  unsigned long long ll;
};

__device__ __forceinline__ int lds_off4(int n, int k) {
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

__global__ __launch_bounds__(512)
void wgrad256_kernel(const unsigned short* __restrict__ dy,
                     const unsigned short* __restrict__ x,
                     float* __restrict__ part,
                     int64_t B, int N, int M, int64_t chunk) {
  // XCD-aware mapping: the dispatcher places linear block b on XCD
  // b % 8 (MI355X_MICROARCH "Workgroup dispatch").  With grid =
  // (splitk, tiles), linear id = slab + splitk*tile, so all tiles of
  // one k-slab land on XCD slab%8 and share their dy/x column reads
  // through that XCD's 4 MB L2 (the old (tiles, splitk) grid put every
  // tile of a slab on a DIFFERENT XCD — zero reuse; operand re-reads
  // made the kernel HBM-bound at ~5.6 TB/s).
  const int tiles_m = (M + W4_BM - 1) / W4_BM;
  const int tile_n = blockIdx.y / tiles_m;
  const int tile_m = blockIdx.y - tile_n * tiles_m;
  const int n0 = tile_n * W4_BN;
  const int m0 = tile_m * W4_BM;
  const int64_t k_begin = (int64_t)blockIdx.x * chunk;
  const int64_t k_end = min(B, k_begin + chunk);

  __shared__ __attribute__((aligned(16))) unsigned char lds_raw[2 * 256 * 128];
  unsigned char* dyT = lds_raw;             // [256 n][64 k] swizzled
  unsigned char* xT = lds_raw + 256 * 128;  // [256 m][64 k]

  const int tid = threadIdx.x;
  const int lane = tid & 63;
  const int wave = tid >> 6;
  const int wn = (wave >> 1) * 64;   // 4 N-quadrants of 64
  const int wm = (wave & 1) * 128;   // 2 M-halves of 128

  // staging map (per tile [64 k][256 cols]): 512 threads, thread =
  // 4 k-rows x 8 cols (4 x 16 B loads -> 8 packed b64 transposed writes)
  const int st_kg = (tid >> 5) * 4;   // 0,4,...,60
  const int st_c = (tid & 31) * 8;    // 0,8,...,248

  f32x4_v4 acc[4][8];
#pragma unroll
  for (int i = 0; i < 4; ++i)
#pragma unroll
    for (int j = 0; j < 8; ++j) acc[i][j] = {0.f, 0.f, 0.f, 0.f};

  const int a_row = lane & 15;
  const int a_k = (lane >> 4) * 8;
  const bool m_edge = (m0 + W4_BM) > M;
  const bool in0 = !m_edge || (m0 + st_c + 4) <= M;
  const bool in1 = !m_edge || (m0 + st_c + 8) <= M;

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
        U16x4c p{{rdy[0][h][j], rdy[1][h][j], rdy[2][h][j],
                  rdy[3][h][j]}};
        *reinterpret_cast<unsigned long long*>(
            dyT + lds_off4(st_c + h * 4 + j, st_kg)) = p.ll;
        U16x4c q{{rx[0][h][j], rx[1][h][j], rx[2][h][j], rx[3][h][j]}};
        *reinterpret_cast<unsigned long long*>(
            xT + lds_off4(st_c + h * 4 + j, st_kg)) = q.ll;
      }
  };

  load_step(k_begin);
  for (int64_t k0 = k_begin; k0 < k_end; k0 += W4_BK) {
    write_step();
    __syncthreads();
    if (k0 + W4_BK < k_end)
      load_step(k0 + W4_BK);  // in flight under the MFMA phase
#pragma unroll
    for (int kh = 0; kh < 2; ++kh) {
      bf16x8_v4 a[4], b[8];
#pragma unroll
      for (int i = 0; i < 4; ++i)
        a[i] = *reinterpret_cast<const bf16x8_v4*>(
            dyT + lds_off4(wn + i * 16 + a_row, kh * 32 + a_k));
#pragma unroll
      for (int j = 0; j < 8; ++j)
        b[j] = *reinterpret_cast<const bf16x8_v4*>(
            xT + lds_off4(wm + j * 16 + a_row, kh * 32 + a_k));
#pragma unroll
      for (int i = 0; i < 4; ++i)
#pragma unroll
        for (int j = 0; j < 8; ++j)
          acc[i][j] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
              a[i], b[j], acc[i][j], 0, 0, 0);
    }
    __syncthreads();
  }

  float* out = part + (int64_t)blockIdx.x * N * M;
  const int c_col = lane & 15;
  const int c_row = (lane >> 4) * 4;
#pragma unroll
  for (int i = 0; i < 4; ++i)
#pragma unroll
    for (int j = 0; j < 8; ++j) {
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

torch::Tensor wgrad_nt256(torch::Tensor dy, torch::Tensor x,
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
  TORCH_CHECK(N % W4_BN == 0 && M % 8 == 0 && B % W4_BK == 0,
              "wgrad_nt256 needs N % 256 == 0, M % 8 == 0, B % 64 == 0");
  const int tiles = (N / W4_BN) * ((M + W4_BM - 1) / W4_BM);
  if (splitk <= 0)
    // 8-wave blocks: 256 blocks = one full residency wave; sk > 32 cliffs
    splitk = std::min<int64_t>(
        32, std::max<int64_t>(1, 256 / std::max(1, tiles)));
  int64_t chunk = ((B + splitk - 1) / splitk + W4_BK - 1) / W4_BK * W4_BK;
  splitk = (B + chunk - 1) / chunk;
  auto part = torch::empty({splitk, N, M},
                           dy.options().dtype(torch::kFloat32));
  auto stream = c10::hip::getCurrentHIPStream().stream();
  dim3 grid(splitk, tiles);
  hipLaunchKernelGGL(wgrad256_kernel, grid, dim3(512), 0, stream,
                     reinterpret_cast<unsigned short*>(dy.data_ptr()),
                     reinterpret_cast<unsigned short*>(x.data_ptr()),
                     part.data_ptr<float>(), B, N, M, chunk);
  // fused split-K reduce (+bf16 cast when the consumer is a
  // bf16 parameter) — replaces torch reduce + .to elementwise
  return reduce_splitk(part, out_bf16);
}
