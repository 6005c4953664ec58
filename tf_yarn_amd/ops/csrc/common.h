// Common device helpers for tf_yarn_amd CDNA4 (gfx950) kernels.
//
// Conventions (per the MI355X HIP guide):
//  * wavefront = 64 lanes; block sizes are multiples of 64 (default 256)
//  * memory-bound kernels vectorize to 16 B/lane (float4 / 8x bf16) and
//    grid-stride with the grid capped (~256 CUs x 8 blocks)
//  * bf16 is loaded/stored as packed ushort vectors; bf16->f32 is an exact
//    bit shift, f32->bf16 rounds to nearest-even
#pragma once

#include <hip/hip_runtime.h>
#include <cstdint>

#define MIYARN_BLOCK 256
#define MIYARN_MAX_BLOCKS 2048  // 256 CUs * 8 blocks/CU

static inline int miyarn_grid_cap(int64_t n_items, int64_t cap) {
  int64_t blocks = (n_items + MIYARN_BLOCK - 1) / MIYARN_BLOCK;
  if (blocks > cap) blocks = cap;
  if (blocks < 1) blocks = 1;
  return static_cast<int>(blocks);
}

static inline int miyarn_grid(int64_t n_items) {
  return miyarn_grid_cap(n_items, MIYARN_MAX_BLOCKS);
}

// ---- bf16 <-> f32 ----------------------------------------------------------

__device__ __forceinline__ float bf16_to_f32(unsigned short u) {
  union { unsigned int i; float f; } v;
  v.i = static_cast<unsigned int>(u) << 16;
  return v.f;
}

__device__ __forceinline__ unsigned short f32_to_bf16(float f) {
  union { float f; unsigned int i; } v;
  v.f = f;
  unsigned int x = v.i;
  // round-to-nearest-even on the truncated 16 bits
  unsigned int rounding = 0x7fffu + ((x >> 16) & 1u);
  x += rounding;
  return static_cast<unsigned short>(x >> 16);
}

// Native fp32 global atomic add: plain atomicAdd lowers to a CAS loop on
// ROCm without -munsafe-fp-atomics; unsafeAtomicAdd emits
// global_atomic_add_f32 directly (correct on HBM; the "unsafe" caveat is
// about fine-grained host memory, which these kernels never touch).
__device__ __forceinline__ void f32_atomic_add(float* p, float v) {
  unsafeAtomicAdd(p, v);
}

// packed vectors: 16 B per lane
typedef __attribute__((ext_vector_type(4))) float f32x4;
typedef __attribute__((ext_vector_type(8))) unsigned short bf16x8;
typedef __attribute__((ext_vector_type(4))) unsigned short bf16x4;

// Generic element accessors so kernels template over float / bf16 storage.
struct F32Io {
  using scalar_t = float;
  __device__ static float load(const float* p, int64_t i) { return p[i]; }
  __device__ static void store(float* p, int64_t i, float v) { p[i] = v; }
};

struct Bf16Io {
  using scalar_t = unsigned short;
  __device__ static float load(const unsigned short* p, int64_t i) {
    return bf16_to_f32(p[i]);
  }
  __device__ static void store(unsigned short* p, int64_t i, float v) {
    p[i] = f32_to_bf16(v);
  }
};

// Vector I/O for one 4-element quad (16 B fp32 / 8 B bf16 per lane) —
// scalar bf16 loads are ~2-2.5x slower than packed (guide G13).
template <typename Io> struct QuadIo;
template <> struct QuadIo<F32Io> {
  __device__ static void load4(const float* p, int64_t q, float (&v)[4]) {
    f32x4 x = reinterpret_cast<const f32x4*>(p)[q];
#pragma unroll
    for (int j = 0; j < 4; ++j) v[j] = x[j];
  }
  __device__ static void store4(float* p, int64_t q, const float (&v)[4]) {
    f32x4 x;
#pragma unroll
    for (int j = 0; j < 4; ++j) x[j] = v[j];
    reinterpret_cast<f32x4*>(p)[q] = x;
  }
};
template <> struct QuadIo<Bf16Io> {
  __device__ static void load4(const unsigned short* p, int64_t q,
                               float (&v)[4]) {
    bf16x4 x = reinterpret_cast<const bf16x4*>(p)[q];
#pragma unroll
    for (int j = 0; j < 4; ++j) v[j] = bf16_to_f32(x[j]);
  }
  __device__ static void store4(unsigned short* p, int64_t q,
                                const float (&v)[4]) {
    bf16x4 x;
#pragma unroll
    for (int j = 0; j < 4; ++j) x[j] = f32_to_bf16(v[j]);
    reinterpret_cast<bf16x4*>(p)[q] = x;
  }
};
