// Standalone scatter ceiling probe: atomic vs plain-store (plain is
// correctness-INVALID under collisions — measurement only).
#include <hip/hip_runtime.h>
#include <cstdio>
#include <vector>
#include <random>

#define CHECK(x) do { auto e = (x); if (e) { printf("err %d @%d\n", e, __LINE__); return 1; } } while (0)

typedef __attribute__((ext_vector_type(4))) float f32x4;

__device__ __forceinline__ void f32aa(float* p, float v) { unsafeAtomicAdd(p, v); }

template <int MODE>  // 0 = atomic, 1 = plain rmw, 2 = store only
__global__ void scatter_kernel(float* table, const long* ids,
                               const unsigned short* g, long n, int dim) {
  const long dvec = dim >> 2;
  const long total = n * dvec;
  const long stride = (long)gridDim.x * blockDim.x;
  for (long t = blockIdx.x * (long)blockDim.x + threadIdx.x; t < total;
       t += stride) {
    const long row = t / dvec;
    const long c4 = t - row * dvec;
    float* dst = table + ids[row] * dim + c4 * 4;
    const unsigned short* gp = g + row * dim + c4 * 4;
    float gv[4];
#pragma unroll
    for (int j = 0; j < 4; ++j) {
      union { unsigned int i; float f; } u;
      u.i = (unsigned)gp[j] << 16;
      gv[j] = u.f;
    }
    if (MODE == 0) {
#pragma unroll
      for (int j = 0; j < 4; ++j) f32aa(dst + j, gv[j]);
    } else if (MODE == 1) {
      f32x4 cur = *reinterpret_cast<f32x4*>(dst);
#pragma unroll
      for (int j = 0; j < 4; ++j) cur[j] += gv[j];
      *reinterpret_cast<f32x4*>(dst) = cur;
    } else {
      f32x4 o;
#pragma unroll
      for (int j = 0; j < 4; ++j) o[j] = gv[j];
      *reinterpret_cast<f32x4*>(dst) = o;
    }
  }
}

int main() {
  const long rows = 26L * 1000000, n = 65536L * 26;
  const int dim = 16;
  float* table; long* ids; unsigned short* g;
  CHECK(hipMalloc(&table, rows * dim * 4));
  CHECK(hipMalloc(&ids, n * 8));
  CHECK(hipMalloc(&g, n * dim * 2));
  std::vector<long> h_ids(n);
  std::mt19937_64 rng(1);
  for (long i = 0; i < n; ++i) h_ids[i] = rng() % rows;
  CHECK(hipMemcpy(ids, h_ids.data(), n * 8, hipMemcpyHostToDevice));
  CHECK(hipMemset(table, 0, rows * dim * 4));
  CHECK(hipMemset(g, 0x3f, n * dim * 2));
  dim3 grid(8192), block(256);
  auto bench = [&](auto kern, const char* name) {
    for (int i = 0; i < 3; ++i)
      hipLaunchKernelGGL(kern, grid, block, 0, 0, table, ids, g, n, dim);
    CHECK(hipDeviceSynchronize());
    hipEvent_t a, b; hipEventCreate(&a); hipEventCreate(&b);
    hipEventRecord(a);
    for (int i = 0; i < 20; ++i)
      hipLaunchKernelGGL(kern, grid, block, 0, 0, table, ids, g, n, dim);
    hipEventRecord(b);
    CHECK(hipDeviceSynchronize());
    float ms; hipEventElapsedTime(&ms, a, b);
    double us = ms * 1000 / 20;
    double gb = n * dim * (2.0 + 8.0) / 1e9;  // grad read + rmw
    printf("%s: %.1f us (%.0f GB/s eff)\n", name, us, gb / us * 1e6);
    return 0;
  };
  bench(scatter_kernel<0>, "atomic   ");
  bench(scatter_kernel<1>, "plain rmw");
  bench(scatter_kernel<2>, "store    ");
  return 0;
}
