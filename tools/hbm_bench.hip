// SPDX-License-Identifier: MIT
// hbm_bench — standalone HBM3E streaming-kernel variant sweep used to
// tune the fill/copy kernels (grid size x vector width x nt policy).
// Build: hipcc --offload-arch=gfx950 -O3 tools/hbm_bench.hip -o build/tools/hbm_bench
#include <hip/hip_runtime.h>

#include <cstdio>
#include <cstdint>
#include <cstdlib>

#define CHECK(x)                                                        \
  do {                                                                  \
    hipError_t e_ = (x);                                                \
    if (e_ != hipSuccess) {                                             \
      fprintf(stderr, "%s: %s\n", #x, hipGetErrorString(e_));           \
      exit(1);                                                          \
    }                                                                   \
  } while (0)

typedef unsigned int u32v4 __attribute__((ext_vector_type(4)));

template <int UNROLL, bool NT>
__global__ void k_copy_v(u32v4* __restrict__ dst,
                         const u32v4* __restrict__ src, uint64_t nvec) {
  uint64_t stride = (uint64_t)gridDim.x * blockDim.x * UNROLL;
  uint64_t base = ((uint64_t)blockIdx.x * blockDim.x + threadIdx.x) * UNROLL;
  for (uint64_t i = base; i + UNROLL - 1 < nvec; i += stride) {
    u32v4 v[UNROLL];
#pragma unroll
    for (int u = 0; u < UNROLL; u++)
      v[u] = NT ? __builtin_nontemporal_load(&src[i + u]) : src[i + u];
#pragma unroll
    for (int u = 0; u < UNROLL; u++) {
      if (NT)
        __builtin_nontemporal_store(v[u], &dst[i + u]);
      else
        dst[i + u] = v[u];
    }
  }
}

// strided-by-wave variant: each thread's UNROLL accesses are 64 vecs
// apart so one wave covers UNROLL contiguous KBs per iteration
template <int UNROLL, bool NT>
__global__ void k_copy_w(u32v4* __restrict__ dst,
                         const u32v4* __restrict__ src, uint64_t nvec) {
  uint64_t stride = (uint64_t)gridDim.x * blockDim.x * UNROLL;
  uint64_t wave = ((uint64_t)blockIdx.x * blockDim.x + threadIdx.x) / 64;
  uint64_t lane = threadIdx.x & 63;
  uint64_t base = wave * 64 * UNROLL + lane;
  for (uint64_t i = base; i + 64 * (UNROLL - 1) < nvec; i += stride) {
    u32v4 v[UNROLL];
#pragma unroll
    for (int u = 0; u < UNROLL; u++)
      v[u] = NT ? __builtin_nontemporal_load(&src[i + 64 * u])
                : src[i + 64 * u];
#pragma unroll
    for (int u = 0; u < UNROLL; u++) {
      if (NT)
        __builtin_nontemporal_store(v[u], &dst[i + 64 * u]);
      else
        dst[i + 64 * u] = v[u];
    }
  }
}

template <typename K>
double bench(K kernel, int grid, u32v4* dst, const u32v4* src,
             uint64_t nvec, int iters) {
  hipEvent_t a, b;
  CHECK(hipEventCreate(&a));
  CHECK(hipEventCreate(&b));
  kernel<<<grid, 256>>>(dst, src, nvec);
  CHECK(hipDeviceSynchronize());
  CHECK(hipEventRecord(a));
  for (int i = 0; i < iters; i++) kernel<<<grid, 256>>>(dst, src, nvec);
  CHECK(hipEventRecord(b));
  CHECK(hipDeviceSynchronize());
  float ms = 0;
  CHECK(hipEventElapsedTime(&ms, a, b));
  hipEventDestroy(a);
  hipEventDestroy(b);
  return 2.0 * nvec * 16 * iters / (ms / 1e3) / 1e9;  // GB/s r+w
}

#include "../rocnrdma_amd/ops/csrc/p2p_pattern.h"

template <int UNROLL, bool NT>
__global__ void k_fill_w(uint64_t* __restrict__ buf, uint64_t nwords,
                         uint64_t seed) {
  typedef unsigned long long u64v2 __attribute__((ext_vector_type(2)));
  uint64_t nvec = nwords / 2;
  uint64_t stride = (uint64_t)gridDim.x * blockDim.x * UNROLL;
  uint64_t wave = ((uint64_t)blockIdx.x * blockDim.x + threadIdx.x) / 64;
  uint64_t lane = threadIdx.x & 63;
  uint64_t base = wave * 64 * UNROLL + lane;
  u64v2* out = (u64v2*)buf;
  for (uint64_t i = base; i + 64 * (UNROLL - 1) < nvec; i += stride) {
#pragma unroll
    for (int u = 0; u < UNROLL; u++) {
      uint64_t j = i + 64 * u;
      u64v2 v = {rocp2p_pattern_word(seed, j * 2),
                 rocp2p_pattern_word(seed, j * 2 + 1)};
      if (NT)
        __builtin_nontemporal_store(v, &out[j]);
      else
        out[j] = v;
    }
  }
}

template <typename K>
double bench_fill(K kernel, int grid, int block, uint64_t* buf,
                  uint64_t nwords, int iters) {
  hipEvent_t a, b;
  CHECK(hipEventCreate(&a));
  CHECK(hipEventCreate(&b));
  kernel<<<grid, block>>>(buf, nwords, 1);
  CHECK(hipDeviceSynchronize());
  CHECK(hipEventRecord(a));
  for (int i = 0; i < iters; i++) kernel<<<grid, block>>>(buf, nwords, 1);
  CHECK(hipEventRecord(b));
  CHECK(hipDeviceSynchronize());
  float ms = 0;
  CHECK(hipEventElapsedTime(&ms, a, b));
  return 8.0 * nwords * iters / (ms / 1e3) / 1e9;
}

template <typename K>
double bench_b(K kernel, int grid, int block, u32v4* dst, const u32v4* src,
               uint64_t nvec, int iters) {
  hipEvent_t a, b;
  CHECK(hipEventCreate(&a));
  CHECK(hipEventCreate(&b));
  kernel<<<grid, block>>>(dst, src, nvec);
  CHECK(hipDeviceSynchronize());
  CHECK(hipEventRecord(a));
  for (int i = 0; i < iters; i++) kernel<<<grid, block>>>(dst, src, nvec);
  CHECK(hipEventRecord(b));
  CHECK(hipDeviceSynchronize());
  float ms = 0;
  CHECK(hipEventElapsedTime(&ms, a, b));
  return 2.0 * nvec * 16 * iters / (ms / 1e3) / 1e9;
}

int main(int argc, char** argv) {
  uint64_t mb = argc > 1 ? strtoull(argv[1], 0, 0) : 2048;
  int iters = argc > 2 ? atoi(argv[2]) : 8;
  uint64_t bytes = mb << 20;
  uint64_t nvec = bytes / 16;
  u32v4 *src, *dst;
  CHECK(hipMalloc(&src, bytes));
  CHECK(hipMalloc(&dst, bytes));
  CHECK(hipMemset(src, 7, bytes));

  int grids[] = {4096, 8192, 16384};
  int blocks[] = {256, 512, 1024};
  for (int g : grids)
    for (int b : blocks)
      printf("copy grid %5d blk %4d  wavu2nt %7.0f  wavu4nt %7.0f  "
             "wavu8nt %7.0f\n",
             g, b, bench_b(k_copy_w<2, true>, g, b, dst, src, nvec, iters),
             bench_b(k_copy_w<4, true>, g, b, dst, src, nvec, iters),
             bench_b(k_copy_w<8, true>, g, b, dst, src, nvec, iters));
  uint64_t nwords = bytes / 8;
  for (int g : grids)
    printf("fill grid %5d blk 256  u1 %7.0f  u1nt %7.0f  wavu4nt %7.0f  "
           "wavu8nt %7.0f\n",
           g, bench_fill(k_fill_w<1, false>, g, 256, (uint64_t*)dst, nwords, iters),
           bench_fill(k_fill_w<1, true>, g, 256, (uint64_t*)dst, nwords, iters),
           bench_fill(k_fill_w<4, true>, g, 256, (uint64_t*)dst, nwords, iters),
           bench_fill(k_fill_w<8, true>, g, 256, (uint64_t*)dst, nwords, iters));
  return 0;
}
