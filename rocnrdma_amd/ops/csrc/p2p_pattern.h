// SPDX-License-Identifier: MIT
// splitmix64 payload pattern — single definition shared by the GPU
// kernels (p2p_kernels.hip), the C++ harness backends and the probe
// CLI.  Must stay bit-identical to rocnrdma_amd/utils/pattern.py.
#pragma once
#ifdef __cplusplus
#include <cstdint>
#else
#include <stdint.h>
#endif

#if defined(__HIPCC__) || defined(__HIP_DEVICE_COMPILE__)
#define ROCP2P_HD __host__ __device__ inline
#else
#define ROCP2P_HD inline
#endif

ROCP2P_HD uint64_t rocp2p_sm64_mix(uint64_t x) {
  x = (x ^ (x >> 30)) * 0xBF58476D1CE4E5B9ULL;
  x = (x ^ (x >> 27)) * 0x94D049BB133111EBULL;
  return x ^ (x >> 31);
}

#define ROCP2P_SM64_GOLDEN 0x9E3779B97F4A7C15ULL

// word i of the pattern for a given seed
ROCP2P_HD uint64_t rocp2p_pattern_word(uint64_t seed, uint64_t i) {
  return rocp2p_sm64_mix(seed + (i + 1) * ROCP2P_SM64_GOLDEN);
}
