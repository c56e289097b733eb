// Shared device/host helpers for the persia_amd HIP kernels (gfx950/CDNA4).
//
// Hash/init math MUST stay bit-identical to the Python oracle in
// persia_amd/core/hashing.py and core/store.py (tests compare them).
#pragma once

#include <hip/hip_runtime.h>
#include <cstdint>

#define PA_BUCKET_SIZE 8
#define PA_PROBE_BUCKETS 4
#define PA_EMPTY_KEY 0ull
#define PA_WAVE 64

__host__ __device__ __forceinline__ uint64_t pa_splitmix64(uint64_t x) {
  uint64_t z = x + 0x9E3779B97F4A7C15ull;
  z = (z ^ (z >> 30)) * 0xBF58476D1CE4E5B9ull;
  z = (z ^ (z >> 27)) * 0x94D049BB133111EBull;
  return z ^ (z >> 31);
}

__host__ __device__ __forceinline__ uint64_t pa_splitmix64_inv(uint64_t z) {
  z = z ^ (z >> 31) ^ (z >> 62);
  z = z * 0x319642B2D24D8EC3ull;
  z = z ^ (z >> 27) ^ (z >> 54);
  z = z * 0x96DE1B173F119089ull;
  z = z ^ (z >> 30) ^ (z >> 60);
  return z - 0x9E3779B97F4A7C15ull;
}

// seed for per-sign row init (core/hashing.py init_seed_for)
__host__ __device__ __forceinline__ uint64_t pa_init_seed(uint64_t sign) {
  return pa_splitmix64(sign ^ 0xA0761D6478BD642Full);
}

// uniform [0,1) from top 24 bits — double math to match the numpy oracle
__host__ __device__ __forceinline__ double pa_u01(uint64_t u) {
  return (double)(u >> 40) * (1.0 / 16777216.0);
}

// bounded-uniform init of column c (0-based) of a row, given the row's seed.
// lo/hi stay double end-to-end so results are bit-identical to the numpy
// oracle (core/store.py row_init).
__host__ __device__ __forceinline__ float pa_init_val(uint64_t seed, int c,
                                                      double lo, double hi) {
  const uint64_t u = pa_splitmix64(seed ^ (uint64_t)(c + 1));
  return (float)(lo + (hi - lo) * pa_u01(u));
}

#define PA_CHECK(cond, msg) TORCH_CHECK(cond, msg)
