// Common helpers for the dgl_operator_amd HIP/CDNA4 (gfx950) kernels.
//
// Conventions (per /opt/skills/guides/cdna_hip_programming.md):
//   * wavefront = 64 lanes; block sizes are multiples of 64
//   * grids sized >> 256 workgroups to fill 8 XCDs x 32 CUs
//   * memory-bound kernels vectorize to float4 (16 B/lane) where layout allows
#pragma once

#include <hip/hip_runtime.h>

#include <cstdint>

#define DOA_CHECK_HIP(expr)                                                   \
  do {                                                                        \
    hipError_t _e = (expr);                                                   \
    if (_e != hipSuccess) {                                                   \
      TORCH_CHECK(false, "HIP error: ", hipGetErrorString(_e), " at ",        \
                  __FILE__, ":", __LINE__);                                   \
    }                                                                         \
  } while (0)

namespace doa {

constexpr int kWave = 64;

// accumulate 2-byte float types in fp32
template <typename T>
struct AccT { using type = T; };
template <>
struct AccT<c10::Half> { using type = float; };
template <>
struct AccT<c10::BFloat16> { using type = float; };

inline int64_t ceil_div(int64_t a, int64_t b) { return (a + b - 1) / b; }

// Grid sizing: enough blocks to fill the chip several times over, but capped
// so tiny launches stay tiny.
inline int grid_for(int64_t work_items, int block_size, int max_blocks = 65535) {
  int64_t b = ceil_div(work_items, block_size);
  if (b < 1) b = 1;
  if (b > max_blocks) b = max_blocks;
  return static_cast<int>(b);
}

// ---------------------------------------------------------------------------
// Stateless counter-based RNG (splitmix64 finalizer): reproducible per
// (seed, index) without any global state — used by the neighbor sampler.
// ---------------------------------------------------------------------------
__device__ __host__ inline uint64_t splitmix64(uint64_t x) {
  x += 0x9E3779B97f4A7C15ull;
  x = (x ^ (x >> 30)) * 0xBF58476D1CE4E5B9ull;
  x = (x ^ (x >> 27)) * 0x94D049BB133111EBull;
  return x ^ (x >> 31);
}

// uniform integer in [0, n) from a counter pair
__device__ inline uint64_t rand_below(uint64_t seed, uint64_t ctr, uint64_t n) {
  return splitmix64(seed ^ (ctr * 0xD1342543DE82EF95ull)) % n;
}

}  // namespace doa
