// Common device helpers for daft_amd HIP kernels (gfx950 / CDNA4).
// Wave width is 64 on CDNA4 — hard-coded per the platform guide.
#pragma once

#include <hip/hip_runtime.h>
#include <cstdint>

#define WAVE 64
#define DEV_INLINE __device__ __forceinline__

// MI355X: 256 CUs; cap memory-bound grids at ~8 blocks/CU and grid-stride.
constexpr int kMaxBlocks = 2048;

static inline int grid_1d(int64_t n, int block, int items_per_thread = 1) {
  int64_t work = (n + (int64_t)block * items_per_thread - 1) /
                 ((int64_t)block * items_per_thread);
  if (work < 1) work = 1;
  if (work > kMaxBlocks) work = kMaxBlocks;
  return (int)work;
}

DEV_INLINE uint64_t splitmix64(uint64_t x) {
  x += 0x9E3779B97F4A7C15ull;
  x = (x ^ (x >> 30)) * 0xBF58476D1CE4E5B9ull;
  x = (x ^ (x >> 27)) * 0x94D049BB133111EBull;
  return x ^ (x >> 31);
}

DEV_INLINE uint64_t hash_bytes_dev(const uint8_t* p, int64_t len) {
  uint64_t h = 0xcbf29ce484222325ull ^ (uint64_t)len;
  int64_t i = 0;
  for (; i + 8 <= len; i += 8) {
    uint64_t w;
    __builtin_memcpy(&w, p + i, 8);
    h = splitmix64(h ^ w);
  }
  if (i < len) {
    uint64_t last = 0;
    for (int j = 0; i < len; ++i, ++j) last |= (uint64_t)p[i] << (8 * j);
    h = splitmix64(h ^ last);
  }
  return h;
}

#define HIP_CHECK(expr)                                                    \
  do {                                                                     \
    hipError_t _e = (expr);                                                \
    if (_e != hipSuccess) {                                                \
      TORCH_CHECK(false, "HIP error: ", hipGetErrorString(_e));            \
    }                                                                      \
  } while (0)
