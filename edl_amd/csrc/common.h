// Common helpers for edl_amd CDNA4 (gfx950) kernels.
// Native HIP — written for MI355X only (wave64, 256 CUs / 8 XCDs).
#pragma once
#include <hip/hip_runtime.h>
#include <hip/hip_bf16.h>

#define EDL_HIP_CHECK(expr)                                                   \
  do {                                                                        \
    hipError_t _e = (expr);                                                   \
    if (_e != hipSuccess) {                                                   \
      throw std::runtime_error(std::string("HIP error: ") +                   \
                               hipGetErrorString(_e) + " at " __FILE__ ":" +  \
                               std::to_string(__LINE__));                     \
    }                                                                         \
  } while (0)

constexpr int kWave = 64;  // CDNA wavefront

// Grid sizing for memory-bound elementwise kernels (guide G11): cap blocks,
// grid-stride the rest. 256 CUs * 8 blocks.
static inline int elementwise_grid(long long n, int block) {
  long long want = (n + block - 1) / block;
  long long cap = 256LL * 8;
  return (int)(want < cap ? want : cap);
}

__device__ __forceinline__ float wave_reduce_sum(float v) {
#pragma unroll
  for (int off = 32; off > 0; off >>= 1) v += __shfl_down(v, off, 64);
  return v;
}

__device__ __forceinline__ float wave_reduce_max(float v) {
#pragma unroll
  for (int off = 32; off > 0; off >>= 1) v = fmaxf(v, __shfl_down(v, off, 64));
  return v;
}
