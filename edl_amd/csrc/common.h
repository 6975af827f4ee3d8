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

// XCD-aware bijective blockIdx remap (guide T1): the dispatcher places
// block b on XCD b%8; remapping gives each XCD a CONTIGUOUS tile range so
// neighbouring tiles (sharing operand panels) hit the same private L2.
__device__ __forceinline__ int xcd_swizzle(int orig, int nwg) {
  const int nx = 8;
  const int q = nwg / nx, r = nwg % nx;
  const int xcd = orig % nx, local = orig / nx;
  return (xcd < r ? xcd * (q + 1) : r * (q + 1) + (xcd - r) * q) + local;
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
