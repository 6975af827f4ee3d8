// 3x3 stride-2 pad-1 max pool, NHWC bf16 (the ResNet stem pool).
//
// torch's max_pool_{forward,backward}_nhwc measured 38 + 86 us/step at
// bs32 (~0.6 TB/s on the backward scatter). Here: fwd emits a per-element
// tap index (u8, 0..8) so the backward is a bounded GATHER — each input
// pixel belongs to <= 4 windows (stride 2), dx = sum of dy where the
// window's argmax tap points back at this pixel. Both kernels stream
// 8-channel octets with 16-B loads/stores.
#include "common.h"

using bf16 = __hip_bfloat16;

struct F8m {
  float v[8];
};

__device__ __forceinline__ F8m mload8(const bf16* p) {
  const uint4 raw = *reinterpret_cast<const uint4*>(
      __builtin_assume_aligned(p, 16));
  F8m o;
  const ushort* u = reinterpret_cast<const ushort*>(&raw);
#pragma unroll
  for (int i = 0; i < 8; ++i) {
    union { unsigned u32; float f; } c;
    c.u32 = ((unsigned)u[i]) << 16;
    o.v[i] = c.f;
  }
  return o;
}

__device__ __forceinline__ void mstore8(bf16* p, const F8m& x) {
  uint4 raw;
  ushort* u = reinterpret_cast<ushort*>(&raw);
#pragma unroll
  for (int i = 0; i < 8; ++i)
    u[i] = (ushort)(__hip_bfloat16_raw(__float2bfloat16(x.v[i])).x);
  *reinterpret_cast<uint4*>(__builtin_assume_aligned(p, 16)) = raw;
}

// fwd: y[n,oh,ow,c] = max_{ky,kx} x[n, 2oh-1+ky, 2ow-1+kx, c]; idx = the
// winning tap (ky*3+kx) per channel.
extern "C" __global__ void maxpool3x3s2_fwd_kernel(
    const bf16* __restrict__ x, bf16* __restrict__ y,
    unsigned char* __restrict__ idx, const int N, const int H, const int W,
    const int Ho, const int Wo, const int C) {
  const int c8 = C >> 3;
  const long long total = (long long)N * Ho * Wo * c8;
  long long i = (long long)blockIdx.x * blockDim.x + threadIdx.x;
  const long long stride = (long long)gridDim.x * blockDim.x;
  for (; i < total; i += stride) {
    const int oct = (int)(i % c8);
    long long t = i / c8;
    const int ow = (int)(t % Wo);
    t /= Wo;
    const int oh = (int)(t % Ho);
    const int n = (int)(t / Ho);
    float best[8];
    unsigned char bidx[8];
#pragma unroll
    for (int k = 0; k < 8; ++k) {
      best[k] = -3.4e38f;
      bidx[k] = 0;
    }
    const int h0 = 2 * oh - 1, w0 = 2 * ow - 1;
#pragma unroll
    for (int ky = 0; ky < 3; ++ky) {
      const int h = h0 + ky;
      if (h < 0 || h >= H) continue;
      for (int kx = 0; kx < 3; ++kx) {
        const int w = w0 + kx;
        if (w < 0 || w >= W) continue;
        F8m v = mload8(x + (((long long)n * H + h) * W + w) * C + oct * 8);
#pragma unroll
        for (int k = 0; k < 8; ++k) {
          if (v.v[k] > best[k]) {
            best[k] = v.v[k];
            bidx[k] = (unsigned char)(ky * 3 + kx);
          }
        }
      }
    }
    F8m o;
#pragma unroll
    for (int k = 0; k < 8; ++k) o.v[k] = best[k];
    const long long eoff = (((long long)n * Ho + oh) * Wo + ow) * C + oct * 8;
    mstore8(y + eoff, o);
    unsigned long long packed = 0;
#pragma unroll
    for (int k = 0; k < 8; ++k)
      packed |= ((unsigned long long)bidx[k]) << (8 * k);
    *reinterpret_cast<unsigned long long*>(
        __builtin_assume_aligned(idx + eoff, 8)) = packed;
  }
}

// bwd: dx[n,h,w,c] = sum over the <=4 windows containing (h,w) of
// dy[window] where idx[window] selects this pixel's tap.
extern "C" __global__ void maxpool3x3s2_bwd_kernel(
    const bf16* __restrict__ dy, const unsigned char* __restrict__ idx,
    bf16* __restrict__ dx, const int N, const int H, const int W,
    const int Ho, const int Wo, const int C) {
  const int c8 = C >> 3;
  const long long total = (long long)N * H * W * c8;
  long long i = (long long)blockIdx.x * blockDim.x + threadIdx.x;
  const long long stride = (long long)gridDim.x * blockDim.x;
  for (; i < total; i += stride) {
    const int oct = (int)(i % c8);
    long long t = i / c8;
    const int w = (int)(t % W);
    t /= W;
    const int h = (int)(t % H);
    const int n = (int)(t / H);
    float acc[8] = {0, 0, 0, 0, 0, 0, 0, 0};
    // windows: oh with 2oh-1 <= h <= 2oh+1  ->  oh in [(h-1+1)/2, (h+1)/2]
    const int oh_lo = (h - 1 + 1) / 2, oh_hi = (h + 1) / 2;
    const int ow_lo = (w - 1 + 1) / 2, ow_hi = (w + 1) / 2;
    for (int oh = oh_lo; oh <= oh_hi; ++oh) {
      if (oh < 0 || oh >= Ho) continue;
      const int ky = h - (2 * oh - 1);
      for (int ow = ow_lo; ow <= ow_hi; ++ow) {
        if (ow < 0 || ow >= Wo) continue;
        const int kx = w - (2 * ow - 1);
        const unsigned char tap = (unsigned char)(ky * 3 + kx);
        const long long eoff =
            (((long long)n * Ho + oh) * Wo + ow) * C + oct * 8;
        const unsigned long long packed =
            *reinterpret_cast<const unsigned long long*>(
                __builtin_assume_aligned(idx + eoff, 8));
        // load dy octet only if any channel selected this tap
        unsigned long long match = 0;
#pragma unroll
        for (int k = 0; k < 8; ++k)
          if (((packed >> (8 * k)) & 0xff) == tap) match |= 1ull << k;
        if (match) {
          F8m g = mload8(dy + eoff);
#pragma unroll
          for (int k = 0; k < 8; ++k)
            if ((match >> k) & 1) acc[k] += g.v[k];
        }
      }
    }
    F8m o;
#pragma unroll
    for (int k = 0; k < 8; ++k) o.v[k] = acc[k];
    mstore8(dx + (((long long)n * H + h) * W + w) * C + oct * 8, o);
  }
}

extern "C" void launch_maxpool3x3s2_fwd(const void* x, void* y,
                                        unsigned char* idx, int N, int H,
                                        int W, int Ho, int Wo, int C,
                                        hipStream_t s) {
  const long long total = (long long)N * Ho * Wo * (C >> 3);
  hipLaunchKernelGGL(maxpool3x3s2_fwd_kernel,
                     dim3(elementwise_grid(total, 256)), dim3(256), 0, s,
                     (const bf16*)x, (bf16*)y, idx, N, H, W, Ho, Wo, C);
}

extern "C" void launch_maxpool3x3s2_bwd(const void* dy,
                                        const unsigned char* idx, void* dx,
                                        int N, int H, int W, int Ho, int Wo,
                                        int C, hipStream_t s) {
  const long long total = (long long)N * H * W * (C >> 3);
  hipLaunchKernelGGL(maxpool3x3s2_bwd_kernel,
                     dim3(elementwise_grid(total, 256)), dim3(256), 0, s,
                     (const bf16*)dy, idx, (bf16*)dx, N, H, W, Ho, Wo, C);
}
