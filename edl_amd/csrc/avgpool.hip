// 2x2 stride-2 average pool (ceil_mode, no padding), NHWC bf16 — the
// ResNet-vd shortcut downsample (VdShortcut). torch's NHWC
// avg_pool2d_backward measured 95 us/dispatch; with kernel==stride each
// input position feeds exactly ONE window, so backward is elementwise.
// Divisor = number of VALID elements in the (clamped) window — torch
// semantics for unpadded ceil_mode pooling.
#include "common.h"

using bf16 = __hip_bfloat16;

extern "C" __global__ void avgpool2x2_fwd_kernel(
    const bf16* __restrict__ x, bf16* __restrict__ y, const int H, const int W,
    const int Ho, const int Wo, const int C) {
  const int c8 = C >> 3;
  const long long total = (long long)gridDim.y * Ho * Wo * c8;
  const int n = blockIdx.y;
  const long long img_in = (long long)H * W * C;
  long long i = (long long)blockIdx.x * blockDim.x + threadIdx.x;
  const long long stride = (long long)gridDim.x * blockDim.x;
  const long long img_elems = (long long)Ho * Wo * c8;
  for (; i < img_elems; i += stride) {
    const int oct = (int)(i % c8);
    const long long pix = i / c8;
    const int wo = (int)(pix % Wo), ho = (int)(pix / Wo);
    const int h0 = ho * 2, w0 = wo * 2;
    const int hn = min(2, H - h0), wn = min(2, W - w0);
    float acc[8] = {0};
    for (int dh = 0; dh < hn; ++dh)
      for (int dw = 0; dw < wn; ++dw) {
        const bf16* src = x + n * img_in +
                          (((long long)(h0 + dh) * W) + (w0 + dw)) * C + oct * 8;
        const uint4 raw = *reinterpret_cast<const uint4*>(__builtin_assume_aligned(src, 16));
        const ushort* u = reinterpret_cast<const ushort*>(&raw);
#pragma unroll
        for (int k = 0; k < 8; ++k) {
          union { unsigned u32; float f; } cv;
          cv.u32 = ((unsigned)u[k]) << 16;
          acc[k] += cv.f;
        }
      }
    const float inv = 1.0f / (float)(hn * wn);
    uint4 out;
    ushort* ou = reinterpret_cast<ushort*>(&out);
#pragma unroll
    for (int k = 0; k < 8; ++k)
      ou[k] = (ushort)__hip_bfloat16_raw(__float2bfloat16(acc[k] * inv)).x;
    bf16* dst = y + (long long)n * Ho * Wo * C + pix * C + oct * 8;
    *reinterpret_cast<uint4*>(__builtin_assume_aligned(dst, 16)) = out;
  }
  (void)total;
}

extern "C" __global__ void avgpool2x2_bwd_kernel(
    const bf16* __restrict__ dy, bf16* __restrict__ dx, const int H, const int W,
    const int Ho, const int Wo, const int C) {
  const int c8 = C >> 3;
  const int n = blockIdx.y;
  long long i = (long long)blockIdx.x * blockDim.x + threadIdx.x;
  const long long stride = (long long)gridDim.x * blockDim.x;
  const long long img_elems = (long long)H * W * c8;
  for (; i < img_elems; i += stride) {
    const int oct = (int)(i % c8);
    const long long pix = i / c8;
    const int w = (int)(pix % W), h = (int)(pix / W);
    const int ho = h >> 1, wo = w >> 1;
    const int hn = min(2, H - ho * 2), wn = min(2, W - wo * 2);
    const float inv = 1.0f / (float)(hn * wn);
    const bf16* src = dy + ((long long)n * Ho * Wo + (long long)ho * Wo + wo) * C
                      + oct * 8;
    const uint4 raw = *reinterpret_cast<const uint4*>(__builtin_assume_aligned(src, 16));
    const ushort* u = reinterpret_cast<const ushort*>(&raw);
    uint4 out;
    ushort* ou = reinterpret_cast<ushort*>(&out);
#pragma unroll
    for (int k = 0; k < 8; ++k) {
      union { unsigned u32; float f; } cv;
      cv.u32 = ((unsigned)u[k]) << 16;
      ou[k] = (ushort)__hip_bfloat16_raw(__float2bfloat16(cv.f * inv)).x;
    }
    bf16* dst = dx + (long long)n * H * W * C + pix * C + oct * 8;
    *reinterpret_cast<uint4*>(__builtin_assume_aligned(dst, 16)) = out;
  }
}

extern "C" void launch_avgpool2x2_fwd(const void* x, void* y, int N, int H,
                                      int W, int Ho, int Wo, int C,
                                      hipStream_t s) {
  const long long per = (long long)Ho * Wo * (C >> 3);
  int gx = (int)((per + 255) / 256);
  if (gx > 1024) gx = 1024;
  hipLaunchKernelGGL(avgpool2x2_fwd_kernel, dim3(gx, N), dim3(256), 0, s,
                     (const bf16*)x, (bf16*)y, H, W, Ho, Wo, C);
}

extern "C" void launch_avgpool2x2_bwd(const void* dy, void* dx, int N, int H,
                                      int W, int Ho, int Wo, int C,
                                      hipStream_t s) {
  const long long per = (long long)H * W * (C >> 3);
  int gx = (int)((per + 255) / 256);
  if (gx > 1024) gx = 1024;
  hipLaunchKernelGGL(avgpool2x2_bwd_kernel, dim3(gx, N), dim3(256), 0, s,
                     (const bf16*)dy, (bf16*)dx, H, W, Ho, Wo, C);
}
