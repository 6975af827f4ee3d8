// bf16 2-D transpose with zero-padded output columns:
//   out[C, Mp] = in[M, C]^T, out[:, M..Mp) = 0   (Mp = M rounded up to 64)
//
// Lets conv wgrad (a TN GEMM, reduction over the huge M) run on the
// existing bt-layout MFMA kernel: dW = gemm_bt(dyT, xT). hipBLASLt's TN
// heuristics measured 272 us on these shapes (profiles/).
//
// 64x64 bf16 tiles through LDS; 16B vectorized loads and stores; +8-byte
// row padding in LDS kills write/read bank conflicts.
#include "common.h"

using bf16 = __hip_bfloat16;

extern "C" __global__ void transpose_pad_kernel(
    const bf16* __restrict__ in, bf16* __restrict__ out, const int M,
    const int C, const int Mp) {
  // tile: 64 rows (m) x 64 cols (c); block 256 threads
  __shared__ bf16 tile[64][64 + 4];
  const int tm0 = blockIdx.x * 64;  // m tile origin
  const int tc0 = blockIdx.y * 64;  // c tile origin
  // load: each thread 16B = 8 c-elems; 8 threads per row; 32 rows per pass
  const int lr = threadIdx.x >> 3;        // 0..31
  const int lc = (threadIdx.x & 7) * 8;   // 0..56
  for (int half = 0; half < 2; ++half) {
    const int m = tm0 + half * 32 + lr;
    if (m < M && tc0 + lc < C) {
      const bf16* src = in + (long long)m * C + tc0 + lc;
      *reinterpret_cast<uint4*>(&tile[half * 32 + lr][lc]) =
          *reinterpret_cast<const uint4*>(__builtin_assume_aligned(src, 16));
    } else {
      uint4 z = {0, 0, 0, 0};
      *reinterpret_cast<uint4*>(&tile[half * 32 + lr][lc]) = z;
    }
  }
  __syncthreads();
  // store: thread covers 8 m-elems of one c row
  for (int half = 0; half < 2; ++half) {
    const int c = tc0 + half * 32 + lr;
    if (c < C) {
      bf16 v[8];
#pragma unroll
      for (int i = 0; i < 8; ++i) v[i] = tile[lc + i][half * 32 + lr];
      bf16* dst = out + (long long)c * Mp + tm0 + lc;
      *reinterpret_cast<uint4*>(__builtin_assume_aligned(dst, 16)) = *reinterpret_cast<const uint4*>(v);
    }
  }
}

extern "C" void launch_transpose_pad(const void* in, void* out, int M, int C,
                                     int Mp, hipStream_t s) {
  dim3 grid((Mp + 63) / 64, (C + 63) / 64);
  hipLaunchKernelGGL(transpose_pad_kernel, grid, dim3(256), 0, s,
                     (const bf16*)in, (bf16*)out, M, C, Mp);
}

// Shifted 9-way transpose for 3x3 wgrad:
//   out[(s*C + c)][m] = xp[prow(m) + shift(s)][c],  out[:, M..Mp) = 0
// where xp is the zero-halo padded NHWC image (pad_nhwc) and prow(m) is
// the conv3x3 output-position -> padded-row map (stride folded in). The
// result is the B operand of dW3 = gemm_bt_splitk(dyT, out).
extern "C" __global__ void shift9_transpose_kernel(
    const bf16* __restrict__ xp, bf16* __restrict__ out, const int M,
    const int C, const int Mp, const int HW_out, const int W_out, const int Hp,
    const int Wp, const int stride_hw) {
  __shared__ bf16 tile[64][64 + 4];
  const int s = blockIdx.z;  // 0..8
  const int shift = ((s / 3) * Wp + (s % 3)) * C;
  const int tm0 = blockIdx.x * 64;
  const int tc0 = blockIdx.y * 64;
  const int lr = threadIdx.x >> 3;
  const int lc = (threadIdx.x & 7) * 8;
  for (int half = 0; half < 2; ++half) {
    const int m = tm0 + half * 32 + lr;
    uint4 v = {0, 0, 0, 0};
    if (m < M && tc0 + lc < C) {
      const int n_img = m / HW_out;
      const int rem = m % HW_out;
      const int h = rem / W_out, w = rem % W_out;
      const long long prow =
          ((long long)n_img * Hp + h * stride_hw) * Wp + w * stride_hw;
      v = *reinterpret_cast<const uint4*>(
          __builtin_assume_aligned(xp + prow * C + shift + tc0 + lc, 16));
    }
    *reinterpret_cast<uint4*>(&tile[half * 32 + lr][lc]) = v;
  }
  __syncthreads();
  for (int half = 0; half < 2; ++half) {
    const int c = tc0 + half * 32 + lr;
    if (c < C) {
      bf16 v[8];
#pragma unroll
      for (int i = 0; i < 8; ++i) v[i] = tile[lc + i][half * 32 + lr];
      bf16* dst = out + ((long long)s * C + c) * Mp + tm0 + lc;
      *reinterpret_cast<uint4*>(__builtin_assume_aligned(dst, 16)) = *reinterpret_cast<const uint4*>(v);
    }
  }
}

extern "C" void launch_shift9_transpose(const void* xp, void* out, int M, int C,
                                        int Mp, int HW_out, int W_out, int Hp,
                                        int Wp, int stride, hipStream_t s) {
  dim3 grid((Mp + 63) / 64, (C + 63) / 64, 9);
  hipLaunchKernelGGL(shift9_transpose_kernel, grid, dim3(256), 0, s,
                     (const bf16*)xp, (bf16*)out, M, C, Mp, HW_out, W_out, Hp,
                     Wp, stride);
}

// One-pass dgrad-weight repack from the channels-last mirror:
//   out[ci][s][co] = in[co][tap_map(s)][ci]
// in: [Co, 9, Ci] bf16 (the channels-last bucket mirror viewed s-major);
// out: [Ci, 9, Co] bf16. mode 0: tap_map(s) = 8-s (the 180-degree
// rotation + Cin<->Cout transpose of the stride-1 dgrad weights — was a
// flip kernel + permute copy + cast per conv per step); mode 1: the
// stride-2 dgrad parity-class tap order (conv.py _S2D_TAP_ORDER).
__constant__ int S2D_ORDER[9] = {4, 3, 5, 1, 7, 0, 2, 6, 8};
// _S2D_TAP_ORDER[j] = (dy,dx) -> tap index dy*3+dx:
// [(1,1)=4,(1,0)=3,(1,2)=5,(0,1)=1,(2,1)=7,(0,0)=0,(0,2)=2,(2,0)=6,(2,2)=8]

extern "C" __global__ void repack_dgrad_w3_kernel(
    const __hip_bfloat16* __restrict__ in, __hip_bfloat16* __restrict__ out,
    const int Co, const int Ci, const int mode) {
  __shared__ __hip_bfloat16 tile[64][64 + 8];
  // grid: x = (Ci/64)*(Co/64) tiles, y = tap s
  const int s = blockIdx.y;
  const int tap = mode == 0 ? 8 - s : S2D_ORDER[s];
  const int tiles_co = Co >> 6;
  const int ci0 = (blockIdx.x / tiles_co) << 6;
  const int co0 = (blockIdx.x % tiles_co) << 6;
  const int lane = threadIdx.x & 63;
  const int quad = threadIdx.x >> 6;
  // load: rows = co (strided), cols = ci contiguous, 16 B per thread
  {
    const int c8 = lane & 7;        // which 8-ci chunk
    const int rr = (lane >> 3) + quad * 8;  // row 0..31; two passes
    for (int half = 0; half < 2; ++half) {
      const int r = rr + half * 32;
      const __hip_bfloat16* src =
          in + ((long long)(co0 + r) * 9 + tap) * Ci + ci0 + c8 * 8;
      *reinterpret_cast<uint4*>(
          __builtin_assume_aligned(&tile[r][c8 * 8], 16)) =
          *reinterpret_cast<const uint4*>(__builtin_assume_aligned(src, 16));
    }
  }
  __syncthreads();
  // store: rows = ci (strided), cols = co contiguous
  {
    const int c8 = lane & 7;
    const int rr = (lane >> 3) + quad * 8;
    for (int half = 0; half < 2; ++half) {
      const int r = rr + half * 32;  // ci row
      __hip_bfloat16* dst =
          out + ((long long)(ci0 + r) * 9 + s) * Co + co0 + c8 * 8;
      uint4 v;
      __hip_bfloat16* pv = reinterpret_cast<__hip_bfloat16*>(&v);
#pragma unroll
      for (int k = 0; k < 8; ++k) pv[k] = tile[c8 * 8 + k][r];
      *reinterpret_cast<uint4*>(__builtin_assume_aligned(dst, 16)) = v;
    }
  }
}

extern "C" void launch_repack_dgrad_w3(const void* in, void* out, int Co,
                                       int Ci, int mode, hipStream_t s) {
  const dim3 grid((Ci >> 6) * (Co >> 6), 9);
  hipLaunchKernelGGL(repack_dgrad_w3_kernel, grid, dim3(256), 0, s,
                     (const __hip_bfloat16*)in, (__hip_bfloat16*)out, Co, Ci,
                     mode);
}

// fp32 -> bf16 cast that also returns the source to ZERO: consumes a
// pooled split-K partial buffer (atomic-fold target, must start zeroed)
// and hands it back clean — replaces a separate FillFunctor zeroing
// launch per split-K conv (6-25 MB each). One float4 read + bf16x4
// store + float4 zero store per thread per iteration.
extern "C" __global__ void cast_bf16_zero_kernel(
    float* __restrict__ src, __hip_bfloat16* __restrict__ dst,
    const long long n4) {
  typedef __hip_bfloat16 bf16;
  long long i = (long long)blockIdx.x * blockDim.x + threadIdx.x;
  const long long stride = (long long)gridDim.x * blockDim.x;
  const float4 z = {0.0f, 0.0f, 0.0f, 0.0f};
  for (; i < n4; i += stride) {
    float4* ps = reinterpret_cast<float4*>(
        __builtin_assume_aligned(src + i * 4, 16));
    const float4 v = *ps;
    ulong1 packed;
    bf16* pv = reinterpret_cast<bf16*>(&packed);
    pv[0] = __float2bfloat16(v.x);
    pv[1] = __float2bfloat16(v.y);
    pv[2] = __float2bfloat16(v.z);
    pv[3] = __float2bfloat16(v.w);
    *reinterpret_cast<ulong1*>(__builtin_assume_aligned(dst + i * 4, 8)) =
        packed;
    *ps = z;
  }
}

extern "C" void launch_cast_bf16_zero(float* src, void* dst, long long n,
                                      hipStream_t s) {
  const long long n4 = n / 4;  // callers guarantee n % 4 == 0 (C % 64 == 0)
  int grid = (int)((n4 + 255) / 256);
  if (grid > 4096) grid = 4096;
  hipLaunchKernelGGL(cast_bf16_zero_kernel, dim3(grid), dim3(256), 0, s, src,
                     (__hip_bfloat16*)dst, n4);
}
