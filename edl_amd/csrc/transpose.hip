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
          *reinterpret_cast<const uint4*>(src);
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
      *reinterpret_cast<uint4*>(dst) = *reinterpret_cast<const uint4*>(v);
    }
  }
}

extern "C" void launch_transpose_pad(const void* in, void* out, int M, int C,
                                     int Mp, hipStream_t s) {
  dim3 grid((Mp + 63) / 64, (C + 63) / 64);
  hipLaunchKernelGGL(transpose_pad_kernel, grid, dim3(256), 0, s,
                     (const bf16*)in, (bf16*)out, M, C, Mp);
}
