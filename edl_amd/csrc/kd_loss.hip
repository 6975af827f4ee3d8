// Fused knowledge-distillation soft-label cross-entropy (forward+backward).
//
// loss = mean_i [ logsumexp(s_i) - sum_j softmax(t_i)_j * s_ij ]
// dL/ds_ij = (softmax(s_i)_j - softmax(t_i)_j) * gout / B
//
// Replaces the reference's soft-label cross_entropy on teacher predictions
// (reference example/distill/resnet/train_with_fleet.py:254-259,
// soft_label=True) with ONE kernel per direction instead of
// softmax+log_softmax+mul+sum chains. One 256-thread workgroup per row
// (C ~= 1000 classes); wave shuffle + LDS cross-wave reduction; bf16 or
// f32 logits, f32 math.
#include "common.h"

template <typename T>
__device__ __forceinline__ float to_f32(T v);
template <>
__device__ __forceinline__ float to_f32<float>(float v) { return v; }
template <>
__device__ __forceinline__ float to_f32<__hip_bfloat16>(__hip_bfloat16 v) {
  return __bfloat162float(v);
}
template <typename T>
__device__ __forceinline__ T from_f32(float v);
template <>
__device__ __forceinline__ float from_f32<float>(float v) { return v; }
template <>
__device__ __forceinline__ __hip_bfloat16 from_f32<__hip_bfloat16>(float v) {
  return __float2bfloat16(v);
}

// block-level reduce over 4 waves (256 threads)
__device__ __forceinline__ float block_reduce(float v, float* lds, bool do_max) {
  const int lane = threadIdx.x & 63, wid = threadIdx.x >> 6;
  v = do_max ? wave_reduce_max(v) : wave_reduce_sum(v);
  if (lane == 0) lds[wid] = v;
  __syncthreads();
  if (threadIdx.x < 4) {
    float x = lds[threadIdx.x];
#pragma unroll
    for (int off = 1; off < 4; off <<= 1) {
      float o = __shfl_down(x, off, 64);
      x = do_max ? fmaxf(x, o) : x + o;
    }
    if (threadIdx.x == 0) lds[0] = x;
  }
  __syncthreads();
  float r = lds[0];
  __syncthreads();
  return r;
}

template <typename T>
__global__ void kd_ce_fwd_kernel(const T* __restrict__ s, const T* __restrict__ t,
                                 float* __restrict__ loss_per_row, const int C) {
  __shared__ float lds[8];
  const int row = blockIdx.x;
  const T* srow = s + (long long)row * C;
  const T* trow = t + (long long)row * C;

  float smax = -1e30f, tmax = -1e30f;
  for (int j = threadIdx.x; j < C; j += blockDim.x) {
    smax = fmaxf(smax, to_f32(srow[j]));
    tmax = fmaxf(tmax, to_f32(trow[j]));
  }
  smax = block_reduce(smax, lds, true);
  tmax = block_reduce(tmax, lds, true);

  float ssum = 0.f, tsum = 0.f, tdot = 0.f;
  for (int j = threadIdx.x; j < C; j += blockDim.x) {
    float sv = to_f32(srow[j]), tv = to_f32(trow[j]);
    ssum += __expf(sv - smax);
    float te = __expf(tv - tmax);
    tsum += te;
    tdot += te * sv;  // unnormalised sum_j exp(t_j - tmax) * s_j
  }
  ssum = block_reduce(ssum, lds, false);
  tsum = block_reduce(tsum, lds, false);
  tdot = block_reduce(tdot, lds, false);
  if (threadIdx.x == 0) {
    float lse = smax + __logf(ssum);
    loss_per_row[row] = lse - tdot / tsum;
  }
}

template <typename T>
__global__ void kd_ce_bwd_kernel(const T* __restrict__ s, const T* __restrict__ t,
                                 T* __restrict__ ds, const float gout_over_B,
                                 const int C) {
  __shared__ float lds[8];
  const int row = blockIdx.x;
  const T* srow = s + (long long)row * C;
  const T* trow = t + (long long)row * C;
  T* drow = ds + (long long)row * C;

  float smax = -1e30f, tmax = -1e30f;
  for (int j = threadIdx.x; j < C; j += blockDim.x) {
    smax = fmaxf(smax, to_f32(srow[j]));
    tmax = fmaxf(tmax, to_f32(trow[j]));
  }
  smax = block_reduce(smax, lds, true);
  tmax = block_reduce(tmax, lds, true);
  float ssum = 0.f, tsum = 0.f;
  for (int j = threadIdx.x; j < C; j += blockDim.x) {
    ssum += __expf(to_f32(srow[j]) - smax);
    tsum += __expf(to_f32(trow[j]) - tmax);
  }
  ssum = block_reduce(ssum, lds, false);
  tsum = block_reduce(tsum, lds, false);
  const float sinv = 1.f / ssum, tinv = 1.f / tsum;
  for (int j = threadIdx.x; j < C; j += blockDim.x) {
    float ps = __expf(to_f32(srow[j]) - smax) * sinv;
    float pt = __expf(to_f32(trow[j]) - tmax) * tinv;
    drow[j] = from_f32<T>((ps - pt) * gout_over_B);
  }
}

extern "C" void launch_kd_ce_fwd_f32(const float* s, const float* t, float* lpr,
                                     int B, int C, hipStream_t stream) {
  hipLaunchKernelGGL((kd_ce_fwd_kernel<float>), dim3(B), dim3(256), 0, stream,
                     s, t, lpr, C);
}
extern "C" void launch_kd_ce_bwd_f32(const float* s, const float* t, float* ds,
                                     float gob, int B, int C, hipStream_t stream) {
  hipLaunchKernelGGL((kd_ce_bwd_kernel<float>), dim3(B), dim3(256), 0, stream,
                     s, t, ds, gob, C);
}
extern "C" void launch_kd_ce_fwd_bf16(const void* s, const void* t, float* lpr,
                                      int B, int C, hipStream_t stream) {
  hipLaunchKernelGGL((kd_ce_fwd_kernel<__hip_bfloat16>), dim3(B), dim3(256), 0, stream,
                     (const __hip_bfloat16*)s, (const __hip_bfloat16*)t, lpr, C);
}
extern "C" void launch_kd_ce_bwd_bf16(const void* s, const void* t, void* ds,
                                      float gob, int B, int C, hipStream_t stream) {
  hipLaunchKernelGGL((kd_ce_bwd_kernel<__hip_bfloat16>), dim3(B), dim3(256), 0, stream,
                     (const __hip_bfloat16*)s, (const __hip_bfloat16*)t,
                     (__hip_bfloat16*)ds, gob, C);
}
