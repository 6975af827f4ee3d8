// Hand-written bf16 MFMA GEMM, B-transposed layout:
//     C[M, N] = A[M, K] @ B[N, K]^T        (all row-major, bf16 in/out)
//
// This is the conv1x1 shape on NHWC: fwd  y = x2d @ W^T   (W: [Cout, Cin])
//                                    dgrad dx = dy2d @ (W^T)^T via a [Cin,
//                                    Cout] transposed-weight copy — the same
//                                    kernel.  MIOpen's igemm ran these at
//                                    ~60 TF and hipBLASLt's heuristic picks
//                                    were worse (rocprof gpurun_out/prof3/4);
//                                    dense bf16 peak is 2.5 PF.
//
// Structure (cdna_hip_programming.md §5 "step-3" class):
//   * 256 threads = 4 waves; per-wave 64x64 output; v_mfma_f32_16x16x32_bf16
//   * BK = 64; A/B tiles staged to LDS by global_load_lds_dwordx4 (16 B),
//     double-buffered; one __syncthreads per K-tile (its implicit vmcnt(0)
//     drains the in-flight glds of the next tile — the 2-phase recipe)
//   * LDS XOR swizzle (byte slot ^= row&7) on the glds SOURCE address and
//     the ds_read address (rule 21: linear dest + inverse-swizzled source)
//     -> conflict-free ds_read_b128 fragments
//   * row-clamped A loads for the M tail (C-write is row-guarded)
//   * two tile shapes: 128x128 (N % 128 == 0) and 256x64
#include "common.h"

using bf16 = __hip_bfloat16;
typedef __bf16 bf16x8 __attribute__((ext_vector_type(8)));
typedef float f32x4 __attribute__((ext_vector_type(4)));

extern __shared__ __attribute__((aligned(16))) char smem[];

template <int BM, int BN, int WAVES_M, int WAVES_N, bool SPLITK = false,
          bool V2 = false>
__global__ __launch_bounds__(256, 2) void gemm_bt_kernel(
    const bf16* __restrict__ A, const bf16* __restrict__ B,
    void* __restrict__ C_any, const int M, const int N, const int K,
    float* __restrict__ bn_part = nullptr, const int ldsb = 0) {
  // bn_part: per-channel sum/sumsq partials of the bf16-rounded output,
  // [tiles_m, 2N] — lets the following BatchNorm skip its stats kernel
  // (see conv3x3.hip BN_PART)
  // SPLITK: grid.y k-slices; each block atomically folds its fp32 partial
  // tile into C (fp32, pre-zeroed). Wgrad-shaped GEMMs (tiny [Cout, Cin]
  // output, K = M ~ 1e5) otherwise serialize on 1-2 blocks.
  bf16* C = SPLITK ? nullptr : (bf16*)C_any;
  float* Cf = SPLITK ? (float*)C_any : nullptr;
  constexpr int BK = 64;
  constexpr int A_BYTES = BM * BK * 2;  // row stride 128 B
  constexpr int B_BYTES = BN * BK * 2;
  // LDS: [A0 | B0 | A1 | B1]
  char* lds = smem;
  const int lane = threadIdx.x & 63;
  const int wave = threadIdx.x >> 6;

  // tile coordinates (grid.x = tiles_m * tiles_n, n fastest; XCD-swizzled)
  const int tiles_n = N / BN;
  const int bid = xcd_swizzle(blockIdx.x, gridDim.x);
  const int tile_m = bid / tiles_n;
  const int tile_n = bid % tiles_n;
  const int m0 = tile_m * BM;
  const int n0 = tile_n * BN;

  const int wm = (wave / WAVES_N) * 64;  // wave's sub-tile origin
  const int wn = (wave % WAVES_N) * 64;

  f32x4 acc[4][4] = {};

  // ---- staging: each wave stages its share of A and B chunks ----
  // chunk = 1 KiB = 8 rows x 8 slots(16 B); lane l -> row l/8, slot l%8
  constexpr int A_CHUNKS = A_BYTES / 1024;
  constexpr int B_CHUNKS = B_BYTES / 1024;
  auto stage = [&](int buf, int kt) {
    const long long k0 = (long long)kt * BK;
    char* abase = lds + buf * (A_BYTES + B_BYTES);
    char* bbase = abase + A_BYTES;
#pragma unroll
    for (int i = 0; i < A_CHUNKS / 4; ++i) {
      const int ch = wave * (A_CHUNKS / 4) + i;
      const int r = ch * 8 + (lane >> 3);
      const int slot = lane & 7;
      const int gslot = slot ^ (r & 7);
      long long grow = m0 + r;
      if (grow >= M) grow = M - 1;  // clamped dup row; C-write is guarded
      const bf16* src = A + grow * K + k0 + gslot * 8;
      __builtin_amdgcn_global_load_lds(
          (const __attribute__((address_space(1))) void*)src,
          (__attribute__((address_space(3))) void*)(abase + ch * 1024), 16, 0, 0);
    }
#pragma unroll
    for (int i = 0; i < B_CHUNKS / 4; ++i) {
      const int ch = wave * (B_CHUNKS / 4) + i;
      const int r = ch * 8 + (lane >> 3);
      const int slot = lane & 7;
      const int gslot = slot ^ (r & 7);
      const bf16* src = B + (long long)(n0 + r) * K + k0 + gslot * 8;
      __builtin_amdgcn_global_load_lds(
          (const __attribute__((address_space(1))) void*)src,
          (__attribute__((address_space(3))) void*)(bbase + ch * 1024), 16, 0, 0);
    }
  };

  // ---- fragment reads (swizzled ds_read_b128) ----
  auto read_a = [&](int buf, int mf, int kk) -> bf16x8 {
    const char* abase = lds + buf * (A_BYTES + B_BYTES);
    const int r = wm + mf * 16 + (lane & 15);
    const int c = kk * 4 + (lane >> 4);
    return *(const __attribute__((address_space(3))) bf16x8*)(
        (const __attribute__((address_space(3))) char*)(abase) + r * 128 +
        ((c ^ (r & 7)) << 4));
  };
  auto read_b = [&](int buf, int nf, int kk) -> bf16x8 {
    const char* bbase = lds + buf * (A_BYTES + B_BYTES) + A_BYTES;
    const int r = wn + nf * 16 + (lane & 15);
    const int c = kk * 4 + (lane >> 4);
    return *(const __attribute__((address_space(3))) bf16x8*)(
        (const __attribute__((address_space(3))) char*)(bbase) + r * 128 +
        ((c ^ (r & 7)) << 4));
  };

  // partial staging: issue only chunk-range [part*PARTS..] of the tile
  auto stage_part = [&](int buf, int kt, int part, int nparts) {
    const long long k0 = (long long)kt * BK;
    char* abase = lds + buf * (A_BYTES + B_BYTES);
    char* bbase = abase + A_BYTES;
    // chunk ranges per part: [part*tot/nparts, (part+1)*tot/nparts)
    const int a_tot = A_CHUNKS / 4, b_tot = B_CHUNKS / 4;
#pragma unroll
    for (int i = part * a_tot / nparts; i < (part + 1) * a_tot / nparts; ++i) {
      const int ch = wave * (A_CHUNKS / 4) + i;
      const int r = ch * 8 + (lane >> 3);
      const int gslot = (lane & 7) ^ (r & 7);
      long long grow = m0 + r;
      if (grow >= M) grow = M - 1;
      const bf16* src = A + grow * K + k0 + gslot * 8;
      __builtin_amdgcn_global_load_lds(
          (const __attribute__((address_space(1))) void*)src,
          (__attribute__((address_space(3))) void*)(abase + ch * 1024), 16, 0, 0);
    }
#pragma unroll
    for (int i = part * b_tot / nparts; i < (part + 1) * b_tot / nparts; ++i) {
      const int ch = wave * (B_CHUNKS / 4) + i;
      const int r = ch * 8 + (lane >> 3);
      const int gslot = (lane & 7) ^ (r & 7);
      const bf16* src = B + (long long)(n0 + r) * K + k0 + gslot * 8;
      __builtin_amdgcn_global_load_lds(
          (const __attribute__((address_space(1))) void*)src,
          (__attribute__((address_space(3))) void*)(bbase + ch * 1024), 16, 0, 0);
    }
  };

  int kt_begin = 0, kt_end = K / BK;
  if (SPLITK) {
    const int per = (kt_end + gridDim.y - 1) / gridDim.y;
    kt_begin = blockIdx.y * per;
    kt_end = min(kt_end, kt_begin + per);
    if (kt_begin >= kt_end) return;
  }
  stage(0, kt_begin);
  __syncthreads();

  // v2 main loop: 4 phases per K-tile — each phase prefetches a quarter of
  // the NEXT tile, reads this phase's fragments, and runs an MFMA cluster
  // under s_setprio(1); waves free-run across phases (role diversity), one
  // explicit vmcnt(0) + raw barrier at the tile boundary.
  if constexpr (V2) {
    for (int kt = kt_begin; kt < kt_end; ++kt) {
      const int cur = (kt - kt_begin) & 1;
      const bool pre = kt + 1 < kt_end;
#pragma unroll
      for (int ph = 0; ph < 4; ++ph) {
        const int kk = ph >> 1, mh = ph & 1;
        if (pre) stage_part(cur ^ 1, kt + 1, ph, 4);
        bf16x8 a[2], b[4];
#pragma unroll
        for (int m2 = 0; m2 < 2; ++m2) a[m2] = read_a(cur, mh * 2 + m2, kk);
#pragma unroll
        for (int nf = 0; nf < 4; ++nf) b[nf] = read_b(cur, nf, kk);
        __builtin_amdgcn_s_setprio(1);
#pragma unroll
        for (int m2 = 0; m2 < 2; ++m2)
#pragma unroll
          for (int nf = 0; nf < 4; ++nf)
            acc[mh * 2 + m2][nf] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
                a[m2], b[nf], acc[mh * 2 + m2][nf], 0, 0, 0);
        __builtin_amdgcn_s_setprio(0);
      }
      // tile boundary: next tile's stages must land; all waves done
      // reading buf cur before it is restaged next iteration
      asm volatile("s_waitcnt vmcnt(0)" ::: "memory");
      __builtin_amdgcn_s_barrier();
    }
  } else
  for (int kt = kt_begin; kt < kt_end; ++kt) {
    const int cur = (kt - kt_begin) & 1;
    if (kt + 1 < kt_end) stage(cur ^ 1, kt + 1);
#pragma unroll
    for (int kk = 0; kk < 2; ++kk) {
      bf16x8 a[4], b[4];
#pragma unroll
      for (int mf = 0; mf < 4; ++mf) a[mf] = read_a(cur, mf, kk);
#pragma unroll
      for (int nf = 0; nf < 4; ++nf) b[nf] = read_b(cur, nf, kk);
#pragma unroll
      for (int mf = 0; mf < 4; ++mf)
#pragma unroll
        for (int nf = 0; nf < 4; ++nf)
          acc[mf][nf] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
              a[mf], b[nf], acc[mf][nf], 0, 0, 0);
    }
    __syncthreads();
  }

  // ---- epilogue: C[m][n] bf16; frag C/D map col=lane&15,
  // row=(lane>>4)*4+reg ----
  const int cn = lane & 15;
  const int r4 = (lane >> 4) * 4;
  float ls[4] = {0, 0, 0, 0}, lq[4] = {0, 0, 0, 0};
#pragma unroll
  for (int mf = 0; mf < 4; ++mf) {
    if (!SPLITK && ldsb) {
      // EDL_BT_STORE_LDS: stage the wave's 64x64 bf16 tile in LDS
      // ([64][72] row-major, 16-B pad breaks the b128 bank pattern), then
      // store full 128-B lines below. Intra-wave only: no __syncthreads
      // needed, just a lgkmcnt wait before the cross-lane reads.
      bf16* wtile = (bf16*)smem + wave * (64 * 72);
#pragma unroll
      for (int reg = 0; reg < 4; ++reg) {
        const int m = m0 + wm + mf * 16 + r4 + reg;
#pragma unroll
        for (int nf = 0; nf < 4; ++nf) {
          const bf16 yb = __float2bfloat16(acc[mf][nf][reg]);
          wtile[(mf * 16 + r4 + reg) * 72 + nf * 16 + cn] = yb;
          if (bn_part != nullptr && m < M) {
            const float yv = __bfloat162float(yb);
            ls[nf] += yv;
            lq[nf] = fmaf(yv, yv, lq[nf]);
          }
        }
      }
      continue;
    }
#pragma unroll
    for (int reg = 0; reg < 4; ++reg) {
      const int m = m0 + wm + mf * 16 + r4 + reg;
      if (m < M) {
        if (SPLITK) {
          float* crow = Cf + (long long)m * N + n0 + wn + cn;
#pragma unroll
          for (int nf = 0; nf < 4; ++nf) {
            atomicAdd(&crow[nf * 16], acc[mf][nf][reg]);
          }
        } else {
          bf16* crow = C + (long long)m * N + n0 + wn + cn;
#pragma unroll
          for (int nf = 0; nf < 4; ++nf) {
            const bf16 yb = __float2bfloat16(acc[mf][nf][reg]);
            crow[nf * 16] = yb;
            if (bn_part != nullptr) {
              const float yv = __bfloat162float(yb);
              ls[nf] += yv;
              lq[nf] = fmaf(yv, yv, lq[nf]);
            }
          }
        }
      }
    }
  }
  if (!SPLITK && ldsb) {
    asm volatile("s_waitcnt lgkmcnt(0)" ::: "memory");
    bf16* wtile = (bf16*)smem + wave * (64 * 72);
    const int rrow = lane >> 3, rcol = (lane & 7) * 8;
#pragma unroll
    for (int it = 0; it < 8; ++it) {
      const int r = it * 8 + rrow;
      const bf16x8 v = *(const bf16x8*)__builtin_assume_aligned(
          wtile + r * 72 + rcol, 16);
      const int m = m0 + wm + r;
      if (m < M)
        *(bf16x8*)__builtin_assume_aligned(
            C + (long long)m * N + n0 + wn + rcol, 16) = v;
    }
  }

  if (!SPLITK && bn_part != nullptr) {
    float* bsum = (float*)smem;  // staging buffers dead past the loop
    __syncthreads();
    for (int i = threadIdx.x; i < 2 * BN; i += blockDim.x) bsum[i] = 0.0f;
    __syncthreads();
#pragma unroll
    for (int nf = 0; nf < 4; ++nf) {
      // fold the 4 lanes sharing column cn (lane, +16, +32, +48): ONE
      // LDS atomic per address per wave instead of a 4-way serialized
      // same-address conflict
      ls[nf] += __shfl_down(ls[nf], 32);
      ls[nf] += __shfl_down(ls[nf], 16);
      lq[nf] += __shfl_down(lq[nf], 32);
      lq[nf] += __shfl_down(lq[nf], 16);
    }
    if (lane < 16) {
#pragma unroll
      for (int nf = 0; nf < 4; ++nf) {
        atomicAdd(&bsum[wn + nf * 16 + cn], ls[nf]);
        atomicAdd(&bsum[BN + wn + nf * 16 + cn], lq[nf]);
      }
    }
    __syncthreads();
    const int tiles_m_g = gridDim.x / tiles_n;
    const int cap = tiles_m_g < 192 ? tiles_m_g : 192;
    float* dst = bn_part + (long long)(tile_m % cap) * 2 * N + n0;
    if (tiles_m_g > cap) {  // pre-zeroed by the caller
      for (int i = threadIdx.x; i < BN; i += blockDim.x) {
        atomicAdd(&dst[i], bsum[i]);
        atomicAdd(&dst[N + i], bsum[BN + i]);
      }
    } else {
      for (int i = threadIdx.x; i < BN; i += blockDim.x) {
        dst[i] = bsum[i];
        dst[N + i] = bsum[BN + i];
      }
    }
  }
}

static bool use_v2() {
  // Measured (gpurun_out/gb_v1 vs gb_v2): the 4-phase interleave is 3-16%
  // SLOWER than the plain 2-phase loop on every conv shape — consistent
  // with the guide's finding that phase-splitting pays only as part of the
  // full 8-phase/256-tile co-design. Kept for A/B; default OFF.
  static const bool v = []() {
    const char* e = getenv("EDL_GEMM_V2");
    return e != nullptr && e[0] == '1';
  }();
  return v;
}

extern "C" int gemm_bt_tiles_m(int M, int N) {
  const int bm = N % 128 == 0 ? 128 : 256;
  return (M + bm - 1) / bm;
}

static int use_store_lds() {
  // EDL_BT_STORE_LDS=1: bounce epilogue stores through LDS (dwordx4
  // full-line stores instead of 64 2-B stores per thread). Opt-in until
  // measured on the flagship step.
  static const int v = []() {
    const char* e = getenv("EDL_BT_STORE_LDS");
    return e ? atoi(e) : 0;
  }();
  return v;
}

extern "C" void launch_gemm_bt(const void* A, const void* B, void* C, int M,
                               int N, int K, float* bn_part, hipStream_t s) {
  if (N % 128 == 0) {
    constexpr int BM = 128, BN = 128;
    const int grid = ((M + BM - 1) / BM) * (N / BN);
    const int lds_bytes = 2 * (BM * 64 * 2 + BN * 64 * 2);
    if (use_v2())
      hipLaunchKernelGGL((gemm_bt_kernel<BM, BN, 2, 2, false, true>), dim3(grid),
                         dim3(256), lds_bytes, s, (const bf16*)A, (const bf16*)B,
                         C, M, N, K, bn_part, use_store_lds());
    else
      hipLaunchKernelGGL((gemm_bt_kernel<BM, BN, 2, 2>), dim3(grid), dim3(256),
                         lds_bytes, s, (const bf16*)A, (const bf16*)B, C, M, N,
                         K, bn_part, use_store_lds());
  } else {  // N % 64 == 0
    constexpr int BM = 256, BN = 64;
    const int grid = ((M + BM - 1) / BM) * (N / BN);
    const int lds_bytes = 2 * (BM * 64 * 2 + BN * 64 * 2);
    if (use_v2())
      hipLaunchKernelGGL((gemm_bt_kernel<BM, BN, 4, 1, false, true>), dim3(grid),
                         dim3(256), lds_bytes, s, (const bf16*)A, (const bf16*)B,
                         C, M, N, K, bn_part, use_store_lds());
    else
      hipLaunchKernelGGL((gemm_bt_kernel<BM, BN, 4, 1>), dim3(grid), dim3(256),
                         lds_bytes, s, (const bf16*)A, (const bf16*)B, C, M, N,
                         K, bn_part, use_store_lds());
  }
}

extern "C" void launch_gemm_bt_splitk(const void* A, const void* B, float* Cf32,
                                      int M, int N, int K, int splitk,
                                      hipStream_t s) {
  // wgrad shapes: small M (= Cout) and N (= Cin); 128x64 tiles via the
  // 256x64 path would waste rows — use 128x128 when possible, else 256x64.
  if (M >= 128 && N % 128 == 0) {
    constexpr int BM = 128, BN = 128;
    const dim3 grid(((M + BM - 1) / BM) * (N / BN), splitk);
    const int lds_bytes = 2 * (BM * 64 * 2 + BN * 64 * 2);
    hipLaunchKernelGGL((gemm_bt_kernel<BM, BN, 2, 2, true>), grid, dim3(256),
                       lds_bytes, s, (const bf16*)A, (const bf16*)B, Cf32,
                       M, N, K);
  } else {
    constexpr int BM = 256, BN = 64;
    const dim3 grid(((M + BM - 1) / BM) * (N / BN), splitk);
    const int lds_bytes = 2 * (BM * 64 * 2 + BN * 64 * 2);
    hipLaunchKernelGGL((gemm_bt_kernel<BM, BN, 4, 1, true>), grid, dim3(256),
                       lds_bytes, s, (const bf16*)A, (const bf16*)B, Cf32,
                       M, N, K);
  }
}
