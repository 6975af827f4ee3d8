// Fused BatchNorm(+residual Add)+ReLU, NHWC bf16, training fwd + bwd.
//
// Replaces the reference's fuse_bn_act_ops / cudnn_batchnorm_spatial_persistent
// knobs (reference train_with_fleet.py:374, train_pretrain.sh:20) — and, on
// MI355X, torch-autocast's fp32 BatchNorm path, which pays TWO full-tensor
// casts (bf16->fp32->bf16: the SubTensorOpWithCast* kernels = ~13% of step
// time in rocprof) plus 5 MIOpen kernels per BN. Here: bf16 reads with fp32
// math, 3 lean kernels fwd / 3 bwd, ReLU and the bottleneck's residual add
// folded in, dgamma/dbeta written straight into fp32 grad bucket views.
//
// Layout: x is NHWC [M rows, C channels], C % 8 == 0 (all ResNet/ResNeXt
// widths). Each thread owns 8 consecutive channels (one ushort4x2 = 16 B
// load, guide G13); a block's 256 threads cover 2048 channels' worth of a
// row-group; blocks grid-stride over rows and atomically fold per-channel
// fp32 partials (distinct addresses — L2-rate, not contended).
#include "common.h"

using bf16 = __hip_bfloat16;

struct F8 {
  float v[8];
};

__device__ __forceinline__ F8 load8(const bf16* p) {
  // 16-byte vector load of 8 bf16. assume_aligned is LOAD-BEARING: from
  // bf16 pointer arithmetic LLVM infers align 2 and lowers the uint4
  // load as ushort+dwordx2+dword+ushort pieces (4 VMEM ops, seen in the
  // .s of the r2 reduce loop — the BN kernels ran 5x off roofline).
  const uint4 raw = *reinterpret_cast<const uint4*>(
      __builtin_assume_aligned(p, 16));
  F8 o;
  const ushort* u = reinterpret_cast<const ushort*>(&raw);
#pragma unroll
  for (int i = 0; i < 8; ++i) {
    union { unsigned u32; float f; } c;
    c.u32 = ((unsigned)u[i]) << 16;
    o.v[i] = c.f;
  }
  return o;
}

__device__ __forceinline__ void store8(bf16* p, const F8& x) {
  uint4 raw;
  ushort* u = reinterpret_cast<ushort*>(&raw);
#pragma unroll
  for (int i = 0; i < 8; ++i) {
    u[i] = (ushort)(__hip_bfloat16_raw(__float2bfloat16(x.v[i])).x);
  }
  *reinterpret_cast<uint4*>(__builtin_assume_aligned(p, 16)) = raw;
}

// ---------------- forward ----------------

// K1: per-channel sum and sum-of-squares PARTIALS (training stats).
// Two-stage: each block LDS-reduces its rows and STORES its 2C partials to
// partial[block][2C]; the finalize kernel reduces over blocks. No global
// atomics at all (single-stage atomics measured 17-40 us/dispatch fixed
// cost: same-word serialization at ~88 adds/us for small C, grid x 2C
// traffic for large C).
template <int ST, bool CONTIG = false>
__global__ void bn_stats_kernel(
    const bf16* __restrict__ x, float* __restrict__ partial,  // [grid, 2C]
    const long long M, const int C) {
  __shared__ float lsum[2 * 2048];
  const int c8 = C >> 3;  // channel-octet count
  const int tpr = c8;     // threads per row-slice (each owns 8 channels)
  long long row0, rstride, row_end;
  if (CONTIG) {
    const long long rows_blk = (M + gridDim.x - 1) / gridDim.x;
    const long long blk0 = blockIdx.x * rows_blk;
    row0 = blk0 + threadIdx.x / tpr;
    rstride = blockDim.x / tpr;
    row_end = blk0 + rows_blk < M ? blk0 + rows_blk : M;
  } else {
    const int tid = blockIdx.x * blockDim.x + threadIdx.x;
    row0 = tid / tpr;
    rstride = ((long long)gridDim.x * blockDim.x) / tpr;
    row_end = M;
  }
  const int lane_c = (int)(threadIdx.x % tpr);   // which channel octet
  const int c0 = lane_c * 8;

  for (int i = threadIdx.x; i < 2 * C; i += blockDim.x) lsum[i] = 0.0f;
  __syncthreads();

  float s[8] = {0}, q[8] = {0};
  long long r = row0;
  // ST independent row streams: the grid is capped (finalize reads the
  // partials serially), so per-thread in-flight bytes must cover HBM
  // latency — 1 stream measured ~1.2 TB/s, latency-bound
  for (; r + (ST - 1) * rstride < row_end; r += ST * rstride) {
    F8 v[ST];
#pragma unroll
    for (int u = 0; u < ST; ++u) v[u] = load8(x + (r + u * rstride) * C + c0);
#pragma unroll
    for (int u = 0; u < ST; ++u) {
#pragma unroll
      for (int i = 0; i < 8; ++i) {
        s[i] += v[u].v[i];
        q[i] = fmaf(v[u].v[i], v[u].v[i], q[i]);
      }
    }
  }
  for (; r < row_end; r += rstride) {
    F8 v = load8(x + r * C + c0);
#pragma unroll
    for (int i = 0; i < 8; ++i) {
      s[i] += v.v[i];
      q[i] = fmaf(v.v[i], v.v[i], q[i]);
    }
  }
#pragma unroll
  for (int i = 0; i < 8; ++i) {
    atomicAdd(&lsum[c0 + i], s[i]);
    atomicAdd(&lsum[C + c0 + i], q[i]);
  }
  __syncthreads();
  float* out = partial + (long long)blockIdx.x * 2 * C;
  for (int i = threadIdx.x; i < 2 * C; i += blockDim.x) out[i] = lsum[i];
}

// K2-v2: strip-parallel fwd finalize — 256 threads = 64 channels x 4
// strips over the block range (v1 ran C threads total: at C=64 a single
// wave serially read every partial, latency-bound).
extern "C" __global__ void bn_finalize_v2_kernel(
    float* __restrict__ partial, const int nblocks, const int zero_src,
    const float* __restrict__ gamma,
    const float* __restrict__ beta, float* __restrict__ mean_out,
    float* __restrict__ invstd_out, float* __restrict__ scale_out,
    float* __restrict__ shift_out, float* __restrict__ running_mean,
    float* __restrict__ running_var, const float momentum, const float eps,
    const long long M, const int C) {
  __shared__ float accs[4][64];
  __shared__ float accq[4][64];
  const int lane = threadIdx.x & 63;
  const int strip = threadIdx.x >> 6;
  const int c = blockIdx.x * 64 + lane;
  const long long st = 2 * C;
  float sa[4] = {0}, qa[4] = {0};
  if (c < C) {
    int b = strip;
    for (; b + 12 < nblocks; b += 16) {
#pragma unroll
      for (int u = 0; u < 4; ++u) {
        const long long i0 = (long long)(b + 4 * u) * st + c;
        sa[u] += partial[i0];
        qa[u] += partial[i0 + C];
        if (zero_src) { partial[i0] = 0.0f; partial[i0 + C] = 0.0f; }
      }
    }
    for (; b < nblocks; b += 4) {
      const long long i0 = (long long)b * st + c;
      sa[0] += partial[i0];
      qa[0] += partial[i0 + C];
      if (zero_src) { partial[i0] = 0.0f; partial[i0 + C] = 0.0f; }
    }
  }
  accs[strip][lane] = (sa[0] + sa[1]) + (sa[2] + sa[3]);
  accq[strip][lane] = (qa[0] + qa[1]) + (qa[2] + qa[3]);
  __syncthreads();
  if (strip != 0 || c >= C) return;
  const float s = (accs[0][lane] + accs[1][lane]) +
                  (accs[2][lane] + accs[3][lane]);
  const float q = (accq[0][lane] + accq[1][lane]) +
                  (accq[2][lane] + accq[3][lane]);
  const float inv_m = 1.0f / (float)M;
  const float mean = s * inv_m;
  const float var = fmaxf(q * inv_m - mean * mean, 0.0f);
  const float invstd = rsqrtf(var + eps);
  mean_out[c] = mean;
  invstd_out[c] = invstd;
  const float g = gamma[c];
  scale_out[c] = g * invstd;
  shift_out[c] = beta[c] - g * invstd * mean;
  if (running_mean != nullptr) {
    const float ub = (M > 1) ? var * (float)M / (float)(M - 1) : var;
    running_mean[c] = fmaf(momentum, mean - running_mean[c], running_mean[c]);
    running_var[c] = fmaf(momentum, ub - running_var[c], running_var[c]);
  }
}

// K2: reduce partials over blocks, finalize mean/invstd, update running
// stats. One thread per channel; thread c's reads of partial[b][c] are
// coalesced across the warp for each fixed b.
extern "C" __global__ void bn_finalize_kernel(
    float* __restrict__ partial, const int nblocks, const int zero_src,
    const float* __restrict__ gamma,
    const float* __restrict__ beta, float* __restrict__ mean_out,
    float* __restrict__ invstd_out, float* __restrict__ scale_out,
    float* __restrict__ shift_out, float* __restrict__ running_mean,
    float* __restrict__ running_var, const float momentum, const float eps,
    const long long M, const int C) {
  const int c = blockIdx.x * blockDim.x + threadIdx.x;
  if (c >= C) return;
  // 8 independent accumulators: a single s+= chain leaves every load
  // latency-exposed (measured 125 us at nblocks=1024 vs ~5 us unrolled)
  float sa[8] = {0}, qa[8] = {0};
  const long long st = 2 * C;
  int b = 0;
  for (; b + 8 <= nblocks; b += 8) {
#pragma unroll
    for (int u = 0; u < 8; ++u) {
      const long long i0 = (long long)(b + u) * st + c;
      sa[u] += partial[i0];
      qa[u] += partial[i0 + C];
      if (zero_src) { partial[i0] = 0.0f; partial[i0 + C] = 0.0f; }
    }
  }
  for (; b < nblocks; ++b) {
    const long long i0 = (long long)b * st + c;
    sa[0] += partial[i0];
    qa[0] += partial[i0 + C];
    if (zero_src) { partial[i0] = 0.0f; partial[i0 + C] = 0.0f; }
  }
  float s = 0.0f, q = 0.0f;
#pragma unroll
  for (int u = 0; u < 8; ++u) {
    s += sa[u];
    q += qa[u];
  }
  const float inv_m = 1.0f / (float)M;
  const float mean = s * inv_m;
  const float var = fmaxf(q * inv_m - mean * mean, 0.0f);
  const float invstd = rsqrtf(var + eps);
  mean_out[c] = mean;
  invstd_out[c] = invstd;
  const float g = gamma[c];
  scale_out[c] = g * invstd;
  shift_out[c] = beta[c] - g * invstd * mean;
  if (running_mean != nullptr) {
    // unbiased variance for the running estimate (torch semantics)
    const float ub = (M > 1) ? var * (float)M / (float)(M - 1) : var;
    running_mean[c] = fmaf(momentum, mean - running_mean[c], running_mean[c]);
    running_var[c] = fmaf(momentum, ub - running_var[c], running_var[c]);
  }
}

// K3: y = relu?(x*scale + shift [+ res]); NHWC vectorized by channel octets.
// MASK: also emit a 1-bit-per-channel ReLU mask (one byte per octet) so the
// backward kernels never re-read y (16x fewer bytes on that stream).
template <bool RELU, bool ADD, bool MASK = false>
__global__ void bn_apply_kernel(
    const bf16* __restrict__ x, const bf16* __restrict__ res,
    bf16* __restrict__ y, unsigned char* __restrict__ mask,
    const float* __restrict__ scale,
    const float* __restrict__ shift, const long long M, const int C) {
  const int c8 = C >> 3;
  // every model width is pow2: i%/c8 as mask/shift (the generic 64-bit
  // divmod lowers to a ~100-instruction sequence per octet)
  const bool p2 = (c8 & (c8 - 1)) == 0;
  const int c8l = 31 - __clz((unsigned)c8);
  const long long total = M * c8;
  long long i = (long long)blockIdx.x * blockDim.x + threadIdx.x;
  const long long stride = (long long)gridDim.x * blockDim.x;
  for (; i < total; i += stride) {
    const int oct = p2 ? (int)(i & (c8 - 1)) : (int)(i % c8);
    const long long eoff = (p2 ? (i >> c8l) : (i / c8)) * C + oct * 8;
    F8 v = load8(x + eoff);
    const int c0 = oct * 8;
#pragma unroll
    for (int k = 0; k < 8; ++k) v.v[k] = fmaf(v.v[k], scale[c0 + k], shift[c0 + k]);
    if (ADD) {
      F8 rv = load8(res + eoff);
#pragma unroll
      for (int k = 0; k < 8; ++k) v.v[k] += rv.v[k];
    }
    if (RELU) {
      unsigned char mb = 0;
#pragma unroll
      for (int k = 0; k < 8; ++k) {
        if (v.v[k] > 0.0f) mb |= (1u << k);
        v.v[k] = fmaxf(v.v[k], 0.0f);
      }
      if (MASK) mask[i] = mb;
    }
    store8(y + eoff, v);
  }
}

// ---------------- backward ----------------

// B1: per-channel reductions s1 = sum(dy_eff), s2 = sum(dy_eff * xhat),
// where dy_eff applies the fused ReLU mask (bitmask from the fwd apply).
// ST = independent row streams (4 or 8): more outstanding loads per
// thread for latency hiding at small grids.
// CONTIG: each block owns a CONTIGUOUS row range instead of striding the
// whole tensor by grid*rows — one streaming region per block (DRAM-page
// and L2-friendly) instead of grid interleaved streams.
template <bool RELU, int ST = 4, bool CONTIG = false>
__global__ void bn_bwd_reduce_kernel(
    const bf16* __restrict__ dy, const unsigned char* __restrict__ mask,
    const bf16* __restrict__ x, const float* __restrict__ mean,
    const float* __restrict__ invstd, float* __restrict__ sums,  // [2, C]
    const long long M, const int C) {
  __shared__ float lsum[2 * 2048];
  const int c8 = C >> 3;
  const int tpr = c8;
  long long row0, rstride, row_end;
  if (CONTIG) {
    const long long rows_blk = (M + gridDim.x - 1) / gridDim.x;
    const long long blk0 = blockIdx.x * rows_blk;
    row0 = blk0 + threadIdx.x / tpr;
    rstride = blockDim.x / tpr;
    row_end = blk0 + rows_blk < M ? blk0 + rows_blk : M;
  } else {
    const int tid = blockIdx.x * blockDim.x + threadIdx.x;
    row0 = tid / tpr;
    rstride = ((long long)gridDim.x * blockDim.x) / tpr;
    row_end = M;
  }
  const int lane_c = (int)(threadIdx.x % tpr);
  const int c0 = lane_c * 8;

  for (int i = threadIdx.x; i < 2 * C; i += blockDim.x) lsum[i] = 0.0f;
  __syncthreads();

  float mu[8], is[8];
#pragma unroll
  for (int i = 0; i < 8; ++i) {
    mu[i] = mean[c0 + i];
    is[i] = invstd[c0 + i];
  }
  float s1[8] = {0}, s2[8] = {0};
  long long r = row0;
  // ST row streams x 3 tensors = up to 3*ST outstanding 16-B loads
  for (; r + (ST - 1) * rstride < row_end; r += ST * rstride) {
#pragma unroll
    for (int u = 0; u < ST / 2; ++u) {
      const long long r0 = r + 2 * u * rstride, r1 = r + (2 * u + 1) * rstride;
      const long long e0 = r0 * C + c0, e1 = r1 * C + c0;
      F8 g0 = load8(dy + e0), g1 = load8(dy + e1);
      F8 x0 = load8(x + e0), x1 = load8(x + e1);
      if (RELU) {
        const unsigned m0 = mask[r0 * c8 + lane_c];
        const unsigned m1 = mask[r1 * c8 + lane_c];
#pragma unroll
        for (int i = 0; i < 8; ++i) {
          g0.v[i] = (m0 >> i) & 1 ? g0.v[i] : 0.0f;
          g1.v[i] = (m1 >> i) & 1 ? g1.v[i] : 0.0f;
        }
      }
#pragma unroll
      for (int i = 0; i < 8; ++i) {
        s1[i] += g0.v[i] + g1.v[i];
        s2[i] = fmaf(g0.v[i], (x0.v[i] - mu[i]) * is[i],
                     fmaf(g1.v[i], (x1.v[i] - mu[i]) * is[i], s2[i]));
      }
    }
  }
  for (; r < row_end; r += rstride) {
    const long long eoff = r * C + c0;
    F8 g = load8(dy + eoff);
    F8 xv = load8(x + eoff);
    if (RELU) {
      const unsigned mb = mask[r * c8 + lane_c];
#pragma unroll
      for (int i = 0; i < 8; ++i) g.v[i] = (mb >> i) & 1 ? g.v[i] : 0.0f;
    }
#pragma unroll
    for (int i = 0; i < 8; ++i) {
      s1[i] += g.v[i];
      s2[i] = fmaf(g.v[i], (xv.v[i] - mu[i]) * is[i], s2[i]);
    }
  }
#pragma unroll
  for (int i = 0; i < 8; ++i) {
    atomicAdd(&lsum[c0 + i], s1[i]);
    atomicAdd(&lsum[C + c0 + i], s2[i]);
  }
  __syncthreads();
  float* out = sums + (long long)blockIdx.x * 2 * C;  // [grid, 2C] partials
  for (int i = threadIdx.x; i < 2 * C; i += blockDim.x) out[i] = lsum[i];
}

// B1b-v2: strip-parallel partial reduction — 256 threads = 64 channel
// indices x 4 strips over the block range (4x the parallelism of v1,
// which ran only 2C threads total: HALF A WAVE at C=64 doing the whole
// latency-bound serial partial read).
extern "C" __global__ void bn_bwd_finalize_v2_kernel(
    const float* __restrict__ partial, const int nblocks,
    float* __restrict__ sums, const int C, float* __restrict__ db_acc,
    float* __restrict__ dg_acc) {
  __shared__ float acc[4][64];
  const int lane = threadIdx.x & 63;
  const int strip = threadIdx.x >> 6;
  const int i = blockIdx.x * 64 + lane;
  const long long st = 2 * C;
  float sa[8] = {0};
  if (i < 2 * C) {
    int b = strip;
    for (; b + 28 < nblocks; b += 32) {
#pragma unroll
      for (int u = 0; u < 8; ++u)
        sa[u] += partial[(long long)(b + 4 * u) * st + i];
    }
    for (; b < nblocks; b += 4) sa[0] += partial[(long long)b * st + i];
  }
  acc[strip][lane] = ((sa[0] + sa[1]) + (sa[2] + sa[3])) +
                     ((sa[4] + sa[5]) + (sa[6] + sa[7]));
  __syncthreads();
  if (strip == 0 && i < 2 * C) {
    const float v = (acc[0][lane] + acc[1][lane]) +
                    (acc[2][lane] + acc[3][lane]);
    sums[i] = v;
    if (i < C) {
      if (db_acc) db_acc[i] += v;
    } else if (dg_acc) {
      dg_acc[i - C] += v;
    }
  }
}

// B1b: reduce bwd partials over blocks -> sums[2C] (= [dbeta; dgamma]).
extern "C" __global__ void bn_bwd_finalize_kernel(
    const float* __restrict__ partial, const int nblocks,
    float* __restrict__ sums, const int C, float* __restrict__ db_acc,
    float* __restrict__ dg_acc) {
  // db_acc/dg_acc: optional direct-grad targets (the param's bucket-view
  // gradient) — accumulating here removes the per-param AccumulateGrad
  // add kernels at world 1 (~106 tiny launches/step across ResNet50's
  // BN layers). Single writer per channel, so a plain += suffices.
  const int i = blockIdx.x * blockDim.x + threadIdx.x;
  if (i >= 2 * C) return;
  const long long st = 2 * C;
  float sa[8] = {0};
  int b = 0;
  for (; b + 8 <= nblocks; b += 8) {
#pragma unroll
    for (int u = 0; u < 8; ++u) sa[u] += partial[(long long)(b + u) * st + i];
  }
  for (; b < nblocks; ++b) sa[0] += partial[(long long)b * st + i];
  const float v = ((sa[0] + sa[1]) + (sa[2] + sa[3])) +
                  ((sa[4] + sa[5]) + (sa[6] + sa[7]));
  sums[i] = v;
  if (i < C) {
    if (db_acc) db_acc[i] += v;
  } else if (dg_acc) {
    dg_acc[i - C] += v;
  }
}

// (no separate dgamma/dbeta kernel: the bwd-reduce workspace IS [dbeta; dgamma]
// — sums[0:C] = s1 = dbeta, sums[C:2C] = s2 = dgamma; Python returns views.)

// B3: dx = scale * (dy_eff - s1/M - xhat * s2/M); optional dres = dy_eff.
template <bool RELU, bool ADD, bool TRAINING>
__global__ void bn_bwd_dx_kernel(
    const bf16* __restrict__ dy, const unsigned char* __restrict__ mask,
    const bf16* __restrict__ x, const float* __restrict__ mean,
    const float* __restrict__ invstd, const float* __restrict__ gamma,
    const float* __restrict__ sums, bf16* __restrict__ dx,
    bf16* __restrict__ dres, const long long M, const int C) {
  const int c8 = C >> 3;
  const bool p2 = (c8 & (c8 - 1)) == 0;  // pow2 fast path (see bn_apply)
  const int c8l = 31 - __clz((unsigned)c8);
  const long long total = M * c8;
  const float inv_m = 1.0f / (float)M;
  long long i = (long long)blockIdx.x * blockDim.x + threadIdx.x;
  const long long stride = (long long)gridDim.x * blockDim.x;
  for (; i < total; i += stride) {
    const int oct = p2 ? (int)(i & (c8 - 1)) : (int)(i % c8);
    const int c0 = oct * 8;
    const long long eoff = (p2 ? (i >> c8l) : (i / c8)) * C + c0;
    F8 g = load8(dy + eoff);
    if (RELU) {
      const unsigned mb = mask[i];
#pragma unroll
      for (int k = 0; k < 8; ++k) g.v[k] = (mb >> k) & 1 ? g.v[k] : 0.0f;
    }
    if (ADD) store8(dres + eoff, g);
    F8 xv = load8(x + eoff);
    F8 o;
#pragma unroll
    for (int k = 0; k < 8; ++k) {
      const int c = c0 + k;
      const float is = invstd[c];
      if (TRAINING) {
        const float xhat = (xv.v[k] - mean[c]) * is;
        o.v[k] = gamma[c] * is *
                 (g.v[k] - sums[c] * inv_m - xhat * sums[C + c] * inv_m);
      } else {
        o.v[k] = gamma[c] * is * g.v[k];
      }
    }
    store8(dx + eoff, o);
  }
}


// B-fused: reduce + finalize + dx in ONE launch (EDL_BN_BWD_FUSED).
// The three-kernel backward pays two launch/drain boundaries per BN layer
// (53 layers in resnet50_vd). Here the grid rendezvouses in device memory:
// every block writes its partial row and bumps an arrival counter; the
// first F blocks wait for full arrival, then finalize a channel SLICE each
// (parallel finalize — a single finalizer block measured 45+ us on 512
// partial rows) and bump a done counter; everyone spins RELAXED (an
// acquire per poll invalidates L2 every iteration — measured as an ~85 us
// rendezvous storm) with ONE acquire fence on exit. grid <= 512 with
// __launch_bounds__(256, 2) guarantees co-residency on 256 CUs (the spin
// cannot deadlock); all spins are iteration-bounded regardless. ws = a
// persistent int[4] {arrive, done, depart, pad}: the last block OUT
// resets it, so hipGraph replays and back-to-back layers need no host
// zeroing.
template <bool RELU, bool ADD>
__global__ __launch_bounds__(256, 2) void bn_bwd_fused_kernel(
    const bf16* __restrict__ dy, const unsigned char* __restrict__ mask,
    const bf16* __restrict__ x, const float* __restrict__ mean,
    const float* __restrict__ invstd, const float* __restrict__ gamma,
    float* __restrict__ partial, float* __restrict__ sums,
    float* __restrict__ db_acc, float* __restrict__ dg_acc,
    bf16* __restrict__ dx, bf16* __restrict__ dres, int* __restrict__ ws,
    const int nfin, const long long M, const int C) {
  __shared__ float lsum[2 * 2048];
  const int c8 = C >> 3;
  const int tpr = c8;
  const int tid = blockIdx.x * blockDim.x + threadIdx.x;
  const long long rstride = ((long long)gridDim.x * blockDim.x) / tpr;
  const int lane_c = (int)(threadIdx.x % tpr);
  const int c0 = lane_c * 8;

  for (int i = threadIdx.x; i < 2 * C; i += blockDim.x) lsum[i] = 0.0f;
  __syncthreads();

  // ---- phase 1: per-block channel reduction (bn_bwd_reduce, ST=4) ----
  float mu[8], is[8];
#pragma unroll
  for (int i = 0; i < 8; ++i) {
    mu[i] = mean[c0 + i];
    is[i] = invstd[c0 + i];
  }
  float s1[8] = {0}, s2[8] = {0};
  long long r = tid / tpr;
  for (; r + 3 * rstride < M; r += 4 * rstride) {
#pragma unroll
    for (int u = 0; u < 2; ++u) {
      const long long r0 = r + 2 * u * rstride, r1 = r + (2 * u + 1) * rstride;
      const long long e0 = r0 * C + c0, e1 = r1 * C + c0;
      F8 g0 = load8(dy + e0), g1 = load8(dy + e1);
      F8 x0 = load8(x + e0), x1 = load8(x + e1);
      if (RELU) {
        const unsigned m0 = mask[r0 * c8 + lane_c];
        const unsigned m1 = mask[r1 * c8 + lane_c];
#pragma unroll
        for (int i = 0; i < 8; ++i) {
          g0.v[i] = (m0 >> i) & 1 ? g0.v[i] : 0.0f;
          g1.v[i] = (m1 >> i) & 1 ? g1.v[i] : 0.0f;
        }
      }
#pragma unroll
      for (int i = 0; i < 8; ++i) {
        s1[i] += g0.v[i] + g1.v[i];
        s2[i] = fmaf(g0.v[i], (x0.v[i] - mu[i]) * is[i],
                     fmaf(g1.v[i], (x1.v[i] - mu[i]) * is[i], s2[i]));
      }
    }
  }
  for (; r < M; r += rstride) {
    const long long eoff = r * C + c0;
    F8 g = load8(dy + eoff);
    F8 xv = load8(x + eoff);
    if (RELU) {
      const unsigned mb = mask[r * c8 + lane_c];
#pragma unroll
      for (int i = 0; i < 8; ++i) g.v[i] = (mb >> i) & 1 ? g.v[i] : 0.0f;
    }
#pragma unroll
    for (int i = 0; i < 8; ++i) {
      s1[i] += g.v[i];
      s2[i] = fmaf(g.v[i], (xv.v[i] - mu[i]) * is[i], s2[i]);
    }
  }
#pragma unroll
  for (int i = 0; i < 8; ++i) {
    atomicAdd(&lsum[c0 + i], s1[i]);
    atomicAdd(&lsum[C + c0 + i], s2[i]);
  }
  __syncthreads();
  {
    float* out = partial + (long long)blockIdx.x * 2 * C;
    for (int i = threadIdx.x; i < 2 * C; i += blockDim.x) out[i] = lsum[i];
  }

  // ---- rendezvous ----
  __threadfence();
  if (threadIdx.x == 0)
    __hip_atomic_fetch_add(&ws[0], 1, __ATOMIC_RELEASE,
                           __HIP_MEMORY_SCOPE_AGENT);
  const int g = gridDim.x;
  if ((int)blockIdx.x < nfin) {
    // finalizer: wait for all partials, then reduce a channel slice
    if (threadIdx.x == 0) {
      for (long long it = 0; it < 50000000LL; ++it) {
        if (__hip_atomic_load(&ws[0], __ATOMIC_RELAXED,
                              __HIP_MEMORY_SCOPE_AGENT) == g)
          break;
        __builtin_amdgcn_s_sleep(2);
      }
    }
    __syncthreads();
    __builtin_amdgcn_fence(__ATOMIC_ACQUIRE, "agent");
    const long long st = 2 * C;
    const int i0 = blockIdx.x * 2 * C / nfin;
    const int i1 = (blockIdx.x + 1) * 2 * C / nfin;
    for (int i = i0 + threadIdx.x; i < i1; i += blockDim.x) {
      float sa[4] = {0, 0, 0, 0};
      int b = 0;
      for (; b + 4 <= g; b += 4)
#pragma unroll
        for (int u = 0; u < 4; ++u)
          sa[u] += partial[(long long)(b + u) * st + i];
      for (; b < g; ++b) sa[0] += partial[(long long)b * st + i];
      const float v = (sa[0] + sa[1]) + (sa[2] + sa[3]);
      sums[i] = v;
      if (i < C) {
        if (db_acc) db_acc[i] += v;
      } else if (dg_acc) {
        dg_acc[i - C] += v;
      }
    }
    __syncthreads();
    __threadfence();
    if (threadIdx.x == 0)
      __hip_atomic_fetch_add(&ws[1], 1, __ATOMIC_RELEASE,
                             __HIP_MEMORY_SCOPE_AGENT);
  }
  // everyone: wait for all slices, one acquire fence on exit
  if (threadIdx.x == 0) {
    for (long long it = 0; it < 50000000LL; ++it) {
      if (__hip_atomic_load(&ws[1], __ATOMIC_RELAXED,
                            __HIP_MEMORY_SCOPE_AGENT) == nfin)
        break;
      __builtin_amdgcn_s_sleep(2);
    }
    const int depart = __hip_atomic_fetch_add(&ws[2], 1, __ATOMIC_ACQ_REL,
                                              __HIP_MEMORY_SCOPE_AGENT);
    if (depart == g - 1) {  // everyone passed both waits: reset
      __hip_atomic_store(&ws[0], 0, __ATOMIC_RELAXED, __HIP_MEMORY_SCOPE_AGENT);
      __hip_atomic_store(&ws[1], 0, __ATOMIC_RELAXED, __HIP_MEMORY_SCOPE_AGENT);
      __hip_atomic_store(&ws[2], 0, __ATOMIC_RELAXED, __HIP_MEMORY_SCOPE_AGENT);
    }
  }
  __syncthreads();
  __builtin_amdgcn_fence(__ATOMIC_ACQUIRE, "agent");

  // ---- phase 2: dx (+dres), training form (bn_bwd_dx) ----
  const float inv_m = 1.0f / (float)M;
  const long long total = M * c8;
  long long i2 = (long long)blockIdx.x * blockDim.x + threadIdx.x;
  const long long stride2 = (long long)gridDim.x * blockDim.x;
  for (; i2 < total; i2 += stride2) {
    const int oct = (int)(i2 % c8);
    const int cc0 = oct * 8;
    const long long eoff = (i2 / c8) * C + cc0;
    F8 g2 = load8(dy + eoff);
    if (RELU) {
      const unsigned mb = mask[i2];
#pragma unroll
      for (int k = 0; k < 8; ++k) g2.v[k] = (mb >> k) & 1 ? g2.v[k] : 0.0f;
    }
    if (ADD) store8(dres + eoff, g2);
    F8 xv = load8(x + eoff);
    F8 o;
#pragma unroll
    for (int k = 0; k < 8; ++k) {
      const int c = cc0 + k;
      const float isv = invstd[c];
      const float xhat = (xv.v[k] - mean[c]) * isv;
      o.v[k] = gamma[c] * isv *
               (g2.v[k] - sums[c] * inv_m - xhat * sums[C + c] * inv_m);
    }
    store8(dx + eoff, o);
  }
}

// ---------------- launchers ----------------

static long long env_ll(const char* name, long long dflt) {
  const char* e = getenv(name);
  return e ? atoll(e) : dflt;
}

static int bn_grid_common(long long M, int C, long long cap_env) {
  const int c8 = C >> 3;
  long long rows_per_block = 256 / c8 > 0 ? 256 / c8 : 1;
  long long want = (M + rows_per_block - 1) / rows_per_block;
  long long cap = 131072 / (2 * (long long)C);
  if (cap > cap_env) cap = cap_env;
  if (cap < 8) cap = 8;
  long long g = want < cap ? want : cap;
  return (int)(g > 0 ? g : 1);
}

extern "C" int bn_stats_grid(long long M, int C) {
  // cover the tensor with enough blocks for bandwidth, but keep the
  // partial buffer (grid x 2C fp32) around <= 1 MiB.
  // cap bounds the finalize kernels' serial partial-read loop (they were
  // latency-bound at ~12-14 us with 512-block partials; ~5 us at 192).
  // EDL_BN_GRID_CAP overrides for A/B (192 = measured best tradeoff).
  return bn_grid_common(M, C, env_ll("EDL_BN_GRID_CAP", 192));
}

extern "C" int bn_bwd_grid(long long M, int C) {
  // the BWD reduce grid is decoupled from the fwd one: with the
  // strip-parallel finalize_v2 the partial-read cost no longer binds the
  // grid, so the reduce can run wider for bandwidth.
  return bn_grid_common(M, C,
                        env_ll("EDL_BN_BWD_GRID_CAP",
                               env_ll("EDL_BN_GRID_CAP", 192)));
}

extern "C" void launch_bn_stats(const void* x, float* partial, int grid,
                                long long M, int C, hipStream_t s) {
  const bool st8 = env_ll("EDL_BN_STATS_STREAMS", 4) >= 8;
  const bool contig = env_ll("EDL_BN_CONTIG", 0) != 0;
  if (contig) {
    if (st8)
      hipLaunchKernelGGL((bn_stats_kernel<8, true>), dim3(grid), dim3(256), 0,
                         s, (const bf16*)x, partial, M, C);
    else
      hipLaunchKernelGGL((bn_stats_kernel<4, true>), dim3(grid), dim3(256), 0,
                         s, (const bf16*)x, partial, M, C);
  } else if (st8) {
    hipLaunchKernelGGL((bn_stats_kernel<8>), dim3(grid), dim3(256), 0, s,
                       (const bf16*)x, partial, M, C);
  } else {
    hipLaunchKernelGGL((bn_stats_kernel<4>), dim3(grid), dim3(256), 0, s,
                       (const bf16*)x, partial, M, C);
  }
}

extern "C" void launch_bn_finalize(float* partial, int nblocks, int zero_src,
                                   const float* gamma,
                                   const float* beta, float* mean, float* invstd,
                                   float* scale, float* shift, float* rmean,
                                   float* rvar, float momentum, float eps,
                                   long long M, int C, hipStream_t s) {
  // zero_src: store 0 back after each read — returns a pooled pre_part
  // buffer to the pool clean, killing the per-step FillFunctor launches
  if (env_ll("EDL_BN_FIN_V2", 1)) {
    hipLaunchKernelGGL(bn_finalize_v2_kernel, dim3((C + 63) / 64), dim3(256),
                       0, s, partial, nblocks, zero_src, gamma, beta, mean,
                       invstd, scale, shift, rmean, rvar, momentum, eps, M, C);
    return;
  }
  hipLaunchKernelGGL(bn_finalize_kernel, dim3((C + 255) / 256), dim3(256), 0, s,
                     partial, nblocks, zero_src, gamma, beta, mean, invstd,
                     scale, shift, rmean, rvar, momentum, eps, M, C);
}

extern "C" void launch_bn_bwd_finalize(const float* partial, int nblocks,
                                       float* sums, int C, float* db_acc,
                                       float* dg_acc, hipStream_t s) {
  if (env_ll("EDL_BN_FIN_V2", 1)) {
    hipLaunchKernelGGL(bn_bwd_finalize_v2_kernel, dim3((2 * C + 63) / 64),
                       dim3(256), 0, s, partial, nblocks, sums, C, db_acc,
                       dg_acc);
    return;
  }
  hipLaunchKernelGGL(bn_bwd_finalize_kernel, dim3((2 * C + 255) / 256), dim3(256),
                     0, s, partial, nblocks, sums, C, db_acc, dg_acc);
}

extern "C" void launch_bn_apply(const void* x, const void* res, void* y,
                                unsigned char* mask,
                                const float* scale, const float* shift,
                                long long M, int C, bool relu, bool add,
                                hipStream_t s) {
  const long long total = M * (C >> 3);
  const int grid = elementwise_grid(total, 256);
  const bf16* xb = (const bf16*)x;
  const bf16* rb = (const bf16*)res;
  bf16* yb = (bf16*)y;
  if (relu && add) {
    if (mask)
      hipLaunchKernelGGL((bn_apply_kernel<true, true, true>), dim3(grid),
                         dim3(256), 0, s, xb, rb, yb, mask, scale, shift, M, C);
    else
      hipLaunchKernelGGL((bn_apply_kernel<true, true>), dim3(grid), dim3(256),
                         0, s, xb, rb, yb, nullptr, scale, shift, M, C);
  } else if (relu) {
    if (mask)
      hipLaunchKernelGGL((bn_apply_kernel<true, false, true>), dim3(grid),
                         dim3(256), 0, s, xb, nullptr, yb, mask, scale, shift, M, C);
    else
      hipLaunchKernelGGL((bn_apply_kernel<true, false>), dim3(grid), dim3(256),
                         0, s, xb, nullptr, yb, nullptr, scale, shift, M, C);
  } else if (add) {
    hipLaunchKernelGGL((bn_apply_kernel<false, true>), dim3(grid), dim3(256), 0,
                       s, xb, rb, yb, nullptr, scale, shift, M, C);
  } else {
    hipLaunchKernelGGL((bn_apply_kernel<false, false>), dim3(grid), dim3(256), 0,
                       s, xb, nullptr, yb, nullptr, scale, shift, M, C);
  }
}

extern "C" void launch_bn_bwd_reduce(const void* dy, const unsigned char* mask,
                                     const void* x,
                                     const float* mean, const float* invstd,
                                     float* partial, int grid, long long M, int C,
                                     bool relu, hipStream_t s) {
  const bool st8 = env_ll("EDL_BN_BWD_STREAMS", 4) >= 8;
  const bool contig = env_ll("EDL_BN_CONTIG", 0) != 0;
#define RCASE(R, ST, CG)                                                       \
  hipLaunchKernelGGL((bn_bwd_reduce_kernel<R, ST, CG>), dim3(grid), dim3(256), \
                     0, s, (const bf16*)dy, (R) ? mask : nullptr,              \
                     (const bf16*)x, mean, invstd, partial, M, C)
  if (relu) {
    if (contig) { if (st8) RCASE(true, 8, true); else RCASE(true, 4, true); }
    else if (st8) RCASE(true, 8, false);
    else RCASE(true, 4, false);
  } else {
    if (contig) { if (st8) RCASE(false, 8, true); else RCASE(false, 4, true); }
    else if (st8) RCASE(false, 8, false);
    else RCASE(false, 4, false);
  }
#undef RCASE
}

extern "C" void launch_bn_bwd_dx(const void* dy, const unsigned char* mask,
                                 const void* x,
                                 const float* mean, const float* invstd,
                                 const float* gamma, const float* sums, void* dx,
                                 void* dres, long long M, int C, bool relu,
                                 bool add, bool training, hipStream_t s) {
  const long long total = M * (C >> 3);
  const int grid = elementwise_grid(total, 256);
#define CASE(R, A, T)                                                          \
  hipLaunchKernelGGL((bn_bwd_dx_kernel<R, A, T>), dim3(grid), dim3(256), 0, s, \
                     (const bf16*)dy, mask, (const bf16*)x, mean,              \
                     invstd, gamma, sums, (bf16*)dx, (bf16*)dres, M, C)
  if (training) {
    if (relu && add) CASE(true, true, true);
    else if (relu) CASE(true, false, true);
    else if (add) CASE(false, true, true);
    else CASE(false, false, true);
  } else {
    if (relu && add) CASE(true, true, false);
    else if (relu) CASE(true, false, false);
    else if (add) CASE(false, true, false);
    else CASE(false, false, false);
  }
#undef CASE
}

extern "C" void launch_bn_bwd_fused(const void* dy, const unsigned char* mask,
                                    const void* x, const float* mean,
                                    const float* invstd, const float* gamma,
                                    float* partial, float* sums, float* db_acc,
                                    float* dg_acc, void* dx, void* dres,
                                    int* ws, int grid, int nfin, long long M,
                                    int C, bool relu, bool add, hipStream_t s) {
#define FCASE(R, A)                                                            \
  hipLaunchKernelGGL((bn_bwd_fused_kernel<R, A>), dim3(grid), dim3(256), 0, s, \
                     (const bf16*)dy, (R) ? mask : nullptr, (const bf16*)x,    \
                     mean, invstd, gamma, partial, sums, db_acc, dg_acc,       \
                     (bf16*)dx, (bf16*)dres, ws, nfin, M, C)
  if (relu && add) FCASE(true, true);
  else if (relu) FCASE(true, false);
  else if (add) FCASE(false, true);
  else FCASE(false, false);
#undef FCASE
}
