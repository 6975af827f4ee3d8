// Hand-written bf16 MFMA GEMM, TN layout (both operands k-major):
//     C[N1, N2] += A[K, N1]^T @ B[K, N2]      (A, B row-major bf16,
//                                              C fp32, split-K atomics)
//
// This is the conv WGRAD shape taken DIRECTLY on the activation layout:
// dW[Cout, Cin] = dY[M, Cout]^T @ X[M, Cin] with K = M = N*H*W ~ 1e5.
// The previous path materialized transpose_pad(dY) and transpose_pad(X)
// into [C, Mp] buffers first so the BT kernel could stream k-contiguous
// rows — a full extra HBM round-trip per operand (~1.3 ms/step across
// ResNet50_vd, profiles/r01_step5_final.txt: transpose_pad 26 ms +
// shift9_transpose 8 ms per 20-step window). Here the operands are read
// in their native [M, C] layout and the "transpose" happens on the LDS
// read side:
//
//   * staging is IDENTICAL to gemm_bt (global_load_lds 16 B chunks,
//     lane-linear dest, XOR slot swizzle on the SOURCE — rule 21), but
//     tiles are [BK=64 k-rows x BN n-cols] slabs of A and B;
//   * MFMA a/b fragments want 8 CONSECUTIVE k per lane at fixed n — in a
//     [k][n] tile that is a column walk, so each fragment is assembled
//     with 8 swizzled ds_read_u16 instead of one ds_read_b128. The row
//     swizzle (tn_swz: slot ^ (k&7), plus slot bit 2 XOR k bit 3) spreads
//     a fragment's k-groups over distinct 32-dword bank sets.
//     Wgrad GEMMs are HBM-bound by ~6x (e.g. M=100352, 256x64 out:
//     64 MB read vs 3.3 GFLOP), so the extra LDS traffic sits inside the
//     HBM shadow — measured end to end before enabling by default.
//   * every wave owns a 64x64 output sub-tile; configs with fewer than 4
//     sub-tiles use the surplus waves as INTRA-BLOCK k-splits (KW): each
//     k-group double-buffers its own LDS slab and walks chunks
//     c0+g, c0+g+KW, ... — groups share only the block-wide barrier, so
//     all groups run the same iteration count (idle tail iterations just
//     hit the barrier).
//   * K tail (M % 64): the owning group stages the partial chunk with
//     guarded 16 B loads + ds_write_b128 to the same lane-linear/swizzled
//     layout, zero-filling out-of-range rows (a clamped-duplicate row
//     would be ADDED into the reduction here, unlike gemm_bt's M tail).
//
// Replaces (for conv wgrad): transpose_pad + gemm_bt_splitk
// (reference example/distill/resnet/train_with_fleet.py conv backward via
// Paddle; SURVEY.md §2.4 conv bwd row).
#include "common.h"

using bf16 = __hip_bfloat16;
typedef __bf16 bf16x8 __attribute__((ext_vector_type(8)));
typedef float f32x4 __attribute__((ext_vector_type(4)));

extern __shared__ __attribute__((aligned(16))) char smem[];

// Row-constant slot swizzle: r&7 rotates 16B slots per row (glds rule 21);
// the extra (r>>3)&1 on slot bit 2 moves k-rows 8 apart onto different
// 32-dword bank sets — ds_read_u16 banks on (a/4)%32, where a pure 16B-slot
// bit-3 difference is invisible and the transposed fragment reads (lanes
// 0-15 vs 16-31 of a half) would otherwise 2-way conflict.
__device__ __forceinline__ int tn_swz(int r) {
  return (r & 7) ^ (((r >> 3) & 1) << 2);
}


// G3: B is a VIRTUAL [M, 9*Cin] operand gathered from the padded NHWC
// input xpad [N, Hp, Wp, Cin] — row m = (n, oy, ox) of dY, col block
// (sidx, ci): value xpad[n, oy*stride+sy, ox*stride+sx, ci]. A col tile
// stays inside ONE (sy,sx) shift because BN2 divides Cin, so the whole
// block shares (sy, sx, ci0) and a tile row is a contiguous BN2-channel
// read — the same glds staging as the plain kernel with a computed row
// base. This removes the shift9_transpose + transpose_pad(dy)
// materializations from conv3x3 wgrad entirely.
// TR: fragment reads via ds_read_b64_tr_b16 (gfx950 hardware transpose
// read). The u16 column-walk costs 128 ds_read_u16 + 64 v_perm per
// 32-MFMA block — the kernel measured 55% issue-stall (PMC r2c14). With
// TR the LDS image is subtiled per 16-col n-block as two 512-element
// regions (region r holds taps k%8 in [4r, 4r+4)):
//   elem(k, n) = nb*1024 + (k&4)*128 + (k>>3)*64 + (k&3)*16 + (n&15)
// and ONE tr read per 4-k half delivers each lane its column — 32 reads,
// zero packing VALU. Staging stays glds lane-linear: chunk ch = region
// (nb = ch>>1, r = ch&1), lane l holds k = (l>>3)*8 + 4r + ((l>>1)&3),
// n = nb*16 + (l&1)*8.
// DEEP (KW==1 only): 3 LDS buffers, raw s_barrier + COUNTED vmcnt so two
// staged k-tiles stay in flight across barriers (the __syncthreads form
// drains every glds per iteration — guide 'pipelining across barriers';
// costs 96 KB LDS -> 1 block/CU).
template <int BN1, int BN2, int KW, bool G3 = false, bool TR = false,
          bool DEEP = false, bool G3S = false>
__global__ __launch_bounds__(256, 2) void gemm_tn_kernel(
    const bf16* __restrict__ A, const bf16* __restrict__ B,
    float* __restrict__ C, const int N1, const int N2, const int K,
    const int Ho = 0, const int Wo = 0, const int Hp = 0, const int Wp = 0,
    const int Cin = 0, const int stride = 1, const int perm_cin = 0) {
  constexpr int BK = 64;
  constexpr int W1 = BN1 / 64, W2 = BN2 / 64;  // wave sub-tiles per dim
  constexpr int GW = W1 * W2;                  // waves per k-group
  static_assert(GW * KW == 4, "4 waves per block");
  constexpr int A_BYTES = BK * BN1 * 2;
  constexpr int B_BYTES = BK * BN2 * 2;
  constexpr int AB = A_BYTES + B_BYTES;        // one buffer of one group
  constexpr int SLOTS_A = BN1 / 8;             // 16 B slots per k-row
  constexpr int SLOTS_B = BN2 / 8;

  const int lane = threadIdx.x & 63;
  const int wave = threadIdx.x >> 6;
  const int grp = wave / GW;   // k-split group
  const int wg = wave % GW;    // wave within group
  const int wm = (wg / W2) * 64;
  const int wn = (wg % W2) * 64;

  const int tiles_n2 = N2 / BN2;
  const int bid = xcd_swizzle(blockIdx.x, gridDim.x);
  const int t1 = (bid / tiles_n2) * BN1;
  const int t2 = (bid % tiles_n2) * BN2;

  // chunk range of this block (grid.y = splitk)
  const int nchunks = (K + BK - 1) / BK;
  const int per = (nchunks + gridDim.y - 1) / gridDim.y;
  const int c0 = blockIdx.y * per;
  const int c1 = min(nchunks, c0 + per);
  if (c0 >= c1) return;
  const int iters = (c1 - c0 + KW - 1) / KW;  // UNIFORM across groups

  char* gbase = smem + grp * 2 * AB;

  // G3 block-uniform gather parameters (one 3x3 shift per col tile)
  int g3_sy = 0, g3_sx = 0, g3_ci0 = 0;
  if constexpr (G3) {
    const int sidx = t2 / Cin;
    g3_sy = sidx / 3;
    g3_sx = sidx % 3;
    g3_ci0 = t2 % Cin;
  }
  // G3S (small-Cin stems): Cin is the POW2-padded channel count (>= 8),
  // so a 16-B slot stays inside one tap but a 64-col tile spans several —
  // (tap, channel) are computed per column: tap = n2 >> log2(Cin). Taps
  // past 8 (virtual N2 padding) clamp to 8: in-bounds reads, garbage
  // columns that the caller slices away.
  const int cin_l2 = G3S ? (31 - __builtin_clz((unsigned)Cin)) : 0;
  // base address of virtual B row k (the +col offset is added by callers)
  auto b_row = [&](long long k) -> const bf16* {
    if constexpr (!G3) return B + k * N2 + t2;
    const int m = (int)k;
    const int hw = Ho * Wo;
    const int n = m / hw, rem = m % hw;
    const int oy = rem / Wo, ox = rem % Wo;
    return B + ((long long)(n * Hp + oy * stride + g3_sy) * Wp + ox * stride +
                g3_sx) *
                   Cin +
           g3_ci0;
  };
  // G3S: full per-column source address (ncol = pre-swizzle col offset)
  auto b_src = [&](long long k, int ncol) -> const bf16* {
    const int m = (int)k;
    const int hw = Ho * Wo;
    const int n = m / hw, rem = m % hw;
    const int oy = rem / Wo, ox = rem % Wo;
    int n2 = t2 + ncol;
    // virtual-padding cols clamp to the LAST real slot (slot-aligned:
    // 9*CinP is a multiple of 8) — in-bounds, 16-B aligned, discarded
    const int n2max = 9 * Cin - 8;
    if (n2 > n2max) n2 = n2max;
    const int tap = n2 >> cin_l2;
    const int cc = n2 - (tap << cin_l2);
    return B + ((long long)(n * Hp + oy * stride + tap / 3) * Wp +
                ox * stride + tap % 3) *
                   Cin +
           cc;
  };

  f32x4 acc[4][4] = {};

  // ---- staging: group's GW wave(s) stage one [BK x BN] slab pair ----
  // chunk = 1 KiB = (64/SLOTS) k-rows x SLOTS slots; lane-linear dest,
  // swizzled source (gslot = slot ^ tn_swz(r))
  // TR source mapping: chunk ch, lane l -> (k, n-offset) of the subtiled
  // image (see template comment)
  auto tr_kn = [&](int ch, int& k, int& noff) {
    const int nb = ch >> 1, reg = ch & 1;
    k = (lane >> 3) * 8 + reg * 4 + ((lane >> 1) & 3);
    noff = nb * 16 + (lane & 1) * 8;
  };

  auto stage = [&](int buf, int ct) {
    const long long k0 = (long long)ct * BK;
    char* abase = gbase + buf * AB;
    char* bbase = abase + A_BYTES;
    constexpr int ACH = A_BYTES / 1024, BCH = B_BYTES / 1024;
#pragma unroll
    for (int i = 0; i < ACH / GW; ++i) {
      const int ch = wg * (ACH / GW) + i;
      const bf16* src;
      if constexpr (TR) {
        int k, noff;
        tr_kn(ch, k, noff);
        src = A + (k0 + k) * N1 + t1 + noff;
      } else {
        const int r = ch * (64 / SLOTS_A) + lane / SLOTS_A;
        const int gslot = (lane % SLOTS_A) ^ tn_swz(r);
        src = A + (k0 + r) * N1 + t1 + gslot * 8;
      }
      __builtin_amdgcn_global_load_lds(
          (const __attribute__((address_space(1))) void*)src,
          (__attribute__((address_space(3))) void*)(abase + ch * 1024), 16, 0, 0);
    }
#pragma unroll
    for (int i = 0; i < BCH / GW; ++i) {
      const int ch = wg * (BCH / GW) + i;
      const bf16* src;
      if constexpr (TR) {
        int k, noff;
        tr_kn(ch, k, noff);
        src = G3S ? b_src(k0 + k, noff) : b_row(k0 + k) + noff;
      } else {
        const int r = ch * (64 / SLOTS_B) + lane / SLOTS_B;
        const int gslot = (lane % SLOTS_B) ^ tn_swz(r);
        src = G3S ? b_src(k0 + r, gslot * 8) : b_row(k0 + r) + gslot * 8;
      }
      __builtin_amdgcn_global_load_lds(
          (const __attribute__((address_space(1))) void*)src,
          (__attribute__((address_space(3))) void*)(bbase + ch * 1024), 16, 0, 0);
    }
  };

  // K-tail chunk: guarded loads -> registers -> ds_write_b128 to the
  // exact lane-linear dest glds would have written; OOB rows become 0
  auto stage_tail = [&](int buf, int ct) {
    const long long k0 = (long long)ct * BK;
    char* abase = gbase + buf * AB;
    char* bbase = abase + A_BYTES;
    constexpr int ACH = A_BYTES / 1024, BCH = B_BYTES / 1024;
#pragma unroll
    for (int i = 0; i < ACH / GW; ++i) {
      const int ch = wg * (ACH / GW) + i;
      int r, noff;
      if constexpr (TR) {
        tr_kn(ch, r, noff);
      } else {
        r = ch * (64 / SLOTS_A) + lane / SLOTS_A;
        noff = ((lane % SLOTS_A) ^ tn_swz(r)) * 8;
      }
      bf16x8 v = {};
      if (k0 + r < K) v = *(const bf16x8*)(A + (k0 + r) * N1 + t1 + noff);
      *(__attribute__((address_space(3))) bf16x8*)(
          (__attribute__((address_space(3))) char*)(abase) + ch * 1024 +
          lane * 16) = v;
    }
#pragma unroll
    for (int i = 0; i < BCH / GW; ++i) {
      const int ch = wg * (BCH / GW) + i;
      int r, noff;
      if constexpr (TR) {
        tr_kn(ch, r, noff);
      } else {
        r = ch * (64 / SLOTS_B) + lane / SLOTS_B;
        noff = ((lane % SLOTS_B) ^ tn_swz(r)) * 8;
      }
      bf16x8 v = {};
      if (k0 + r < K)
        v = *(const bf16x8*)(G3S ? b_src(k0 + r, noff)
                                 : b_row(k0 + r) + noff);
      *(__attribute__((address_space(3))) bf16x8*)(
          (__attribute__((address_space(3))) char*)(bbase) + ch * 1024 +
          lane * 16) = v;
    }
  };

  // ---- transposed fragment reads: 8 swizzled u16 column-walk ----
  // All addressing is hoisted 32-BIT integer offsets against one
  // addrspace(3) base: the lambda-local generic-pointer form cost a
  // v_mad_u64/v_lshl_add_u64 chain PER ELEMENT (PMC: 55% of wave cycles
  // were issue stalls). rb is a multiple of 8, so tn_swz(rb+j) =
  // j ^ ((rb>>3)&1)<<2 — the per-j XOR folds to one constant.
  const __attribute__((address_space(3))) char* lds3 =
      (const __attribute__((address_space(3))) char*)smem;
  const unsigned gbase_off = (unsigned)(grp * 2 * AB);
  auto read_a = [&](int buf, int mf, int kk) -> bf16x8 {
    const int n = wm + mf * 16 + (lane & 15);
    const int slot = n >> 3;
    const int rb = kk * 32 + (lane >> 4) * 8;
    const unsigned s0 = (unsigned)(slot ^ (((rb >> 3) & 1) << 2));
    const unsigned base = gbase_off + (unsigned)buf * AB +
                          (unsigned)rb * (SLOTS_A * 16) +
                          (unsigned)((n & 7) * 2);
    bf16x8 v;
#pragma unroll
    for (int j = 0; j < 8; ++j) {
      const unsigned off = base + (unsigned)j * (SLOTS_A * 16) +
                           ((s0 ^ (unsigned)j) << 4);
      v[j] = *(const __attribute__((address_space(3))) __bf16*)(lds3 + off);
    }
    return v;
  };
  auto read_b = [&](int buf, int nf, int kk) -> bf16x8 {
    const int n = wn + nf * 16 + (lane & 15);
    const int slot = n >> 3;
    const int rb = kk * 32 + (lane >> 4) * 8;
    const unsigned s0 = (unsigned)(slot ^ (((rb >> 3) & 1) << 2));
    const unsigned base = gbase_off + (unsigned)buf * AB + A_BYTES +
                          (unsigned)rb * (SLOTS_B * 16) +
                          (unsigned)((n & 7) * 2);
    bf16x8 v;
#pragma unroll
    for (int j = 0; j < 8; ++j) {
      const unsigned off = base + (unsigned)j * (SLOTS_B * 16) +
                           ((s0 ^ (unsigned)j) << 4);
      v[j] = *(const __attribute__((address_space(3))) __bf16*)(lds3 + off);
    }
    return v;
  };

  auto do_stage = [&](int buf, int i) {
    const int ct = c0 + grp + i * KW;
    if (ct >= c1) return;
    const int valid = min(BK, K - ct * BK);
    if (valid == BK)
      stage(buf, ct);
    else
      stage_tail(buf, ct);
  };

  // TR per-lane base byte addresses (buf/kk/frag offsets added per read;
  // all uniform parts folded here — the read is ONE ds_read_b64_tr_b16)
  const unsigned tr_a_base =
      gbase_off + (unsigned)((wm >> 4) * 2048) + (unsigned)lane * 8;
  const unsigned tr_b_base =
      gbase_off + A_BYTES + (unsigned)((wn >> 4) * 2048) + (unsigned)lane * 8;
  auto tr_read = [&](unsigned addr) -> unsigned long long {
    unsigned long long d;
    asm volatile("ds_read_b64_tr_b16 %0, %1" : "=v"(d) : "v"(addr));
    return d;
  };

  static_assert(!DEEP || KW == 1, "DEEP needs KW == 1");
  static_assert(!DEEP || (A_BYTES + B_BYTES) / 1024 / GW == 8,
                "DEEP wait literals assume 8 glds per wave per stage");

  if constexpr (DEEP) {
    do_stage(0, 0);
    if (iters > 1) do_stage(1, 1);
  } else {
    do_stage(0, 0);
    __syncthreads();  // implicit vmcnt(0)+lgkmcnt(0) drains glds/ds stores
  }

  for (int i = 0; i < iters; ++i) {
    int cur;
    if constexpr (DEEP) {
      // own stage(i) must have landed; stage(i+1) may stay in flight
      if (i + 1 < iters)
        asm volatile("s_waitcnt vmcnt(8)" ::: "memory");
      else
        asm volatile("s_waitcnt vmcnt(0)" ::: "memory");
      asm volatile("s_waitcnt lgkmcnt(0)" ::: "memory");  // tail ds_writes
      __builtin_amdgcn_s_barrier();
      if (i + 2 < iters) do_stage((i + 2) % 3, i + 2);
      cur = i % 3;
    } else {
      cur = i & 1;
      if (i + 1 < iters) do_stage(cur ^ 1, i + 1);
    }
    if (c0 + grp + i * KW < c1) {
#pragma unroll
      for (int kk = 0; kk < 2; ++kk) {
        bf16x8 a[4], b[4];
        if constexpr (TR) {
          union Q { unsigned long long q[2]; bf16x8 v; };
          Q qa[4], qb[4];
          const unsigned ab =
              tr_a_base + (unsigned)cur * AB + (unsigned)(kk * 512);
          const unsigned bb =
              tr_b_base + (unsigned)cur * AB + (unsigned)(kk * 512);
#pragma unroll
          for (int mf = 0; mf < 4; ++mf) {
            qa[mf].q[0] = tr_read(ab + mf * 2048);
            qa[mf].q[1] = tr_read(ab + mf * 2048 + 1024);
          }
#pragma unroll
          for (int nf = 0; nf < 4; ++nf) {
            qb[nf].q[0] = tr_read(bb + nf * 2048);
            qb[nf].q[1] = tr_read(bb + nf * 2048 + 1024);
          }
          // hipcc does not count asm LDS reads: drain them, then fence
          // the MFMAs below the wait (guide rule 18)
          asm volatile("s_waitcnt lgkmcnt(0)" ::: "memory");
          __builtin_amdgcn_sched_barrier(0);
#pragma unroll
          for (int f = 0; f < 4; ++f) {
            a[f] = qa[f].v;
            b[f] = qb[f].v;
          }
        } else {
#pragma unroll
          for (int mf = 0; mf < 4; ++mf) a[mf] = read_a(cur, mf, kk);
#pragma unroll
          for (int nf = 0; nf < 4; ++nf) b[nf] = read_b(cur, nf, kk);
        }
#pragma unroll
        for (int mf = 0; mf < 4; ++mf)
#pragma unroll
          for (int nf = 0; nf < 4; ++nf)
            acc[mf][nf] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
                a[mf], b[nf], acc[mf][nf], 0, 0, 0);
      }
    }
    if constexpr (!DEEP) __syncthreads();
  }
  if constexpr (DEEP) __syncthreads();  // drain before epilogue reuse

  // ---- KW > 1: fold the k-group partials in LDS first — without this
  // every group wave atomicAdds the SAME 64x64 tile (KW x the atomic
  // traffic, all on one small hot region; measured 0.53x on the
  // (100352,64,64) wgrad). The tile buffers are dead past the last
  // barrier, so the reduction aliases smem freely. ----
  if constexpr (KW > 1) {
    float* red = (float*)smem;
    if (grp > 0) {
      float* dst = red + ((grp - 1) * GW + wg) * 4096;
#pragma unroll
      for (int mf = 0; mf < 4; ++mf)
#pragma unroll
        for (int nf = 0; nf < 4; ++nf)
#pragma unroll
          for (int reg = 0; reg < 4; ++reg)
            dst[(mf * 16 + nf * 4 + reg) * 64 + lane] = acc[mf][nf][reg];
    }
    __syncthreads();
    if (grp > 0) return;
    for (int j = 1; j < KW; ++j) {
      const float* src = red + ((j - 1) * GW + wg) * 4096;
#pragma unroll
      for (int mf = 0; mf < 4; ++mf)
#pragma unroll
        for (int nf = 0; nf < 4; ++nf)
#pragma unroll
          for (int reg = 0; reg < 4; ++reg)
            acc[mf][nf][reg] += src[(mf * 16 + nf * 4 + reg) * 64 + lane];
    }
  }

  // ---- epilogue: C fp32 [N1, N2], atomic fold (splitk partials);
  // D frag: row (n1) = (lane>>4)*4+reg, col (n2) = lane&15 ----
  const int cn = lane & 15;
  const int r4 = (lane >> 4) * 4;
#pragma unroll
  for (int mf = 0; mf < 4; ++mf) {
#pragma unroll
    for (int reg = 0; reg < 4; ++reg) {
      const int n1 = t1 + wm + mf * 16 + r4 + reg;
      if (n1 < N1) {
        if (perm_cin > 0) {
          // direct-grad layout: C is the conv weight [Cout, Cin, 3, 3];
          // our n2 = s*Cin + ci (s-major) -> offset (n1*Cin + ci)*9 + s
#pragma unroll
          for (int nf = 0; nf < 4; ++nf) {
            const int n2 = t2 + wn + cn + nf * 16;
            const int sidx = n2 / perm_cin, ci = n2 % perm_cin;
            atomicAdd(C + ((long long)n1 * perm_cin + ci) * 9 + sidx,
                      acc[mf][nf][reg]);
          }
        } else {
          float* crow = C + (long long)n1 * N2 + t2 + wn + cn;
#pragma unroll
          for (int nf = 0; nf < 4; ++nf) {
            atomicAdd(&crow[nf * 16], acc[mf][nf][reg]);
          }
        }
      }
    }
  }
}

static bool tn_use_tr() {
  static const bool v = []() {
    const char* e = getenv("EDL_TN_TR");
    return !e || atoi(e) != 0;  // default ON (A/B: EDL_TN_TR=0)
  }();
  return v;
}

static bool tn_use_deep() {
  static const bool v = []() {
    const char* e = getenv("EDL_TN_DEEP");
    return e && atoi(e) != 0;  // opt-in experiment (3-buffer pipeline)
  }();
  return v;
}

extern "C" void launch_gemm_tn_splitk(const void* A, const void* B, float* C,
                                      int N1, int N2, int K, int splitk,
                                      hipStream_t s) {
  const bool b1 = N1 % 128 == 0, b2 = N2 % 128 == 0;
  const int BN1 = b1 ? 128 : 64, BN2 = b2 ? 128 : 64;
  const dim3 grid((N1 / BN1) * (N2 / BN2), splitk);
  const int KW = 4 / ((BN1 / 64) * (BN2 / 64));
  const int lds = KW * 2 * (64 * BN1 * 2 + 64 * BN2 * 2);
#define TN_LAUNCH(B1, B2, W)                                                  do {                                                                          if (tn_use_tr())                                                              hipLaunchKernelGGL((gemm_tn_kernel<B1, B2, W, false, true>), grid,                             dim3(256), lds, s, (const bf16*)A, (const bf16*)B,                          C, N1, N2, K);                                         else                                                                          hipLaunchKernelGGL((gemm_tn_kernel<B1, B2, W, false, false>), grid,                            dim3(256), lds, s, (const bf16*)A, (const bf16*)B,                          C, N1, N2, K);                                       } while (0)
  if (b1 && b2) {
    if (tn_use_deep()) {
      const int lds3 = 3 * (64 * 128 * 2 + 64 * 128 * 2);
      if (tn_use_tr())
        hipLaunchKernelGGL((gemm_tn_kernel<128, 128, 1, false, true, true>),
                           grid, dim3(256), lds3, s, (const bf16*)A,
                           (const bf16*)B, C, N1, N2, K);
      else
        hipLaunchKernelGGL((gemm_tn_kernel<128, 128, 1, false, false, true>),
                           grid, dim3(256), lds3, s, (const bf16*)A,
                           (const bf16*)B, C, N1, N2, K);
    } else {
      TN_LAUNCH(128, 128, 1);
    }
  }
  else if (b1) TN_LAUNCH(128, 64, 2);
  else if (b2) TN_LAUNCH(64, 128, 2);
  else  // KW=4 spills 28 B scratch under TR (16 live u64 reads + acc)
    hipLaunchKernelGGL((gemm_tn_kernel<64, 64, 4, false, false>), grid,
                       dim3(256), lds, s, (const bf16*)A, (const bf16*)B, C,
                       N1, N2, K);
#undef TN_LAUNCH
}

extern "C" void launch_gemm_tn3x3_splitk(const void* dy, const void* xpad,
                                         float* C, int Cout, int Cin, int M,
                                         int Ho, int Wo, int Hp, int Wp,
                                         int stride, int splitk, int perm,
                                         hipStream_t s) {
  // dW3[Cout, 9*Cin] = dY[M, Cout]^T @ gather3x3(xpad); BN2 must divide
  // Cin so each col tile sits inside one (sy, sx) shift
  const int N2 = 9 * Cin;
  const bool b1 = Cout % 128 == 0, b2 = Cin % 128 == 0;
  const int BN1 = b1 ? 128 : 64, BN2 = b2 ? 128 : 64;
  const dim3 grid((Cout / BN1) * (N2 / BN2), splitk);
  const int lds = (4 / ((BN1 / 64) * (BN2 / 64))) * 2 *
                  (64 * BN1 * 2 + 64 * BN2 * 2);
#define TN3_LAUNCH(B1, B2, W)                                                   do {                                                                            if (tn_use_tr())                                                                hipLaunchKernelGGL((gemm_tn_kernel<B1, B2, W, true, true>), grid,                                dim3(256), lds, s, (const bf16*)dy,                                           (const bf16*)xpad, C, Cout, N2, M, Ho, Wo, Hp, Wp,                            Cin, stride, perm);                                      else                                                                            hipLaunchKernelGGL((gemm_tn_kernel<B1, B2, W, true, false>), grid,                               dim3(256), lds, s, (const bf16*)dy,                                           (const bf16*)xpad, C, Cout, N2, M, Ho, Wo, Hp, Wp,                            Cin, stride, perm);                                    } while (0)
  if (b1 && b2) {
    if (tn_use_deep()) {
      const int lds3 = 3 * (64 * 128 * 2 + 64 * 128 * 2);
      if (tn_use_tr())
        hipLaunchKernelGGL((gemm_tn_kernel<128, 128, 1, true, true, true>),
                           grid, dim3(256), lds3, s, (const bf16*)dy,
                           (const bf16*)xpad, C, Cout, N2, M, Ho, Wo, Hp, Wp,
                           Cin, stride, perm);
      else
        hipLaunchKernelGGL((gemm_tn_kernel<128, 128, 1, true, false, true>),
                           grid, dim3(256), lds3, s, (const bf16*)dy,
                           (const bf16*)xpad, C, Cout, N2, M, Ho, Wo, Hp, Wp,
                           Cin, stride, perm);
    } else {
      TN3_LAUNCH(128, 128, 1);
    }
  }
  else if (b1) TN3_LAUNCH(128, 64, 2);
  else if (b2) TN3_LAUNCH(64, 128, 2);
  else
    hipLaunchKernelGGL((gemm_tn_kernel<64, 64, 4, true, false>), grid,
                       dim3(256), lds, s, (const bf16*)dy, (const bf16*)xpad,
                       C, Cout, N2, M, Ho, Wo, Hp, Wp, Cin, stride, perm);
#undef TN3_LAUNCH
}

extern "C" void launch_gemm_tn3x3_small(const void* dy, const void* xpad,
                                        float* C, int N1v, int N2v, int M,
                                        int Ho, int Wo, int Hp, int Wp,
                                        int CinP, int stride, int splitk,
                                        hipStream_t s) {
  // Deep-stem wgrad (VERDICT r1 #4 tail): dW[Cout<=64, 9*CinP] with
  // CinP = pow2-padded channels (>= 8, x channel-padded by the caller),
  // dy pre-padded to N1v=64 columns, N2v = 64*ceil(9*CinP/64) with
  // tap-clamped garbage columns the caller slices away (G3S).
  const dim3 grid((N1v / 64) * (N2v / 64), splitk);
  const int lds = 4 * 2 * (64 * 64 * 2 + 64 * 64 * 2);
  hipLaunchKernelGGL((gemm_tn_kernel<64, 64, 4, true, false, false, true>),
                     grid, dim3(256), lds, s, (const bf16*)dy,
                     (const bf16*)xpad, C, N1v, N2v, M, Ho, Wo, Hp, Wp, CinP,
                     stride, 0);
}
