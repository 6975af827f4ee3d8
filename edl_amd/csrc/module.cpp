// Python bindings for the edl_amd CDNA4 kernel layer (edl_amd._C).
// Compiled by hipcc directly (no hipify, no CUDA path) — see build_hip.py.
#include <unordered_map>
#include <unordered_set>
#include <torch/extension.h>

#include <c10/hip/HIPStream.h>
#include <hip/hip_runtime.h>

// launchers from the .hip translation units
extern "C" void launch_fused_sgd_f32(float*, const float*, float*, float, float,
                                     float, float, long long, const float*,
                                     hipStream_t);
extern "C" void launch_kd_ce_fwd_f32(const float*, const float*, float*, int, int,
                                     hipStream_t);
extern "C" void launch_kd_ce_bwd_f32(const float*, const float*, float*, float, int,
                                     int, hipStream_t);
extern "C" void launch_kd_ce_fwd_bf16(const void*, const void*, float*, int, int,
                                      hipStream_t);
extern "C" void launch_kd_ce_bwd_bf16(const void*, const void*, void*, float, int,
                                      int, hipStream_t);
extern "C" int bn_stats_grid(long long, int);
extern "C" int bn_bwd_grid(long long, int);
extern "C" void launch_bn_stats(const void*, float*, int, long long, int,
                                hipStream_t);
extern "C" void launch_bn_finalize(float*, int, int, const float*, const float*,
                                   float*, float*, float*, float*, float*, float*,
                                   float, float, long long, int, hipStream_t);
extern "C" void launch_bn_apply(const void*, const void*, void*, unsigned char*,
                                const float*, const float*, long long, int,
                                bool, bool, hipStream_t);
extern "C" void launch_bn_bwd_fused(const void*, const unsigned char*,
                                    const void*, const float*, const float*,
                                    const float*, float*, float*, float*,
                                    float*, void*, void*, int*, int, int,
                                    long long, int, bool, bool, hipStream_t);
extern "C" void launch_bn_bwd_reduce(const void*, const unsigned char*,
                                     const void*, const float*, const float*,
                                     float*, int, long long, int, bool,
                                     hipStream_t);
extern "C" void launch_bn_bwd_finalize(const float*, int, float*, int, float*,
                                       float*, hipStream_t);
extern "C" void launch_gemm_bt(const void*, const void*, void*, int, int, int,
                               float*, hipStream_t);
extern "C" int gemm_bt_tiles_m(int, int);
extern "C" int conv3x3_tiles_m(int, int);
extern "C" void launch_pad_nhwc(const void*, void*, int, int, int, int, int, int,
                                hipStream_t);
extern "C" void launch_conv3x3(const void*, const void*, void*, int, int, int,
                               int, int, int, int, int, float*, int, float*,
                               hipStream_t);
extern "C" int conv3x3_pick_splitk(int, int, int);
extern "C" void launch_conv3x3_grouped(const void*, const void*, void*, int, int,
                                       int, int, int, int, int, int, int,
                                       hipStream_t);
extern "C" void launch_conv3x3s2_dgrad(const void*, const void*, void*, int,
                                       int, int, int, int, int, int,
                                       hipStream_t);
extern "C" void launch_conv3x3_small(const void*, const void*, void*, int, int,
                                     int, int, int, int, int, int, float*,
                                     hipStream_t);
extern "C" void launch_gemm_tn3x3_small(const void*, const void*, float*,
                                        int, int, int, int, int, int, int,
                                        int, int, int, hipStream_t);
extern "C" void launch_pad_nhwc_cpad(const void*, void*, int, int, int, int,
                                     int, int, int, hipStream_t);
extern "C" void launch_cast_bf16_zero(float*, void*, long long, hipStream_t);
extern "C" void launch_repack_dgrad_w3(const void*, void*, int, int, int,
                                       hipStream_t);
extern "C" void launch_maxpool3x3s2_fwd(const void*, void*, unsigned char*,
                                        int, int, int, int, int, int,
                                        hipStream_t);
extern "C" void launch_maxpool3x3s2_bwd(const void*, const unsigned char*,
                                        void*, int, int, int, int, int, int,
                                        hipStream_t);
extern "C" void launch_transpose_pad(const void*, void*, int, int, int,
                                     hipStream_t);
extern "C" void launch_gemm_bt_splitk(const void*, const void*, float*, int, int,
                                      int, int, hipStream_t);
extern "C" void launch_shift9_transpose(const void*, void*, int, int, int, int,
                                        int, int, int, int, hipStream_t);
extern "C" void launch_gemm_tn_splitk(const void*, const void*, float*, int,
                                      int, int, int, hipStream_t);
extern "C" void launch_gemm_tn3x3_splitk(const void*, const void*, float*, int,
                                         int, int, int, int, int, int, int,
                                         int, int, hipStream_t);
extern "C" void launch_avgpool2x2_fwd(const void*, void*, int, int, int, int,
                                      int, int, hipStream_t);
extern "C" void launch_avgpool2x2_bwd(const void*, void*, int, int, int, int,
                                      int, int, hipStream_t);
extern "C" void launch_bn_bwd_dx(const void*, const unsigned char*, const void*,
                                 const float*, const float*, const float*,
                                 const float*, void*, void*, long long, int,
                                 bool, bool, bool, hipStream_t);

namespace {

hipStream_t cur_stream() {
  return c10::hip::getCurrentHIPStream().stream();
}

void fused_sgd(torch::Tensor p, torch::Tensor g, torch::Tensor m, double lr,
               double momentum, double weight_decay, double grad_scale,
               c10::optional<torch::Tensor> lr_dev) {
  TORCH_CHECK(p.is_cuda() && g.is_cuda() && m.is_cuda(), "fused_sgd: GPU tensors required");
  TORCH_CHECK(p.scalar_type() == torch::kFloat32 && g.scalar_type() == torch::kFloat32 &&
                  m.scalar_type() == torch::kFloat32,
              "fused_sgd: fp32 only");
  TORCH_CHECK(p.is_contiguous() && g.is_contiguous() && m.is_contiguous(),
              "fused_sgd: contiguous flat buffers required");
  TORCH_CHECK(p.numel() == g.numel() && p.numel() == m.numel(), "fused_sgd: size mismatch");
  const float* lrp = nullptr;
  if (lr_dev.has_value()) {
    TORCH_CHECK(lr_dev->is_cuda() && lr_dev->scalar_type() == torch::kFloat32 &&
                    lr_dev->numel() == 1,
                "fused_sgd: lr_dev must be a fp32 GPU scalar");
    lrp = lr_dev->data_ptr<float>();
  }
  launch_fused_sgd_f32(p.data_ptr<float>(), g.data_ptr<float>(), m.data_ptr<float>(),
                       (float)lr, (float)momentum, (float)weight_decay,
                       (float)grad_scale, (long long)p.numel(), lrp, cur_stream());
}

torch::Tensor kd_ce_forward(torch::Tensor s, torch::Tensor t) {
  TORCH_CHECK(s.is_cuda() && t.is_cuda(), "kd_ce: GPU tensors required");
  TORCH_CHECK(s.dim() == 2 && t.sizes() == s.sizes(), "kd_ce: [B,C] logits expected");
  TORCH_CHECK(s.scalar_type() == t.scalar_type(), "kd_ce: dtype mismatch");
  auto sc = s.contiguous();
  auto tc = t.contiguous();
  const int B = (int)s.size(0), C = (int)s.size(1);
  auto lpr = torch::empty({B}, s.options().dtype(torch::kFloat32));
  if (s.scalar_type() == torch::kFloat32) {
    launch_kd_ce_fwd_f32(sc.data_ptr<float>(), tc.data_ptr<float>(),
                         lpr.data_ptr<float>(), B, C, cur_stream());
  } else if (s.scalar_type() == torch::kBFloat16) {
    launch_kd_ce_fwd_bf16(sc.data_ptr(), tc.data_ptr(), lpr.data_ptr<float>(), B, C,
                          cur_stream());
  } else {
    TORCH_CHECK(false, "kd_ce: fp32/bf16 only");
  }
  return lpr;
}

torch::Tensor kd_ce_backward(torch::Tensor s, torch::Tensor t, double gout_over_B) {
  auto sc = s.contiguous();
  auto tc = t.contiguous();
  const int B = (int)s.size(0), C = (int)s.size(1);
  auto ds = torch::empty_like(sc);
  if (s.scalar_type() == torch::kFloat32) {
    launch_kd_ce_bwd_f32(sc.data_ptr<float>(), tc.data_ptr<float>(), ds.data_ptr<float>(),
                         (float)gout_over_B, B, C, cur_stream());
  } else if (s.scalar_type() == torch::kBFloat16) {
    launch_kd_ce_bwd_bf16(sc.data_ptr(), tc.data_ptr(), ds.data_ptr(),
                          (float)gout_over_B, B, C, cur_stream());
  } else {
    TORCH_CHECK(false, "kd_ce: fp32/bf16 only");
  }
  return ds;
}

static std::unordered_set<const void*>& pooled_ptr_set() {
  static std::unordered_set<const void*> ptrs;
  return ptrs;
}

static torch::Tensor part_pool_get(int rows, int64_t cols,
                                   torch::TensorOptions opts) {
  // Capped BN-stats partial buffers are accumulated with atomics, so they
  // must start zeroed. Pool them per (device, shape): zeroed ONCE here,
  // and bn_finalize(zero_src=1) stores zeros back after reading, so reuse
  // needs no per-step fill launch. If a producer's partials are ever
  // dropped unconsumed, the Python side zeroes them (bnrelu.forward).
  static std::unordered_map<int64_t, torch::Tensor> pool;
  const int64_t key = ((int64_t)opts.device().index() << 48) |
                      ((int64_t)rows << 28) | cols;
  auto it = pool.find(key);
  if (it != pool.end()) return it->second;
  auto t = torch::zeros({rows, cols}, opts);
  pool.emplace(key, t);
  pooled_ptr_set().insert(t.data_ptr());
  return t;
}

// x: NHWC bf16 viewed as [M, C] contiguous (channels_last 4-D collapses to
// this). gamma/beta/running stats: fp32 [C].
void check_bn_inputs_eval(const torch::Tensor& x, int64_t C) {
  // eval apply is elementwise — no channel cap (ResNeXt101_32x16d
  // stage-4 BNs are 4096-wide)
  TORCH_CHECK(x.is_cuda() && x.scalar_type() == torch::kBFloat16,
              "bn: bf16 GPU tensor required");
  TORCH_CHECK(C % 8 == 0, "bn: C % 8 == 0 required, got ", C);
}

void check_bn_inputs(const torch::Tensor& x, int64_t C) {
  check_bn_inputs_eval(x, C);
  // the training stats/reduce kernels stage 2*C fp32 in LDS
  TORCH_CHECK(C <= 2048, "bn: C <= 2048 supported");
}

std::vector<torch::Tensor> bn_fwd_train(torch::Tensor x, torch::Tensor gamma,
                                        torch::Tensor beta, torch::Tensor rmean,
                                        torch::Tensor rvar, double momentum,
                                        double eps,
                                        c10::optional<torch::Tensor> res,
                                        bool relu,
                                        c10::optional<torch::Tensor> pre_part) {
  // x: [M, C] contiguous view of an NHWC tensor (Python side reshapes)
  const int64_t C = gamma.numel();
  TORCH_CHECK(x.dim() == 2 && x.size(1) == C && x.is_contiguous(),
              "bn: x must be a contiguous [M, C] view");
  check_bn_inputs(x, C);
  const long long M = x.numel() / C;
  auto opts = gamma.options().dtype(torch::kFloat32);
  const int grid = bn_stats_grid(M, (int)C);
  auto partial = torch::empty({grid, 2 * C}, opts);
  auto mean = torch::empty({C}, opts);
  auto invstd = torch::empty({C}, opts);
  auto scale = torch::empty({C}, opts);
  auto shift = torch::empty({C}, opts);
  auto y = torch::empty_like(x);
  // 1-bit/channel ReLU mask for the backward (one byte per channel octet)
  auto msk = relu ? torch::empty({M, C / 8}, x.options().dtype(torch::kUInt8))
                  : torch::empty({0}, x.options().dtype(torch::kUInt8));
  auto s = cur_stream();
  int fin_grid = grid;
  if (pre_part.has_value()) {
    // stats already folded into the producing conv's epilogue
    // ([tiles_m, 2C] fp32, conv3x3.hip BN_PART) — skip the stats kernel
    // (a full activation re-read)
    TORCH_CHECK(pre_part->is_contiguous() &&
                    pre_part->scalar_type() == torch::kFloat32 &&
                    pre_part->dim() == 2 && pre_part->size(1) == 2 * C,
                "bn: pre_part [tiles, 2C] fp32");
    partial = *pre_part;
    fin_grid = (int)pre_part->size(0);
  } else {
    launch_bn_stats(x.data_ptr(), partial.data_ptr<float>(), grid, M, (int)C,
                    s);
  }
  // zero-after-read ONLY for pooled buffers (capped partials): transient
  // uncapped partials are discarded, zeroing them is pure write waste
  const int zero_src =
      pre_part.has_value() &&
      pooled_ptr_set().count((const void*)partial.data_ptr());
  launch_bn_finalize(partial.data_ptr<float>(), fin_grid,
                     zero_src, gamma.data_ptr<float>(),
                     beta.data_ptr<float>(), mean.data_ptr<float>(),
                     invstd.data_ptr<float>(), scale.data_ptr<float>(),
                     shift.data_ptr<float>(),
                     rmean.defined() ? rmean.data_ptr<float>() : nullptr,
                     rvar.defined() ? rvar.data_ptr<float>() : nullptr,
                     (float)momentum, (float)eps, M, (int)C, s);
  launch_bn_apply(x.data_ptr(), res.has_value() ? res->data_ptr() : nullptr,
                  y.data_ptr(),
                  relu ? (unsigned char*)msk.data_ptr() : nullptr,
                  scale.data_ptr<float>(), shift.data_ptr<float>(),
                  M, (int)C, relu, res.has_value(), s);
  return {y, mean, invstd, msk};
}

torch::Tensor bn_fwd_eval(torch::Tensor x, torch::Tensor scale, torch::Tensor shift,
                          c10::optional<torch::Tensor> res, bool relu) {
  const int64_t C = scale.numel();
  check_bn_inputs_eval(x, C);
  const long long M = x.numel() / C;
  auto y = torch::empty_like(x);
  launch_bn_apply(x.data_ptr(), res.has_value() ? res->data_ptr() : nullptr,
                  y.data_ptr(), nullptr,
                  scale.data_ptr<float>(), shift.data_ptr<float>(),
                  M, (int)C, relu, res.has_value(), cur_stream());
  return y;
}

std::vector<torch::Tensor> bn_bwd(torch::Tensor dy, c10::optional<torch::Tensor> msk,
                                  torch::Tensor x,
                                  torch::Tensor mean, torch::Tensor invstd,
                                  torch::Tensor gamma, bool relu, bool add,
                                  bool training,
                                  c10::optional<torch::Tensor> dgamma_acc,
                                  c10::optional<torch::Tensor> dbeta_acc) {
  const int64_t C = gamma.numel();
  check_bn_inputs(x, C);
  const long long M = x.numel() / C;
  auto opts = gamma.options().dtype(torch::kFloat32);
  const int grid = bn_bwd_grid(M, (int)C);
  auto sums = torch::empty({2, C}, opts);
  auto dx = torch::empty_like(x);
  auto dres = add ? torch::empty_like(x) : torch::Tensor();
  auto s = cur_stream();
  auto dyc = dy.is_contiguous() ? dy : dy.contiguous();
  TORCH_CHECK(!relu || (msk.has_value() && msk->numel() == M * (C / 8)),
              "bn_bwd: relu path needs the fwd mask");
  const unsigned char* mp =
      relu ? (const unsigned char*)msk->data_ptr() : nullptr;
  // direct-grad mode: accumulate dgamma/dbeta straight into the params'
  // bucket-view gradients (skips the per-param AccumulateGrad kernels)
  float* dg_acc = nullptr;
  float* db_acc = nullptr;
  if (dgamma_acc.has_value()) {
    TORCH_CHECK(dgamma_acc->numel() == C && dgamma_acc->is_contiguous() &&
                    dgamma_acc->scalar_type() == torch::kFloat32,
                "bn_bwd: dgamma_acc fp32 [C]");
    dg_acc = dgamma_acc->data_ptr<float>();
  }
  if (dbeta_acc.has_value()) {
    TORCH_CHECK(dbeta_acc->numel() == C && dbeta_acc->is_contiguous() &&
                    dbeta_acc->scalar_type() == torch::kFloat32,
                "bn_bwd: dbeta_acc fp32 [C]");
    db_acc = dbeta_acc->data_ptr<float>();
  }
  // read per call (not static) so tests can A/B via os.environ
  const char* fenv = getenv("EDL_BN_BWD_FUSED");
  const bool fused_bwd = fenv ? atoi(fenv) != 0 : false;
  if (fused_bwd && training) {
    // one launch: reduce + last-block finalize + dx (grid-wide flag sync);
    // ws = persistent per-device {arrive, flag, depart} ints, self-resetting
    static std::unordered_map<int, torch::Tensor> ws_pool;
    const int dev = (int)x.device().index();
    auto wit = ws_pool.find(dev);
    if (wit == ws_pool.end())
      wit = ws_pool.emplace(dev, torch::zeros({4},
          x.options().dtype(torch::kInt32))).first;
    // grid: big enough for the dx phase to saturate HBM (the reduce-grid
    // cap of 192 starved phase 2 by 3-7x in the v1 cut), bounded at 512 =
    // the co-residency guarantee of __launch_bounds__(256, 2) on 256 CUs
    // (the rendezvous spins cannot deadlock)
    const long long total8 = M * (C / 8);
    int fgrid = (int)((total8 + 1023) / 1024);
    if (fgrid < grid) fgrid = grid;
    if (fgrid > 512) fgrid = 512;
    // parallel finalize: slice 2C across up to 8 blocks, >=256 ch each
    int nfin = (int)(2 * C) / 256;
    if (nfin < 1) nfin = 1;
    if (nfin > 8) nfin = 8;
    auto fpartial = torch::empty({fgrid, 2 * C}, opts);
    launch_bn_bwd_fused(dyc.data_ptr(), mp, x.data_ptr(),
                        mean.data_ptr<float>(), invstd.data_ptr<float>(),
                        gamma.data_ptr<float>(), fpartial.data_ptr<float>(),
                        sums.data_ptr<float>(), db_acc, dg_acc, dx.data_ptr(),
                        add ? dres.data_ptr() : nullptr,
                        wit->second.data_ptr<int>(), fgrid, nfin, M, (int)C,
                        relu, add, s);
    return {dx, sums[1], sums[0], dres};
  }
  auto partial = torch::empty({grid, 2 * C}, opts);
  launch_bn_bwd_reduce(dyc.data_ptr(), mp, x.data_ptr(),
                       mean.data_ptr<float>(), invstd.data_ptr<float>(),
                       partial.data_ptr<float>(), grid, M, (int)C, relu, s);
  launch_bn_bwd_finalize(partial.data_ptr<float>(), grid, sums.data_ptr<float>(),
                         (int)C, db_acc, dg_acc, s);
  launch_bn_bwd_dx(dyc.data_ptr(), mp, x.data_ptr(),
                   mean.data_ptr<float>(), invstd.data_ptr<float>(),
                   gamma.data_ptr<float>(), sums.data_ptr<float>(), dx.data_ptr(),
                   add ? dres.data_ptr() : nullptr, M, (int)C, relu, add, training,
                   s);
  // dbeta = sums[0], dgamma = sums[1] (views of the reduce workspace)
  return {dx, sums[1], sums[0], dres};
}

torch::Tensor gemm_bt(torch::Tensor a, torch::Tensor b) {
  // C[M, N] = A[M, K] @ B[N, K]^T, bf16 row-major
  TORCH_CHECK(a.is_cuda() && b.is_cuda(), "gemm_bt: GPU tensors required");
  TORCH_CHECK(a.scalar_type() == torch::kBFloat16 &&
                  b.scalar_type() == torch::kBFloat16,
              "gemm_bt: bf16 only");
  TORCH_CHECK(a.dim() == 2 && b.dim() == 2 && a.size(1) == b.size(1),
              "gemm_bt: [M,K] x [N,K]");
  auto ac = a.contiguous();
  auto bc = b.contiguous();
  const int M = (int)a.size(0), K = (int)a.size(1), N = (int)b.size(0);
  TORCH_CHECK(K % 64 == 0 && N % 64 == 0, "gemm_bt: K,N % 64 == 0");
  auto c = torch::empty({M, N}, a.options());
  launch_gemm_bt(ac.data_ptr(), bc.data_ptr(), c.data_ptr(), M, N, K, nullptr,
                 cur_stream());
  return c;
}


std::vector<torch::Tensor> gemm_bt_stats(torch::Tensor a, torch::Tensor b) {
  // like gemm_bt, additionally returning BN stats partials [tiles_m, 2N]
  // of the bf16-rounded output (the following BN skips its stats kernel)
  TORCH_CHECK(a.is_cuda() && b.is_cuda(), "gemm_bt: GPU tensors required");
  TORCH_CHECK(a.scalar_type() == torch::kBFloat16 &&
                  b.scalar_type() == torch::kBFloat16,
              "gemm_bt: bf16 only");
  TORCH_CHECK(a.dim() == 2 && b.dim() == 2 && a.size(1) == b.size(1),
              "gemm_bt: [M,K] x [N,K]");
  auto ac = a.contiguous();
  auto bc = b.contiguous();
  const int M = (int)a.size(0), K = (int)a.size(1), N = (int)b.size(0);
  TORCH_CHECK(K % 64 == 0 && N % 64 == 0, "gemm_bt: K,N % 64 == 0");
  auto c = torch::empty({M, N}, a.options());
  const int tiles_m = gemm_bt_tiles_m(M, N);
  const int rows = std::min(tiles_m, 192);
  auto part = tiles_m > rows
                  ? part_pool_get(rows, 2 * N,
                                  a.options().dtype(torch::kFloat32))
                  : torch::empty({rows, 2 * N},
                                 a.options().dtype(torch::kFloat32));
  launch_gemm_bt(ac.data_ptr(), bc.data_ptr(), c.data_ptr(), M, N, K,
                 part.data_ptr<float>(), cur_stream());
  return {c, part};
}

torch::Tensor conv3x3_grouped_fwd(torch::Tensor x, torch::Tensor w3g,
                                  int64_t stride) {
  // grouped 3x3: w3g [Cout, 9*gw] where gw = the gemm group width —
  // channels-per-group when >= 64 (exact, zero wasted MFMA), else 64
  // with a block-diagonal zero-padded repack (conv.py _repack_w3_grouped)
  TORCH_CHECK(x.is_cuda() && x.dim() == 4 &&
                  x.scalar_type() == torch::kBFloat16 &&
                  x.is_contiguous(torch::MemoryFormat::ChannelsLast),
              "conv3x3g: 4-D channels_last bf16");
  const int Nimg = (int)x.size(0), Cin = (int)x.size(1);
  const int H = (int)x.size(2), W = (int)x.size(3);
  const int Cout = (int)w3g.size(0);
  TORCH_CHECK(w3g.size(1) % 9 == 0 && w3g.is_contiguous(), "conv3x3g: w3g");
  const int gw = (int)(w3g.size(1) / 9);
  TORCH_CHECK(gw % 64 == 0 && Cin % gw == 0, "conv3x3g: gw % 64, Cin % gw");
  TORCH_CHECK(Cin % 64 == 0 && Cout % 64 == 0 && Cin == Cout,
              "conv3x3g: C % 64, equal in/out");
  const int Hp = H + 2, Wp = W + 2;
  const int Hout = (H + 2 - 3) / (int)stride + 1;
  const int Wout = (W + 2 - 3) / (int)stride + 1;
  const long long M = (long long)Nimg * Hout * Wout;
  auto s = cur_stream();
  auto xp = torch::empty({(long long)Nimg * Hp * Wp * Cin}, x.options());
  launch_pad_nhwc(x.data_ptr(), xp.data_ptr(), Nimg, H, W, Hp, Wp, Cin, s);
  auto y = torch::empty({M, Cout}, x.options());
  launch_conv3x3_grouped(xp.data_ptr(), w3g.data_ptr(), y.data_ptr(), (int)M,
                         Cout, Cin, Hout * Wout, Wout, Hp, Wp, (int)stride,
                         gw, s);
  return y;
}

torch::Tensor conv3x3_small_fwd(torch::Tensor x, torch::Tensor w3s,
                                int64_t cout_real, int64_t cpt,
                                int64_t stride) {
  // stem conv (small channels): x [N, Cin, H, W] channels_last bf16 with
  // Cin <= cpt (cpt in {16,32,64}); w3s [64, TAPS_PAD*cpt] zero-padded
  // repack (conv.py _repack_w3_small). Returns y2d [M, cout_real].
  TORCH_CHECK(x.is_cuda() && x.dim() == 4 &&
                  x.scalar_type() == torch::kBFloat16 &&
                  x.is_contiguous(torch::MemoryFormat::ChannelsLast),
              "conv3x3s: 4-D channels_last bf16");
  TORCH_CHECK(stride == 1 || stride == 2, "conv3x3s: stride 1 or 2");
  TORCH_CHECK(cpt == 16 || cpt == 32 || cpt == 64, "conv3x3s: cpt 16/32/64");
  const int Nimg = (int)x.size(0), Cin = (int)x.size(1);
  const int H = (int)x.size(2), W = (int)x.size(3);
  TORCH_CHECK(Cin <= cpt, "conv3x3s: Cin <= cpt");
  const int taps_pad = cpt == 16 ? 12 : (cpt == 32 ? 10 : 9);
  TORCH_CHECK(w3s.is_contiguous() && w3s.size(0) == 64 &&
                  w3s.size(1) == taps_pad * cpt,
              "conv3x3s: w3s [64, taps_pad*cpt]");
  TORCH_CHECK(cout_real >= 1 && cout_real <= 64, "conv3x3s: cout <= 64");
  const int Hp = H + 2, Wp = W + 2;
  const int Hout = (H - 1) / (int)stride + 1;
  const int Wout = (W - 1) / (int)stride + 1;
  const long long M = (long long)Nimg * Hout * Wout;
  auto s = cur_stream();
  auto xp = torch::empty({(long long)Nimg * Hp * Wp * cpt}, x.options());
  if (Cin == cpt) {
    launch_pad_nhwc(x.data_ptr(), xp.data_ptr(), Nimg, H, W, Hp, Wp, Cin, s);
  } else {
    launch_pad_nhwc_cpad(x.data_ptr(), xp.data_ptr(), Nimg, H, W, Hp, Wp,
                         Cin, (int)cpt, s);
  }
  auto y = torch::empty({M, cout_real}, x.options());
  launch_conv3x3_small(xp.data_ptr(), w3s.data_ptr(), y.data_ptr(), (int)M,
                       (int)cout_real, (int)cpt, Hout * Wout, Wout, Hp, Wp,
                       (int)stride, nullptr, s);
  return y;
}

std::vector<torch::Tensor> conv3x3_small_fwd_stats(torch::Tensor x,
                                                   torch::Tensor w3s,
                                                   int64_t cout_real,
                                                   int64_t cpt,
                                                   int64_t stride) {
  // conv3x3_small_fwd + BN stats partials [tiles_m, 2*cout_real]
  TORCH_CHECK(x.is_cuda() && x.dim() == 4 &&
                  x.scalar_type() == torch::kBFloat16 &&
                  x.is_contiguous(torch::MemoryFormat::ChannelsLast),
              "conv3x3s: 4-D channels_last bf16");
  TORCH_CHECK(stride == 1 || stride == 2, "conv3x3s: stride 1 or 2");
  TORCH_CHECK(cpt == 16 || cpt == 32 || cpt == 64, "conv3x3s: cpt 16/32/64");
  const int Nimg = (int)x.size(0), Cin = (int)x.size(1);
  const int H = (int)x.size(2), W = (int)x.size(3);
  TORCH_CHECK(Cin <= cpt, "conv3x3s: Cin <= cpt");
  const int taps_pad = cpt == 16 ? 12 : (cpt == 32 ? 10 : 9);
  TORCH_CHECK(w3s.is_contiguous() && w3s.size(0) == 64 &&
                  w3s.size(1) == taps_pad * cpt,
              "conv3x3s: w3s [64, taps_pad*cpt]");
  TORCH_CHECK(cout_real >= 1 && cout_real <= 64, "conv3x3s: cout <= 64");
  const int Hp = H + 2, Wp = W + 2;
  const int Hout = (H - 1) / (int)stride + 1;
  const int Wout = (W - 1) / (int)stride + 1;
  const long long M = (long long)Nimg * Hout * Wout;
  auto s = cur_stream();
  auto xp = torch::empty({(long long)Nimg * Hp * Wp * cpt}, x.options());
  if (Cin == cpt) {
    launch_pad_nhwc(x.data_ptr(), xp.data_ptr(), Nimg, H, W, Hp, Wp, Cin, s);
  } else {
    launch_pad_nhwc_cpad(x.data_ptr(), xp.data_ptr(), Nimg, H, W, Hp, Wp,
                         Cin, (int)cpt, s);
  }
  auto y = torch::empty({M, cout_real}, x.options());
  const int tiles_m = (int)((M + 255) / 256);
  const int rows = std::min(tiles_m, 192);
  auto bpart = tiles_m > rows
                   ? part_pool_get(rows, 2 * cout_real,
                                   x.options().dtype(torch::kFloat32))
                   : torch::empty({rows, 2 * cout_real},
                                  x.options().dtype(torch::kFloat32));
  launch_conv3x3_small(xp.data_ptr(), w3s.data_ptr(), y.data_ptr(), (int)M,
                       (int)cout_real, (int)cpt, Hout * Wout, Wout, Hp, Wp,
                       (int)stride, bpart.data_ptr<float>(), s);
  return {y, bpart};
}

torch::Tensor conv3x3s2_dgrad(torch::Tensor dy, torch::Tensor wcat,
                              int64_t H, int64_t W) {
  // stride-2 3x3 same-pad dgrad (parity decomposition, conv3x3.hip).
  // dy: [N, Cout, Ho, Wo] channels_last bf16; wcat: [Cin, 9*Cout]
  // (conv.py _repack_w3_s2dgrad). Returns dx2d [N*H*W, Cin] bf16.
  TORCH_CHECK(dy.is_cuda() && dy.dim() == 4 &&
                  dy.scalar_type() == torch::kBFloat16 &&
                  dy.is_contiguous(torch::MemoryFormat::ChannelsLast),
              "s2dgrad: 4-D channels_last bf16 dy");
  const int Nimg = (int)dy.size(0), Cout = (int)dy.size(1);
  const int Ho = (int)dy.size(2), Wo = (int)dy.size(3);
  const int Cin = (int)(wcat.size(0));
  TORCH_CHECK(wcat.is_contiguous() && wcat.size(1) == 9 * Cout,
              "s2dgrad: wcat [Cin, 9*Cout]");
  TORCH_CHECK(Cin % 64 == 0 && Cout % 64 == 0, "s2dgrad: C % 64");
  TORCH_CHECK((H + 1) / 2 == Ho && (W + 1) / 2 == Wo, "s2dgrad: shape");
  const int Hop = Ho + 2, Wop = Wo + 2;
  auto s = cur_stream();
  auto dyp = torch::empty({(long long)Nimg * Hop * Wop * Cout}, dy.options());
  launch_pad_nhwc(dy.data_ptr(), dyp.data_ptr(), Nimg, Ho, Wo, Hop, Wop, Cout,
                  s);
  auto dx = torch::empty({(long long)Nimg * H * W, Cin}, dy.options());
  launch_conv3x3s2_dgrad(dyp.data_ptr(), wcat.data_ptr(), dx.data_ptr(), Nimg,
                         (int)H, (int)W, Cin, Cout, Hop, Wop, s);
  return dx;
}

torch::Tensor repack_dgrad_w3(torch::Tensor w_smaj, int64_t ci,
                              int64_t mode) {
  // w_smaj: [Co, 9*Ci] bf16 contiguous (the channels-last mirror's
  // s-major view) -> [Ci, 9*Co]: mode 0 = rotated stride-1 dgrad
  // weights, mode 1 = the stride-2 parity-class wcat. One kernel
  // instead of flip + permute-copy + cast per conv per step.
  TORCH_CHECK(w_smaj.is_cuda() && w_smaj.scalar_type() == torch::kBFloat16 &&
                  w_smaj.is_contiguous() && w_smaj.dim() == 2 &&
                  w_smaj.size(1) == 9 * ci,
              "repack_dgrad_w3: [Co, 9*Ci] bf16");
  const int Co = (int)w_smaj.size(0);
  TORCH_CHECK(Co % 64 == 0 && ci % 64 == 0, "repack_dgrad_w3: C % 64");
  auto out = torch::empty({ci, (long long)9 * Co}, w_smaj.options());
  launch_repack_dgrad_w3(w_smaj.data_ptr(), out.data_ptr(), Co, (int)ci,
                         (int)mode, cur_stream());
  return out;
}

std::vector<torch::Tensor> maxpool3x3s2_fwd(torch::Tensor x) {
  TORCH_CHECK(x.is_cuda() && x.dim() == 4 &&
                  x.scalar_type() == torch::kBFloat16 &&
                  x.is_contiguous(torch::MemoryFormat::ChannelsLast),
              "maxpool3x3s2: 4-D channels_last bf16");
  const int N = (int)x.size(0), C = (int)x.size(1);
  const int H = (int)x.size(2), W = (int)x.size(3);
  TORCH_CHECK(C % 8 == 0, "maxpool3x3s2: C % 8");
  const int Ho = (H + 1) / 2, Wo = (W + 1) / 2;
  auto y = torch::empty({N, C, Ho, Wo},
                        x.options().memory_format(torch::MemoryFormat::ChannelsLast));
  auto idx = torch::empty({(long long)N * Ho * Wo * C},
                          x.options().dtype(torch::kUInt8));
  launch_maxpool3x3s2_fwd(x.data_ptr(), y.data_ptr(),
                          idx.data_ptr<unsigned char>(), N, H, W, Ho, Wo, C,
                          cur_stream());
  return {y, idx};
}

torch::Tensor maxpool3x3s2_bwd(torch::Tensor dy, torch::Tensor idx,
                               int64_t H, int64_t W) {
  TORCH_CHECK(dy.is_cuda() && dy.dim() == 4 &&
                  dy.scalar_type() == torch::kBFloat16 &&
                  dy.is_contiguous(torch::MemoryFormat::ChannelsLast),
              "maxpool3x3s2_bwd: 4-D channels_last bf16");
  const int N = (int)dy.size(0), C = (int)dy.size(1);
  const int Ho = (int)dy.size(2), Wo = (int)dy.size(3);
  auto dx = torch::empty({N, C, (long long)H, (long long)W},
                         dy.options().memory_format(torch::MemoryFormat::ChannelsLast));
  launch_maxpool3x3s2_bwd(dy.data_ptr(), idx.data_ptr<unsigned char>(),
                          dx.data_ptr(), N, (int)H, (int)W, Ho, Wo, C,
                          cur_stream());
  return dx;
}

torch::Tensor avgpool2x2_fwd(torch::Tensor x) {
  // x: 4-D channels_last bf16; 2x2 stride-2 ceil_mode pool
  TORCH_CHECK(x.is_cuda() && x.dim() == 4 &&
                  x.scalar_type() == torch::kBFloat16 &&
                  x.is_contiguous(torch::MemoryFormat::ChannelsLast),
              "avgpool2x2: 4-D channels_last bf16");
  const int N = (int)x.size(0), C = (int)x.size(1);
  const int H = (int)x.size(2), W = (int)x.size(3);
  TORCH_CHECK(C % 8 == 0, "avgpool2x2: C % 8");
  const int Ho = (H + 1) / 2, Wo = (W + 1) / 2;
  auto y = torch::empty({N, C, Ho, Wo},
                        x.options().memory_format(torch::MemoryFormat::ChannelsLast));
  launch_avgpool2x2_fwd(x.data_ptr(), y.data_ptr(), N, H, W, Ho, Wo, C,
                        cur_stream());
  return y;
}

torch::Tensor avgpool2x2_bwd(torch::Tensor dy, int64_t H, int64_t W) {
  const int N = (int)dy.size(0), C = (int)dy.size(1);
  const int Ho = (int)dy.size(2), Wo = (int)dy.size(3);
  auto dyc = dy.contiguous(torch::MemoryFormat::ChannelsLast);
  auto dx = torch::empty({N, C, H, W},
                         dy.options().memory_format(torch::MemoryFormat::ChannelsLast));
  launch_avgpool2x2_bwd(dyc.data_ptr(), dx.data_ptr(), N, (int)H, (int)W, Ho,
                        Wo, C, cur_stream());
  return dx;
}

torch::Tensor conv3x3_wgrad_operand(torch::Tensor x, int64_t stride) {
  // x: 4-D channels_last bf16 input of the conv -> [9*Cin, Mp] shifted
  // transpose (pads internally).
  TORCH_CHECK(x.is_cuda() && x.dim() == 4 &&
                  x.scalar_type() == torch::kBFloat16 &&
                  x.is_contiguous(torch::MemoryFormat::ChannelsLast),
              "wgrad_operand: 4-D channels_last bf16");
  const int Nimg = (int)x.size(0), Cin = (int)x.size(1);
  const int H = (int)x.size(2), W = (int)x.size(3);
  const int Hp = H + 2, Wp = W + 2;
  const int Hout = (H - 1) / (int)stride + 1;
  const int Wout = (W - 1) / (int)stride + 1;
  const int M = Nimg * Hout * Wout;
  const int Mp = (M + 63) / 64 * 64;
  auto s = cur_stream();
  auto xp = torch::empty({(long long)Nimg * Hp * Wp * Cin}, x.options());
  launch_pad_nhwc(x.data_ptr(), xp.data_ptr(), Nimg, H, W, Hp, Wp, Cin, s);
  auto out = torch::empty({(long long)9 * Cin, Mp}, x.options());
  launch_shift9_transpose(xp.data_ptr(), out.data_ptr(), M, Cin, Mp,
                          Hout * Wout, Wout, Hp, Wp, (int)stride, s);
  return out;
}

torch::Tensor gemm_bt_splitk(torch::Tensor a, torch::Tensor b, int64_t splitk) {
  // fp32 C[M,N] = A[M,K] @ B[N,K]^T with grid.y k-slices (atomic combine).
  TORCH_CHECK(a.is_cuda() && b.is_cuda() && a.dim() == 2 && b.dim() == 2 &&
                  a.size(1) == b.size(1),
              "gemm_bt_splitk: [M,K] x [N,K] GPU");
  TORCH_CHECK(a.scalar_type() == torch::kBFloat16 &&
                  b.scalar_type() == torch::kBFloat16,
              "gemm_bt_splitk: bf16 only");
  auto ac = a.contiguous();
  auto bc = b.contiguous();
  const int M = (int)a.size(0), K = (int)a.size(1), N = (int)b.size(0);
  TORCH_CHECK(K % 64 == 0 && N % 64 == 0, "gemm_bt_splitk: K,N % 64");
  if (splitk <= 0) {
    // pick splitk so total blocks ~ 2x CUs, bounded by k-tiles
    const int tiles = ((M + 127) / 128) * ((N + 63) / 64);
    splitk = std::max<int64_t>(1, 512 / std::max(1, tiles));
    splitk = std::min<int64_t>(splitk, K / 64);
  }
  auto c = torch::zeros({M, N}, a.options().dtype(torch::kFloat32));
  launch_gemm_bt_splitk(ac.data_ptr(), bc.data_ptr(), c.data_ptr<float>(), M, N,
                        K, (int)splitk, cur_stream());
  return c;
}

torch::Tensor gemm_tn_splitk(torch::Tensor a, torch::Tensor b, int64_t splitk,
                             c10::optional<torch::Tensor> out) {
  // fp32 C[N1,N2] = A[K,N1]^T @ B[K,N2] — conv wgrad WITHOUT the
  // transpose_pad materializations (operands in native activation layout)
  TORCH_CHECK(a.is_cuda() && b.is_cuda() && a.dim() == 2 && b.dim() == 2 &&
                  a.size(0) == b.size(0),
              "gemm_tn_splitk: [K,N1] x [K,N2] GPU");
  TORCH_CHECK(a.scalar_type() == torch::kBFloat16 &&
                  b.scalar_type() == torch::kBFloat16,
              "gemm_tn_splitk: bf16 only");
  auto ac = a.contiguous();
  auto bc = b.contiguous();
  const int K = (int)a.size(0), N1 = (int)a.size(1), N2 = (int)b.size(1);
  TORCH_CHECK(N1 % 64 == 0 && N2 % 64 == 0, "gemm_tn_splitk: N1,N2 % 64");
  const int nchunks = (K + 63) / 64;
  if (splitk <= 0) {
    const int tiles = (N1 / (N1 % 128 == 0 ? 128 : 64)) *
                      (N2 / (N2 % 128 == 0 ? 128 : 64));
    splitk = std::max<int64_t>(1, 512 / std::max(1, tiles));
  }
  splitk = std::min<int64_t>(splitk, nchunks);
  torch::Tensor c;
  if (out.has_value()) {
    // direct-grad mode: ACCUMULATE into the caller's fp32 buffer (the
    // param's bucket-view gradient, pre-zeroed by reducer.zero_grad) —
    // no fresh zeros tensor, no AccumulateGrad add afterwards
    c = *out;
    TORCH_CHECK(c.is_contiguous() && c.scalar_type() == torch::kFloat32 &&
                    c.numel() == (int64_t)N1 * N2,
                "gemm_tn_splitk: out fp32 contiguous [N1,N2]");
  } else {
    c = torch::zeros({N1, N2}, a.options().dtype(torch::kFloat32));
  }
  launch_gemm_tn_splitk(ac.data_ptr(), bc.data_ptr(), c.data_ptr<float>(), N1,
                        N2, K, (int)splitk, cur_stream());
  return c;
}

torch::Tensor gemm_tn3x3_splitk(torch::Tensor dy2d, torch::Tensor x,
                                int64_t stride, int64_t splitk,
                                c10::optional<torch::Tensor> out) {
  // conv3x3 wgrad, fully direct: fp32 dW3[Cout, 9*Cin] = dY^T @
  // gather3x3(pad(x)) — no transpose_pad / shift9 materializations.
  // dy2d: [M, Cout] bf16 (NHWC-flattened dY); x: 4-D channels_last bf16.
  TORCH_CHECK(dy2d.is_cuda() && dy2d.dim() == 2 &&
                  dy2d.scalar_type() == torch::kBFloat16,
              "gemm_tn3x3: dy2d [M,Cout] bf16");
  TORCH_CHECK(x.is_cuda() && x.dim() == 4 &&
                  x.scalar_type() == torch::kBFloat16 &&
                  x.is_contiguous(torch::MemoryFormat::ChannelsLast),
              "gemm_tn3x3: x 4-D channels_last bf16");
  const int Nimg = (int)x.size(0), Cin = (int)x.size(1);
  const int H = (int)x.size(2), W = (int)x.size(3);
  const int Hp = H + 2, Wp = W + 2;
  const int Ho = (H - 1) / (int)stride + 1, Wo = (W - 1) / (int)stride + 1;
  const int M = Nimg * Ho * Wo;
  const int Cout = (int)dy2d.size(1);
  TORCH_CHECK((int)dy2d.size(0) == M, "gemm_tn3x3: dy2d rows != N*Ho*Wo");
  TORCH_CHECK(Cin % 64 == 0 && Cout % 64 == 0, "gemm_tn3x3: C % 64");
  auto dyc = dy2d.contiguous();
  auto s = cur_stream();
  auto xp = torch::empty({(long long)Nimg * Hp * Wp * Cin}, x.options());
  launch_pad_nhwc(x.data_ptr(), xp.data_ptr(), Nimg, H, W, Hp, Wp, Cin, s);
  const int nchunks = (M + 63) / 64;
  if (splitk <= 0) {
    const int tiles = (Cout / (Cout % 128 == 0 ? 128 : 64)) *
                      (9 * Cin / (Cin % 128 == 0 ? 128 : 64));
    splitk = std::max<int64_t>(1, 512 / std::max(1, tiles));
  }
  splitk = std::min<int64_t>(splitk, nchunks);
  torch::Tensor c;
  int perm = 0;
  if (out.has_value()) {
    // direct-grad mode: accumulate straight into the conv weight's
    // bucket-view gradient. A 2-D [Cout, 9*Cin] out (channels-last
    // bucket storage) takes the kernel's NATIVE s-major epilogue —
    // coalesced; a [Cout, Cin, 3, 3] out takes the perm remap
    // (stride-9 scatter, kept for the standard layout).
    c = *out;
    TORCH_CHECK(c.is_contiguous() && c.scalar_type() == torch::kFloat32 &&
                    c.numel() == (int64_t)Cout * Cin * 9,
                "gemm_tn3x3_splitk: out fp32 contiguous");
    perm = (c.dim() == 2 && c.size(1) == 9 * Cin) ? 0 : Cin;
  } else {
    c = torch::zeros({Cout, 9 * Cin}, x.options().dtype(torch::kFloat32));
  }
  launch_gemm_tn3x3_splitk(dyc.data_ptr(), xp.data_ptr(), c.data_ptr<float>(),
                           Cout, Cin, M, Ho, Wo, Hp, Wp, (int)stride,
                           (int)splitk, perm, s);
  return c;
}

torch::Tensor transpose_pad(torch::Tensor x) {
  TORCH_CHECK(x.is_cuda() && x.dim() == 2 &&
                  x.scalar_type() == torch::kBFloat16 && x.is_contiguous(),
              "transpose_pad: contiguous 2-D bf16 GPU tensor");
  const int M = (int)x.size(0), C = (int)x.size(1);
  TORCH_CHECK(C % 8 == 0, "transpose_pad: C % 8");
  const int Mp = (M + 63) / 64 * 64;
  auto y = torch::empty({C, Mp}, x.options());
  launch_transpose_pad(x.data_ptr(), y.data_ptr(), M, C, Mp, cur_stream());
  return y;
}

torch::Tensor conv3x3_fwd(torch::Tensor x, torch::Tensor w3, int64_t stride) {
  // x: 4-D channels_last bf16 [N, C, H, W]; w3: [Cout, 9*Cin] bf16
  TORCH_CHECK(x.is_cuda() && x.dim() == 4 &&
                  x.scalar_type() == torch::kBFloat16,
              "conv3x3: 4-D bf16 GPU tensor required");
  TORCH_CHECK(x.is_contiguous(torch::MemoryFormat::ChannelsLast),
              "conv3x3: channels_last required");
  TORCH_CHECK(stride == 1 || stride == 2, "conv3x3: stride 1 or 2");
  const int Nimg = (int)x.size(0), Cin = (int)x.size(1);
  const int H = (int)x.size(2), W = (int)x.size(3);
  const int Cout = (int)w3.size(0);
  TORCH_CHECK(w3.size(1) == 9 * Cin && w3.is_contiguous(), "conv3x3: w3 shape");
  TORCH_CHECK(Cin % 64 == 0 && Cout % 64 == 0, "conv3x3: C % 64");
  const int Hp = H + 2, Wp = W + 2;
  const int Hout = (H + 2 - 3) / (int)stride + 1;
  const int Wout = (W + 2 - 3) / (int)stride + 1;
  const long long M = (long long)Nimg * Hout * Wout;
  auto s = cur_stream();
  auto xp = torch::empty({(long long)Nimg * Hp * Wp * Cin}, x.options());
  launch_pad_nhwc(x.data_ptr(), xp.data_ptr(), Nimg, H, W, Hp, Wp, Cin, s);
  const int splitk = conv3x3_pick_splitk((int)M, Cout, Cin);
  if (splitk > 1) {
    // CU-starved deep-K shape: fp32 split-K partials + cast
    // pooled pre-zeroed fold target; cast_bf16_zero returns it clean
    auto part = part_pool_get((int)M, Cout, x.options().dtype(torch::kFloat32));
    launch_conv3x3(xp.data_ptr(), w3.data_ptr(), nullptr, (int)M, Cout, Cin,
                   Hout * Wout, Wout, Hp, Wp, (int)stride,
                   part.data_ptr<float>(), splitk, nullptr, s);
    auto y = torch::empty({M, Cout}, x.options());
    launch_cast_bf16_zero(part.data_ptr<float>(), y.data_ptr(),
                          (long long)M * Cout, s);
    return y;
  }
  auto y = torch::empty({M, Cout}, x.options());
  launch_conv3x3(xp.data_ptr(), w3.data_ptr(), y.data_ptr(), (int)M, Cout, Cin,
                 Hout * Wout, Wout, Hp, Wp, (int)stride, nullptr, 1, nullptr,
                 s);
  return y;
}

std::vector<torch::Tensor> conv3x3_fwd_stats(torch::Tensor x, torch::Tensor w3,
                                             int64_t stride) {
  // conv3x3_fwd + BN stats partials of the output ([tiles_m, 2*Cout]).
  // Split-K shapes can't fold stats (sumsq is nonlinear over partial
  // sums) — the second return is then an undefined tensor (Python None).
  TORCH_CHECK(x.is_cuda() && x.dim() == 4 &&
                  x.scalar_type() == torch::kBFloat16,
              "conv3x3: 4-D bf16 GPU tensor required");
  TORCH_CHECK(x.is_contiguous(torch::MemoryFormat::ChannelsLast),
              "conv3x3: channels_last required");
  TORCH_CHECK(stride == 1 || stride == 2, "conv3x3: stride 1 or 2");
  const int Nimg = (int)x.size(0), Cin = (int)x.size(1);
  const int H = (int)x.size(2), W = (int)x.size(3);
  const int Cout = (int)w3.size(0);
  TORCH_CHECK(w3.size(1) == 9 * Cin && w3.is_contiguous(), "conv3x3: w3 shape");
  TORCH_CHECK(Cin % 64 == 0 && Cout % 64 == 0, "conv3x3: C % 64");
  const int Hp = H + 2, Wp = W + 2;
  const int Hout = (H + 2 - 3) / (int)stride + 1;
  const int Wout = (W + 2 - 3) / (int)stride + 1;
  const long long M = (long long)Nimg * Hout * Wout;
  auto s = cur_stream();
  auto xp = torch::empty({(long long)Nimg * Hp * Wp * Cin}, x.options());
  launch_pad_nhwc(x.data_ptr(), xp.data_ptr(), Nimg, H, W, Hp, Wp, Cin, s);
  const int splitk = conv3x3_pick_splitk((int)M, Cout, Cin);
  if (splitk > 1) {
    auto part = part_pool_get((int)M, Cout, x.options().dtype(torch::kFloat32));
    launch_conv3x3(xp.data_ptr(), w3.data_ptr(), nullptr, (int)M, Cout, Cin,
                   Hout * Wout, Wout, Hp, Wp, (int)stride,
                   part.data_ptr<float>(), splitk, nullptr, s);
    auto y = torch::empty({M, Cout}, x.options());
    launch_cast_bf16_zero(part.data_ptr<float>(), y.data_ptr(),
                          (long long)M * Cout, s);
    return {y, torch::Tensor()};
  }
  auto y = torch::empty({M, Cout}, x.options());
  const int tiles_m = conv3x3_tiles_m((int)M, Cout);
  const int rows = std::min(tiles_m, 192);
  auto bpart = tiles_m > rows
                   ? part_pool_get(rows, 2 * Cout,
                                   x.options().dtype(torch::kFloat32))
                   : torch::empty({rows, 2 * Cout},
                                  x.options().dtype(torch::kFloat32));
  launch_conv3x3(xp.data_ptr(), w3.data_ptr(), y.data_ptr(), (int)M, Cout, Cin,
                 Hout * Wout, Wout, Hp, Wp, (int)stride, nullptr, 1,
                 bpart.data_ptr<float>(), s);
  return {y, bpart};
}

torch::Tensor gemm_tn3x3_small(torch::Tensor dy2d, torch::Tensor x,
                               int64_t stride, int64_t splitk) {
  // Deep-stem conv wgrad, fully in-repo (closes VERDICT r1 #4's last
  // MIOpen rows): fp32 dW[64, N2v] = dY^T @ gather3x3(cpad(x)) via the
  // G3S small-Cin gather (gemm_tn.hip). dy2d: [M, 64] bf16 with the real
  // Cout columns first (caller zero-pads); x: 4-D channels_last bf16,
  // Cin <= 64. Returns [64, N2v]; caller slices [:Cout, :9*CinP] and
  // re-lays to [Cout, Cin, 3, 3].
  TORCH_CHECK(x.is_cuda() && x.dim() == 4 &&
                  x.scalar_type() == torch::kBFloat16 &&
                  x.is_contiguous(torch::MemoryFormat::ChannelsLast),
              "tn3x3s: x 4-D channels_last bf16");
  TORCH_CHECK(stride == 1 || stride == 2, "tn3x3s: stride 1 or 2");
  const int Nimg = (int)x.size(0), Cin = (int)x.size(1);
  const int H = (int)x.size(2), W = (int)x.size(3);
  TORCH_CHECK(Cin >= 1 && Cin <= 64, "tn3x3s: Cin <= 64");
  const int Hp = H + 2, Wp = W + 2;
  const int Hout = (H - 1) / (int)stride + 1;
  const int Wout = (W - 1) / (int)stride + 1;
  const long long M = (long long)Nimg * Hout * Wout;
  TORCH_CHECK(dy2d.is_cuda() && dy2d.is_contiguous() && dy2d.dim() == 2 &&
                  dy2d.scalar_type() == torch::kBFloat16 &&
                  dy2d.size(0) == M && dy2d.size(1) == 64,
              "tn3x3s: dy2d [M, 64] bf16");
  int CinP = 8;
  while (CinP < Cin) CinP <<= 1;
  const int N2 = 9 * CinP;
  const int N2v = (N2 + 63) / 64 * 64;
  auto s = cur_stream();
  auto xp = torch::empty({(long long)Nimg * Hp * Wp * CinP}, x.options());
  if (Cin == CinP) {
    launch_pad_nhwc(x.data_ptr(), xp.data_ptr(), Nimg, H, W, Hp, Wp, Cin, s);
  } else {
    launch_pad_nhwc_cpad(x.data_ptr(), xp.data_ptr(), Nimg, H, W, Hp, Wp,
                         Cin, CinP, s);
  }
  if (splitk <= 0) {
    const int tiles = N2v / 64;
    splitk = std::max<int64_t>(1, 512 / std::max(1, tiles));
    const int nch = (int)((M + 63) / 64);
    splitk = std::min<int64_t>(splitk, std::max(1, nch));
  }
  auto c = torch::zeros({64, N2v}, x.options().dtype(torch::kFloat32));
  launch_gemm_tn3x3_small(dy2d.data_ptr(), xp.data_ptr(), c.data_ptr<float>(),
                          64, N2v, (int)M, Hout, Wout, Hp, Wp, CinP,
                          (int)stride, (int)splitk, s);
  return c;
}

}  // namespace

PYBIND11_MODULE(TORCH_EXTENSION_NAME, m) {
  m.def("fused_sgd", &fused_sgd,
        "fused flat momentum-SGD update (p,g,m flat fp32; folds grad_scale)",
        py::arg("p"), py::arg("g"), py::arg("m"), py::arg("lr"),
        py::arg("momentum"), py::arg("weight_decay"), py::arg("grad_scale"),
        py::arg("lr_dev") = py::none());
  m.def("kd_ce_forward", &kd_ce_forward, "KD soft-label CE forward -> per-row loss");
  m.def("kd_ce_backward", &kd_ce_backward, "KD soft-label CE backward -> dlogits");
  m.def("bn_fwd_train", &bn_fwd_train,
        "fused NHWC bf16 BN(+add)+ReLU train fwd -> (y, mean, invstd, mask)",
        py::arg("x"), py::arg("gamma"), py::arg("beta"), py::arg("rmean"),
        py::arg("rvar"), py::arg("momentum"), py::arg("eps"), py::arg("res"),
        py::arg("relu"), py::arg("pre_part") = py::none());
  m.def("bn_fwd_eval", &bn_fwd_eval, "fused NHWC bf16 BN(+add)+ReLU eval fwd");
  m.def("bn_bwd", &bn_bwd,
        "fused BN(+add)+ReLU bwd -> (dx, dgamma, dbeta, dres?)",
        pybind11::arg("dy"), pybind11::arg("msk"), pybind11::arg("x"),
        pybind11::arg("mean"), pybind11::arg("invstd"), pybind11::arg("gamma"),
        pybind11::arg("relu"), pybind11::arg("add"), pybind11::arg("training"),
        pybind11::arg("dgamma_acc") = pybind11::none(),
        pybind11::arg("dbeta_acc") = pybind11::none());
  m.def("gemm_bt", &gemm_bt, "bf16 MFMA GEMM: C[M,N] = A[M,K] @ B[N,K]^T");
  m.def("conv3x3_fwd", &conv3x3_fwd,
        "implicit-GEMM 3x3 same-pad conv (stride 1/2) -> y2d [M, Cout]");
  m.def("conv3x3_fwd_stats", &conv3x3_fwd_stats,
        "conv3x3_fwd + BN stats partials -> (y2d, part-or-None)");
  m.def("gemm_bt_stats", &gemm_bt_stats,
        "gemm_bt + BN stats partials -> (C, part)");
  m.def("transpose_pad", &transpose_pad,
        "bf16 [M,C] -> [C, ceil64(M)] transpose with zero pad");
  m.def("gemm_bt_splitk", &gemm_bt_splitk,
        "split-K bt GEMM -> fp32 C (wgrad shapes)", pybind11::arg("a"),
        pybind11::arg("b"), pybind11::arg("splitk") = 0);
  m.def("gemm_tn3x3_splitk", &gemm_tn3x3_splitk,
        "direct conv3x3 wgrad: fp32 dW3[Cout,9Cin] = dY^T @ gather3x3(pad(x))",
        pybind11::arg("dy2d"), pybind11::arg("x"), pybind11::arg("stride"),
        pybind11::arg("splitk") = 0, pybind11::arg("out") = pybind11::none());
  m.def("gemm_tn3x3_small", &gemm_tn3x3_small, py::arg("dy2d"),
        py::arg("x"), py::arg("stride"), py::arg("splitk") = 0,
        "stem wgrad: dY^T @ gather3x3(cpad(x)) -> fp32 [64, N2v] (G3S)");
  m.def("gemm_tn_splitk", &gemm_tn_splitk,
        "split-K TN GEMM: fp32 C[N1,N2] = A[K,N1]^T @ B[K,N2] (direct wgrad)",
        pybind11::arg("a"), pybind11::arg("b"), pybind11::arg("splitk") = 0,
        pybind11::arg("out") = pybind11::none());
  m.def("conv3x3_wgrad_operand", &conv3x3_wgrad_operand,
        "padded shifted transpose of conv3x3 input -> [9*Cin, Mp]");
  m.def("avgpool2x2_fwd", &avgpool2x2_fwd, "2x2/s2 ceil avg pool (NHWC bf16)");
  m.def("conv3x3_grouped_fwd", &conv3x3_grouped_fwd,
        "grouped (16ch/group) 3x3 conv fwd -> y2d [M, Cout]");
  m.def("conv3x3s2_dgrad", &conv3x3s2_dgrad,
        "stride-2 3x3 same-pad dgrad (parity implicit GEMM) -> dx2d [N*H*W, Cin]");
  m.def("conv3x3_small_fwd", &conv3x3_small_fwd,
        "small-channel 3x3 conv (deep stem) -> y2d [M, cout_real]");
  m.def("conv3x3_small_fwd_stats", &conv3x3_small_fwd_stats,
        "stem conv + BN stats partials -> (y2d, part)");
  m.def("repack_dgrad_w3", &repack_dgrad_w3,
        "dgrad weight repack [Co,9Ci]->[Ci,9Co] (mode 0 rot / 1 s2 order)");
  m.def("maxpool3x3s2_fwd", &maxpool3x3s2_fwd,
        "3x3/s2/p1 max pool NHWC bf16 -> (y, tap idx)");
  m.def("maxpool3x3s2_bwd", &maxpool3x3s2_bwd,
        "3x3/s2/p1 max pool backward (bounded gather via tap idx)");
  m.def("avgpool2x2_bwd", &avgpool2x2_bwd, "2x2/s2 ceil avg pool backward");
  m.attr("_arch") = "gfx950";
}
