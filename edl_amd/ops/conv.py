"""Convolution dispatch for MI355X.

1x1 convolutions are plain GEMMs: on NHWC input, conv1x1 is
y[M, Cout] = x[M, Cin] @ W^T — routed through torch.matmul (hipBLASLt on
ROCm), which runs far closer to the MFMA roofline than MIOpen's igemm
path at these shapes (igemm measured ~2-4% of bf16 peak at bs32; rocprof
gpurun_out/prof3). Other shapes stay on F.conv2d (MIOpen) until the
hand-written implicit-GEMM kernels land.

Weight layout stays nn.Conv2d-compatible ([Cout, Cin, kh, kw]) so state
dicts interchange."""
import torch
import torch.nn as nn
import torch.nn.functional as F


import os

from . import available, ext

# 1x1 path selector: "hip" (our MFMA gemm_bt kernel), "matmul"
# (hipBLASLt/rocBLAS via torch.matmul), "miopen" (F.conv2d).
_CONV1X1 = os.environ.get("EDL_CONV1X1", "hip")

# wgrad path: "tn" = direct TN split-K kernel on the native [M, C]
# activation layout (no transpose_pad materializations); "bt" = the
# transpose_pad + gemm_bt_splitk pipeline (A/B fallback).
_WGRAD = os.environ.get("EDL_WGRAD", "tn")
# 3x3 gather wgrad wins where the pixel dim is large relative to C
# (stage 1/2: 1.04-1.48x) but trails the materialized pipeline on the
# deep small-HW shapes (Cin>=256: 0.90-0.96x, gemm_bench --wgrad3) — the
# per-row div/mod gather addressing amortizes worse at small M. Route by
# Cin; override with EDL_WGRAD3_TN_MAXC (0 = never, 9999 = always).
_WGRAD3_MAXC = int(os.environ.get("EDL_WGRAD3_TN_MAXC", "128"))


class _Conv1x1Hip(torch.autograd.Function):
    """y2d = x2d @ W^T via the gemm_bt MFMA kernel; dgrad reuses the same
    kernel on a (per-step cached) transposed-weight copy; wgrad runs the
    direct TN split-K kernel on the native [M, C] operands (gemm_tn.hip),
    optionally accumulating straight into the bucket-view grad."""

    @staticmethod
    def forward(ctx, x2d, w_param, w_bf16, wt_cached, grad_tgt,
                part_out=None):
        # w_param: the fp32 (or bf16) parameter view — the DIFFERENTIABLE
        # input; w_bf16/wt_cached: per-step cached compute copies. Grads
        # come back in fp32 straight from the split-K kernel: no cast
        # nodes on the weight path at all. grad_tgt (direct-grad mode,
        # world 1): the weight's bucket-view .grad — backward accumulates
        # into it in-kernel and returns None, skipping AccumulateGrad.
        ctx.save_for_backward(x2d)
        ctx.wt = wt_cached  # [Cin, Cout] bf16, derived per weight epoch
        ctx.w_dtype = w_param.dtype
        ctx.grad_tgt = grad_tgt
        if part_out is not None:
            y2d, part = ext().gemm_bt_stats(x2d, w_bf16)
            part_out.append(part)
            return y2d
        return ext().gemm_bt(x2d, w_bf16)

    @staticmethod
    def backward(ctx, dy2d):
        (x2d,) = ctx.saved_tensors
        dy2d = dy2d.contiguous()
        e = ext()
        dx = e.gemm_bt(dy2d, ctx.wt)
        # wgrad: TN with reduction over huge M. Default: the direct TN
        # split-K kernel reading dY/X in their native [M, C] layout (the
        # "transpose" happens in the LDS fragment reads — gemm_tn.hip).
        # Fallback "bt": transpose-pad both operands and run the split-K
        # bt kernel; that path costs an extra HBM round-trip per operand
        # (~1.3 ms/step, profiles/r01_step5_final.txt). hipBLASLt's TN
        # picks measured 272 us on these shapes, a non-split bt kernel
        # serialized on 1-2 blocks (24 ms/step end-to-end).
        if _WGRAD == "tn":
            if ctx.grad_tgt is not None:
                e.gemm_tn_splitk(
                    dy2d, x2d, 0,
                    ctx.grad_tgt.view(ctx.grad_tgt.shape[0], -1))
                return dx, None, None, None, None, None
            dw = e.gemm_tn_splitk(dy2d, x2d, 0)
        else:
            dw = e.gemm_bt_splitk(e.transpose_pad(dy2d), e.transpose_pad(x2d), 0)
        if dw.dtype != ctx.w_dtype:
            dw = dw.to(ctx.w_dtype)
        return dx, dw, None, None, None, None


_CONV3X3 = os.environ.get("EDL_CONV3X3", "hip")

# Grouped-conv routing: groups with >= this many channels-per-group run
# the in-repo grouped kernel (teacher fwd); below it, MIOpen. Measured
# r2 (gpurun_out/r2c3/teacher_fwd_b16.log): routing EVERYTHING (16) wins
# — ResNeXt101 teacher fwd 13.4 -> 5.5 ms/bs16 (2.4x vs all-MIOpen);
# even the 4x-zero cpg=16 block-diagonal beats MIOpen once the rest of
# the net runs in-repo kernels.
_GROUPED_MINC = int(os.environ.get("EDL_CONV3X3_GROUPED_MINC", "16"))

# Fold BN stats partials into the conv fwd epilogues (the following BN
# skips its stats kernel — a full activation re-read, ~0.53 ms/step).
_BN_STATS_FUSED = os.environ.get("EDL_BN_STATS_FUSED", "1") == "1"

# Weight-derived tensors (bf16 casts, transposed/repacked layouts) are
# immutable within one optimizer step; FusedSGD.step() bumps this epoch and
# Conv2dFast caches per-epoch, removing ~100 small cast/copy kernels per
# step (profiles/r01_step4).
_weight_epoch = 0


def bump_weight_epoch():
    global _weight_epoch
    _weight_epoch += 1


def _repack_w3(weight):
    """[Cout, Cin, 3, 3] -> s-major [Cout, 9*Cin] bf16 contiguous."""
    co, ci = weight.shape[0], weight.shape[1]
    return weight.permute(0, 2, 3, 1).reshape(co, 9 * ci).to(
        torch.bfloat16).contiguous()


class _Conv3x3Hip(torch.autograd.Function):
    """3x3 same-pad conv on the implicit-GEMM kernel.

    dgrad (stride 1) is ANOTHER 3x3 stride-1 conv with rotated/transposed
    weights on the same kernel; stride-2 dgrad goes through torch.nn.grad
    (MIOpen); wgrad runs the direct TN gather kernel (Cin<=128) or the
    shift9+split-K pipeline (deeper layers)."""

    @staticmethod
    def forward(ctx, x, w_param, w_bf16, w3_cached, w3rot_cached, stride,
                grad_tgt, part_out=None):
        ctx.save_for_backward(x, w_bf16)
        ctx.stride = stride
        ctx.w3rot = w3rot_cached  # s1: rotated fwd repack; s2: s2dgrad wcat
        ctx.w_dtype = w_param.dtype
        ctx.grad_tgt = grad_tgt
        if part_out is not None:
            y2d, part = ext().conv3x3_fwd_stats(x, w3_cached, stride)
            if part is not None:  # split-K shapes return None
                part_out.append(part)
        else:
            y2d = ext().conv3x3_fwd(x, w3_cached, stride)
        n, _, h, w = x.shape
        ho = (h - 1) // stride + 1
        wo = (w - 1) // stride + 1
        co = w_param.shape[0]
        return y2d.view(n, ho, wo, co).permute(0, 3, 1, 2)

    @staticmethod
    def backward(ctx, dy):
        x, weight = ctx.saved_tensors
        stride = ctx.stride
        dy = dy.contiguous(memory_format=torch.channels_last)
        n, _, h, w = x.shape
        if stride == 1:
            dx2d = ext().conv3x3_fwd(dy.to(torch.bfloat16), ctx.w3rot, 1)
            dx = dx2d.view(n, h, w, x.shape[1]).permute(0, 3, 1, 2)
        elif ctx.w3rot is not None:
            # stride 2: parity-decomposed implicit-GEMM dgrad (4 class
            # launches, exact work) — replaces MIOpen's igemm_bwd
            dx2d = ext().conv3x3s2_dgrad(dy.to(torch.bfloat16), ctx.w3rot, h, w)
            dx = dx2d.view(n, h, w, x.shape[1]).permute(0, 3, 1, 2)
        else:
            dx = torch.nn.grad.conv2d_input(
                list(x.shape), weight, dy, stride=(stride, stride),
                padding=(1, 1))
        # wgrad: dW3[Cout, 9Cin] = dY^T @ shift9(x). Default: the direct
        # TN gather kernel reading dY and pad(x) in native NHWC layout
        # (gemm_tn.hip G3 path); fallback "bt": materialize
        # transpose_pad(dy) [Cout, Mp] + shift9_transpose(x) [9Cin, Mp]
        # and run the split-K bt kernel.
        e = ext()
        n = dy.shape[0]
        co = weight.shape[0]
        ci = weight.shape[1]
        dy2d = dy.permute(0, 2, 3, 1).reshape(-1, co)
        if _WGRAD == "tn" and ci <= _WGRAD3_MAXC:
            if ctx.grad_tgt is not None:
                # accumulate straight into the bucket-view grad, laid out
                # [Cout, Cin, 3, 3] (kernel epilogue remaps)
                e.gemm_tn3x3_splitk(dy2d.to(torch.bfloat16), x, stride, 0,
                                    ctx.grad_tgt)
                return dx, None, None, None, None, None, None, None
            dw3 = e.gemm_tn3x3_splitk(dy2d.to(torch.bfloat16), x, stride, 0)
        else:
            dw3 = e.gemm_bt_splitk(
                e.transpose_pad(dy2d.to(torch.bfloat16)),
                e.conv3x3_wgrad_operand(x, stride), 0)
        dw = dw3.view(co, 3, 3, ci).permute(0, 3, 1, 2)  # fp32
        if dw.dtype != ctx.w_dtype:
            dw = dw.to(ctx.w_dtype)
        return dx, dw, None, None, None, None, None, None


def _small_cpt(cin):
    """gemm channel width for the small (stem) kernel."""
    if cin <= 16:
        return 16
    if cin <= 32:
        return 32
    return 64


def _repack_w3_small(weight, cpt):
    """[Cout<=64, Cin<=cpt, 3, 3] -> [64, TAPS_PAD*cpt] bf16 for the stem
    kernel: rows padded to 64 (guarded store), taps padded (12/10/9 for
    cpt 16/32/64 — the pad taps' weights are zero, the A-gather re-reads
    tap 0 there), channels padded to cpt."""
    co, ci = weight.shape[0], weight.shape[1]
    taps_pad = {16: 12, 32: 10, 64: 9}[cpt]
    out = torch.zeros(64, taps_pad, cpt, dtype=torch.bfloat16,
                      device=weight.device)
    out[:co, :9, :ci] = (weight.detach().permute(0, 2, 3, 1)
                         .reshape(co, 9, ci).to(torch.bfloat16))
    return out.reshape(64, taps_pad * cpt).contiguous()


class _Conv3x3SmallHip(torch.autograd.Function):
    """Deep-stem 3x3 conv (Cin in {3..64}, Cout <= 64) on the small-channel
    implicit-GEMM kernel: fwd + dgrad in-repo (replaces MIOpen's stem
    igemm/naive kernels and their find phase — VERDICT r1 #4); wgrad runs
    as ONE plain library GEMM on an unfold im2col (hipBLASLt — a plain
    GEMM per the north star's library-GEMM allowance)."""

    @staticmethod
    def forward(ctx, x, w_param, w3s, w3srot, stride, cout, part_out=None):
        ctx.save_for_backward(x)
        ctx.stride = stride
        ctx.w3srot = w3srot  # None when dx is not needed (stem conv0)
        ctx.w_dtype = w_param.dtype
        ctx.w_shape = tuple(w_param.shape)
        cpt = _small_cpt(x.shape[1])
        if part_out is not None:
            y2d, part = ext().conv3x3_small_fwd_stats(x, w3s, cout, cpt,
                                                      stride)
            part_out.append(part)
        else:
            y2d = ext().conv3x3_small_fwd(x, w3s, cout, cpt, stride)
        n, _, h, w = x.shape
        ho = (h - 1) // stride + 1
        wo = (w - 1) // stride + 1
        return y2d.view(n, ho, wo, cout).permute(0, 3, 1, 2)

    @staticmethod
    def backward(ctx, dy):
        (x,) = ctx.saved_tensors
        co, ci = ctx.w_shape[0], ctx.w_shape[1]
        dy = dy.contiguous(memory_format=torch.channels_last)
        dyb = dy.to(torch.bfloat16)
        dx = None
        if ctx.needs_input_grad[0] and ctx.w3srot is not None:
            # dgrad = stride-1 small conv of dy with rotated/transposed
            # weights (stem convs are stride 1 wherever dx is needed)
            n, _, h, w = x.shape
            cpt = _small_cpt(co)
            dx2d = ext().conv3x3_small_fwd(dyb, ctx.w3srot, ci, cpt, 1)
            dx = dx2d.view(n, h, w, ci).permute(0, 3, 1, 2)
        # wgrad. Default: torch.nn.grad.conv2d_weight (MIOpen igemm_wrw,
        # ~51 us/call steady). The unfold+hipBLASLt route is kept opt-in
        # (EDL_STEM_WGRAD=unfold) but MEASURED NEGATIVE: hipBLASLt runs
        # the [32, 401k] @ [401k, 288] reduction GEMM at ~918 us — no
        # split-K pick at tiny M,N (gpurun_out/r2c7, bench 3221 -> 2566).
        route = os.environ.get("EDL_STEM_WGRAD", "miopen")
        if route == "tn":
            # in-repo G3S gather wgrad (gemm_tn3x3_small): dY cols padded
            # to 64, x channel-padded to pow2 inside the binding; output
            # [64, N2v] sliced back to [co, ci, 3, 3]
            dy2d = dyb.permute(0, 2, 3, 1).reshape(-1, co)
            if co < 64:
                dy2d = F.pad(dy2d, (0, 64 - co))
            cfull = ext().gemm_tn3x3_small(dy2d.contiguous(), x, ctx.stride)
            cinp = max(8, 1 << max(0, (ci - 1).bit_length()))
            dw = (cfull[:co, :9 * cinp].view(co, 3, 3, cinp)
                  .permute(0, 3, 1, 2)[:, :ci].contiguous())
        elif route == "unfold":
            dy2d = dyb.permute(0, 2, 3, 1).reshape(-1, co)
            xu = F.unfold(x, 3, padding=1, stride=ctx.stride)
            xu2d = xu.permute(0, 2, 1).reshape(-1, ci * 9)  # [M, ci*9]
            dw = (dy2d.t() @ xu2d).float().view(co, ci, 3, 3)
        else:
            dw = torch.nn.grad.conv2d_weight(
                x, ctx.w_shape, dyb, stride=(ctx.stride, ctx.stride),
                padding=(1, 1))
        if dw.dtype != ctx.w_dtype:
            dw = dw.to(ctx.w_dtype)
        return dx, dw, None, None, None, None, None


_S2D_TAP_ORDER = [(1, 1), (1, 0), (1, 2), (0, 1), (2, 1),
                  (0, 0), (0, 2), (2, 0), (2, 2)]


def _repack_w3_s2dgrad(weight):
    """[Cout, Cin, 3, 3] -> wcat [Cin, 9*Cout] bf16 for the stride-2 dgrad
    parity kernel (conv3x3.hip conv3x3s2_dgrad_kernel): parity classes'
    tap slabs contiguous — class (h&1,w&1) uses taps with dy=(h+1)&1,
    dxx=(w+1)&1; column bases [0, Co, 3Co, 5Co]."""
    co, ci = weight.shape[0], weight.shape[1]
    w = weight.detach()
    slabs = torch.stack([w[:, :, dy, dx] for dy, dx in _S2D_TAP_ORDER])
    return (slabs.permute(2, 0, 1).reshape(ci, 9 * co)
            .to(torch.bfloat16).contiguous())


def _repack_w3_grouped(weight):
    """Grouped 3x3 weight [Cout, cpg, 3, 3] -> s-major [Cout, 9*gw] bf16
    for the grouped implicit-GEMM kernel (conv3x3.hip GROUPED mode).

    cpg >= 64 (ResNeXt101_32x16d stages 3/4 — ~80% of its grouped FLOPs):
    gw = cpg, a plain s-major repack — each 64-wide n-tile consumes
    exactly its own group's input channels, ZERO wasted MFMA.

    cpg 16/32 (stages 1/2): gw = 64 block-diagonal — the 64//cpg groups
    of each 64-wide output tile share one contiguous 64-channel input
    block; out-channel n's cpg true input channels sit at lane block
    (n//cpg) % (64//cpg), the rest are zeros (4x/2x MFMA on zeros, still
    ~MFMA rate vs MIOpen's grouped path)."""
    co, cpg = weight.shape[0], weight.shape[1]
    if cpg >= 64:
        return _repack_w3(weight.detach())
    w = weight.detach().to(torch.bfloat16)
    gpt = 64 // cpg  # groups per 64-wide tile
    out = torch.zeros(co, 3, 3, 64, dtype=torch.bfloat16, device=w.device)
    pos = (torch.arange(co, device=w.device) // cpg) % gpt
    for p in range(gpt):
        m = pos == p
        out[m, :, :, p * cpg:(p + 1) * cpg] = w[m].permute(0, 2, 3, 1)
    return out.reshape(co, 9 * 64).contiguous()


class Conv2dFast(nn.Conv2d):
    """nn.Conv2d drop-in; 1x1/group-1 convs on CUDA bypass MIOpen."""

    def _grad_tgt(self):
        """Direct-grad mode (reducer sets _edl_direct_grad at world 1):
        the weight's bucket-view gradient, or None. fp32 params only —
        the wgrad kernels accumulate fp32."""
        w = self.weight
        if (getattr(w, "_edl_direct_grad", False) and w.grad is not None
                and w.grad.dtype == torch.float32 and w.grad.is_contiguous()
                and torch.is_grad_enabled() and self.training):
            return w.grad
        return None

    def _cached(self, key, fn):
        # frozen weights (teacher serving: requires_grad False) cache
        # FOREVER — the global weight epoch is bumped by the STUDENT's
        # optimizer and would otherwise re-repack every teacher conv's
        # weights each student step (~400 MB of copies for ResNeXt101)
        epoch = _weight_epoch if self.weight.requires_grad else -1
        cache = getattr(self, "_w_cache", None)
        if cache is None or cache[0] != epoch:
            cache = (epoch, {})
            self._w_cache = cache
        d = cache[1]
        if key not in d:
            d[key] = fn()
        return d[key]

    def prefill_derived(self):
        """Rebuild this conv's per-weight-epoch derived tensors (wt_t /
        dgrad repacks) for the CURRENT epoch. The engine calls this on a
        side HIP stream at step start, so the repack kernels overlap the
        forward pass instead of sitting serially in the step (~49
        transpose/repack launches, ~240 us/step in the r2c27 trace);
        forward then finds them via _cached and launches nothing inline.
        Only keys a previous forward actually used are rebuilt; anything
        missed falls back to the inline path (correctness never depends
        on prefill)."""
        if not self.weight.requires_grad or not self.weight.is_cuda:
            return
        if not available():
            return
        cache = getattr(self, "_w_cache", None)
        if cache is None or not cache[1]:
            return
        old_keys = list(cache[1].keys())
        co, ci = self.out_channels, self.in_channels
        mb = getattr(self.weight, "_edl_bf16", None)
        cl = getattr(self.weight, "_edl_phys_shape", None) is not None
        builders = {}
        if mb is not None and self.kernel_size == (1, 1):
            builders["wt_t"] = lambda: ext().transpose_pad(mb.view(co, ci))
        if mb is not None and cl and self.kernel_size == (3, 3):
            w3v = mb.permute(0, 2, 3, 1).reshape(co, 9 * ci)
            builders["w3rot"] = lambda: ext().repack_dgrad_w3(w3v, ci, 0)
            builders["w3s2d"] = lambda: ext().repack_dgrad_w3(w3v, ci, 1)
        if (self.kernel_size == (3, 3) and self.in_channels <= 64
                and self.out_channels <= 64
                and (self.in_channels % 64 != 0
                     or self.out_channels % 64 != 0)):
            # w3srot only: it is read by BACKWARD (after the engine joins
            # the side stream). w3s is read by FORWARD, which does NOT
            # wait on the prefill stream — prefilling it races.
            builders["w3srot"] = lambda: _repack_w3_small(
                self.weight.detach().permute(1, 0, 2, 3).flip(2, 3),
                _small_cpt(co))
        for k in old_keys:
            b = builders.get(k)
            if b is not None:
                self._cached(k, b)

    def pop_bn_part(self):
        """BN stats partials of the LAST forward (or None) — consumed by
        the following BatchNorm to skip its stats kernel."""
        p = getattr(self, "_bn_part", None)
        self._bn_part = None
        return p

    def forward(self, x, bn_stats=False):
        self._bn_part = None
        bn_stats = bn_stats and _BN_STATS_FUSED
        if (
            _CONV3X3 == "hip"
            and x.is_cuda
            and available()
            and x.dtype == torch.bfloat16
            and self.kernel_size == (3, 3)
            and self.groups == 1
            and self.padding == (1, 1)
            and self.stride[0] == self.stride[1]
            and self.stride[0] in (1, 2)
            and self.bias is None
            and self.in_channels % 64 == 0
            and self.out_channels % 64 == 0
        ):
            if not x.is_contiguous(memory_format=torch.channels_last):
                x = x.contiguous(memory_format=torch.channels_last)
            co, ci = self.out_channels, self.in_channels
            cl = getattr(self.weight, "_edl_phys_shape", None) is not None
            mb = getattr(self.weight, "_edl_bf16", None)
            if mb is not None:  # engine bucket mirror: zero-cost bf16 copy
                w_bf16 = mb
                wsrc = mb
            else:
                w_bf16 = self._cached("w_bf16", lambda: self.weight.detach().to(
                    torch.bfloat16).contiguous())
                wsrc = self.weight.detach()
            if cl and mb is not None:
                # channels-last bucket: s-major w3 IS the mirror (a view)
                w3 = mb.permute(0, 2, 3, 1).reshape(co, 9 * ci)
            else:
                w3 = self._cached("w3", lambda: _repack_w3(wsrc))
            if cl and mb is not None:
                # one-pass kernel repack from the s-major mirror view
                mode = 0 if self.stride[0] == 1 else 1
                w3rot = self._cached(
                    "w3rot" if mode == 0 else "w3s2d",
                    lambda: ext().repack_dgrad_w3(w3, ci, mode))
            elif self.stride[0] == 1:
                w3rot = self._cached("w3rot", lambda: _repack_w3(
                    wsrc.permute(1, 0, 2, 3).flip(2, 3)))
            else:
                w3rot = self._cached("w3s2d",
                                     lambda: _repack_w3_s2dgrad(wsrc))
            # direct-grad (world 1): ONLY with channels-last bucket
            # storage — the wgrad's s-major [Cout, 9Cin] epilogue then IS
            # the grad-bucket layout (coalesced accumulate). With the
            # standard layout the stride-9 scatter measured slower than
            # the AccumulateGrad add it saves (round 1).
            grad_tgt = None
            w = self.weight
            if (cl and getattr(w, "_edl_direct_grad", False)
                    and w.grad is not None and w.grad.dtype == torch.float32
                    and ci <= _WGRAD3_MAXC
                    and torch.is_grad_enabled() and self.training):
                grad_tgt = w.grad.permute(0, 2, 3, 1).reshape(co, 9 * ci)
            holder = [] if bn_stats else None
            y = _Conv3x3Hip.apply(x, self.weight, w_bf16, w3, w3rot,
                                  self.stride[0], grad_tgt, holder)
            if holder:
                self._bn_part = holder[0]
            return y
        if (
            # deep-stem 3x3s (3->32->32->64): small-channel kernel
            _CONV3X3 == "hip"
            and os.environ.get("EDL_CONV3X3_SMALL", "1") == "1"
            and x.is_cuda
            and available()
            and (x.dtype == torch.bfloat16
                 or (x.dtype == torch.float32 and torch.is_autocast_enabled()))
            and self.kernel_size == (3, 3)
            and self.groups == 1
            and self.padding == (1, 1)
            and self.stride[0] == self.stride[1]
            and self.stride[0] in (1, 2)
            and self.bias is None
            and self.in_channels <= 64
            and self.out_channels <= 64
            and (self.in_channels % 64 != 0 or self.out_channels % 64 != 0)
        ):
            if x.dtype != torch.bfloat16:
                x = x.to(torch.bfloat16)  # the cast autocast would do
            if not x.is_contiguous(memory_format=torch.channels_last):
                x = x.contiguous(memory_format=torch.channels_last)
            cpt = _small_cpt(self.in_channels)
            wsrc = self.weight.detach()
            w3s = self._cached("w3s", lambda: _repack_w3_small(wsrc, cpt))
            w3srot = None
            if self.stride[0] == 1 and x.requires_grad:
                # dgrad weights: transpose Cin<->Cout + 180° rotate
                w3srot = self._cached("w3srot", lambda: _repack_w3_small(
                    wsrc.permute(1, 0, 2, 3).flip(2, 3),
                    _small_cpt(self.out_channels)))
            holder = [] if bn_stats else None
            y = _Conv3x3SmallHip.apply(x, self.weight, w3s, w3srot,
                                       self.stride[0], self.out_channels,
                                       holder)
            if holder:
                self._bn_part = holder[0]
            return y
        if (
            # Grouped teacher 3x3s. cpg >= _GROUPED_MINC runs the in-repo
            # grouped implicit-GEMM kernel (cpg >= 64: exact, zero wasted
            # MFMA — ResNeXt stage 3/4; cpg 16/32: block-diagonal with
            # 4x/2x zero work). Default MINC=32: the r1-measured loss was
            # the cpg=16-only 4x-waste path (distill 486 -> 415 img/s);
            # EDL_CONV3X3_GROUPED_MINC to A/B (16 = all, 9999 = none).
            _CONV3X3 == "hip"
            and x.is_cuda
            and available()
            and x.dtype == torch.bfloat16
            and self.kernel_size == (3, 3)
            and self.groups > 1
            and self.in_channels == self.out_channels
            and self.in_channels // self.groups >= _GROUPED_MINC
            and self.in_channels // self.groups in (16, 32, 64, 128)
            and self.in_channels % 64 == 0
            and self.padding == (1, 1)
            and self.stride[0] == self.stride[1]
            and self.stride[0] in (1, 2)
            and self.bias is None
            and not (torch.is_grad_enabled() and
                     (x.requires_grad or self.weight.requires_grad))
        ):
            # grouped teacher convs: eval/inference-only fast path
            if not x.is_contiguous(memory_format=torch.channels_last):
                x = x.contiguous(memory_format=torch.channels_last)
            w3g = self._cached("w3g", lambda: _repack_w3_grouped(self.weight))
            y2d = ext().conv3x3_grouped_fwd(x, w3g, self.stride[0])
            n, _, h, w = x.shape
            ho = (h - 1) // self.stride[0] + 1
            wo = (w - 1) // self.stride[0] + 1
            return y2d.view(n, ho, wo, self.out_channels).permute(0, 3, 1, 2)
        if (
            _CONV1X1 != "miopen"
            and x.is_cuda
            and self.kernel_size == (1, 1)
            and self.groups == 1
            and self.padding == (0, 0)
            and self.bias is None
        ):
            n, c, h, w = x.shape
            sh, sw = self.stride
            if sh != 1 or sw != 1:
                x = x[:, :, ::sh, ::sw].contiguous(memory_format=torch.channels_last)
                n, c, h, w = x.shape
            if not x.is_contiguous(memory_format=torch.channels_last):
                x = x.contiguous(memory_format=torch.channels_last)
            x2d = x.permute(0, 2, 3, 1).reshape(n * h * w, c)
            wt = self.weight.view(self.out_channels, c)
            if (
                _CONV1X1 == "hip"
                and available()
                and x2d.dtype == torch.bfloat16
                and c % 64 == 0
                and self.out_channels % 64 == 0
            ):
                mb = getattr(self.weight, "_edl_bf16", None)
                if mb is not None:  # engine bucket mirror: free bf16 view
                    w_bf16 = mb.view(self.out_channels, c)
                    # LDS-tiled transpose kernel instead of torch's
                    # permute-copy (one per 1x1 layer per step);
                    # transpose_pad is exact here since Cout % 64 == 0
                    wt_t = self._cached("wt_t", lambda: ext().transpose_pad(
                        mb.view(self.out_channels, c)))
                else:
                    w_bf16 = self._cached("w_bf16", lambda: self.weight.detach()
                                          .view(self.out_channels, c)
                                          .to(torch.bfloat16).contiguous())
                    wt_t = self._cached("wt_t", lambda: self.weight.detach().view(
                        self.out_channels, c).to(torch.bfloat16).t().contiguous())
                holder = [] if bn_stats else None
                y2d = _Conv1x1Hip.apply(x2d, wt, w_bf16, wt_t,
                                        self._grad_tgt(), holder)
                if holder:
                    self._bn_part = holder[0]
            else:
                y2d = x2d @ wt.t()
            return (
                y2d.view(n, h, w, self.out_channels).permute(0, 3, 1, 2)
            )
        return super().forward(x)
