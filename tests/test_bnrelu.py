"""Fused BN(+Add)+ReLU: CPU fallback semantics + GPU kernel numerics vs a
plain PyTorch fp32 reference of the same op."""
import pytest
import torch
import torch.nn.functional as F

from edl_amd.ops.bnrelu import BNAddReLU2d, BNReLU2d


def _ref_bn(x, bn, relu=True, res=None, training=True):
    """fp32 torch reference with explicit batch stats (matches kernel math)."""
    y = F.batch_norm(x.float(), bn.running_mean.clone(), bn.running_var.clone(),
                     bn.weight.float(), bn.bias.float(), training, bn.momentum, bn.eps)
    if res is not None:
        y = y + res.float()
    if relu:
        y = F.relu(y)
    return y


def test_cpu_fallback_matches_torch_bn():
    torch.manual_seed(0)
    bn = BNReLU2d(16)
    ref = torch.nn.BatchNorm2d(16)
    ref.load_state_dict({k: v for k, v in bn.state_dict().items()}, strict=False)
    x = torch.randn(4, 16, 8, 8)
    y = bn(x)
    yr = F.relu(ref(x))
    assert torch.allclose(y, yr, atol=1e-5)
    # running stats updated identically
    assert torch.allclose(bn.running_mean, ref.running_mean, atol=1e-6)
    assert torch.allclose(bn.running_var, ref.running_var, atol=1e-5)


def test_state_dict_compatible_with_batchnorm():
    bn = BNReLU2d(8)
    sd = torch.nn.BatchNorm2d(8).state_dict()
    bn.load_state_dict(sd)  # must accept BatchNorm2d layout


@pytest.mark.gpu
class TestFusedGPU:
    @pytest.fixture(autouse=True)
    def _gpu(self):
        if not torch.cuda.is_available():
            pytest.skip("no GPU")

    def _mk(self, C, N=4, H=9, W=9, add=False, act=True, seed=0):
        torch.manual_seed(seed)
        cls = BNAddReLU2d if add else BNReLU2d
        m = (cls(C) if add else cls(C, act=act)).cuda()
        with torch.no_grad():
            m.weight.mul_(1.5).add_(0.1)
            m.bias.add_(0.05)
        x = (torch.randn(N, C, H, W, device="cuda") * 2).to(torch.bfloat16)
        x = x.contiguous(memory_format=torch.channels_last).requires_grad_(True)
        res = None
        if add:
            res = torch.randn(N, C, H, W, device="cuda").to(torch.bfloat16)
            res = res.contiguous(memory_format=torch.channels_last).requires_grad_(True)
        return m, x, res

    @pytest.mark.parametrize("C", [8, 64, 256, 2048])
    @pytest.mark.parametrize("act", [True, False])
    def test_forward_train(self, C, act):
        m, x, _ = self._mk(C, act=act)
        y = m(x)
        assert y.dtype == torch.bfloat16
        ref = _ref_bn(x.detach(), m, relu=act)
        assert torch.allclose(y.float(), ref, atol=3e-2, rtol=3e-2), \
            (y.float() - ref).abs().max().item()

    def test_running_stats_update(self):
        m, x, _ = self._mk(64)
        rm0, rv0 = m.running_mean.clone(), m.running_var.clone()
        bn = torch.nn.BatchNorm2d(64).cuda()
        with torch.no_grad():
            bn.weight.copy_(m.weight)
            bn.bias.copy_(m.bias)
        m(x)
        bn(x.detach().float())
        assert torch.allclose(m.running_mean, bn.running_mean, atol=1e-3)
        assert torch.allclose(m.running_var, bn.running_var, atol=1e-2)
        assert not torch.allclose(m.running_mean, rm0)
        assert not torch.allclose(m.running_var, rv0)

    @pytest.mark.parametrize("add", [False, True])
    def test_backward(self, add):
        C = 64
        m, x, res = self._mk(C, add=add)
        y = m(x, res) if add else m(x)
        g = torch.randn_like(y.float()).to(torch.bfloat16).contiguous(
            memory_format=torch.channels_last)
        y.backward(g)

        # fp32 reference with autograd
        xr = x.detach().float().requires_grad_(True)
        rr = res.detach().float().requires_grad_(True) if add else None
        wr = m.weight.detach().clone().requires_grad_(True)
        br = m.bias.detach().clone().requires_grad_(True)
        yr = F.batch_norm(xr, None, None, wr, br, True, 0.0, m.eps)
        if add:
            yr = yr + rr
        yr = F.relu(yr)
        yr.backward(g.float())

        assert torch.allclose(x.grad.float(), xr.grad, atol=5e-2, rtol=5e-2), \
            (x.grad.float() - xr.grad).abs().max().item()
        assert torch.allclose(m.weight.grad, wr.grad, atol=2e-1, rtol=2e-2), \
            (m.weight.grad - wr.grad).abs().max().item()
        assert torch.allclose(m.bias.grad, br.grad, atol=2e-1, rtol=2e-2)
        if add:
            assert torch.allclose(res.grad.float(), rr.grad, atol=5e-2, rtol=5e-2)

    def test_eval_uses_running_stats(self):
        m, x, _ = self._mk(64)
        m(x)  # one training step to move stats
        m.eval()
        with torch.no_grad():
            y = m(x)
            ref = _ref_bn(x, m, relu=True, training=False)
        assert torch.allclose(y.float(), ref, atol=3e-2, rtol=3e-2)

    def test_resnet50vd_fused_forward_close_to_fallback(self):
        """Whole-model: fused kernels vs the fp32 fallback path."""
        from edl_amd.models import resnet50_vd

        torch.manual_seed(0)
        model = resnet50_vd(num_classes=100).cuda().to(
            memory_format=torch.channels_last).eval()
        x = torch.randn(2, 3, 64, 64, device="cuda")
        with torch.no_grad():
            with torch.autocast("cuda", dtype=torch.bfloat16):
                y_fused = model(x.to(torch.bfloat16).contiguous(
                    memory_format=torch.channels_last))
            y_ref = model(x)  # fp32, fallback path inside BNReLU2d
        assert torch.allclose(y_fused.float(), y_ref, atol=0.5, rtol=0.1), \
            (y_fused.float() - y_ref).abs().max().item()


@pytest.mark.gpu
class TestConvFusedStats:
    def _gpu(self):
        if not torch.cuda.is_available():
            pytest.skip("no GPU")

    @pytest.mark.parametrize("shape", [
        (2, 64, 64, 14, 1),    # 1x1 (gemm_bt epilogue)
        (2, 64, 128, 14, 3),   # 3x3 (conv3x3 epilogue)
        (2, 3, 32, 16, 3),     # stem (small kernel epilogue)
    ])
    def test_conv_epilogue_stats_match_bn_stats(self, shape, monkeypatch):
        """BN fed conv-folded stats partials must match BN computing its
        own stats on the same bf16 output (train mode, running stats and
        y compared)."""
        self._gpu()
        import copy

        import edl_amd.ops.conv as conv_mod
        from edl_amd.models.resnet_vd import ConvBN

        n, ci, co, hw, k = shape
        torch.manual_seed(31)
        m1 = ConvBN(ci, co, k).cuda().train()
        m1.conv.to(torch.bfloat16)
        m2 = copy.deepcopy(m1)
        x = torch.randn(n, ci, hw, hw, device="cuda").to(torch.bfloat16)
        x = x.contiguous(memory_format=torch.channels_last)

        monkeypatch.setattr(conv_mod, "_BN_STATS_FUSED", True)
        y1 = m1(x)
        monkeypatch.setattr(conv_mod, "_BN_STATS_FUSED", False)
        y2 = m2(x)
        assert torch.allclose(y1.float(), y2.float(), atol=1e-2, rtol=1e-2), \
            (y1.float() - y2.float()).abs().max().item()
        assert torch.allclose(m1.bn.running_mean, m2.bn.running_mean,
                              atol=1e-4, rtol=1e-4)
        assert torch.allclose(m1.bn.running_var, m2.bn.running_var,
                              atol=1e-4, rtol=1e-4)

    @pytest.mark.parametrize("shape", [
        (4, 64, 64, 112, 1),   # gemm_bt epilogue, tiles_m=196 > 192 cap
        (4, 64, 64, 112, 3),   # conv3x3 epilogue, capped
        (2, 3, 32, 160, 3),    # stem small-kernel epilogue, capped
    ])
    def test_capped_pooled_partials_stay_clean(self, shape, monkeypatch):
        """Shapes with tiles_m > 192 accumulate stats into a POOLED zeroed
        buffer with atomics; bn_finalize(zero_src) must return it to the
        pool clean. Two successive steps must both match the unfused-stats
        reference — step 2 is wrong if the pool came back dirty."""
        self._gpu()
        import copy

        import edl_amd.ops.conv as conv_mod
        from edl_amd.models.resnet_vd import ConvBN

        n, ci, co, hw, k = shape
        torch.manual_seed(33)
        m1 = ConvBN(ci, co, k).cuda().train()
        m1.conv.to(torch.bfloat16)
        m2 = copy.deepcopy(m1)
        for it in range(2):
            x = torch.randn(n, ci, hw, hw, device="cuda").to(torch.bfloat16)
            x = x.contiguous(memory_format=torch.channels_last)
            monkeypatch.setattr(conv_mod, "_BN_STATS_FUSED", True)
            y1 = m1(x)
            monkeypatch.setattr(conv_mod, "_BN_STATS_FUSED", False)
            y2 = m2(x)
            assert torch.allclose(y1.float(), y2.float(), atol=1e-2,
                                  rtol=1e-2), (it, (y1.float() - y2.float()).abs().max().item())
            assert torch.allclose(m1.bn.running_mean, m2.bn.running_mean,
                                  atol=1e-4, rtol=1e-4), it
            assert torch.allclose(m1.bn.running_var, m2.bn.running_var,
                                  atol=1e-4, rtol=1e-4), it

    def test_bottleneck_trains_with_fused_stats(self, monkeypatch):
        """Full bottleneck fwd+bwd with conv-folded stats: finite grads,
        same trajectory as the unfused-stats path."""
        self._gpu()
        import copy

        import edl_amd.ops.conv as conv_mod
        from edl_amd.models.resnet_vd import BottleneckVd

        torch.manual_seed(32)
        b1 = BottleneckVd(64, 64, stride=1, if_first=True).cuda().train()
        for mod in b1.modules():
            if isinstance(mod, conv_mod.Conv2dFast):
                mod.to(torch.bfloat16)
        b2 = copy.deepcopy(b1)
        x = torch.randn(2, 64, 14, 14, device="cuda").to(torch.bfloat16)
        x = x.contiguous(memory_format=torch.channels_last)

        monkeypatch.setattr(conv_mod, "_BN_STATS_FUSED", True)
        y1 = b1(x)
        y1.float().square().mean().backward()
        monkeypatch.setattr(conv_mod, "_BN_STATS_FUSED", False)
        y2 = b2(x)
        y2.float().square().mean().backward()
        assert torch.allclose(y1.float(), y2.float(), atol=1e-2, rtol=1e-2)
        for p1, p2 in zip(b1.parameters(), b2.parameters()):
            if p1.grad is not None:
                assert torch.allclose(p1.grad.float(), p2.grad.float(),
                                      atol=5e-2, rtol=5e-2), p1.shape


@pytest.mark.gpu
class TestFusedBwdSingleLaunch:
    """EDL_BN_BWD_FUSED: reduce+finalize+dx as ONE kernel with a grid-wide
    flag rendezvous (bnrelu.hip bn_bwd_fused_kernel). Two back-to-back
    calls verify the self-resetting workspace."""

    @pytest.mark.parametrize("relu,add", [(True, False), (True, True),
                                          (False, False)])
    @pytest.mark.parametrize("mc", [(4 * 56 * 56, 64), (4 * 7 * 7, 2048)])
    def test_matches_unfused(self, relu, add, mc, monkeypatch):
        if not torch.cuda.is_available():
            pytest.skip("no GPU")
        import os

        from edl_amd.ops import ext

        M, C = mc
        torch.manual_seed(40)
        for it in range(2):  # second call checks the ws reset
            x2d = torch.randn(M, C, device="cuda").to(torch.bfloat16)
            gamma = torch.rand(C, device="cuda") + 0.5
            beta = torch.randn(C, device="cuda")
            res = (torch.randn(M, C, device="cuda").to(torch.bfloat16)
                   if add else None)
            y, mean, invstd, mask = ext().bn_fwd_train(
                x2d, gamma, beta, torch.zeros(C, device="cuda"),
                torch.ones(C, device="cuda"), 0.1, 1e-5, res, relu, None)
            dy = torch.randn(M, C, device="cuda").to(torch.bfloat16)

            monkeypatch.setenv("EDL_BN_BWD_FUSED", "0")
            r0 = ext().bn_bwd(dy, mask, x2d, mean, invstd, gamma, relu, add,
                              True, None, None)
            monkeypatch.setenv("EDL_BN_BWD_FUSED", "1")
            r1 = ext().bn_bwd(dy, mask, x2d, mean, invstd, gamma, relu, add,
                              True, None, None)
            monkeypatch.setenv("EDL_BN_BWD_FUSED", "0")
            names = ["dx", "dgamma", "dbeta", "dres"]
            for n, a, b in zip(names, r0, r1):
                if a is None or (hasattr(a, "numel") and a.numel() == 0):
                    continue
                assert torch.allclose(a.float(), b.float(), atol=2e-2,
                                      rtol=2e-2), (it, n)
