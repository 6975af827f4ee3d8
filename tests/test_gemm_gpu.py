"""gemm_bt MFMA kernel numerics vs torch fp32 reference (GPU)."""
import pytest
import torch

pytestmark = pytest.mark.gpu


@pytest.fixture(autouse=True)
def _gpu():
    if not torch.cuda.is_available():
        pytest.skip("no GPU")


# M values exercise both tile shapes and the row-clamped tail
@pytest.mark.parametrize("M,N,K", [
    (256, 128, 64), (512, 256, 128), (100, 128, 256),   # 128x128 tile + tail
    (300, 64, 64), (1000, 64, 512),                      # 256x64 tile + tail
    (3136, 512, 128), (784, 1024, 256),                  # real conv shapes
])
def test_gemm_bt_numerics(M, N, K):
    from edl_amd import ops

    torch.manual_seed(0)
    # asymmetric operands (guide: symmetric B passes transposed kernels)
    a = (torch.randn(M, K, device="cuda") * 2).to(torch.bfloat16)
    b = (torch.randn(N, K, device="cuda") + torch.arange(K, device="cuda") * 0.01
         ).to(torch.bfloat16)
    c = ops.ext().gemm_bt(a, b)
    ref = a.float() @ b.float().t()
    err = (c.float() - ref).abs()
    scale = ref.abs().mean().clamp(min=1)
    assert (err / scale).max() < 0.05, (err.max().item(), scale.item())


def test_conv1x1_hip_matches_matmul():
    from edl_amd.ops.conv import Conv2dFast

    torch.manual_seed(1)
    conv = Conv2dFast(128, 256, 1, bias=False).cuda().to(torch.bfloat16)
    x = torch.randn(4, 128, 14, 14, device="cuda").to(torch.bfloat16)
    x = x.contiguous(memory_format=torch.channels_last).requires_grad_(True)
    y = conv(x)
    g = torch.randn_like(y)
    y.backward(g)

    xr = x.detach().float().requires_grad_(True)
    wr = conv.weight.detach().float().requires_grad_(True)
    yr = torch.nn.functional.conv2d(xr, wr)
    yr.backward(g.float())

    assert torch.allclose(y.float(), yr.detach(), atol=0.5, rtol=0.05), \
        (y.float() - yr.detach()).abs().max().item()
    assert torch.allclose(x.grad.float(), xr.grad, atol=0.5, rtol=0.05)
    assert torch.allclose(conv.weight.grad.float(), wr.grad, atol=2.0, rtol=0.05), \
        (conv.weight.grad.float() - wr.grad).abs().max().item()


@pytest.mark.parametrize("stride", [1, 2])
@pytest.mark.parametrize("shape", [(2, 64, 16, 16, 64), (2, 128, 14, 14, 256),
                                   (3, 64, 9, 11, 64)])
def test_conv3x3_fwd_numerics(stride, shape):
    import torch.nn.functional as F

    from edl_amd.ops.conv import Conv2dFast

    n, cin, h, w, cout = shape
    torch.manual_seed(0)
    conv = Conv2dFast(cin, cout, 3, stride=stride, padding=1, bias=False
                      ).cuda().to(torch.bfloat16)
    x = (torch.randn(n, cin, h, w, device="cuda") * 1.5).to(torch.bfloat16)
    x = x.contiguous(memory_format=torch.channels_last)
    y = conv(x)
    ref = F.conv2d(x.float(), conv.weight.float(), stride=stride, padding=1)
    assert y.shape == ref.shape
    err = (y.float() - ref).abs()
    scale = ref.abs().mean().clamp(min=0.5)
    assert (err / scale).max() < 0.1, (err.max().item(), scale.item())


@pytest.mark.parametrize("stride", [1, 2])
def test_conv3x3_backward_numerics(stride):
    import torch.nn.functional as F

    from edl_amd.ops.conv import Conv2dFast

    torch.manual_seed(1)
    conv = Conv2dFast(64, 128, 3, stride=stride, padding=1, bias=False
                      ).cuda().to(torch.bfloat16)
    x = torch.randn(2, 64, 12, 12, device="cuda").to(torch.bfloat16)
    x = x.contiguous(memory_format=torch.channels_last).requires_grad_(True)
    y = conv(x)
    g = torch.randn_like(y).contiguous(memory_format=torch.channels_last)
    y.backward(g)

    xr = x.detach().float().requires_grad_(True)
    wr = conv.weight.detach().float().requires_grad_(True)
    yr = F.conv2d(xr, wr, stride=stride, padding=1)
    yr.backward(g.float())
    assert torch.allclose(x.grad.float(), xr.grad, atol=0.5, rtol=0.1), \
        (x.grad.float() - xr.grad).abs().max().item()
    assert torch.allclose(conv.weight.grad.float(), wr.grad, atol=2.0, rtol=0.1), \
        (conv.weight.grad.float() - wr.grad).abs().max().item()


@pytest.mark.parametrize("M,C", [(256, 64), (1000, 128), (100, 64), (1568, 512)])
def test_transpose_pad(M, C):
    from edl_amd import ops

    torch.manual_seed(2)
    x = (torch.randn(M, C, device="cuda")).to(torch.bfloat16)
    y = ops.ext().transpose_pad(x)
    Mp = (M + 63) // 64 * 64
    assert y.shape == (C, Mp)
    assert torch.equal(y[:, :M], x.t().contiguous())
    assert (y[:, M:] == 0).all()


@pytest.mark.parametrize("M,N,K,splitk", [
    (64, 64, 6400, 0), (128, 256, 4096, 8), (256, 64, 1280, 4),
    (512, 128, 64, 0),
])
def test_gemm_bt_splitk_numerics(M, N, K, splitk):
    from edl_amd import ops

    torch.manual_seed(3)
    a = (torch.randn(M, K, device="cuda") * 0.5).to(torch.bfloat16)
    b = (torch.randn(N, K, device="cuda") * 0.5).to(torch.bfloat16)
    c = ops.ext().gemm_bt_splitk(a, b, splitk)
    assert c.dtype == torch.float32
    ref = a.float() @ b.float().t()
    err = (c - ref).abs()
    scale = ref.abs().mean().clamp(min=1)
    assert (err / scale).max() < 0.05, (err.max().item(), scale.item())


@pytest.mark.parametrize("stride", [1, 2])
@pytest.mark.parametrize("cpg", [16, 32, 64, 128])
def test_conv3x3_grouped_eval_numerics(stride, cpg, monkeypatch):
    """Grouped 3x3 inference fast path vs F.conv2d at every ResNeXt
    channels-per-group width: 16/32 run the block-diagonal repack, 64/128
    the exact per-group dense engine (zero wasted MFMA)."""
    import torch.nn.functional as F

    import edl_amd.ops.conv as conv_mod
    from edl_amd.ops.conv import Conv2dFast

    monkeypatch.setattr(conv_mod, "_GROUPED_MINC", 16)  # route every width

    torch.manual_seed(5)
    C = max(128, cpg * 2)
    groups = C // cpg
    conv = Conv2dFast(C, C, 3, stride=stride, padding=1, groups=groups,
                      bias=False).cuda().to(torch.bfloat16)
    x = torch.randn(2, C, 14, 14, device="cuda").to(torch.bfloat16)
    x = x.contiguous(memory_format=torch.channels_last)
    with torch.no_grad():
        y = conv(x)
        ref = F.conv2d(x.float(), conv.weight.float(), stride=stride,
                       padding=1, groups=groups)
    assert y.shape == ref.shape
    err = (y.float() - ref).abs()
    scale = ref.abs().mean().clamp(min=0.2)
    assert (err / scale).max() < 0.1, (err.max().item(), scale.item())


def test_conv3x3_grouped_training_falls_back():
    """With grad enabled the grouped fast path must NOT engage (fwd-only)."""
    from edl_amd.ops.conv import Conv2dFast

    conv = Conv2dFast(64, 64, 3, padding=1, groups=4, bias=False
                      ).cuda().to(torch.bfloat16)
    x = torch.randn(1, 64, 8, 8, device="cuda").to(torch.bfloat16)
    x = x.contiguous(memory_format=torch.channels_last).requires_grad_(True)
    y = conv(x)  # MIOpen path; must be differentiable
    y.sum().backward()
    assert x.grad is not None and conv.weight.grad is not None


# K values exercise all four (BN1, BN2) tile configs, the K tail
# (guarded stage_tail path), single-partial-chunk K, and splitk clamping
@pytest.mark.parametrize("K,N1,N2,splitk", [
    (4096, 64, 64, 0),        # (64,64,KW=4) config, 128 KB LDS
    (100352, 64, 64, 0),      # real stage-1 wgrad shape
    (4096, 64, 256, 0),       # (64,128,KW=2)
    (4096, 256, 64, 0),       # (128,64,KW=2)
    (6272, 256, 1024, 0),     # (128,128,KW=1), real stage-3 shape
    (1000, 128, 128, 0),      # K tail (1000 % 64 = 40)
    (63, 64, 64, 0),          # single partial chunk
    (130, 64, 128, 8),        # splitk > nchunks -> clamped
    (1568, 512, 2048, 0),     # real stage-4 shape
])
def test_gemm_tn_splitk_numerics(K, N1, N2, splitk):
    from edl_amd import ops

    torch.manual_seed(2)
    a = (torch.randn(K, N1, device="cuda") * 1.5).to(torch.bfloat16)
    b = (torch.randn(K, N2, device="cuda") +
         torch.arange(N2, device="cuda") * 0.003).to(torch.bfloat16)
    c = ops.ext().gemm_tn_splitk(a, b, splitk)
    assert c.dtype == torch.float32 and c.shape == (N1, N2)
    ref = a.float().t() @ b.float()
    err = (c - ref).abs()
    scale = ref.abs().mean().clamp(min=1)
    assert (err / scale).max() < 0.05, (err.max().item(), scale.item())


def test_gemm_tn_matches_bt_pipeline():
    """Direct TN wgrad == transpose_pad + gemm_bt_splitk on the same data."""
    from edl_amd import ops

    e = ops.ext()
    torch.manual_seed(3)
    K, N1, N2 = 12544, 128, 512
    a = torch.randn(K, N1, device="cuda").to(torch.bfloat16)
    b = torch.randn(K, N2, device="cuda").to(torch.bfloat16)
    c_tn = e.gemm_tn_splitk(a, b, 0)
    c_bt = e.gemm_bt_splitk(e.transpose_pad(a), e.transpose_pad(b), 0)
    assert torch.allclose(c_tn, c_bt, atol=0.5, rtol=0.02), \
        (c_tn - c_bt).abs().max().item()


@pytest.mark.parametrize("stride", [1, 2])
@pytest.mark.parametrize("shape", [(2, 64, 16, 16, 64), (2, 128, 14, 14, 256),
                                   (3, 64, 9, 11, 64)])
def test_gemm_tn3x3_matches_bt_pipeline(stride, shape):
    """Direct-gather conv3x3 wgrad == transpose_pad + shift9 + bt pipeline."""
    from edl_amd import ops

    e = ops.ext()
    torch.manual_seed(4)
    n, ci, h, w, co = shape
    x = torch.randn(n, ci, h, w, device="cuda").to(torch.bfloat16).contiguous(
        memory_format=torch.channels_last)
    ho = (h - 1) // stride + 1
    wo = (w - 1) // stride + 1
    dy2d = torch.randn(n * ho * wo, co, device="cuda").to(torch.bfloat16)
    c_tn = e.gemm_tn3x3_splitk(dy2d, x, stride, 0)
    c_bt = e.gemm_bt_splitk(e.transpose_pad(dy2d),
                            e.conv3x3_wgrad_operand(x, stride), 0)
    assert c_tn.shape == c_bt.shape == (co, 9 * ci)
    assert torch.allclose(c_tn, c_bt, atol=0.5, rtol=0.02), \
        (c_tn - c_bt).abs().max().item()


def test_gemm_tn3x3_accumulate_out_layout():
    """out= path accumulates in the conv-weight [Cout,Cin,3,3] layout."""
    from edl_amd import ops

    e = ops.ext()
    torch.manual_seed(5)
    n, ci, h, w, co = 2, 64, 10, 10, 64
    x = torch.randn(n, ci, h, w, device="cuda").to(torch.bfloat16).contiguous(
        memory_format=torch.channels_last)
    dy2d = torch.randn(n * h * w, co, device="cuda").to(torch.bfloat16)
    native = e.gemm_tn3x3_splitk(dy2d, x, 1, 0)
    out = torch.full((co, ci, 3, 3), 0.25, device="cuda")
    e.gemm_tn3x3_splitk(dy2d, x, 1, 0, out)
    ref = native.view(co, 3, 3, ci).permute(0, 3, 1, 2) + 0.25
    assert torch.allclose(out, ref, atol=1e-3, rtol=1e-3), \
        (out - ref).abs().max().item()


@pytest.mark.parametrize("shape", [(2, 128, 128, 14), (2, 256, 256, 28),
                                   (1, 512, 512, 14), (2, 64, 128, 13)])
def test_conv3x3_s2_dgrad_numerics(shape):
    """Stride-2 3x3 dgrad parity kernel vs torch.nn.grad.conv2d_input
    (fp32 reference). Covers even and odd H/W and the Cin%128 launch
    split (replaces MIOpen's igemm_bwd — VERDICT r1 #4)."""
    import torch.nn.functional as F

    from edl_amd import ops
    from edl_amd.ops.conv import _repack_w3_s2dgrad

    n, ci, co, hw = shape
    torch.manual_seed(11)
    w = (torch.randn(co, ci, 3, 3, device="cuda") * 0.2)
    x_shape = [n, ci, hw, hw]
    ho = (hw + 1) // 2
    dy = (torch.randn(n, co, ho, ho, device="cuda") * 0.5).to(torch.bfloat16)
    dy = dy.contiguous(memory_format=torch.channels_last)

    wcat = _repack_w3_s2dgrad(w)
    dx2d = ops.ext().conv3x3s2_dgrad(dy, wcat, hw, hw)
    dx = dx2d.view(n, hw, hw, ci).permute(0, 3, 1, 2)

    ref = torch.nn.grad.conv2d_input(
        x_shape, w.float(), dy.float(), stride=(2, 2), padding=(1, 1))
    err = (dx.float() - ref).abs()
    scale = ref.abs().mean().clamp(min=0.05)
    assert (err / scale).max() < 0.1, (err.max().item(), scale.item())


def test_conv3x3_s2_full_backward_matches_miopen():
    """End-to-end Conv2dFast stride-2 backward (hip dgrad + wgrad) vs the
    fp32 autograd reference."""
    from edl_amd.ops.conv import Conv2dFast

    torch.manual_seed(13)
    conv = Conv2dFast(128, 128, 3, stride=2, padding=1, bias=False).cuda()
    x = (torch.randn(2, 128, 28, 28, device="cuda") * 0.5).to(torch.bfloat16)
    x = x.contiguous(memory_format=torch.channels_last).requires_grad_(True)
    y = conv(x)
    g = torch.randn_like(y)
    y.backward(g)

    xr = x.detach().float().requires_grad_(True)
    wr = conv.weight.detach().float().requires_grad_(True)
    yr = torch.nn.functional.conv2d(xr, wr, stride=2, padding=1)
    yr.backward(g.float())

    for got, ref in ((x.grad.float(), xr.grad), (conv.weight.grad.float(), wr.grad)):
        err = (got - ref).abs()
        scale = ref.abs().mean().clamp(min=0.02)
        assert (err / scale).max() < 0.12, (err.max().item(), scale.item())


@pytest.mark.parametrize("shape", [
    (2, 3, 32, 2, 64),    # stem conv0: 3->32 s2
    (2, 32, 32, 1, 33),   # stem conv1: 32->32 s1 (odd HW)
    (2, 32, 64, 1, 28),   # stem conv2: 32->64 s1
])
def test_conv3x3_small_fwd_numerics(shape):
    """Deep-stem small-channel conv fwd vs F.conv2d fp32."""
    import torch.nn.functional as F

    from edl_amd.ops.conv import Conv2dFast

    n, ci, co, stride, hw = shape
    torch.manual_seed(21)
    conv = Conv2dFast(ci, co, 3, stride=stride, padding=1, bias=False
                      ).cuda().to(torch.bfloat16)
    x = (torch.randn(n, ci, hw, hw, device="cuda")).to(torch.bfloat16)
    x = x.contiguous(memory_format=torch.channels_last)
    with torch.no_grad():
        y = conv(x)
        ref = F.conv2d(x.float(), conv.weight.float(), stride=stride, padding=1)
    assert y.shape == ref.shape
    err = (y.float() - ref).abs()
    scale = ref.abs().mean().clamp(min=0.2)
    assert (err / scale).max() < 0.1, (err.max().item(), scale.item())


@pytest.mark.parametrize("shape", [(2, 32, 32, 30), (2, 32, 64, 14)])
def test_conv3x3_small_backward_numerics(shape):
    """Stem conv backward: in-repo dgrad + unfold-GEMM wgrad vs fp32."""
    import torch.nn.functional as F

    from edl_amd.ops.conv import Conv2dFast

    n, ci, co, hw = shape
    torch.manual_seed(22)
    conv = Conv2dFast(ci, co, 3, padding=1, bias=False).cuda().to(torch.bfloat16)
    x = (torch.randn(n, ci, hw, hw, device="cuda") * 0.5).to(torch.bfloat16)
    x = x.contiguous(memory_format=torch.channels_last).requires_grad_(True)
    y = conv(x)
    g = torch.randn_like(y).contiguous(memory_format=torch.channels_last)
    y.backward(g)

    xr = x.detach().float().requires_grad_(True)
    wr = conv.weight.detach().float().requires_grad_(True)
    yr = F.conv2d(xr, wr, padding=1)
    yr.backward(g.float())
    for got, ref in ((x.grad.float(), xr.grad),
                     (conv.weight.grad.float(), wr.grad)):
        err = (got - ref).abs()
        scale = ref.abs().mean().clamp(min=0.02)
        assert (err / scale).max() < 0.15, (err.max().item(), scale.item())


def test_stem_conv0_wgrad_no_dx():
    """conv0 (3ch input, requires_grad False): weight grad flows, no dx."""
    from edl_amd.ops.conv import Conv2dFast

    conv = Conv2dFast(3, 32, 3, stride=2, padding=1, bias=False
                      ).cuda().to(torch.bfloat16)
    x = torch.randn(2, 3, 32, 32, device="cuda").to(torch.bfloat16)
    x = x.contiguous(memory_format=torch.channels_last)
    y = conv(x)
    y.float().sum().backward()
    assert conv.weight.grad is not None
    assert torch.isfinite(conv.weight.grad.float()).all()


@pytest.mark.parametrize("mode", [0, 1])
@pytest.mark.parametrize("co,ci", [(64, 64), (128, 64), (256, 128)])
def test_repack_dgrad_w3_kernel(mode, co, ci):
    """One-pass dgrad weight repack vs the torch reference chain."""
    from edl_amd import ops
    from edl_amd.ops.conv import _repack_w3, _repack_w3_s2dgrad

    torch.manual_seed(17)
    w = torch.randn(co, ci, 3, 3, device="cuda")
    w_smaj = w.permute(0, 2, 3, 1).reshape(co, 9 * ci).to(
        torch.bfloat16).contiguous()
    got = ops.ext().repack_dgrad_w3(w_smaj, ci, mode)
    wb = w.to(torch.bfloat16).float().to(torch.bfloat16)  # match rounding
    if mode == 0:
        ref = _repack_w3(w.to(torch.bfloat16).float()
                         .permute(1, 0, 2, 3).flip(2, 3))
    else:
        ref = _repack_w3_s2dgrad(w.to(torch.bfloat16).float())
    assert got.shape == ref.shape
    assert torch.equal(got.float(), ref.float()), \
        (got.float() - ref.float()).abs().max().item()


@pytest.mark.gpu
def test_conv3x3_splitk_pooled_partials_two_steps():
    """Split-K conv fwd folds fp32 partials into a POOLED pre-zeroed buffer
    and cast_bf16_zero returns it clean — two successive calls on the same
    stage-3 shape must both match torch (a dirty pool corrupts call 2)."""
    if not torch.cuda.is_available():
        pytest.skip("no GPU")
    import torch.nn.functional as F

    from edl_amd.ops.conv import Conv2dFast

    torch.manual_seed(41)
    # bs32 stage-3: M=6272 tiles ~ 52-98 -> split-K route (conv3x3_pick_splitk)
    m = Conv2dFast(1024, 256, 3, padding=1, bias=False).cuda().to(torch.bfloat16)
    for it in range(2):
        x = torch.randn(32, 1024, 14, 14, device="cuda").to(torch.bfloat16)
        x = x.contiguous(memory_format=torch.channels_last)
        with torch.no_grad():
            y = m(x)
            ref = F.conv2d(x.float(), m.weight.float(), padding=1)
        assert torch.allclose(y.float(), ref, atol=0.5, rtol=5e-2), \
            (it, (y.float() - ref).abs().max().item())


@pytest.mark.gpu
@pytest.mark.parametrize("shape", [
    (4, 3, 32, 64, 2),    # stem conv0: Cin=3 (cpad->8), stride 2
    (4, 32, 32, 32, 1),   # stem conv1
    (4, 32, 64, 32, 1),   # stem conv2 (Cout=64, no dy pad)
])
def test_gemm_tn3x3_small_wgrad_numerics(shape):
    """G3S stem wgrad == torch conv2d_weight (fp32 reference)."""
    if not torch.cuda.is_available():
        pytest.skip("no GPU")
    from edl_amd import ops

    n, ci, co, hw, stride = shape
    torch.manual_seed(50)
    x = torch.randn(n, ci, hw, hw, device="cuda").to(torch.bfloat16)
    x = x.contiguous(memory_format=torch.channels_last)
    ho = (hw - 1) // stride + 1
    dy = torch.randn(n, co, ho, ho, device="cuda").to(torch.bfloat16)
    dy = dy.contiguous(memory_format=torch.channels_last)

    dy2d = dy.permute(0, 2, 3, 1).reshape(-1, co)
    if co < 64:
        import torch.nn.functional as F
        dy2d = F.pad(dy2d, (0, 64 - co))
    cfull = ops.ext().gemm_tn3x3_small(dy2d.contiguous(), x, stride)
    cinp = max(8, 1 << (ci - 1).bit_length())
    dw = (cfull[:co, :9 * cinp].view(co, 3, 3, cinp)
          .permute(0, 3, 1, 2)[:, :ci])

    ref = torch.nn.grad.conv2d_weight(
        x.float(), (co, ci, 3, 3), dy.float(), stride=(stride, stride),
        padding=(1, 1))
    assert torch.allclose(dw, ref, atol=0.5, rtol=2e-2), \
        (dw - ref).abs().max().item()


@pytest.mark.gpu
def test_stem_wgrad_tn_route_matches_default(monkeypatch):
    """EDL_STEM_WGRAD=tn through the module backward (pad + slice glue)
    must match the default route's weight grad."""
    if not torch.cuda.is_available():
        pytest.skip("no GPU")
    import copy

    from edl_amd.ops.conv import Conv2dFast

    torch.manual_seed(51)
    m1 = Conv2dFast(32, 32, 3, padding=1, bias=False).cuda()
    m2 = copy.deepcopy(m1)
    x = torch.randn(4, 32, 32, 32, device="cuda").to(torch.bfloat16)
    x = x.contiguous(memory_format=torch.channels_last).requires_grad_(True)

    monkeypatch.setenv("EDL_STEM_WGRAD", "tn")
    y1 = m1(x)
    y1.float().square().mean().backward()
    monkeypatch.delenv("EDL_STEM_WGRAD")
    x2 = x.detach().clone().requires_grad_(True)
    y2 = m2(x2)
    y2.float().square().mean().backward()
    assert torch.allclose(m1.weight.grad, m2.weight.grad, atol=1e-3,
                          rtol=1e-2), \
        (m1.weight.grad - m2.weight.grad).abs().max().item()
