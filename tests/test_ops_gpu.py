"""GPU numerics tests: HIP kernels vs plain PyTorch fp32 references.

Every test here requires an MI355X (run via gpurun / driver round-end)."""
import pytest
import torch

pytestmark = pytest.mark.gpu

if torch.cuda.is_available():
    from edl_amd import ops
    from edl_amd.ops.functional import kd_soft_cross_entropy
else:
    ops = None


@pytest.fixture(scope="module", autouse=True)
def _require_gpu():
    if not torch.cuda.is_available():
        pytest.skip("no GPU")
    # loud check: the HIP extension must be present on a GPU box
    assert ops.available(), "edl_amd._C missing on a GPU box"


def test_fused_sgd_matches_reference():
    torch.manual_seed(0)
    n = 1 << 20 | 3  # odd tail exercises the scalar path
    p = torch.randn(n, device="cuda")
    g = torch.randn(n, device="cuda")
    m = torch.randn(n, device="cuda")
    pr, gr, mr = p.clone(), g.clone(), m.clone()
    lr, mu, wd, scale = 0.1, 0.9, 1e-4, 0.125

    ops.ext().fused_sgd(p, g, m, lr, mu, wd, scale)
    # fp32 torch reference
    d = gr * scale + wd * pr
    mref = mu * mr + d
    pref = pr - lr * mref
    torch.cuda.synchronize()
    assert torch.allclose(m, mref, atol=1e-6), (m - mref).abs().max().item()
    assert torch.allclose(p, pref, atol=1e-6), (p - pref).abs().max().item()


@pytest.mark.parametrize("dtype", [torch.float32, torch.bfloat16])
@pytest.mark.parametrize("shape", [(32, 1000), (7, 1000), (256, 1000), (4, 13)])
def test_kd_ce_forward_backward(dtype, shape):
    torch.manual_seed(1)
    B, C = shape
    s32 = (torch.randn(B, C, device="cuda") * 4).float()
    t32 = (torch.randn(B, C, device="cuda") * 4).float()
    s = s32.to(dtype).requires_grad_(True)
    t = t32.to(dtype)

    loss = kd_soft_cross_entropy(s, t)
    loss.backward()

    sref = s32.detach().to(dtype).float().requires_grad_(True)
    tref = t32.detach().to(dtype).float()
    lref = -(torch.softmax(tref, 1) * torch.log_softmax(sref, 1)).sum(1).mean()
    lref.backward()

    tol = 1e-5 if dtype == torch.float32 else 2e-2
    assert torch.allclose(loss.float(), lref.detach(), atol=tol, rtol=tol), \
        (loss.item(), lref.item())
    assert torch.allclose(s.grad.float(), sref.grad, atol=tol, rtol=tol), \
        (s.grad.float() - sref.grad).abs().max().item()


def test_kd_ce_extreme_logits_stable():
    """Large-magnitude logits must not overflow (online max-subtraction)."""
    B, C = 16, 1000
    s = (torch.randn(B, C, device="cuda") * 60).requires_grad_(True)
    t = torch.randn(B, C, device="cuda") * 60
    loss = kd_soft_cross_entropy(s, t)
    assert torch.isfinite(loss)
    loss.backward()
    assert torch.isfinite(s.grad).all()


def test_engine_gpu_step():
    from edl_amd.data.synthetic import SyntheticImageNet
    from edl_amd.train.engine import TrainerEngine

    eng = TrainerEngine(model="resnet50_vd", per_device_batch=8, base_lr=0.01,
                        checkpoint_dir=None, use_hip_ops=True,
                        graph_capture=False).setup()
    loader = SyntheticImageNet(8, eng.device, channels_last=True)
    x, y = loader.next()
    l0 = eng.train_step(x, y)
    torch.cuda.synchronize()
    assert torch.isfinite(l0)


def test_engine_graph_capture_subprocess():
    """hipGraph capture of the full step. Run in a FRESH process: capture
    after unrelated CUDA activity in the same process can crash the
    runtime (observed with pytest-ordered tests), and a segfault must not
    take down the whole suite."""
    import json
    import os
    import subprocess
    import sys

    repo = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
    out = subprocess.run(
        [sys.executable, os.path.join(repo, "bench.py"), "--steps", "3",
         "--warmup", "1", "--batch_size", "8", "--graph_capture", "1"],
        capture_output=True, text=True, timeout=600, cwd=repo,
    )
    assert out.returncode == 0, out.stdout + out.stderr
    line = [l for l in out.stdout.splitlines() if l.startswith("{")][-1]
    res = json.loads(line)
    assert res["config"]["graph_capture"] is True
    assert res["value"] > 0


def test_fused_sgd_trains_resnet_gpu(tmp_path):
    """End-to-end: a few engine steps reduce loss on a fixed batch."""
    from edl_amd.train.engine import TrainerEngine

    eng = TrainerEngine(model="resnet18_vd", per_device_batch=8, num_classes=10,
                        base_lr=0.01, use_hip_ops=True, graph_capture=False,
                        checkpoint_dir=None).setup()
    x = torch.randn(8, 3, 64, 64, device="cuda").contiguous(
        memory_format=torch.channels_last)
    y = torch.randint(0, 10, (8,), device="cuda")
    l0 = eng.train_step(x, y).item()
    for _ in range(10):
        loss = eng.train_step(x, y)
    torch.cuda.synchronize()
    assert loss.item() < l0


def test_full_model_memorizes():
    """End-to-end gradient correctness through EVERY custom kernel:
    ResNet50_vd must drive a fixed 64-sample synthetic set to ~zero loss
    (verified: both the custom path and the torch path reach acc 1.0 by
    400 steps; 100-step snapshots differ only within seed-level
    trajectory noise — gpurun_out/of_*)."""
    import json
    import os
    import subprocess
    import sys

    repo = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
    out = subprocess.run(
        [sys.executable, os.path.join(repo, "tools", "overfit_check.py"),
         "--steps", "400"],
        capture_output=True, text=True, timeout=280, cwd=repo)
    assert out.returncode == 0, out.stdout + out.stderr
    res = json.loads([l for l in out.stdout.splitlines() if l.startswith("{")][-1])
    assert res["acc_end"] > 0.95


@pytest.mark.parametrize("H,W", [(56, 56), (7, 7), (9, 11)])
def test_avgpool2x2_numerics(H, W):
    from edl_amd.ops.pool import AvgPool2x2

    torch.manual_seed(4)
    pool = AvgPool2x2().cuda()
    x = torch.randn(3, 64, H, W, device="cuda").to(torch.bfloat16)
    x = x.contiguous(memory_format=torch.channels_last).requires_grad_(True)
    y = pool(x)
    g = torch.randn_like(y).contiguous(memory_format=torch.channels_last)
    y.backward(g)

    xr = x.detach().float().requires_grad_(True)
    yr = torch.nn.AvgPool2d(2, 2, ceil_mode=True)(xr)
    yr.backward(g.float())
    assert y.shape == yr.shape
    assert torch.allclose(y.float(), yr.detach(), atol=2e-2, rtol=2e-2), \
        (y.float() - yr.detach()).abs().max().item()
    assert torch.allclose(x.grad.float(), xr.grad, atol=2e-2, rtol=2e-2), \
        (x.grad.float() - xr.grad).abs().max().item()


@pytest.mark.parametrize("seed", [0, 1])
def test_direct_grad_matches_standard(seed, monkeypatch):
    """World-1 direct-grad mode (kernels accumulate into bucket-view grads,
    autograd sees None) must produce the same gradients as the standard
    AccumulateGrad path."""
    import torch.nn as nn

    from edl_amd.ops.bnrelu import BNReLU2d
    from edl_amd.ops.conv import Conv2dFast
    from edl_amd.train.bucketed_ddp import BucketedAllReducer

    def run(direct):
        monkeypatch.setenv("EDL_DIRECT_GRAD", "1" if direct else "0")
        torch.manual_seed(seed)
        m = nn.Sequential(
            Conv2dFast(64, 128, 1, bias=False),
            BNReLU2d(128),
            Conv2dFast(128, 64, 3, padding=1, bias=False),
            BNReLU2d(64, act=False),
        ).cuda().to(memory_format=torch.channels_last)
        red = BucketedAllReducer(list(m.parameters()), flatten_params=True)
        red.zero_grad()
        torch.manual_seed(123)
        x = (torch.randn(4, 64, 12, 12, device="cuda")
             .to(torch.bfloat16).contiguous(memory_format=torch.channels_last))
        y = x
        for mod in m:
            y = mod(y)
        y.float().pow(2).mean().backward()
        return [p.grad.detach().clone() for p in m.parameters()]

    g_std = run(False)
    g_dir = run(True)
    for a, b in zip(g_std, g_dir):
        assert torch.allclose(a, b, atol=1e-3, rtol=1e-3), \
            (a - b).abs().max().item()


@pytest.mark.parametrize("shape", [(2, 64, 112, 112), (1, 64, 57, 57),
                                   (3, 32, 14, 15)])
def test_maxpool3x3s2_numerics(shape):
    """MaxPool3x3s2 fwd+bwd vs nn.MaxPool2d(3,2,1) fp32 reference."""
    import torch.nn.functional as F

    from edl_amd.ops.pool import MaxPool3x3s2

    n, c, h, w = shape
    torch.manual_seed(9)
    x = torch.randn(n, c, h, w, device="cuda").to(torch.bfloat16)
    x = x.contiguous(memory_format=torch.channels_last).requires_grad_(True)
    y = MaxPool3x3s2()(x)
    g = torch.randn_like(y).contiguous(memory_format=torch.channels_last)
    y.backward(g)

    xr = x.detach().float().requires_grad_(True)
    yr = F.max_pool2d(xr, 3, 2, 1)
    yr.backward(g.float())
    assert y.shape == yr.shape
    assert torch.equal(y.float(), yr.detach())  # max of the same values
    # bwd: when a window's max TIES (bf16 values), routing the grad to a
    # different tied position is an equally valid subgradient — compare
    # conservation (every dy lands exactly once) + mismatch sparsity
    # bf16-rounded dx elements: channel sums accumulate ~sqrt(M)*0.4%
    # rounding noise on top of tie effects — compare loosely
    assert torch.allclose(x.grad.float().sum(dim=(0, 2, 3)),
                          xr.grad.sum(dim=(0, 2, 3)), rtol=5e-2, atol=2.0)
    err = (x.grad.float() - xr.grad).abs()
    frac = float((err > 1e-3).float().mean().item())
    assert frac < 0.05, frac  # only tie sites may differ (bf16 ties ~3%)
