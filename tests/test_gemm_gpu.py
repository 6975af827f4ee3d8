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
