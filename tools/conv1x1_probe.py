"""Compare y / dx / dw of ONE 1x1 conv across paths (miopen / matmul /
hip) against an fp64 reference at training-realistic magnitudes."""
import os
import sys

import torch
import torch.nn.functional as F

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import edl_amd.ops.conv as convmod  # noqa: E402
from edl_amd.ops.conv import Conv2dFast  # noqa: E402


def metrics(a, ref):
    a, ref = a.double().flatten(), ref.double().flatten()
    cos = torch.nn.functional.cosine_similarity(a, ref, dim=0).item()
    rel = ((a - ref).abs().mean() / ref.abs().mean().clamp(min=1e-12)).item()
    return cos, rel


def run(path, x0, w0, g0):
    convmod._CONV1X1 = path
    convmod.bump_weight_epoch()
    conv = Conv2dFast(w0.shape[1], w0.shape[0], 1, bias=False).cuda()
    with torch.no_grad():
        conv.weight.copy_(w0.view_as(conv.weight))
    x = x0.clone().requires_grad_(True)
    with torch.autocast("cuda", torch.bfloat16):
        y = conv(x)
    y.backward(g0.to(y.dtype))
    return y.detach().float(), x.grad.float(), conv.weight.grad.float()


def main():
    torch.manual_seed(0)
    n, cin, cout, hw = 64, 256, 128, 28
    # post-BN-ReLU-like input, kaiming-like weights, CE-scale grads
    x0 = torch.relu(torch.randn(n, cin, hw, hw, device="cuda")) \
        .contiguous(memory_format=torch.channels_last)
    w0 = torch.randn(cout, cin, 1, 1, device="cuda") * (2.0 / cin) ** 0.5
    g0 = torch.randn(n, cout, hw, hw, device="cuda").contiguous(
        memory_format=torch.channels_last) * 1e-3

    # fp64 reference
    xr = x0.double().requires_grad_(True)
    wr = w0.double().requires_grad_(True)
    yr = F.conv2d(xr, wr)
    yr.backward(g0.double())

    for path in ("miopen", "matmul", "hip"):
        y, dx, dw = run(path, x0, w0, g0)
        print("%-7s y: cos=%.6f rel=%.4f | dx: cos=%.6f rel=%.4f | "
              "dw: cos=%.6f rel=%.4f" % (
                  path, *metrics(y, yr.detach()), *metrics(dx, xr.grad),
                  *metrics(dw, wr.grad)))


if __name__ == "__main__":
    main()
