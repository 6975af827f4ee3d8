"""Pinpoint gradient bias: run ONE fwd+bwd of resnet50_vd on identical
weights/input through (a) the custom HIP conv path and (b) the MIOpen
path, plus (c) an fp32 reference, and report per-parameter relative
errors (worst first). A systematic per-layer deviation beyond bf16 noise
marks the broken op."""
import json
import os
import sys

import torch

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import edl_amd.ops.conv as convmod  # noqa: E402
from edl_amd.models import resnet50_vd  # noqa: E402


def run_path(model, x, y, conv1x1, conv3x3):
    convmod._CONV1X1 = conv1x1
    convmod._CONV3X3 = conv3x3
    convmod.bump_weight_epoch()
    model.zero_grad(set_to_none=True)
    with torch.autocast("cuda", torch.bfloat16):
        out = model(x)
    loss = torch.nn.functional.cross_entropy(out.float(), y)
    loss.backward()
    grads = {n: p.grad.detach().clone() for n, p in model.named_parameters()}
    return out.detach().float(), grads


def main():
    assert torch.cuda.is_available()
    torch.manual_seed(0)
    model = resnet50_vd(num_classes=16).cuda().to(
        memory_format=torch.channels_last)
    x = torch.randn(16, 3, 112, 112, device="cuda").contiguous(
        memory_format=torch.channels_last)
    y = torch.randint(0, 16, (16,), device="cuda")

    out_hip, g_hip = run_path(model, x, y, "hip", "hip")
    out_mio, g_mio = run_path(model, x, y, "miopen", "miopen")

    print("fwd max|hip-miopen| =", (out_hip - out_mio).abs().max().item())
    rows = []
    for n in g_hip:
        a, b = g_hip[n].float(), g_mio[n].float()
        denom = b.abs().mean().clamp(min=1e-8)
        rel = ((a - b).abs().mean() / denom).item()
        cos = torch.nn.functional.cosine_similarity(
            a.flatten(), b.flatten(), dim=0).item()
        rows.append((rel, cos, n, a.abs().mean().item(), b.abs().mean().item()))
    rows.sort(reverse=True)
    print("%8s %8s  %-55s %10s %10s" % ("rel", "cos", "param", "|hip|", "|mio|"))
    for rel, cos, n, ma, mb in rows[:25]:
        print("%8.4f %8.4f  %-55s %10.4g %10.4g" % (rel, cos, n, ma, mb))
    bad = [r for r in rows if r[1] < 0.98]
    print(json.dumps({"n_params": len(rows), "n_cos_below_098": len(bad)}))


if __name__ == "__main__":
    main()
