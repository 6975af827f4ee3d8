"""Convergence sanity: ResNet50_vd must memorize a small fixed synthetic
set through the FULL custom path (MFMA convs fwd/dgrad/wgrad, fused BN,
fused SGD, bf16 autocast). Verifies end-to-end gradient correctness beyond
per-op numerics tests.

    python tools/overfit_check.py [--steps 150]
Prints one JSON line {"loss0":..., "loss_end":..., "acc_end":...}.
"""
import argparse
import json
import os
import sys

import torch

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
from edl_amd.train.engine import TrainerEngine  # noqa: E402


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--steps", type=int, default=400)
    ap.add_argument("--batch", type=int, default=64)
    ap.add_argument("--classes", type=int, default=16)
    ap.add_argument("--seed", type=int, default=7)
    args = ap.parse_args()

    eng = TrainerEngine(model="resnet50_vd", per_device_batch=args.batch,
                        num_classes=args.classes, base_lr=0.02,
                        label_smoothing=0.0, weight_decay=0.0,
                        use_hip_ops=torch.cuda.is_available(),
                        graph_capture=False, checkpoint_dir=None).setup()
    dev = eng.device
    torch.manual_seed(args.seed)
    x = torch.randn(args.batch, 3, 112, 112, device=dev)
    if dev.type == "cuda":
        x = x.to(torch.bfloat16).contiguous(memory_format=torch.channels_last)
    y = torch.randint(0, args.classes, (args.batch,), device=dev)
    eng.model.train()
    eng.set_lr(0.02)
    loss0 = None
    curve = []
    for i in range(args.steps):
        loss = eng.train_step(x, y)
        if i == 0:
            loss0 = loss.item()
        if i % 25 == 0 or i == args.steps - 1:
            curve.append(round(loss.item(), 3))
    if dev.type == "cuda":
        torch.cuda.synchronize()
    with torch.no_grad():
        with torch.autocast(dev.type, torch.bfloat16, enabled=dev.type == "cuda"):
            logits = eng.model(x)
        acc = (logits.argmax(1) == y).float().mean().item()
    print(json.dumps({"loss0": round(loss0, 4), "loss_end": round(loss.item(), 4),
                      "acc_end": round(acc, 4), "steps": args.steps,
                      "curve": curve}))
    assert loss.item() < 0.1, "did not learn"
    assert acc > 0.95, "did not memorize"


if __name__ == "__main__":
    main()
