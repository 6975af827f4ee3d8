#!/usr/bin/env python3
"""Teacher (ResNeXt101_32x16d_wsl) forward A/B: grouped-conv routing.

The distill headline (reference 656 img/s shared-GPU, README.md:84) is
bottlenecked by the teacher forward (~50 ms/bs32 on MIOpen in r1). This
sweeps EDL_CONV3X3_GROUPED_MINC — which channels-per-group widths run the
in-repo grouped implicit-GEMM kernel vs MIOpen — and checks numerics of
each variant against the all-MIOpen forward.

    python tools/teacher_fwd_bench.py [--batch 16] [--iters 10]
"""
import argparse
import json
import os
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--batch", type=int, default=16)
    ap.add_argument("--iters", type=int, default=10)
    ap.add_argument("--warmup", type=int, default=4)
    ap.add_argument("--model", default="resnext101_32x16d_wsl")
    args = ap.parse_args()

    import torch

    import edl_amd.ops.conv as conv_mod
    from edl_amd.models import build_model

    torch.manual_seed(7)
    # same prep as distill/teacher_server.py: fp32 params, eval,
    # channels_last, bf16 autocast forward
    m = build_model(args.model).cuda()
    m = m.to(memory_format=torch.channels_last).eval()
    x = torch.randn(args.batch, 3, 224, 224, device="cuda")
    x = x.contiguous(memory_format=torch.channels_last)

    results = {}
    ref = None
    for minc in (9999, 128, 64, 32, 16):
        conv_mod._GROUPED_MINC = minc
        with torch.no_grad(), torch.autocast("cuda", torch.bfloat16):
            for _ in range(args.warmup):
                y = m(x)
            torch.cuda.synchronize()
            t0 = time.monotonic()
            for _ in range(args.iters):
                y = m(x)
            torch.cuda.synchronize()
            dt = (time.monotonic() - t0) / args.iters
        if minc == 9999:
            ref = y.float()
            err = 0.0
        else:
            d = (y.float() - ref).abs()
            err = float((d / ref.abs().mean().clamp(min=0.05)).max().item())
        results["minc%d" % minc] = {
            "ms_per_fwd": round(dt * 1000, 3),
            "img_per_s": round(args.batch / dt, 1),
            "rel_err_vs_miopen": round(err, 5),
        }
        print("[minc=%4d] %7.3f ms/fwd  %8.1f img/s  err=%.4g" %
              (minc, dt * 1000, args.batch / dt, err), flush=True)

    print(json.dumps({"bench": "teacher_fwd", "model": args.model,
                      "batch": args.batch, "results": results}))


if __name__ == "__main__":
    main()
