#!/usr/bin/env python3
"""BN backward microbench — per-shape, per-variant.

Measured r1: BN backward = ~2.7 ms/step of a 10.4 ms step; roofline
(read dy+x+mask twice, write dx once at 6.3 TB/s) is ~0.62 ms. This
sweeps the ResNet50_vd bs32 BN shapes across:
  * EDL_BN_BWD_GRID_CAP   (reduce grid: 192 = r1 default / 384 / 768)
  * EDL_BN_BWD_STREAMS    (4 = r1 / 8 row streams)
  * EDL_BN_FIN_V2         (strip-parallel finalize vs v1)
and reports us + achieved GB/s vs roofline per shape.

    python tools/bn_bwd_bench.py [--iters 20]
"""
import argparse
import itertools
import json
import os
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

# (M, C, relu, add) for ResNet50_vd bs32 — one row per distinct shape,
# weight = how many times it runs per backward
SHAPES = [
    (32 * 112 * 112, 32, True, False, 2),    # stem conv0/conv1 BN
    (32 * 112 * 112, 64, True, False, 1),    # stem conv2 BN
    (32 * 56 * 56, 64, True, False, 6),      # stage1 bottleneck 1x1/3x3
    (32 * 56 * 56, 256, True, True, 3),      # stage1 tail BNAdd
    (32 * 56 * 56, 256, False, False, 1),    # stage1 proj BN
    (32 * 28 * 28, 128, True, False, 8),
    (32 * 28 * 28, 512, True, True, 4),
    (32 * 28 * 28, 512, False, False, 1),
    (32 * 14 * 14, 256, True, False, 12),
    (32 * 14 * 14, 1024, True, True, 6),
    (32 * 14 * 14, 1024, False, False, 1),
    (32 * 7 * 7, 512, True, False, 6),
    (32 * 7 * 7, 2048, True, True, 3),
    (32 * 7 * 7, 2048, False, False, 1),
]


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--iters", type=int, default=20)
    ap.add_argument("--warmup", type=int, default=5)
    args = ap.parse_args()

    import torch

    from edl_amd import ops

    e = ops.ext()
    torch.manual_seed(3)

    # (cap, streams, fin_v2, contig)
    variants = [("192", "4", "0", "0"),  # r1 baseline
                ("192", "4", "1", "0"),  # r2 shipped default
                ("192", "4", "1", "1"),  # contiguous block rows
                ("384", "4", "1", "1"),
                ("768", "4", "1", "1"),
                ("192", "8", "1", "1")]

    tensors = {}
    for (M, C, relu, add, w) in SHAPES:
        dy = torch.randn(M, C, device="cuda").to(torch.bfloat16)
        x = torch.randn(M, C, device="cuda").to(torch.bfloat16)
        mask = torch.randint(0, 256, (M * C // 8,), device="cuda",
                             dtype=torch.uint8) if relu else None
        mean = torch.zeros(C, device="cuda")
        invstd = torch.ones(C, device="cuda")
        gamma = torch.ones(C, device="cuda")
        tensors[(M, C, relu, add)] = (dy, mask, x, mean, invstd, gamma)

    summary = {}
    for cap, st, fv2, contig in variants:
        os.environ["EDL_BN_BWD_GRID_CAP"] = cap
        os.environ["EDL_BN_BWD_STREAMS"] = st
        os.environ["EDL_BN_FIN_V2"] = fv2
        os.environ["EDL_BN_CONTIG"] = contig
        key = "cap%s_st%s_fin%s%s" % (cap, st, "v2" if fv2 == "1" else "v1",
                                      "_contig" if contig == "1" else "")
        total_us = 0.0
        rows = []
        for (M, C, relu, add, w) in SHAPES:
            dy, mask, x, mean, invstd, gamma = tensors[(M, C, relu, add)]
            for _ in range(args.warmup):
                e.bn_bwd(dy, mask, x, mean, invstd, gamma, relu, add, True,
                         None, None)
            torch.cuda.synchronize()
            t0 = time.monotonic()
            for _ in range(args.iters):
                e.bn_bwd(dy, mask, x, mean, invstd, gamma, relu, add, True,
                         None, None)
            torch.cuda.synchronize()
            us = (time.monotonic() - t0) / args.iters * 1e6
            elems = M * C
            gbytes = elems * (4.125 * 2 + 2 + (2 if add else 0)) / 1e9
            rows.append((M, C, relu, add, us, gbytes / (us * 1e-6 + 1e-12)))
            total_us += us * w
        summary[key] = {"total_us_per_step": round(total_us, 1)}
        print("== %s: weighted total %.0f us/step ==" % (key, total_us),
              flush=True)
        for (M, C, relu, add, us, gbps) in rows:
            print("  M=%7d C=%4d relu=%d add=%d  %7.1f us  %6.0f GB/s" %
                  (M, C, relu, add, us, gbps))

    print(json.dumps({"bench": "bn_bwd", "variants": summary}))


if __name__ == "__main__":
    main()
