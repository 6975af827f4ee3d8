"""Microbench: gemm_bt vs torch.matmul (hipBLASLt) on the ResNet50_vd
conv1x1 shapes at bs32. Run on a GPU box:
    python tools/gemm_bench.py
"""
import os
import sys
import time

import torch

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
from edl_amd import ops  # noqa: E402

# (M=N*H*W, N=Cout, K=Cin) for bs32 ResNet50_vd 1x1 convs (fwd)
SHAPES = [
    (32 * 56 * 56, 64, 64),
    (32 * 56 * 56, 256, 64),
    (32 * 56 * 56, 64, 256),
    (32 * 28 * 28, 512, 256),
    (32 * 28 * 28, 128, 512),
    (32 * 14 * 14, 1024, 512),
    (32 * 14 * 14, 256, 1024),
    (32 * 7 * 7, 2048, 1024),
    (32 * 7 * 7, 512, 2048),
]


def bench(fn, iters=30, warmup=10):
    for _ in range(warmup):
        fn()
    torch.cuda.synchronize()
    t0 = time.monotonic()
    for _ in range(iters):
        fn()
    torch.cuda.synchronize()
    return (time.monotonic() - t0) / iters


def main():
    print("%-28s %10s %10s %8s" % ("shape (M,N,K)", "gemm_bt", "hipBLASLt", "ratio"))
    for M, N, K in SHAPES:
        a = torch.randn(M, K, device="cuda").to(torch.bfloat16)
        b = torch.randn(N, K, device="cuda").to(torch.bfloat16)
        flops = 2.0 * M * N * K
        t_hip = bench(lambda: ops.ext().gemm_bt(a, b))
        bt = b.t().contiguous()
        t_blas = bench(lambda: a @ b.t())
        print("%-28s %7.1f TF %7.1f TF %7.2fx" % (
            str((M, N, K)), flops / t_hip / 1e12, flops / t_blas / 1e12,
            t_blas / t_hip))


if __name__ == "__main__":
    main()
