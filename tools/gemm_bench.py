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


def main_wgrad():
    """Wgrad: direct TN kernel vs the transpose_pad + bt_splitk pipeline.
    Shapes are the ResNet50_vd 1x1 wgrads at bs32: C[Cout,Cin] over K=M."""
    e = ops.ext()
    print("%-28s %10s %10s %8s" % ("shape (K,N1,N2)", "tn", "pad+bt", "ratio"))
    for M, Cout, Cin in SHAPES:
        a = torch.randn(M, Cout, device="cuda").to(torch.bfloat16)
        b = torch.randn(M, Cin, device="cuda").to(torch.bfloat16)
        flops = 2.0 * M * Cout * Cin
        t_tn = bench(lambda: e.gemm_tn_splitk(a, b, 0))
        t_bt = bench(lambda: e.gemm_bt_splitk(
            e.transpose_pad(a), e.transpose_pad(b), 0))
        print("%-28s %7.1f TF %7.1f TF %7.2fx" % (
            str((M, Cout, Cin)), flops / t_tn / 1e12, flops / t_bt / 1e12,
            t_bt / t_tn))


def main_wgrad3():
    """conv3x3 wgrad: direct TN gather vs transpose_pad+shift9+bt.
    (N, Cin, H, W, Cout, stride) = bs32 ResNet50_vd 3x3 convs."""
    e = ops.ext()
    shapes = [
        (32, 64, 56, 56, 64, 1),
        (32, 128, 28, 28, 128, 1), (32, 128, 56, 56, 128, 2),
        (32, 256, 14, 14, 256, 1), (32, 256, 28, 28, 256, 2),
        (32, 512, 7, 7, 512, 1), (32, 512, 14, 14, 512, 2),
    ]
    print("%-28s %10s %10s %8s" % ("(N,C,H,W,s)", "tn3x3", "old", "ratio"))
    for n, ci, h, w, co, s in shapes:
        x = torch.randn(n, ci, h, w, device="cuda").to(torch.bfloat16)
        x = x.contiguous(memory_format=torch.channels_last)
        ho, wo = (h - 1) // s + 1, (w - 1) // s + 1
        dy = torch.randn(n * ho * wo, co, device="cuda").to(torch.bfloat16)
        flops = 2.0 * n * ho * wo * co * 9 * ci
        t_tn = bench(lambda: e.gemm_tn3x3_splitk(dy, x, s, 0))
        t_bt = bench(lambda: e.gemm_bt_splitk(
            e.transpose_pad(dy), e.conv3x3_wgrad_operand(x, s), 0))
        print("%-28s %7.1f TF %7.1f TF %7.2fx" % (
            str((n, ci, h, w, s)), flops / t_tn / 1e12, flops / t_bt / 1e12,
            t_bt / t_tn))


if __name__ == "__main__":
    if "--wgrad" in sys.argv:
        main_wgrad()
    elif "--wgrad3" in sys.argv:
        main_wgrad3()
    else:
        main()
