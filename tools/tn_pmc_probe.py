#!/usr/bin/env python3
"""Run the dominant wgrad gemm_tn shapes in a loop for PMC collection."""
import sys, os
sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import torch
from edl_amd import ops

e = ops.ext()
# (K=M, N1=Cout, N2=Cin): stage-2 1x1 expand wgrad + stage-3 deep wgrad
shapes = [(25088, 512, 128), (25088, 128, 512), (6272, 1024, 256),
          (1568, 2048, 1024)]
for M, n1, n2 in shapes:
    a = torch.randn(M, n1, device="cuda").to(torch.bfloat16)
    b = torch.randn(M, n2, device="cuda").to(torch.bfloat16)
    for _ in range(3):
        e.gemm_tn_splitk(a, b, 0)
    torch.cuda.synchronize()
    import time
    t0 = time.monotonic()
    for _ in range(20):
        e.gemm_tn_splitk(a, b, 0)
    torch.cuda.synchronize()
    dt = (time.monotonic() - t0) / 20
    gb = M * (n1 + n2) * 2 / 1e9
    print("shape (%d,%d,%d): %.1f us  %.0f GB/s-algorithmic  %.1f TF" %
          (M, n1, n2, dt * 1e6, gb / dt, 2.0 * M * n1 * n2 / dt / 1e12),
          flush=True)
