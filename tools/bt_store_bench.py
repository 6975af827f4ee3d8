"""Per-shape A/B of the gemm_bt epilogue store path (EDL_BT_STORE_LDS).

Times the flagship 1x1-conv shapes (fwd + dgrad both run gemm_bt) with the
legacy 2-B stores vs the LDS-bounce dwordx4 stores. The env is read once
(static) per process, so each variant runs in its own subprocess.
"""
import json
import os
import subprocess
import sys

SHAPES = [  # (M, N, K): resnet50_vd bs32 1x1 shapes (fwd y=x@W^T)
    (32 * 56 * 56, 64, 64),
    (32 * 56 * 56, 64, 256),
    (32 * 56 * 56, 256, 64),
    (32 * 28 * 28, 128, 512),
    (32 * 28 * 28, 512, 128),
    (32 * 14 * 14, 256, 1024),
    (32 * 14 * 14, 1024, 256),
    (32 * 7 * 7, 512, 2048),
    (32 * 7 * 7, 2048, 512),
]


def worker():
    import torch

    from edl_amd.ops import ext

    torch.manual_seed(0)
    out = []
    for M, N, K in SHAPES:
        a = torch.randn(M, K, device="cuda", dtype=torch.bfloat16)
        b = torch.randn(N, K, device="cuda", dtype=torch.bfloat16)
        for _ in range(3):
            c = ext().gemm_bt(a, b)
        torch.cuda.synchronize()
        st = torch.cuda.Event(True)
        en = torch.cuda.Event(True)
        st.record()
        for _ in range(50):
            c = ext().gemm_bt(a, b)
        en.record()
        torch.cuda.synchronize()
        us = st.elapsed_time(en) * 1000 / 50
        ref = (a.float() @ b.float().t()).to(torch.bfloat16)
        # bf16 accumulation error grows ~sqrt(K) on random data
        tol = 0.06 * (K ** 0.5)
        ok = bool((c.float() - ref.float()).abs().max().item() < tol)
        out.append({"shape": [M, N, K], "us": round(us, 2), "ok": ok})
    print(json.dumps({"ldsb": os.environ.get("EDL_BT_STORE_LDS", "0"),
                      "shapes": out}))


if __name__ == "__main__":
    if len(sys.argv) > 1 and sys.argv[1] == "--worker":
        worker()
        sys.exit(0)
    for v in ("0", "1"):
        repo = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
        env = dict(os.environ, EDL_BT_STORE_LDS=v,
                   PYTHONPATH=repo + os.pathsep + os.environ.get("PYTHONPATH", ""))
        r = subprocess.run([sys.executable, __file__, "--worker"], env=env,
                           cwd=repo, capture_output=True, text=True, timeout=300)
        sys.stdout.write(r.stdout)
        if r.returncode != 0:
            sys.stderr.write(r.stderr[-2000:])
