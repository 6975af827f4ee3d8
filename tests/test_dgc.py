"""DGC sparse all-reduce tests: single-process semantics + 2-rank gloo
exchange + convergence-preserving error feedback."""
import json
import os
import subprocess
import sys

import pytest
import torch

from edl_amd.train.bucketed_ddp import BucketedAllReducer
from edl_amd.train.dgc import DGCCompressor

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))


def test_dgc_disabled_single_process():
    m = torch.nn.Linear(16, 16)
    r = BucketedAllReducer(m.parameters(), bucket_cap_mb=1)
    dgc = DGCCompressor(r, compress_ratio=0.1)
    m(torch.randn(4, 16)).sum().backward()
    g0 = [p.grad.clone() for p in m.parameters()]
    dgc.step()  # world 1: no-op
    for p, g in zip(m.parameters(), g0):
        assert torch.equal(p.grad, g)


WORKER = r"""
import json, os, sys
import torch
import torch.distributed as dist
sys.path.insert(0, os.environ["EDL_REPO"])
from edl_amd.train.bucketed_ddp import BucketedAllReducer
from edl_amd.train.dgc import DGCCompressor

rank = int(os.environ["RANK"]); world = int(os.environ["WORLD_SIZE"])
dist.init_process_group("gloo", rank=rank, world_size=world)
torch.manual_seed(3)
m = torch.nn.Linear(64, 32)
r = BucketedAllReducer(m.parameters(), bucket_cap_mb=1)
dgc = DGCCompressor(r, compress_ratio=0.5)

torch.manual_seed(50 + rank)
x = torch.randn(8, 64)
results = []
for step in range(3):
    r.zero_grad()
    (m(x) ** 2).mean().backward()
    dgc.step()
    # after exchange: buckets hold the SUM over ranks of transmitted values
    results.append(float(sum(b.buffer.abs().sum() for b in r._buckets)))
# error feedback: residual mass is nonzero (half the coords withheld)
res_mass = float(sum(t.abs().sum() for t in dgc._residuals))
# buckets identical across ranks after exchange
flat = torch.cat([b.buffer for b in r._buckets])
gathered = [torch.empty_like(flat) for _ in range(world)]
dist.all_gather(gathered, flat)
same = all(torch.allclose(g, flat, atol=1e-6) for g in gathered)
if rank == 0:
    print("RESULT " + json.dumps({"res_mass": res_mass, "same": same}))
dist.destroy_process_group()
"""


def test_dgc_two_rank_exchange(tmp_path):
    sp = tmp_path / "w.py"
    sp.write_text(WORKER)
    env = dict(os.environ)
    env.update({"EDL_REPO": REPO, "CUDA_VISIBLE_DEVICES": ""})
    out = subprocess.run(
        [sys.executable, "-m", "torch.distributed.run", "--nnodes=1",
         "--nproc-per-node", "2", "--master-addr", "127.0.0.1",
         "--master-port", "29533", "--no-python", sys.executable, str(sp)],
        env=env, capture_output=True, text=True, timeout=180)
    assert out.returncode == 0, out.stdout + out.stderr
    line = [l for l in out.stdout.splitlines() if l.startswith("RESULT ")][0]
    res = json.loads(line[len("RESULT "):])
    assert res["same"], "ranks diverged after DGC exchange"
    assert res["res_mass"] > 0, "error feedback should withhold mass"


def test_dgc_error_feedback_accumulates():
    """A coordinate skipped this step must carry over and eventually send."""
    torch.manual_seed(0)
    m = torch.nn.Linear(8, 8, bias=False)
    r = BucketedAllReducer(m.parameters(), bucket_cap_mb=1)
    dgc = DGCCompressor(r, compress_ratio=0.1)
    # emulate the exchange path locally by forcing 'enabled'
    r.zero_grad()
    (m(torch.ones(2, 8)) ** 2).mean().backward()
    res_before = dgc._residuals[0].clone()
    assert res_before.abs().sum() == 0
    # world=1 -> step() no-ops, so exercise the residual math directly
    dgc._residuals[0].add_(r._buckets[0].buffer)
    n = dgc._residuals[0].numel()
    k = max(1, int(n * dgc.compress_ratio))
    _, idx = torch.topk(dgc._residuals[0].abs(), k, sorted=False)
    dgc._residuals[0][idx] = 0
    remaining = dgc._residuals[0].abs().sum()
    total = r._buckets[0].buffer.abs().sum()
    assert 0 < remaining < total
