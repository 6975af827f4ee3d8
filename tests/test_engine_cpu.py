"""Training-engine tests on CPU: single-process numerics, multi-process
gloo world=2 gradient equivalence, checkpoint resume, bucketing."""
import json
import os
import subprocess
import sys

import pytest
import torch

from edl_amd.train.bucketed_ddp import BucketedAllReducer
from edl_amd.train.engine import TrainerEngine, piecewise_lr

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))


def test_bucketed_reducer_views_and_flatten():
    torch.manual_seed(0)
    m = torch.nn.Sequential(torch.nn.Linear(10, 20), torch.nn.Linear(20, 5))
    before = {n: p.clone() for n, p in m.named_parameters()}
    r = BucketedAllReducer(m.parameters(), bucket_cap_mb=1)
    # params preserved by the re-homing
    for n, p in m.named_parameters():
        assert torch.equal(p.data, before[n])
    x = torch.randn(4, 10)
    m(x).sum().backward()
    # grads are views into bucket buffers
    total = sum(b.buffer.numel() for b in r._buckets)
    assert total == sum(p.numel() for p in m.parameters())
    gnorm = sum(float(b.buffer.abs().sum()) for b in r._buckets)
    assert gnorm > 0
    r.zero_grad()
    for p in m.parameters():
        assert float(p.grad.abs().sum()) == 0.0


def test_fused_sgd_matches_torch_sgd():
    """FusedSGD (torch-fallback path on CPU, same math as the HIP kernel)
    must match torch.optim.SGD with weight decay + momentum."""
    torch.manual_seed(0)
    m1 = torch.nn.Linear(8, 8)
    m2 = torch.nn.Linear(8, 8)
    m2.load_state_dict(m1.state_dict())

    from edl_amd.ops.sgd import FusedSGD

    r = BucketedAllReducer(m1.parameters(), bucket_cap_mb=1)
    opt1 = FusedSGD(m1.parameters(), lr=0.1, momentum=0.9, weight_decay=1e-2,
                    reducer=r)
    opt2 = torch.optim.SGD(m2.parameters(), lr=0.1, momentum=0.9, weight_decay=1e-2)
    x = torch.randn(16, 8)
    for _ in range(5):
        r.zero_grad()
        m1(x).pow(2).mean().backward()
        opt1.step()
        opt2.zero_grad()
        m2(x).pow(2).mean().backward()
        opt2.step()
    for p1, p2 in zip(m1.parameters(), m2.parameters()):
        assert torch.allclose(p1, p2, atol=1e-6), (p1 - p2).abs().max()


def test_fused_sgd_state_dict_roundtrip():
    from edl_amd.ops.sgd import FusedSGD

    torch.manual_seed(0)
    m = torch.nn.Linear(4, 4)
    r = BucketedAllReducer(m.parameters(), bucket_cap_mb=1)
    opt = FusedSGD(m.parameters(), lr=0.1, reducer=r)
    r.zero_grad()
    m(torch.randn(2, 4)).sum().backward()
    opt.step()
    sd = opt.state_dict()
    m2 = torch.nn.Linear(4, 4)
    r2 = BucketedAllReducer(m2.parameters(), bucket_cap_mb=1)
    opt2 = FusedSGD(m2.parameters(), lr=0.5, reducer=r2)
    opt2.load_state_dict(sd)
    assert opt2.param_groups[0]["lr"] == 0.1
    for bk1, bk2 in zip(opt._materialize(), opt2._materialize()):
        assert torch.allclose(bk1["m"], bk2["m"])


def test_kd_loss_reference():
    from edl_amd.ops.functional import kd_soft_cross_entropy

    torch.manual_seed(0)
    s = torch.randn(8, 10, requires_grad=True)
    t = torch.randn(8, 10)
    loss = kd_soft_cross_entropy(s, t)
    # hand reference
    ref = -(torch.softmax(t, 1) * torch.log_softmax(s, 1)).sum(1).mean()
    assert torch.allclose(loss, ref, atol=1e-6)
    loss.backward()
    gref = (torch.softmax(s.detach(), 1) - torch.softmax(t, 1)) / s.shape[0]
    assert torch.allclose(s.grad, gref, atol=1e-6)


def test_piecewise_lr():
    assert piecewise_lr(0.1, 0, warmup_epochs=0) == pytest.approx(0.1)
    assert piecewise_lr(0.1, 35, warmup_epochs=0) == pytest.approx(0.01)
    assert piecewise_lr(0.1, 65, warmup_epochs=0) == pytest.approx(0.001)
    assert piecewise_lr(0.1, 95, warmup_epochs=0) == pytest.approx(1e-4)
    assert piecewise_lr(0.1, 2, warmup_epochs=5, step_in_epoch=0.5) == pytest.approx(0.05)


def test_engine_single_process_trains(tmp_path):
    eng = TrainerEngine(model="resnet18_vd", per_device_batch=2, num_classes=10,
                        dtype="fp32", channels_last=False, use_hip_ops=False,
                        base_lr=0.005, checkpoint_dir=str(tmp_path / "ck"))
    eng.setup()
    x = torch.randn(2, 3, 64, 64)
    y = torch.randint(0, 10, (2,))
    l0 = eng.train_step(x, y).item()
    for _ in range(5):
        loss = eng.train_step(x, y)
    assert loss.item() < l0  # it learns the batch
    eng.save_checkpoint(0, blocking=True)
    eng.ckpt.wait()

    # resume continues from epoch 1
    eng2 = TrainerEngine(model="resnet18_vd", per_device_batch=2, num_classes=10,
                         dtype="fp32", channels_last=False, use_hip_ops=False,
                         checkpoint_dir=str(tmp_path / "ck"))
    eng2.setup()
    assert eng2.start_epoch == 1
    assert eng2.global_step == eng.global_step
    for p1, p2 in zip(eng.model.parameters(), eng2.model.parameters()):
        assert torch.allclose(p1, p2)


WORKER = r"""
import json, os, sys
import torch
import torch.distributed as dist
sys.path.insert(0, os.environ["EDL_REPO"])
from edl_amd.train.bucketed_ddp import BucketedAllReducer
from edl_amd.ops.sgd import FusedSGD

def main():
    rank = int(os.environ["RANK"]); world = int(os.environ["WORLD_SIZE"])
    dist.init_process_group("gloo", rank=rank, world_size=world)
    torch.manual_seed(7)  # same init on both ranks
    m = torch.nn.Sequential(torch.nn.Linear(10, 64), torch.nn.ReLU(),
                            torch.nn.Linear(64, 4))
    r = BucketedAllReducer(m.parameters(), bucket_cap_mb=1)
    opt = FusedSGD(m.parameters(), lr=0.05, momentum=0.9, weight_decay=0.0,
                   grad_scale=r.grad_scale, reducer=r)
    # rank-dependent data: the reduced gradient must equal the full-batch grad
    torch.manual_seed(100 + rank)
    x = torch.randn(8, 10); y = torch.randn(8, 4)
    for _ in range(3):
        r.zero_grad()
        ((m(x) - y) ** 2).mean().backward()
        r.finalize()
        opt.step()
    out = {"params_sum": float(sum(p.sum() for p in m.parameters()))}
    if rank == 0:
        print("RESULT " + json.dumps(out), flush=True)
    dist.destroy_process_group()

main()
"""


def _run_world(n, script, tmp_path, extra_env=None):
    import torch.distributed.run  # noqa: F401

    sp = tmp_path / "worker.py"
    sp.write_text(script)
    env = dict(os.environ)
    env.update({"EDL_REPO": REPO, "CUDA_VISIBLE_DEVICES": ""})
    env.update(extra_env or {})
    out = subprocess.run(
        [sys.executable, "-m", "torch.distributed.run", "--nnodes=1",
         "--nproc-per-node", str(n), "--master-addr", "127.0.0.1",
         "--master-port", "29531", "--no-python", sys.executable, str(sp)],
        env=env, capture_output=True, text=True, timeout=180,
    )
    assert out.returncode == 0, out.stdout + out.stderr
    for line in out.stdout.splitlines():
        if line.startswith("RESULT "):
            return json.loads(line[len("RESULT "):])
    raise AssertionError("no RESULT line:\n" + out.stdout + out.stderr)


def test_bucketed_allreduce_world2_matches_fullbatch(tmp_path):
    """2-rank DDP with rank-split data must equal single-process training on
    the concatenated batch (the all-reduce average IS the full-batch grad)."""
    res = _run_world(2, WORKER, tmp_path)

    # single-process reference over both ranks' data
    torch.manual_seed(7)
    m = torch.nn.Sequential(torch.nn.Linear(10, 64), torch.nn.ReLU(),
                            torch.nn.Linear(64, 4))
    torch.manual_seed(100)
    x0 = torch.randn(8, 10); y0 = torch.randn(8, 4)
    torch.manual_seed(101)
    x1 = torch.randn(8, 10); y1 = torch.randn(8, 4)
    opt = torch.optim.SGD(m.parameters(), lr=0.05, momentum=0.9)
    for _ in range(3):
        opt.zero_grad()
        # average of per-rank MSE losses == what DDP computes
        loss = (((m(x0) - y0) ** 2).mean() + ((m(x1) - y1) ** 2).mean()) / 2
        loss.backward()
        opt.step()
    with torch.no_grad():
        ref = float(sum(p.sum() for p in m.parameters()))
    assert res["params_sum"] == pytest.approx(ref, abs=1e-4)


def test_mid_epoch_step_resume(tmp_path):
    """Step-level checkpoint: resume INSIDE an epoch (beyond the
    reference's per-epoch granularity)."""
    eng = TrainerEngine(model="resnet18_vd", per_device_batch=2, num_classes=8,
                        dtype="fp32", channels_last=False, use_hip_ops=False,
                        base_lr=0.001, checkpoint_dir=str(tmp_path / "ck"))
    eng.setup()
    x = torch.randn(2, 3, 32, 32)
    y = torch.randint(0, 8, (2,))
    for _ in range(5):
        eng.train_step(x, y)
    eng.save_checkpoint(3, extra={"mid_epoch": True, "step_in_epoch": 5},
                        blocking=True)
    eng.ckpt.wait()

    eng2 = TrainerEngine(model="resnet18_vd", per_device_batch=2, num_classes=8,
                         dtype="fp32", channels_last=False, use_hip_ops=False,
                         base_lr=0.001, checkpoint_dir=str(tmp_path / "ck"))
    eng2.setup()
    assert eng2.start_epoch == 3
    assert eng2.start_step == 5
    assert eng2.global_step == 5
    # epoch-level checkpoints still resume at the NEXT epoch
    eng2.save_checkpoint(3, blocking=True)
    eng2.ckpt.wait()
    eng3 = TrainerEngine(model="resnet18_vd", per_device_batch=2, num_classes=8,
                         dtype="fp32", channels_last=False, use_hip_ops=False,
                         base_lr=0.001, checkpoint_dir=str(tmp_path / "ck"))
    eng3.setup()
    assert eng3.start_epoch == 4 and eng3.start_step == 0


def test_direct_grad_flags_world1_only(monkeypatch):
    """Reducer marks params for in-kernel grad accumulation only at
    world 1 (no collectives); EDL_DIRECT_GRAD=0 disables."""
    import torch

    from edl_amd.train.bucketed_ddp import BucketedAllReducer

    ps = [torch.nn.Parameter(torch.randn(8, 8)) for _ in range(3)]
    BucketedAllReducer(ps, flatten_params=True)
    assert all(p._edl_direct_grad for p in ps)

    monkeypatch.setenv("EDL_DIRECT_GRAD", "0")
    BucketedAllReducer(ps, flatten_params=True)
    assert not any(p._edl_direct_grad for p in ps)


def test_dynamic_loss_scaler_policy():
    from edl_amd.train.engine import DynamicLossScaler

    s = DynamicLossScaler(init_scale=1024.0, growth_interval=3)
    assert s.update(found_inf=False) and s.value == 1024.0
    assert not s.update(found_inf=True)          # overflow: skip + backoff
    assert s.value == 512.0
    for _ in range(3):
        assert s.update(found_inf=False)
    assert s.value == 1024.0                     # grew after interval


def test_fp16_overflow_skips_step():
    """With the scaler forced huge, gradients overflow fp32 -> the update
    is skipped, the scale backs off, params stay put; a sane scale then
    trains normally. (CPU runs fp32 compute; the scaler logic is the
    same one the fp16 GPU path uses.)"""
    import torch

    from edl_amd.train.engine import DynamicLossScaler, TrainerEngine

    eng = TrainerEngine(model="resnet18_vd", per_device_batch=2, dtype="fp32",
                        channels_last=False, use_hip_ops=False,
                        graph_capture=False, num_classes=4).setup()
    eng.scaler = DynamicLossScaler(init_scale=1e38, growth_interval=10)
    x = torch.randn(2, 3, 32, 32)
    y = torch.randint(0, 4, (2,))
    p0 = [p.detach().clone() for p in eng.model.parameters()]
    eng.train_step(x, y)
    assert eng.scaler.value < 1e38               # backed off
    for p, before in zip(eng.model.parameters(), p0):
        assert torch.equal(p.detach(), before)   # step skipped

    eng.scaler.value = 128.0
    eng.train_step(x, y)
    changed = any(not torch.equal(p.detach(), before)
                  for p, before in zip(eng.model.parameters(), p0))
    assert changed                               # normal scaled step applied


def test_channels_last_bucket_storage_equivalence():
    """Conv3x3 weights flagged for channels-last bucket storage
    ([Cout,3,3,Cin] physical, logical view unchanged) must train
    IDENTICALLY to the standard layout: same params after 3 steps, state
    dicts interchange, momentum roundtrips."""
    import torch

    from edl_amd.train.bucketed_ddp import BucketedAllReducer
    from edl_amd.ops.sgd import FusedSGD

    def build(flag):
        torch.manual_seed(0)
        m = torch.nn.Sequential(
            torch.nn.Conv2d(8, 8, 3, padding=1, bias=False),
            torch.nn.Conv2d(8, 4, 1, bias=False))
        if flag:
            w = m[0].weight
            w._edl_phys_shape = (8, 3, 3, 8)
            w._edl_phys_perm = (0, 3, 1, 2)
        red = BucketedAllReducer(m.parameters(), bucket_cap_mb=1)
        opt = FusedSGD(m.parameters(), lr=0.1, momentum=0.9,
                       weight_decay=1e-4, reducer=red)
        return m, red, opt

    m1, r1, o1 = build(False)
    m2, r2, o2 = build(True)
    # same initial logical values
    for p1, p2 in zip(m1.parameters(), m2.parameters()):
        assert torch.equal(p1.data, p2.data)
    torch.manual_seed(5)
    xs = [torch.randn(2, 8, 6, 6) for _ in range(3)]
    for x in xs:
        for m, r, o in ((m1, r1, o1), (m2, r2, o2)):
            r.zero_grad()
            m(x).square().mean().backward()
            o.step()
    for p1, p2 in zip(m1.parameters(), m2.parameters()):
        assert torch.allclose(p1.data, p2.data, atol=1e-6), \
            (p1.data - p2.data).abs().max()
    # state dicts interchange (logical layout) and momentum roundtrips
    sd1 = m1.state_dict()
    sd2 = m2.state_dict()
    for k in sd1:
        assert torch.allclose(sd1[k], sd2[k], atol=1e-6)
        # the live entry may be a bucket view; a clone must round-trip
        # through torch.save-style materialization with correct values
        assert torch.equal(sd2[k].detach().clone().contiguous(), sd2[k])
    os1 = o1.state_dict()
    os2 = o2.state_dict()
    for k in os1["state"]:
        assert torch.allclose(os1["state"][k]["momentum_buffer"],
                              os2["state"][k]["momentum_buffer"], atol=1e-6)
    # load the flagged model from the plain state dict and keep training
    m2.load_state_dict(sd1)
    o2.load_state_dict(os1)
    for m, r, o in ((m1, r1, o1), (m2, r2, o2)):
        r.zero_grad()
        m(xs[0]).square().mean().backward()
        o.step()
    for p1, p2 in zip(m1.parameters(), m2.parameters()):
        assert torch.allclose(p1.data, p2.data, atol=1e-6)
