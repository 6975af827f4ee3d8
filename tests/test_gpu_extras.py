"""GPU coverage for the r1-unmeasured paths: fp16
dynamic-loss-scaling step, the ctr (wide&deep) step, and DGC on CUDA
tensors at world 2 (gloo collectives — RCCL refuses 2 ranks on 1 GPU)."""
import json
import os
import subprocess
import sys

import pytest
import torch

pytestmark = pytest.mark.gpu

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))


@pytest.fixture(autouse=True)
def _gpu():
    if not torch.cuda.is_available():
        pytest.skip("no GPU")


def test_fp16_loss_scaled_step():
    """fp16 + DynamicLossScaler on MI355X: steps apply, loss decreases,
    found_inf path costs ONE host sync (engine r2 fix)."""
    from edl_amd.data.synthetic import SyntheticImageNet
    from edl_amd.train.engine import TrainerEngine

    eng = TrainerEngine(model="resnet18_vd", per_device_batch=8,
                        dtype="fp16", checkpoint_dir=None).setup()
    assert eng.scaler is not None
    loader = SyntheticImageNet(8, eng.device, channels_last=True, seed=5)
    x, y = loader.next()
    losses = []
    for _ in range(6):
        losses.append(float(eng.train_step(x, y).item()))
    assert losses[-1] < losses[0], losses  # memorizing one batch
    assert eng.scaler.value > 0


def test_fp16_overflow_skips_step():
    """A gradient overflow must back the scale off and skip the update."""
    from edl_amd.train.engine import TrainerEngine

    eng = TrainerEngine(model="mnist_mlp", per_device_batch=4, num_classes=10,
                        dtype="fp16", checkpoint_dir=None).setup()
    x = torch.randn(4, 1, 28, 28, device="cuda") * 1e4  # force inf grads
    y = torch.randint(0, 10, (4,), device="cuda")
    s0 = eng.scaler.value
    p0 = next(eng.model.parameters()).detach().clone()
    eng.train_step(x, y)
    assert eng.scaler.value <= s0  # backed off (or unchanged if no inf)
    if eng.scaler.value < s0:  # overflow happened: params untouched
        assert torch.equal(p0, next(eng.model.parameters()).detach())


def test_ctr_wide_and_deep_gpu_step():
    """BASELINE config 5 model: wide&deep trains on GPU (dense-embedding
    all-reduce path; sparse ids through nn.Embedding)."""
    from edl_amd.data.synthetic import SyntheticCTR
    from edl_amd.models import WideAndDeep

    torch.manual_seed(0)
    m = WideAndDeep().cuda()
    opt = torch.optim.SGD(m.parameters(), lr=0.05)
    data = SyntheticCTR(256, torch.device("cuda"))
    dense, sparse, label = data.next()  # fixed batch: loss must memorize
    losses = []
    for _ in range(12):
        loss = torch.nn.functional.binary_cross_entropy_with_logits(
            m(dense, sparse).view(-1), label.view(-1))
        opt.zero_grad()
        loss.backward()
        opt.step()
        losses.append(float(loss.item()))
    assert losses[-1] < losses[0], losses


def test_dgc_two_rank_cuda(tmp_path):
    """DGC compressed exchange at world 2 on CUDA tensors (gloo): ranks
    converge to identical grads and error feedback accumulates."""
    script = os.path.join(REPO, "tests", "_dgc_cuda_worker.py")
    env = dict(os.environ)
    env["PYTHONPATH"] = REPO + os.pathsep + env.get("PYTHONPATH", "")
    r = subprocess.run(
        [sys.executable, "-m", "torch.distributed.run", "--nnodes=1",
         "--nproc-per-node", "2", "--master-addr", "127.0.0.1",
         "--master-port", "29541", script],
        env=env, cwd=REPO, capture_output=True, text=True, timeout=600,
    )
    sys.stderr.write(r.stdout[-2000:] + r.stderr[-1500:])
    assert r.returncode == 0
    import re

    # torchrun can interleave both ranks' lines without a newline
    oks = [json.loads(m) for m in
           re.findall(r'\{"dgc_cuda".*?\}', r.stdout)]
    assert oks and all(v["ok"] for v in oks)


def test_prefill_derived_matches_inline():
    """The side-stream weight-repack prefill (EDL_PREFILL_DERIVED=1,
    opt-in) must produce byte-identical repacks to the inline builders.
    (Trajectory comparison across runs is NOT valid here: the step uses
    fp32 atomics — split-K folds, capped stats partials — so two runs
    diverge in the low bits regardless of prefill.)"""
    import os

    from edl_amd.data.synthetic import SyntheticImageNet
    from edl_amd.ops import ext
    from edl_amd.ops.conv import Conv2dFast
    from edl_amd.train.engine import TrainerEngine

    os.environ["EDL_PREFILL_DERIVED"] = "1"
    try:
        torch.manual_seed(7)
        eng = TrainerEngine(model="resnet18_vd", per_device_batch=8,
                            dtype="bf16", checkpoint_dir=None).setup()
        assert eng._prefill_stream is not None
        loader = SyntheticImageNet(8, eng.device, channels_last=True, seed=9)
        x, y = loader.next()
        losses = [float(eng.train_step(x, y).item()) for _ in range(2)]
        assert all(l == l for l in losses), losses  # finite
        torch.cuda.synchronize()

        # snapshot the mirrors BEFORE the next step: the repacks that
        # step's backward consumes must equal repacks of THESE values
        # (comparing against post-step mirrors is wrong — the optimizer
        # refreshes them at step end)
        snaps = {}
        for m in eng.model.modules():
            if isinstance(m, Conv2dFast):
                mb = getattr(m.weight, "_edl_bf16", None)
                if mb is not None:
                    snaps[id(m)] = mb.detach().clone()
        eng.train_step(x, y)
        torch.cuda.synchronize()

        checked = 0
        for m in eng.model.modules():
            if not isinstance(m, Conv2dFast) or id(m) not in snaps:
                continue
            cache = getattr(m, "_w_cache", None)
            if cache is None:
                continue
            d = cache[1]
            snap = snaps[id(m)]
            co, ci = m.out_channels, m.in_channels
            cl = getattr(m.weight, "_edl_phys_shape", None) is not None
            if "wt_t" in d and m.kernel_size == (1, 1):
                ref = ext().transpose_pad(snap.reshape(co, ci))
                assert torch.equal(d["wt_t"], ref), "wt_t raced"
                checked += 1
            if cl and m.kernel_size == (3, 3):
                w3v = snap.permute(0, 2, 3, 1).reshape(co, 9 * ci).contiguous()
                for key, mode in (("w3rot", 0), ("w3s2d", 1)):
                    if key in d:
                        ref = ext().repack_dgrad_w3(w3v, ci, mode)
                        assert torch.equal(d[key], ref), key + " raced"
                        checked += 1
        assert checked > 0, "prefill never produced a repack to verify"
    finally:
        os.environ.pop("EDL_PREFILL_DERIVED", None)
