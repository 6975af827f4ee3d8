"""End-to-end trainer-script tests under the real launcher on CPU:
fit_a_line (BASELINE config 1: collective on gloo world 2) and ctr."""
import os
import subprocess
import sys

import pytest

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))


def run_edlrun(coord_server, tmp_path, script_args, n_agents=2, timeout=180):
    procs = []
    env = dict(os.environ)
    env.update({
        "PYTHONPATH": REPO + os.pathsep + env.get("PYTHONPATH", ""),
        "EDL_LEASE_TTL": "2",
        "EDL_LEADER_RETRY": "0.5",
        "CUDA_VISIBLE_DEVICES": "",
    })
    for i in range(n_agents):
        logf = open(tmp_path / ("agent%d.log" % i), "wb")
        p = subprocess.Popen(
            [sys.executable, "-m", "edl_amd.launch",
             "--job_id", "script_job", "--store_endpoints", coord_server.endpoint,
             "--nodes_range", "%d:%d" % (n_agents, n_agents),
             "--nproc_per_node", "1",
             "--log_dir", str(tmp_path / ("logs%d" % i)), "--"] + script_args,
            env=env, stdout=logf, stderr=subprocess.STDOUT, cwd=REPO,
            start_new_session=True)
        p._logf = logf
        procs.append(p)
    try:
        for p in procs:
            assert p.wait(timeout=timeout) == 0, \
                (tmp_path / "agent0.log").read_text() + \
                (tmp_path / ("agent%d.log" % (n_agents - 1))).read_text()
    finally:
        for p in procs:
            if p.poll() is None:
                p.kill()
            p._logf.close()


@pytest.mark.parametrize("model", ["fit_a_line", "ctr"])
def test_simple_trainer_world2(coord_server, tmp_path, model):
    run_edlrun(
        coord_server, tmp_path,
        ["-m", "edl_amd.train.train_simple", "--model", model,
         "--num_epochs", "1", "--steps_per_epoch", "20",
         "--checkpoint", str(tmp_path / "ck")],
    )
    assert (tmp_path / "ck" / "checkpoint.0").is_dir()


def test_resnet_trainer_world2_tiny(coord_server, tmp_path):
    """The flagship trainer end-to-end under the launcher on CPU (tiny)."""
    run_edlrun(
        coord_server, tmp_path,
        ["-m", "edl_amd.train.train_resnet", "--model", "resnet18_vd",
         "--batch_size", "2", "--num_epochs", "1", "--steps_per_epoch", "2",
         "--checkpoint", str(tmp_path / "ck"), "--use_hip_ops", "0",
         "--dtype", "fp32"],
        timeout=420,
    )
    assert (tmp_path / "ck" / "checkpoint.0").is_dir()


def test_resume_across_world_sizes(coord_server, tmp_path):
    """Checkpoint written at world 2 resumes at world 1 (elastic restart
    semantics: optimizer state survives re-bucketing, epoch cursor
    continues)."""
    ck = str(tmp_path / "ck")
    run_edlrun(
        coord_server, tmp_path,
        ["-m", "edl_amd.train.train_simple", "--model", "fit_a_line",
         "--num_epochs", "1", "--steps_per_epoch", "10", "--checkpoint", ck],
        n_agents=2,
    )
    assert (tmp_path / "ck" / "checkpoint.0").is_dir()
    # second job id to avoid the SUCCEED short-circuit of the first job
    import json
    import subprocess
    import sys

    env = dict(os.environ)
    env.update({"PYTHONPATH": REPO, "EDL_LEASE_TTL": "2",
                "EDL_LEADER_RETRY": "0.5", "CUDA_VISIBLE_DEVICES": ""})
    logf = open(tmp_path / "resume.log", "wb")
    p = subprocess.Popen(
        [sys.executable, "-m", "edl_amd.launch", "--job_id", "resume_job",
         "--store_endpoints", coord_server.endpoint, "--nodes_range", "1:1",
         "--nproc_per_node", "1", "--log_dir", str(tmp_path / "rlogs"), "--",
         "-m", "edl_amd.train.train_simple", "--model", "fit_a_line",
         "--num_epochs", "2", "--steps_per_epoch", "10", "--checkpoint", ck],
        env=env, stdout=logf, stderr=subprocess.STDOUT, cwd=REPO,
        start_new_session=True)
    assert p.wait(timeout=120) == 0, (tmp_path / "resume.log").read_text()
    logf.close()
    # epoch 1 ran on resume -> checkpoint.1 exists
    assert (tmp_path / "ck" / "checkpoint.1").is_dir()
    with open(tmp_path / "ck" / "checkpoint.1" / "train_status.json") as f:
        assert json.load(f)["epoch_no"] == 1


def test_fit_a_line_elastic_data_plane(coord_server, tmp_path):
    """fit_a_line trained from FILES through the elastic data plane under
    the launcher at world 2 (leader-balanced Reader; BASELINE config 1 with
    real input plumbing)."""
    data = tmp_path / "data"
    data.mkdir()
    import numpy as np

    rng = np.random.RandomState(0)
    for i in range(4):
        rows = []
        for _ in range(32):
            x = rng.randn(13)
            y = x.sum() * 0.5 + 1.0
            rows.append(" ".join("%.5f" % v for v in list(x) + [y]))
        (data / ("part%d.txt" % i)).write_text("\n".join(rows) + "\n")

    run_edlrun(
        coord_server, tmp_path,
        ["-m", "edl_amd.train.train_simple", "--model", "fit_a_line",
         "--num_epochs", "1", "--batch_size", "8",
         "--data_dir", str(data), "--checkpoint", str(tmp_path / "ck")],
    )
    assert (tmp_path / "ck" / "checkpoint.0").is_dir()
    # both ranks consumed a share of the 128 records
    logs = "".join((tmp_path / ("agent%d.log" % i)).read_text() for i in (0, 1))
    import re

    counts = [int(m) for m in re.findall(r"consumed (\d+) records", logs)]
    assert sum(counts) == 128, counts
    assert all(c > 0 for c in counts), counts


def test_resnet_elastic_data_plane(coord_server, tmp_path):
    """ResNet trained from FILES through the elastic data plane at world 2
    (leader-balanced Reader feeding deterministic record->image synthesis;
    reference: DALI file pipeline + utils/data_server.py balancing)."""
    data = tmp_path / "data"
    data.mkdir()
    for i in range(4):
        rows = ["%d rec-%d-%d" % ((i * 20 + j) % 10, i, j) for j in range(20)]
        (data / ("part%d.txt" % i)).write_text("\n".join(rows) + "\n")

    run_edlrun(
        coord_server, tmp_path,
        ["-m", "edl_amd.train.train_resnet", "--model", "resnet50_vd",
         "--num_epochs", "1", "--batch_size", "4", "--image_hw", "32",
         "--steps_per_epoch", "3", "--data_dir", str(data),
         "--checkpoint", str(tmp_path / "ck")],
        timeout=240,
    )
    assert (tmp_path / "ck" / "checkpoint.0").is_dir()
    logs = "".join((tmp_path / ("agent%d.log" % i)).read_text() for i in (0, 1))
    import re

    counts = [int(m) for m in re.findall(r"rank \d+: (\d+) records", logs)]
    assert len(counts) == 2 and sum(counts) == 80, counts
    assert "epoch 0 done" in logs


def test_train_distill_mnist_cpu():
    """mnist distill config (reference example/distill/mnist_distill):
    mnist_cnn student distilled from a served mnist_mlp teacher through
    the real train_distill CLI at world 1 on CPU."""
    import torch

    from edl_amd.distill.teacher_server import TeacherServer, TeacherService
    from edl_amd.models import build_model
    from edl_amd.train import train_distill

    teacher = TeacherService(model=build_model("mnist_mlp"),
                             device=torch.device("cpu"))
    srv = TeacherServer(teacher, host="127.0.0.1", port=0).start()
    try:
        rc = train_distill.main([
            "--model", "mnist_cnn", "--image_shape", "1x28x28",
            "--num_classes", "10", "--batch_size", "8",
            "--steps_per_epoch", "4", "--num_epochs", "1",
            "--teachers", "127.0.0.1:%d" % srv.port, "--require_num", "1",
            "--kd_alpha", "0.7",
        ])
        assert rc == 0
    finally:
        srv.stop()


def test_bench_distill_service_smoke():
    """Service-distill orchestration (BASELINE config 4 shape) on CPU:
    one teacher process + two gloo student ranks, one JSON line out."""
    import json as _json

    out = subprocess.run(
        [sys.executable, os.path.join(REPO, "tools", "bench_distill.py"),
         "--teacher_gpus", "", "--student_gpus", "",
         "--steps", "2", "--warmup", "1", "--batch_size", "2",
         "--teacher_batch_size", "2", "--teacher_model", "resnet18_vd",
         "--student_model", "resnet18_vd"],
        capture_output=True, text=True, timeout=280,
        env=dict(os.environ, CUDA_VISIBLE_DEVICES=""))
    assert out.returncode == 0, out.stdout + out.stderr
    line = [ln for ln in out.stdout.splitlines()
            if ln.startswith("{")][-1]
    d = _json.loads(line)
    assert d["mode"] == "distill_service" and d["n_students"] == 2
    assert d["value"] > 0
