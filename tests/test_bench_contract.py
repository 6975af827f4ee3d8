"""bench.py driver-contract test on CPU (gloo, world 2): the exact
invocation shape the driver uses for SCALE runs, minus the GPUs."""
import json
import os
import re
import subprocess
import sys

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))


def test_bench_world2_gloo_json_contract():
    env = dict(os.environ)
    env["PYTHONPATH"] = REPO + os.pathsep + env.get("PYTHONPATH", "")
    r = subprocess.run(
        [sys.executable, "-m", "torch.distributed.run", "--nnodes=1",
         "--nproc-per-node", "2", "--master-addr", "127.0.0.1",
         "--master-port", "29541", os.path.join(REPO, "bench.py"),
         "--gpus", "2", "--steps", "2", "--warmup", "1",
         "--model", "resnet18_vd", "--batch_size", "2"],
        env=env, cwd=REPO, capture_output=True, text=True, timeout=420,
    )
    sys.stderr.write(r.stdout[-2000:] + r.stderr[-1500:])
    assert r.returncode == 0
    # torchrun can interleave rank output without newlines
    matches = re.findall(r'\{"metric".*?\}\}', r.stdout)
    assert len(matches) == 1, "exactly ONE rank prints the JSON line"
    j = json.loads(matches[0])
    assert j["metric"] == "img/s" and j["value"] > 0
    assert j["n_gpus"] == 2 and j["steps"] == 2 and j["warmup"] == 1
    assert j["scaling"] == "weak" and j["higher_is_better"] is True
    assert j["config"]["global_batch"] == 4  # 2/rank x 2 (whole-job)
    assert j["config"]["parallelism"] == "dp2"
    assert "ms_per_step" in j and "vs_baseline" in j
