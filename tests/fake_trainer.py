"""Env-controlled fake trainer (parity: reference launch_demo.py /
edl_demo.py — exit code from env, used by test_launch.sh:40-66).

Writes one JSON line per run to $EDL_DEMO_OUT so tests can assert which
(world_size, stage, rank) actually ran."""
import json
import os
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

from edl_amd.train.env import TrainerEnv  # noqa: E402


def main():
    env = TrainerEnv()
    out = os.environ.get("EDL_DEMO_OUT")
    rec = {
        "rank": env.global_rank,
        "world": env.world_size,
        "stage": env.cluster_stage,
        "job_stage": env.job_stage,
        "pid": os.getpid(),
        "endpoints": ",".join(env.trainer_endpoints or []),
    }
    print("fake_trainer start: %s" % rec, flush=True)
    if out:
        with open(out, "a") as f:
            f.write(json.dumps(rec) + "\n")
    time.sleep(float(os.environ.get("EDL_DEMO_SLEEP", "0")))
    code = int(os.environ.get("EDL_DEMO_EXIT_CODE", "0"))
    print("fake_trainer exit %d" % code, flush=True)
    sys.exit(code)


if __name__ == "__main__":
    main()
