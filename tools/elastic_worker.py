"""Trainer used by bench_elastic.py: real torch.distributed world (RCCL on
GPU, gloo on CPU) doing tiny all-reduce steps forever; records each step's
(rank, world, ts) to $EDL_STEP_MARKER so the bench can detect the first
step of a new world."""
import json
import os
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import torch  # noqa: E402
import torch.distributed as dist  # noqa: E402

from edl_amd.train import dist as edist  # noqa: E402


def main():
    tenv, device = edist.init_from_env(timeout_s=300)
    marker = os.environ.get("EDL_STEP_MARKER")
    t = torch.ones(1 << 18, device=device)
    step = 0
    t_start = time.time()
    while True:
        # self-terminate when orphaned (our launcher agent died) or after a
        # hard TTL: trainers run in their OWN process group, so killing the
        # agent does not reach them — a leaked worker once ran for hours
        if os.getppid() == 1 or time.time() - t_start > 600:
            return
        if dist.is_initialized():
            dist.all_reduce(t)
        if device.type == "cuda":
            torch.cuda.synchronize()
        step += 1
        if marker:
            with open(marker, "a") as f:
                f.write(json.dumps({"rank": tenv.global_rank,
                                    "world": tenv.world_size, "pid": os.getpid(),
                                    "step": step, "ts": time.time()}) + "\n")
        time.sleep(0.05)


if __name__ == "__main__":
    main()
