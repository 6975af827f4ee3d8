"""Resize-recovery benchmark — BASELINE config 3 / the reference's
headline "< 5 min hot recovery" metric (doc/edl_live_fault_tolerance.md:37).

Starts N launcher agents (one trainer process each) against an in-process
coordination store, lets training reach steady state, then SIGKILLs agents
(or adds them back) and measures the wall time from the membership change
until every surviving trainer has taken its first step in the NEW world
(stop-resume: re-barrier + communicator rebuild + checkpoint reload).

    python tools/bench_elastic.py --start 8 --drop-to 4 --rejoin   # GPU box
    python tools/bench_elastic.py --start 2 --drop-to 1            # CPU ok

Prints one JSON line: {"resize_down_s": ..., "resize_up_s": ...}.
"""
import argparse
import json
import os
import signal
import subprocess
import sys
import tempfile
import time

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
sys.path.insert(0, REPO)

from edl_amd.coord.server import CoordServer  # noqa: E402

MARKER = os.path.join(tempfile.gettempdir(),
                      "edl_elastic_steps.%d.jsonl" % os.getpid())


def spawn_agent(store_ep, job_id, idx, nodes_range, log_dir, gpu=None):
    env = dict(os.environ)
    env.update({
        "PYTHONPATH": REPO + os.pathsep + env.get("PYTHONPATH", ""),
        "EDL_LEASE_TTL": env.get("EDL_LEASE_TTL", "3"),
        "EDL_LEADER_RETRY": "0.5",
        "EDL_STEP_MARKER": MARKER,
        "CUDA_VISIBLE_DEVICES": str(gpu) if gpu is not None else "",
        **({"EDL_FORCE_BACKEND": "gloo"}
           if os.environ.get("EDL_SHARE_GPU0") == "1" else {}),
    })
    return subprocess.Popen(
        [sys.executable, "-m", "edl_amd.launch",
         "--job_id", job_id, "--store_endpoints", store_ep,
         "--nodes_range", nodes_range, "--nproc_per_node", "1",
         "--log_dir", os.path.join(log_dir, "agent%d" % idx),
         os.path.join(REPO, "tools", "elastic_worker.py")],
        env=env, stdout=subprocess.DEVNULL, stderr=subprocess.STDOUT,
        cwd=REPO, start_new_session=True,
    )


def read_steps():
    if not os.path.exists(MARKER):
        return []
    out = []
    with open(MARKER) as f:
        for line in f:
            try:
                out.append(json.loads(line))
            except ValueError:
                pass
    return out


def wait_world_steps(world, min_ranks, deadline_s=180, after_ts=0.0):
    """Wait until >= min_ranks distinct ranks stepped at `world` after ts."""
    deadline = time.monotonic() + deadline_s
    while time.monotonic() < deadline:
        ranks = {s["rank"] for s in read_steps()
                 if s["world"] == world and s["ts"] > after_ts}
        if len(ranks) >= min_ranks:
            return time.time()
        time.sleep(0.05)
    raise TimeoutError("world=%d never stepped" % world)


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--start", type=int, default=2)
    ap.add_argument("--drop-to", type=int, default=1)
    ap.add_argument("--rejoin", action="store_true")
    ap.add_argument("--use-gpus", action="store_true")
    ap.add_argument("--share-gpu0", action="store_true",
                    help="all agents on cuda:0 with gloo collectives "
                         "(1-GPU box: RCCL refuses duplicate devices)")
    args = ap.parse_args()
    if args.share_gpu0:
        os.environ["EDL_SHARE_GPU0"] = "1"

    if os.path.exists(MARKER):
        os.remove(MARKER)
    srv = CoordServer(port=0).start()
    job = "elastic_bench_%d" % os.getpid()
    rng = "%d:%d" % (args.drop_to, args.start)
    log_dir = tempfile.mkdtemp(prefix="edl_elastic_")
    agents = []
    try:
        for i in range(args.start):
            agents.append(spawn_agent(srv.endpoint, job, i, rng, log_dir,
                                      gpu=0 if args.share_gpu0 else (i if args.use_gpus else None)))
        wait_world_steps(args.start, args.start, 300)
        time.sleep(1.0)  # steady state

        # ---- scale down: kill (start - drop_to) agents ----
        t_kill = time.time()
        for p in agents[args.drop_to:]:
            try:
                os.killpg(os.getpgid(p.pid), signal.SIGKILL)
            except ProcessLookupError:
                pass
        t_recovered = wait_world_steps(args.drop_to, args.drop_to, 300,
                                       after_ts=t_kill)
        down_s = t_recovered - t_kill
        result = {"resize_down_s": round(down_s, 2),
                  "from": args.start, "to": args.drop_to}

        if args.rejoin:
            t_join = time.time()
            for i in range(args.drop_to, args.start):
                agents.append(spawn_agent(srv.endpoint, job, 100 + i, rng,
                                          log_dir,
                                          gpu=0 if args.share_gpu0 else (i if args.use_gpus else None)))
            t_back = wait_world_steps(args.start, args.start, 300,
                                      after_ts=t_join)
            result["resize_up_s"] = round(t_back - t_join, 2)
        print(json.dumps(result))
    finally:
        for p in agents:
            try:
                os.killpg(os.getpgid(p.pid), signal.SIGKILL)
            except ProcessLookupError:
                pass
        # trainer workers live in their OWN process groups (procs.py uses
        # start_new_session): kill the exact pids they reported
        for pid in {s["pid"] for s in read_steps() if "pid" in s}:
            try:
                os.kill(pid, signal.SIGKILL)
            except (ProcessLookupError, PermissionError):
                pass
        srv.stop()


if __name__ == "__main__":
    main()
