"""edlrun — the per-node elastic launcher agent.

    python -m edl_amd.launch --nodes_range 1:8 \
        --store_endpoints 127.0.0.1:2379 --job_id myjob  train.py [args...]

Parity: reference collective/launch.py:32-55 + utils/launcher.py:33-261
(the §3.1 state machine in SURVEY.md):

    register resources -> elect leader (leader runs the cluster generator)
    -> barrier -> spawn trainer subprocesses -> watch
    -> on membership change: re-barrier, stop-resume trainers (new stage)
    -> on failure: deregister (lease lapse publishes the loss)

`--standalone` starts an in-process coordination store (single-node jobs,
tests) — the reference required an external etcd.
"""
import argparse
import os
import sys
import time

from .cluster.barrier import barrier
from .cluster.generator import ClusterGenerator
from .cluster.leader import LeaderElector
from .cluster.model import Pod
from .cluster.resource import ResourceRegister, wait_resource_empty
from .cluster.status import (
    Status,
    load_job_status,
    save_job_status,
    save_pod_status,
)
from .cluster.watcher import ClusterWatcher
from .coord.client import CoordClient
from .procs import TrainerProcs
from .train.env import JobEnv
from .utils.errors import EdlBarrierError, EdlPodIDNotExistError
from .utils.log import get_logger

log = get_logger("edl.launch")


def parse_args(argv=None):
    p = argparse.ArgumentParser("edlrun", description="EDL-AMD elastic launcher agent")
    p.add_argument("--job_id", default=None)
    p.add_argument("--store_endpoints", "--etcd_endpoints", dest="store_endpoints", default=None)
    p.add_argument("--nodes_range", default=None, help="min:max elastic node count")
    p.add_argument("--nproc_per_node", type=int, default=None,
                   help="trainer procs per node (default: one per visible GPU)")
    p.add_argument("--log_dir", default=None)
    p.add_argument("--log_level", default=None)
    p.add_argument("--checkpoint_dir", default=None)
    p.add_argument("--standalone", action="store_true",
                   help="run an in-process coordination store")
    p.add_argument("--store_snapshot", default=None,
                   help="standalone store: persist state here so a "
                        "restarted agent resumes the same job keyspace")
    p.add_argument("cmd", nargs=argparse.REMAINDER, help="training script and args")
    args = p.parse_args(argv)
    if args.cmd and args.cmd[0] == "--":
        args.cmd = args.cmd[1:]
    return args


class Launcher:
    def __init__(self, job_env, pod, client, cmd):
        self._env = job_env
        self._pod = pod
        self._client = client
        self._cmd = cmd
        self._resource = None
        self._elector = None
        self._generator = None
        self._watcher = None
        self._procs = None
        self.final_status = Status.FAILED

    # ---- lifecycle (reference launcher.py:58-67 init, 132-246 launch) ----
    def init(self):
        save_pod_status(self._client, self._pod.pod_id, Status.INITIAL)
        self._resource = ResourceRegister(self._client, self._pod).start()
        self._generator = ClusterGenerator(
            self._client, self._pod.pod_id,
            min_nodes=self._env.min_nodes, max_nodes=self._env.max_nodes,
        )
        retry = float(os.environ.get("EDL_LEADER_RETRY", "3"))
        self._elector = LeaderElector(
            self._client, self._pod.pod_id, on_elected=lambda: self._generator.start(),
            retry_interval=retry,
        ).start()

    def launch(self):
        try:
            self._run()
        finally:
            self._exit()

    def _start_stage(self, cluster):
        self._watcher = ClusterWatcher(
            self._env.store_endpoints, self._env.job_id, cluster
        ).start()
        self._procs = TrainerProcs(self._env, cluster, cluster.get_pod(self._pod.pod_id),
                                   self._cmd).start()

    def _stop_stage(self):
        if self._procs:
            self._procs.terminate()
            self._procs = None
        if self._watcher:
            self._watcher.stop()
            self._watcher = None

    def _await_cluster_change(self, factor=3.0):
        """After a trainer failure, wait up to ~factor lease-TTLs for the
        cluster watcher to report a membership change (a peer death that
        broke the communicator). True -> resize; False -> genuine local
        failure."""
        from .coord.tables import ETCD_TTL

        deadline = time.monotonic() + factor * ETCD_TTL + 1.0
        while time.monotonic() < deadline:
            if self._watcher.changed:
                return True
            if self._resource.failed or self._elector.lost:
                return False
            time.sleep(0.2)
        return self._watcher.changed

    def _run(self):
        cluster = barrier(self._client, self._pod.pod_id, timeout=600, allow_join=True)
        log.info("initial barrier done: stage=%s world=%d",
                 cluster.stage, cluster.world_size())
        save_pod_status(self._client, self._pod.pod_id, Status.RUNNING)
        self._start_stage(cluster)

        while True:
            time.sleep(0.3)
            self._procs.tail_rank0()
            alive, failed = self._procs.poll()
            if not alive:
                if not failed:
                    log.info("all trainers exited cleanly")
                    self.final_status = Status.SUCCEED
                    return
                # A trainer died. If a PEER pod just died, our trainer's
                # collective fails near-instantly ("connection closed by
                # peer") while the dead pod's lease takes up to one TTL to
                # lapse — so before declaring the failure OURS, give the
                # failure detector a grace window to publish the membership
                # change; if the cluster changed, this is a resize, not a
                # local fault (stop-resume below).
                if not self._await_cluster_change():
                    log.error("trainer process failed; pod exits FAILED")
                    self.final_status = Status.FAILED
                    return

            if self._resource.failed or self._elector.lost:
                log.error("lost store lease/leadership; stopping trainers")
                self.final_status = Status.FAILED
                return

            if self._watcher.changed:
                log.info("cluster changed; stop-resume resize begins")
                t0 = time.monotonic()
                self._stop_stage()
                try:
                    cluster = barrier(self._client, self._pod.pod_id, timeout=120)
                except EdlPodIDNotExistError as e:
                    # scale-in: we are no longer a member -> clean exit
                    log.info("not in new cluster (%s); exiting", e)
                    self.final_status = Status.SUCCEED
                    return
                except EdlBarrierError as e:
                    log.error("re-barrier failed: %s", e)
                    self.final_status = Status.FAILED
                    return
                self._start_stage(cluster)
                log.info("resize to world=%d done in %.2fs (stage=%s)",
                         cluster.world_size(), time.monotonic() - t0, cluster.stage)

    def _exit(self):
        """Reference launcher.py:99-130: write pod flag; leader waits for
        followers then writes the job flag."""
        self._stop_stage()
        try:
            save_pod_status(self._client, self._pod.pod_id, self.final_status)
            was_leader = self._elector is not None and self._elector.is_leader
            if self._generator:
                self._generator.stop()
            if self._resource:
                self._resource.stop()
            if was_leader:
                wait_resource_empty(self._client, self._pod.pod_id, timeout=30)
                save_job_status(self._client, self.final_status)
                log.info("leader wrote job status %s", self.final_status)
            if self._elector:
                self._elector.stop()
        except Exception as e:  # noqa: BLE001
            log.warning("exit cleanup error: %s", e)


def main(argv=None):
    args = parse_args(argv)
    if args.log_level:
        os.environ["EDL_LOG_LEVEL"] = args.log_level
    if not args.cmd:
        print("edlrun: no training command given", file=sys.stderr)
        return 2

    server = None
    if args.standalone:
        from .coord.server import CoordServer

        server = CoordServer(port=0, snapshot=args.store_snapshot).start()
        args.store_endpoints = server.endpoint
        log.info("standalone coordination store at %s", server.endpoint)

    job_env = JobEnv(args)
    client = CoordClient(job_env.store_endpoints, job_env.job_id)

    # job already finished? (reference launch.py:44-47)
    if load_job_status(client) == Status.SUCCEED:
        log.info("job %s already SUCCEED; exiting", job_env.job_id)
        return 0

    pod = Pod.from_env(job_env)
    log.info("pod %s starting: gpus=%s trainers=%d",
             pod.pod_id, pod.gpus, len(pod.trainers))
    launcher = Launcher(job_env, pod, client, args.cmd)
    launcher.init()
    try:
        launcher.launch()
    except KeyboardInterrupt:
        launcher.final_status = Status.FAILED
        launcher._exit()
    finally:
        if server is not None:
            server.stop()
    return 0 if launcher.final_status == Status.SUCCEED else 1


if __name__ == "__main__":
    sys.exit(main())
