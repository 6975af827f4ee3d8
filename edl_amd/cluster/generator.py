"""Cluster generator — the leader's reconciliation loop.

Parity: reference utils/cluster_generator.py:32-273. Leader-only background
thread (period ~1 s here vs 3 s there; resize latency is a headline metric):

  * first boot: build cluster from resource pods, leader rank 0 (95-134)
  * pods disappeared (lease lapsed) or FAILED -> regenerate with survivors,
    new stage (179-192)
  * new INITIAL pods while world < max_nodes and train not near end ->
    append with new ranks (136-153, 200-215)
  * refuse to go below min_nodes — job holds (255-264)
  * publish via a still-being-leader guarded transaction (224-250)
"""
import threading

from ..coord.tables import ETCD_POD_RANK, LEADER_KEY
from ..utils.log import get_logger
from .model import Cluster, load_cluster, save_cluster
from .resource import load_resource_pods
from .status import Status, TrainStatus, load_pods_status, load_train_statuses

log = get_logger("edl.generator")


class ClusterGenerator:
    def __init__(self, client, pod_id, min_nodes=1, max_nodes=None, period=1.0):
        self._client = client
        self._pod_id = pod_id
        self._min = min_nodes
        self._max = max_nodes
        self._period = period
        self._stop = threading.Event()
        self._thread = None
        self.holding = threading.Event()  # set while below min_nodes

    # ---- lifecycle ----
    def start(self):
        self._thread = threading.Thread(target=self._loop, daemon=True, name="cluster-gen")
        self._thread.start()
        return self

    def stop(self):
        self._stop.set()
        if self._thread:
            self._thread.join(timeout=5.0)

    def _loop(self):
        while not self._stop.wait(self._period):
            try:
                self.generate_once()
            except Exception as e:  # noqa: BLE001
                log.warning("generate_cluster error: %s", e)

    # ---- reconciliation ----
    def _leader_guard(self):
        return (self._client.table_key(ETCD_POD_RANK, LEADER_KEY), self._pod_id)

    def generate_once(self):
        resource = load_resource_pods(self._client)
        if self._pod_id not in resource:
            return None  # our own registration not visible yet
        statuses = load_pods_status(self._client)
        current = load_cluster(self._client)

        # external scale-in requests (cluster/scale.py): drop named pods.
        # The request is cleared only AFTER the resulting cluster publishes
        # (or when it names no known pod): clearing first would silently
        # drop the scale-in if we lose leadership mid-publish — the next
        # leader must still see and apply it (reference pattern:
        # utils/cluster_generator.py:224-250 leader-guarded txn).
        from .scale import clear_scale_request, read_scale_request

        req = read_scale_request(self._client)
        removed_by_request = set()
        if req:
            removed_by_request = set(req.get("remove_pods") or [])
            for pid in removed_by_request:
                resource.pop(pid, None)

        if current is None:
            got = self._first_boot(resource)
            if req and got is not None:
                clear_scale_request(self._client)
            return got

        alive = [p for p in current.pods if p.pod_id in resource
                 and statuses.get(p.pod_id) != Status.FAILED]
        disappeared = len(alive) != len(current.pods)
        # request named no current member and no registered pod -> no-op;
        # clear it so a stale/bogus request doesn't linger forever
        if req and not removed_by_request & (
                {p.pod_id for p in current.pods} | set(resource)):
            clear_scale_request(self._client)
            req = None

        # candidate new pods: in resource, not in cluster, INITIAL status
        member_ids = {p.pod_id for p in current.pods}
        train_statuses = load_train_statuses(self._client).values()
        near_end = any(s in (TrainStatus.NEARTHEEND, TrainStatus.SUCCEED) for s in train_statuses)
        new_pods = []
        if not near_end:
            for pid, pod in sorted(resource.items()):
                if pid in member_ids:
                    continue
                if statuses.get(pid, Status.INITIAL) == Status.INITIAL:
                    new_pods.append(pod)

        if not disappeared and not new_pods:
            return current

        pods = list(alive)
        if self._max is not None:
            room = self._max - len(pods)
            new_pods = new_pods[:max(0, room)]
        pods.extend(new_pods)

        # leader pod must be rank 0 if present
        pods.sort(key=lambda p: (p.pod_id != self._pod_id,))
        if len(pods) < self._min:
            if not self.holding.is_set():
                log.warning(
                    "cluster would have %d pods < min_nodes=%d; holding", len(pods), self._min
                )
            self.holding.set()
            return current
        self.holding.clear()

        nxt = Cluster(pods=pods, job_stage=current.job_stage)
        nxt.new_stage()
        nxt.assign_ranks()
        if save_cluster(self._client, nxt, leader_guard=self._leader_guard()):
            log.info(
                "published cluster stage=%s pods=%s (was %s)",
                nxt.stage, nxt.pod_ids(), current.pod_ids(),
            )
            if req:
                clear_scale_request(self._client)
            return nxt
        log.warning("lost leadership mid-publish; cluster unchanged"
                    " (scale request, if any, stays pending)")
        return current

    def _first_boot(self, resource):
        pods = sorted(resource.values(), key=lambda p: (p.pod_id != self._pod_id, p.pod_id))
        if len(pods) < self._min:
            self.holding.set()
            return None
        self.holding.clear()
        cluster = Cluster(pods=pods)
        cluster.assign_ranks()
        if save_cluster(self._client, cluster, leader_guard=self._leader_guard()):
            log.info("published first cluster stage=%s pods=%s", cluster.stage, cluster.pod_ids())
            return cluster
        return None
