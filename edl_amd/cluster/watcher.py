"""Cluster watcher (parity: reference utils/cluster_watcher.py:23-121).

Watches the published cluster key; `changed` flips True when the stage or
the rank-ordered pod-id list differs from the cluster it was started with.
Uses the store's long-poll wait (sub-second latency) instead of the
reference's 3 s poll — resize-recovery seconds are a headline metric."""
import threading

from ..coord.client import CoordClient
from ..coord.tables import ETCD_CLUSTER, CLUSTER_KEY
from ..utils.errors import EdlStoreError
from .model import Cluster


class ClusterWatcher:
    def __init__(self, endpoints, job_id, base_cluster):
        self._client = CoordClient(endpoints, job_id)
        self._base = base_cluster
        self._lock = threading.Lock()
        self._new = None
        self._changed = threading.Event()
        self._stop = threading.Event()
        self._thread = threading.Thread(target=self._run, daemon=True, name="cluster-watch")

    def start(self):
        self._thread.start()
        return self

    def _run(self):
        key = self._client.table_key(ETCD_CLUSTER, CLUSTER_KEY)
        rev = 0
        while not self._stop.is_set():
            try:
                s = self._client.get(key)
                rev = self._client.rev()
                if s:
                    cur = Cluster.from_json(s)
                    if cur.stage != self._base.stage or not cur.same_members(self._base):
                        with self._lock:
                            self._new = cur
                        self._changed.set()
                        return
                self._client.wait(rev, timeout=5.0)
            except EdlStoreError:
                self._stop.wait(1.0)

    @property
    def changed(self):
        return self._changed.is_set()

    def wait_changed(self, timeout):
        return self._changed.wait(timeout)

    def new_cluster(self):
        with self._lock:
            return self._new

    def stop(self):
        self._stop.set()
        self._client.close()
        self._thread.join(timeout=5.0)
