"""Leader election (parity: reference utils/leader_pod.py:28-165).

Every agent races to put-if-absent its pod_id under rank/nodes/0 bound to
its lease. The winner is the leader and starts the cluster generator; the
losers retry every few seconds — when the leader's lease expires (TTL 15 s)
a follower seizes the key (proven in reference test_leader_pod.py:45-61)."""
import threading

from ..coord.tables import ETCD_POD_RANK, ETCD_TTL, LEADER_KEY
from ..utils.log import get_logger

log = get_logger("edl.leader")


class LeaderElector:
    def __init__(self, client, pod_id, on_elected=None, retry_interval=3.0):
        self._client = client
        self._pod_id = pod_id
        self._on_elected = on_elected
        self._interval = retry_interval
        self._key = client.table_key(ETCD_POD_RANK, LEADER_KEY)
        self._stop = threading.Event()
        self._lost = threading.Event()
        self._is_leader = threading.Event()
        self._lease = None
        self._thread = None

    def start(self):
        self._thread = threading.Thread(target=self._run, daemon=True, name="leader-elect")
        self._thread.start()
        return self

    def _try_seize(self):
        self._lease = self._client.grant(ETCD_TTL)
        acquired, holder = self._client.put_if_absent(self._key, self._pod_id, self._lease)
        if not acquired:
            self._client.revoke(self._lease)
            self._lease = None
            return False
        return True

    def _run(self):
        while not self._stop.is_set():
            if not self._is_leader.is_set():
                try:
                    if self._try_seize():
                        log.info("pod %s became leader", self._pod_id)
                        self._is_leader.set()
                        if self._on_elected:
                            self._on_elected()
                        continue
                except Exception as e:  # noqa: BLE001
                    log.debug("leader seize retry: %s", e)
                self._stop.wait(self._interval)
            else:
                # refresh our leadership lease at TTL/3
                ok = False
                try:
                    ok = self._client.keepalive(self._lease)
                except Exception:  # noqa: BLE001
                    ok = False
                if not ok:
                    log.warning("pod %s lost leadership lease", self._pod_id)
                    self._is_leader.clear()
                    self._lost.set()
                    return
                self._stop.wait(ETCD_TTL / 3.0)

    @property
    def is_leader(self):
        return self._is_leader.is_set()

    @property
    def lost(self):
        """True if we were leader and our lease lapsed — the launcher treats
        this like a register failure and restarts (reference launcher loop)."""
        return self._lost.is_set()

    def leader_id(self):
        return self._client.get(self._key)

    def stop(self):
        self._stop.set()
        if self._thread:
            self._thread.join(timeout=5.0)
        if self._lease is not None:
            try:
                self._client.revoke(self._lease)
            except Exception:  # noqa: BLE001
                pass
