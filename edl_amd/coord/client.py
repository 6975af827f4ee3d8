"""CoordClient — client for the coordd store.

API mirrors the subset of etcd the reference uses
(discovery/etcd_client.py:51-263): get/range/put with leases, put-if-absent
CAS, leader-guarded transactions, keepalive refresh, and prefix watch with
add/remove-server callbacks (etcd_client.py:122-155).

Key layout follows the reference (utils/constants.py:15-39):
    /<job_id>/<table>/nodes/<key>
"""
import threading

from ..utils.errors import EdlStoreError
from ..utils.log import get_logger
from . import protocol

log = get_logger("edl.coord")


class CoordClient:
    def __init__(self, endpoints, job_id="", timeout=6.0, retry_s=None):
        """endpoints: 'host:port' or comma-separated list (first reachable wins).

        retry_s (or env EDL_STORE_RETRY_S): keep retrying a failed RPC for
        this many seconds before raising — rides out a coordd restart on
        the same endpoint (the durability etcd gave the reference; pair
        with the server's --snapshot). Default 0: fail after one
        reconnect attempt (failure-detection paths stay prompt)."""
        import os

        if isinstance(endpoints, str):
            endpoints = [e for e in endpoints.split(",") if e]
        self._endpoints = endpoints
        self._timeout = timeout
        self._retry_s = (float(os.environ.get("EDL_STORE_RETRY_S", "0"))
                         if retry_s is None else float(retry_s))
        self.job_id = job_id
        self._lock = threading.Lock()
        self._sock = None

    # ---- connection ----
    def _connect(self):
        last = None
        for ep in self._endpoints:
            try:
                return protocol.connect(ep, self._timeout)
            except OSError as e:
                last = e
        raise EdlStoreError("cannot reach coordd at %s: %s" % (self._endpoints, last))

    def _call(self, req):
        import time as _time

        with self._lock:
            deadline = _time.monotonic() + self._retry_s
            attempt = 0
            while True:
                try:
                    if self._sock is None:
                        self._sock = self._connect()
                    protocol.send_msg(self._sock, req)
                    resp = protocol.recv_msg(self._sock)
                    break
                except (EdlStoreError, ConnectionError, OSError) as e:
                    if self._sock is not None:
                        try:
                            self._sock.close()
                        except OSError:
                            pass
                        self._sock = None
                    attempt += 1
                    # one immediate reconnect (a half-open socket), then
                    # back off within the retry window (store restart)
                    if attempt > 1:
                        if _time.monotonic() >= deadline:
                            if isinstance(e, EdlStoreError):
                                raise
                            raise EdlStoreError("coordd rpc failed: %s" % e)
                        _time.sleep(0.3)
        if not resp.get("ok"):
            raise EdlStoreError(resp.get("err", "unknown store error"))
        return resp

    def close(self):
        with self._lock:
            if self._sock is not None:
                try:
                    self._sock.close()
                except OSError:
                    pass
                self._sock = None

    # ---- key helpers ----
    def table_key(self, table, key=""):
        root = "/%s/%s/nodes" % (self.job_id, table)
        return "%s/%s" % (root, key) if key else root + "/"

    # ---- ops ----
    def grant(self, ttl):
        return self._call({"op": "grant", "ttl": ttl})["lease"]

    def keepalive(self, lease):
        try:
            self._call({"op": "keepalive", "lease": lease})
            return True
        except EdlStoreError:
            return False

    def revoke(self, lease):
        self._call({"op": "revoke", "lease": lease})

    def put(self, key, val, lease=None):
        return self._call({"op": "put", "key": key, "val": val, "lease": lease})["rev"]

    def get(self, key):
        r = self._call({"op": "get", "key": key})
        return r["val"] if r["found"] else None

    def range(self, prefix):
        """-> list of (key, value), sorted."""
        return [tuple(kv) for kv in self._call({"op": "range", "prefix": prefix})["kvs"]]

    def rev(self):
        return self._call({"op": "ping"})["rev"]

    def delete(self, key):
        return self._call({"op": "delete", "key": key})["deleted"]

    def delete_prefix(self, prefix):
        return self._call({"op": "delete_prefix", "prefix": prefix})["deleted"]

    def put_if_absent(self, key, val, lease=None):
        """-> (acquired, current_value)."""
        r = self._call({"op": "cas", "key": key, "val": val, "lease": lease})
        return r["acquired"], (val if r["acquired"] else r.get("val"))

    def txn_if(self, guard_key, guard_val, puts=(), dels=()):
        """Guarded atomic write. puts: [(key, val) or (key, val, lease)]."""
        r = self._call(
            {
                "op": "txn",
                "guard_key": guard_key,
                "guard_val": guard_val,
                "puts": [list(p) for p in puts],
                "dels": list(dels),
            }
        )
        return r["applied"]

    def wait(self, rev, timeout=30.0):
        """Block until global revision > rev (or timeout). -> (changed, rev)."""
        r = self._call({"op": "wait", "rev": rev, "timeout": timeout})
        return r["changed"], r["rev"]

    def clean(self):
        """Wipe this job's keyspace (reference constants.clean_etcd, 30-39)."""
        self.delete_prefix("/%s/" % self.job_id)


class Watcher:
    """Prefix watcher thread: on any content change under `prefix`, calls
    callback(kvs) with the new sorted (key, value) list.

    Uses its OWN connection (long-poll op would block a shared one).
    Parity: reference etcd watch_service (etcd_client.py:122-155) and the
    3 s polling Watcher (utils/cluster_watcher.py:23-121).
    """

    def __init__(self, endpoints, job_id, prefix, callback, poll_timeout=5.0):
        self._client = CoordClient(endpoints, job_id)
        self._prefix = prefix
        self._callback = callback
        self._poll_timeout = poll_timeout
        self._stop = threading.Event()
        self._thread = threading.Thread(target=self._run, daemon=True, name="coord-watch")
        self._last = None

    def start(self):
        self._thread.start()
        return self

    def _run(self):
        rev = 0
        while not self._stop.is_set():
            try:
                kvs = self._client.range(self._prefix)
                rev = self._client.rev()
                if kvs != self._last:
                    self._last = kvs
                    self._callback(kvs)
                changed, rev = self._client.wait(rev, timeout=self._poll_timeout)
            except EdlStoreError as e:
                if not self._stop.is_set():
                    log.debug("watch retry after store error: %s", e)
                self._stop.wait(1.0)

    def stop(self):
        self._stop.set()
        self._client.close()
        self._thread.join(timeout=5.0)
