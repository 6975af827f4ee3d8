"""coordd — the in-repo coordination store.

Replaces the reference's external etcd (discovery/etcd_client.py:51-263):
a small threaded TCP server holding a revisioned key/value map with

  * leases with TTL + keepalive (etcd leases; reference utils/register.py)
  * put-if-absent CAS   (reference set_server_not_exists, etcd_client.py:177-197)
  * guarded transaction (reference _set_cluster_if_leader etcd txn,
                         cluster_generator.py:224-250; state.py:186-200)
  * long-poll prefix watch via a global revision counter
                         (reference watch_service, etcd_client.py:122-155)

Run standalone:  python -m edl_amd.coord.server --port 2379
Or in-process:   CoordServer(port=0).start()   (tests, --standalone launch)

Scale target is a control plane (≤ tens of agents, 3 s loops), not a data
path, so a lock-around-a-dict threaded server is the right amount of
machinery.
"""
import argparse
import json
import os
import threading
import time
import socket
import socketserver

from ..utils.log import get_logger
from .protocol import send_msg, recv_msg

log = get_logger("edl.coordd")


class _State:
    def __init__(self):
        self.lock = threading.Condition()
        self.kv = {}  # key -> dict(v=str, lease=int|None, ver=int, rev=int)
        self.leases = {}  # id -> dict(ttl=float, deadline=float)
        self.rev = 0
        self.next_lease = 1
        self.dirty = False  # snapshot needed (set by bump)

    # All methods below assume self.lock is held.
    def bump(self):
        self.rev += 1
        self.dirty = True
        self.lock.notify_all()
        return self.rev

    # ---- snapshot persistence (the durability etcd gave the reference:
    # a store restart must not take every job down with it) ----
    def to_snapshot(self):
        now = time.monotonic()
        return {
            "kv": self.kv,
            "leases": {str(i): {"ttl": L["ttl"]} for i, L in self.leases.items()
                       if L["deadline"] > now},
            "rev": self.rev,
            "next_lease": self.next_lease,
        }

    def load_snapshot(self, d):
        self.kv = dict(d.get("kv", {}))
        now = time.monotonic()
        # every surviving lease gets one full-TTL grace window: owners
        # resume keepalives as soon as their client reconnects
        self.leases = {int(i): {"ttl": L["ttl"], "deadline": now + L["ttl"]}
                       for i, L in d.get("leases", {}).items()}
        self.rev = int(d.get("rev", 0))
        self.next_lease = int(d.get("next_lease", 1))

    def alive(self, lease_id):
        L = self.leases.get(lease_id)
        return L is not None and L["deadline"] > time.monotonic()

    def expire_leases(self):
        now = time.monotonic()
        dead = [i for i, L in self.leases.items() if L["deadline"] <= now]
        changed = False
        for i in dead:
            del self.leases[i]
            for k in [k for k, e in self.kv.items() if e["lease"] == i]:
                del self.kv[k]
                changed = True
        if changed or dead:
            self.bump()
        return dead


class _Handler(socketserver.BaseRequestHandler):
    def handle(self):
        st = self.server.state
        self.request.setsockopt(socket.IPPROTO_TCP, socket.TCP_NODELAY, 1)
        try:
            while True:
                req = recv_msg(self.request)
                try:
                    resp = self.dispatch(st, req)
                except Exception as e:  # noqa: BLE001 - marshal to client
                    resp = {"ok": False, "err": "%s: %s" % (type(e).__name__, e)}
                send_msg(self.request, resp)
        except (ConnectionError, OSError):
            pass

    def dispatch(self, st, req):
        op = req.get("op")
        fn = getattr(self, "op_" + op, None)
        if fn is None:
            return {"ok": False, "err": "unknown op %r" % op}
        return fn(st, req)

    # ---- lease ops ----
    def op_grant(self, st, req):
        ttl = float(req["ttl"])
        with st.lock:
            lid = st.next_lease
            st.next_lease += 1
            st.leases[lid] = {"ttl": ttl, "deadline": time.monotonic() + ttl}
        return {"ok": True, "lease": lid}

    def op_keepalive(self, st, req):
        lid = int(req["lease"])
        with st.lock:
            L = st.leases.get(lid)
            if L is None or L["deadline"] <= time.monotonic():
                st.leases.pop(lid, None)
                return {"ok": False, "err": "lease expired"}
            L["deadline"] = time.monotonic() + L["ttl"]
        return {"ok": True}

    def op_revoke(self, st, req):
        lid = int(req["lease"])
        with st.lock:
            if lid in st.leases:
                st.leases[lid]["deadline"] = 0.0
                st.expire_leases()
        return {"ok": True}

    # ---- kv ops ----
    def op_put(self, st, req):
        key, val = req["key"], req["val"]
        lease = req.get("lease")
        with st.lock:
            if lease is not None and not st.alive(lease):
                return {"ok": False, "err": "lease expired"}
            e = st.kv.get(key)
            ver = (e["ver"] + 1) if e else 1
            st.kv[key] = {"v": val, "lease": lease, "ver": ver, "rev": st.rev + 1}
            return {"ok": True, "rev": st.bump()}

    def op_get(self, st, req):
        with st.lock:
            st.expire_leases()
            e = st.kv.get(req["key"])
            if e is None:
                return {"ok": True, "found": False, "rev": st.rev}
            return {"ok": True, "found": True, "val": e["v"], "ver": e["ver"], "rev": st.rev}

    def op_range(self, st, req):
        pfx = req["prefix"]
        with st.lock:
            st.expire_leases()
            kvs = sorted((k, e["v"]) for k, e in st.kv.items() if k.startswith(pfx))
            return {"ok": True, "kvs": kvs, "rev": st.rev}

    def op_delete(self, st, req):
        with st.lock:
            if st.kv.pop(req["key"], None) is not None:
                return {"ok": True, "deleted": 1, "rev": st.bump()}
            return {"ok": True, "deleted": 0, "rev": st.rev}

    def op_delete_prefix(self, st, req):
        pfx = req["prefix"]
        with st.lock:
            ks = [k for k in st.kv if k.startswith(pfx)]
            for k in ks:
                del st.kv[k]
            rev = st.bump() if ks else st.rev
            return {"ok": True, "deleted": len(ks), "rev": rev}

    def op_cas(self, st, req):
        """Put-if-absent, optionally lease-bound. Returns acquired=True/False
        plus the current holder's value when held (leader election)."""
        key, val, lease = req["key"], req["val"], req.get("lease")
        with st.lock:
            st.expire_leases()
            if lease is not None and not st.alive(lease):
                return {"ok": False, "err": "lease expired"}
            e = st.kv.get(key)
            if e is not None:
                return {"ok": True, "acquired": False, "val": e["v"]}
            st.kv[key] = {"v": val, "lease": lease, "ver": 1, "rev": st.rev + 1}
            return {"ok": True, "acquired": True, "rev": st.bump()}

    def op_txn(self, st, req):
        """If kv[guard_key].v == guard_val: apply puts/dels atomically.

        The leader-guard idiom of the reference's etcd transactions
        (cluster_generator.py:224-250, state.py:186-200).
        """
        gk, gv = req["guard_key"], req["guard_val"]
        with st.lock:
            st.expire_leases()
            e = st.kv.get(gk)
            if e is None or e["v"] != gv:
                return {"ok": True, "applied": False}
            for item in req.get("puts", []):
                k, v = item[0], item[1]
                lease = item[2] if len(item) > 2 else None
                if lease is not None and not st.alive(lease):
                    return {"ok": False, "err": "lease expired"}
                old = st.kv.get(k)
                ver = (old["ver"] + 1) if old else 1
                st.kv[k] = {"v": v, "lease": lease, "ver": ver, "rev": st.rev + 1}
            for k in req.get("dels", []):
                st.kv.pop(k, None)
            return {"ok": True, "applied": True, "rev": st.bump()}

    def op_wait(self, st, req):
        """Long-poll: block until the global revision advances past rev, or
        timeout. Deliberately coarse (any mutation wakes every waiter —
        deletes included); the client re-reads its prefix and diffs content.
        Control-plane scale makes spurious wakeups free."""
        rev = int(req["rev"])
        deadline = time.monotonic() + min(float(req.get("timeout", 30.0)), 300.0)
        with st.lock:
            while True:
                st.expire_leases()
                if st.rev > rev:
                    return {"ok": True, "rev": st.rev, "changed": True}
                remain = deadline - time.monotonic()
                if remain <= 0:
                    return {"ok": True, "rev": st.rev, "changed": False}
                st.lock.wait(min(remain, 0.5))

    def op_ping(self, st, req):
        return {"ok": True, "rev": st.rev}


class _TCPServer(socketserver.ThreadingTCPServer):
    allow_reuse_address = True
    daemon_threads = True


class CoordServer:
    """In-process coordination store server."""

    def __init__(self, host="127.0.0.1", port=0, snapshot=None):
        """snapshot: optional path — state is persisted there (atomic
        rename, written by the sweeper when dirty and on stop) and
        reloaded on construction, so a store restart on the same
        endpoint preserves keys and gives leases one TTL of grace."""
        self._srv = _TCPServer((host, port), _Handler)
        self._srv.state = _State()
        self._snapshot = snapshot
        if snapshot and os.path.exists(snapshot):
            with open(snapshot) as f:
                self._srv.state.load_snapshot(json.load(f))
            log.info("coordd: restored %d keys, %d leases from %s",
                     len(self._srv.state.kv), len(self._srv.state.leases),
                     snapshot)
        self._thread = None
        self._sweeper = None
        self._stop = threading.Event()
        self.host, self.port = self._srv.server_address

    @property
    def endpoint(self):
        return "%s:%d" % (self.host, self.port)

    def start(self):
        self._thread = threading.Thread(target=self._srv.serve_forever, daemon=True, name="coordd")
        self._thread.start()
        self._sweeper = threading.Thread(target=self._sweep, daemon=True, name="coordd-sweep")
        self._sweeper.start()
        return self

    def _sweep(self):
        st = self._srv.state
        while not self._stop.wait(0.5):
            with st.lock:
                st.expire_leases()
            self._maybe_snapshot()

    def _maybe_snapshot(self, force=False):
        if not self._snapshot:
            return
        st = self._srv.state
        with st.lock:
            if not (st.dirty or force):
                return
            snap = st.to_snapshot()
            st.dirty = False
        tmp = self._snapshot + ".tmp"
        with open(tmp, "w") as f:
            json.dump(snap, f)
        os.replace(tmp, self._snapshot)

    def stop(self):
        self._stop.set()
        self._srv.shutdown()
        self._srv.server_close()
        self._maybe_snapshot(force=True)


def main():
    ap = argparse.ArgumentParser(description="edl_amd coordination store server")
    ap.add_argument("--host", default="0.0.0.0")
    ap.add_argument("--port", type=int, default=2379)
    ap.add_argument("--snapshot", default=None,
                    help="persist state here; reload on restart")
    args = ap.parse_args()
    srv = CoordServer(args.host, args.port, snapshot=args.snapshot).start()
    log.info("coordd listening on %s", srv.endpoint)
    try:
        while True:
            time.sleep(3600)
    except KeyboardInterrupt:
        srv.stop()


if __name__ == "__main__":
    main()
