"""Elastic data plane: leader-balanced batch redistribution.

Parity: reference utils/data_server.py:31-431 + data_server_client.py —
every pod runs a DataServer caching the batches it produced; the LEADER
instance additionally (a) hands each pod its file-list slice round-robin
(PodsData._init, 118-133) and (b) collects produced batch ids per pod and,
once every unfinished pod has reported, levels the queues to the average by
stealing surplus from fast pods (PodsData.put 171-224). Slow pods then
fetch the stolen batches from the producing pod's cache (GetBatchData,
319-330).

Transport: the distill proto (length-prefixed JSON + raw payloads) instead
of gRPC — one wire format across the repo."""
import pickle
import socket
import socketserver
import threading

from ..distill import proto
from ..utils.log import get_logger

log = get_logger("edl.data_server")


class PodsData:
    """Leader-side balancing state."""

    def __init__(self, file_list, pod_ids):
        self.file_list = list(file_list)
        self.pod_ids = list(pod_ids)
        # round-robin file slices (reference PodsData._init)
        self.slices = {p: [] for p in pod_ids}
        for i, f in enumerate(self.file_list):
            self.slices[pod_ids[i % len(pod_ids)]].append(f)
        self.lock = threading.Lock()
        self.queues = {p: [] for p in pod_ids}      # produced, unassigned
        self.reported = {p: False for p in pod_ids}
        self.finished = {p: False for p in pod_ids}
        self.assignments = {p: [] for p in pod_ids}  # balanced output

    def get_file_list(self, pod_id):
        return self.slices.get(pod_id, [])

    def report(self, pod_id, batch_ids, finished=False):
        """A pod reports newly produced batch ids (owner = pod_id)."""
        with self.lock:
            self.queues[pod_id].extend((pod_id, b) for b in batch_ids)
            self.reported[pod_id] = True
            if finished:
                self.finished[pod_id] = True
            self._maybe_balance()

    def _maybe_balance(self):
        # "finished" = finished PRODUCING (data end); every pod still
        # CONSUMES, so leveling always targets all pods. Gate: every pod
        # has reported at least once (or already hit data end).
        if not all(self.reported[p] or self.finished[p] for p in self.pod_ids):
            return
        total = sum(len(q) for q in self.queues.values())
        if total == 0:
            return
        targets = self.pod_ids
        avg = total // len(targets)
        if avg == 0:
            # too little to level: hand out round-robin by current load
            pool = [item for p in self.pod_ids for item in self.queues[p]]
            for p in self.pod_ids:
                self.queues[p] = []
            order = sorted(targets, key=lambda p: len(self.assignments[p]))
            for j, item in enumerate(pool):
                self.assignments[order[j % len(order)]].append(item)
            return
        # keep up to avg locally (locality), pool the surplus, top up the
        # CURRENTLY least-loaded pod with each pooled item
        pool = []
        for p in self.pod_ids:
            self.assignments[p].extend(self.queues[p][:avg])
            pool.extend(self.queues[p][avg:])
            self.queues[p] = []
        for item in pool:
            p = min(targets, key=lambda q: len(self.assignments[q]))
            self.assignments[p].append(item)

    def take_assignments(self, pod_id, max_n=64):
        with self.lock:
            out = self.assignments.get(pod_id, [])[:max_n]
            self.assignments[pod_id] = self.assignments.get(pod_id, [])[max_n:]
            # done = nothing further will ever arrive for THIS pod
            done = all(self.finished.values()) and \
                not any(self.queues.values()) and \
                not self.assignments.get(pod_id)
            return out, done


class _Handler(socketserver.BaseRequestHandler):
    def handle(self):
        self.request.setsockopt(socket.IPPROTO_TCP, socket.TCP_NODELAY, 1)
        srv = self.server.owner
        try:
            while True:
                header, payloads = proto._recv(self.request)
                op = header["op"]
                meta = header.get("meta", {})
                if op == "get_file_list":
                    fl = srv.pods_data.get_file_list(meta["pod_id"])
                    proto._send(self.request, {"op": "ok", "meta": {"files": fl},
                                               "n_payloads": 0}, [])
                elif op == "report":
                    srv.pods_data.report(meta["pod_id"], meta["batch_ids"],
                                         meta.get("finished", False))
                    proto._send(self.request, {"op": "ok", "n_payloads": 0}, [])
                elif op == "get_meta":
                    items, done = srv.pods_data.take_assignments(meta["pod_id"])
                    proto._send(self.request,
                                {"op": "ok",
                                 "meta": {"items": items, "done": done},
                                 "n_payloads": 0}, [])
                elif op == "get_batch":
                    data = srv.get_batch(meta["batch_id"])
                    proto._send(self.request, {"op": "ok", "n_payloads": 1},
                                [data])
                elif op == "put_batch":  # producer caches its batch
                    srv.put_batch(meta["batch_id"], payloads[0])
                    proto._send(self.request, {"op": "ok", "n_payloads": 0}, [])
                else:
                    proto._send(self.request,
                                {"op": "err", "meta": {"err": "bad op"},
                                 "n_payloads": 0}, [])
        except (ConnectionError, OSError):
            pass


class _TCP(socketserver.ThreadingTCPServer):
    allow_reuse_address = True
    daemon_threads = True


class DataServer:
    """Per-pod data server; the leader's also balances (pods_data set)."""

    def __init__(self, host="127.0.0.1", port=0, file_list=None, pod_ids=None):
        self._srv = _TCP((host, port), _Handler)
        self._srv.owner = self
        self.port = self._srv.server_address[1]
        self.pods_data = PodsData(file_list or [], pod_ids or [])
        self._cache = {}
        self._cache_lock = threading.Lock()
        self._thread = None

    def start(self):
        self._thread = threading.Thread(target=self._srv.serve_forever,
                                        daemon=True, name="data-server")
        self._thread.start()
        return self

    def stop(self):
        self._srv.shutdown()
        self._srv.server_close()

    def put_batch(self, batch_id, blob):
        with self._cache_lock:
            self._cache[batch_id] = blob

    def get_batch(self, batch_id):
        with self._cache_lock:
            return self._cache[batch_id]


class DataClient:
    def __init__(self, endpoint):
        self._sock = proto.connect(endpoint)
        self._lock = threading.Lock()

    def _call(self, op, meta=None, payloads=()):
        with self._lock:
            proto._send(self._sock,
                        {"op": op, "meta": meta or {},
                         "n_payloads": len(payloads)}, list(payloads))
            header, pl = proto._recv(self._sock)
        if header["op"] != "ok":
            raise RuntimeError(header.get("meta", {}).get("err", "data rpc error"))
        return header.get("meta", {}), pl

    def get_file_list(self, pod_id):
        meta, _ = self._call("get_file_list", {"pod_id": pod_id})
        return meta["files"]

    def report(self, pod_id, batch_ids, finished=False):
        self._call("report", {"pod_id": pod_id, "batch_ids": batch_ids,
                              "finished": finished})

    def get_meta(self, pod_id):
        meta, _ = self._call("get_meta", {"pod_id": pod_id})
        return [tuple(i) for i in meta["items"]], meta["done"]

    def put_batch(self, batch_id, obj):
        self._call("put_batch", {"batch_id": batch_id}, [pickle.dumps(obj)])

    def get_batch(self, batch_id):
        _, pl = self._call("get_batch", {"batch_id": batch_id})
        return pickle.loads(pl[0])

    def close(self):
        self._sock.close()
