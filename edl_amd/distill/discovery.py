"""Teacher discovery tier over the coordination store.

Parity map to the reference's two flavors (SURVEY C25 etcd/gRPC,
C26 redis/epoll — neither etcd nor redis exists in the MI355X image, so
ONE store-backed implementation covers both APIs):

  * DiscoveryServer == discovery_server.py + balance_table.BalanceTable:
    registers itself under the `balance` table, shards service names over
    the live discovery servers with the 300-vnode consistent-hash ring
    (balance_table.py:393-464), runs the greedy rebalance for services it
    OWNS, and publishes assignments to service_assign/<service>/<client>.
    (The reference's REDIRECT protocol, distill_discovery.proto Code
    enum, is unnecessary by construction: clients read assignments from
    the store, not from a specific discovery server.)
  * DiscoveryClient == discovery_client.py: lease-bound client
    registration (the 2 s heartbeat is the lease keepalive) + versioned
    assignment reads (90-109)."""
import json
import threading
import time as _time
import uuid

from ..coord.client import CoordClient
from ..coord.register import Register
from ..coord.tables import (
    ETCD_BALANCE,
    ETCD_SERVICE,
    ETCD_SERVICE_ASSIGN,
    ETCD_SERVICE_CLIENTS,
)
from ..utils.errors import EdlStoreError
from ..utils.log import get_logger
from .balance import ConsistentHash, Service

log = get_logger("edl.discovery")


class DiscoveryServer:
    def __init__(self, store_endpoints, job_id="distill", server_id=None, period=1.0):
        self._client = CoordClient(store_endpoints, job_id)
        self.server_id = server_id or uuid.uuid4().hex[:8]
        self._period = period
        self._reg = None
        self._services = {}  # name -> Service
        self._stop = threading.Event()
        self._thread = None

    # ---- lifecycle ----
    def start(self):
        self._reg = Register(
            self._client, self._client.table_key(ETCD_BALANCE, self.server_id), "1"
        ).start()
        self._thread = threading.Thread(target=self._loop, daemon=True,
                                        name="discovery-balance")
        self._thread.start()
        return self

    def stop(self):
        self._stop.set()
        if self._thread:
            self._thread.join(timeout=5)
        if self._reg:
            self._reg.stop()
        self._client.close()

    # ---- the balance loop ----
    def _owned(self, service_name):
        pfx = self._client.table_key(ETCD_BALANCE)
        balancers = sorted(k[len(pfx):] for k, _ in self._client.range(pfx))
        if not balancers:
            return True
        ring = ConsistentHash(balancers)
        return ring.get_node(service_name) == self.server_id

    def _loop(self):
        while not self._stop.wait(self._period):
            try:
                self.balance_once()
            except EdlStoreError as e:
                log.debug("balance retry: %s", e)

    def balance_once(self):
        c = self._client
        # discover all service names from both teachers and clients tables
        spfx = c.table_key(ETCD_SERVICE)
        cpfx = c.table_key(ETCD_SERVICE_CLIENTS)
        names = set()
        servers = {}
        for k, _ in c.range(spfx):
            name, _, ep = k[len(spfx):].partition("/")
            names.add(name)
            servers.setdefault(name, []).append(ep)
        clients = {}
        for k, v in c.range(cpfx):
            name, _, cid = k[len(cpfx):].partition("/")
            names.add(name)
            try:
                req = json.loads(v).get("require", 1)
            except ValueError:
                req = 1
            clients.setdefault(name, {})[cid] = req

        for name in sorted(names):
            if not self._owned(name):
                self._services.pop(name, None)
                continue
            svc = self._services.setdefault(name, Service(name))
            svc.update_servers(servers.get(name, []))
            svc.update_clients(clients.get(name, {}))
            if svc.rebalance():
                for cid, ent in svc.clients.items():
                    key = c.table_key(ETCD_SERVICE_ASSIGN, "%s/%s" % (name, cid))
                    c.put(key, json.dumps(
                        {"version": svc.version, "servers": ent["assigned"]}))
                log.info("service %s rebalanced v%d (%d teachers, %d students)",
                         name, svc.version, len(svc.servers), len(svc.clients))
            # GC assignments of departed clients
            apfx = c.table_key(ETCD_SERVICE_ASSIGN, name + "/")
            for k, _ in c.range(apfx):
                cid = k[len(apfx):]
                if cid not in svc.clients:
                    c.delete(k)


class DiscoveryClient:
    """Student-side: register (lease-bound), heartbeat, read assignments."""

    def __init__(self, store_endpoints, service_name, require=1, job_id="distill",
                 client_id=None):
        self._client = CoordClient(store_endpoints, job_id)
        self.service_name = service_name
        self.client_id = client_id or uuid.uuid4().hex[:12]
        self._require = require
        self._reg = None
        self._version = -1

    def start(self):
        key = self._client.table_key(
            ETCD_SERVICE_CLIENTS, "%s/%s" % (self.service_name, self.client_id))
        self._reg = Register(self._client, key,
                             json.dumps({"require": self._require})).start()
        return self

    def get_servers(self):
        """-> (changed, [teacher endpoints])."""
        key = self._client.table_key(
            ETCD_SERVICE_ASSIGN, "%s/%s" % (self.service_name, self.client_id))
        v = self._client.get(key)
        if not v:
            return False, []
        d = json.loads(v)
        changed = d["version"] != self._version
        self._version = d["version"]
        return changed, d["servers"]

    def stop(self):
        if self._reg:
            self._reg.stop()
        try:
            self._client.delete(self._client.table_key(
                ETCD_SERVICE_CLIENTS, "%s/%s" % (self.service_name, self.client_id)))
        except EdlStoreError:
            pass
        self._client.close()


class RedisFlavorClient:
    """Drop-in for the reference's redis-flavor balance client
    (distill/redis/client.py:24-147): same constructor shape
    (endpoints, service_name, require_num, token), same surface —
    start() -> teacher list, get_teacher_list() -> (is_update, servers),
    get_servers(), stop() — and the same update semantics (a background
    ~2 s heartbeat notices version bumps; is_update latches until read).

    The transport is the coordination store instead of the reference's
    epoll TCP server + redis keys: C26's registry/TTL/rebalance semantics
    are provided by the SAME DiscoveryServer that backs the etcd-flavor
    API (SURVEY C25/C26 — one store-backed impl, two client surfaces)."""

    def __init__(self, endpoints, service_name, require_num, token=None,
                 job_id="distill", heartbeat_s=2.0):
        self._inner = DiscoveryClient(endpoints, service_name,
                                      require=require_num)
        self._heartbeat_s = heartbeat_s
        self.teacher_list = []
        self._is_update = False
        self._version_seen = False
        self._lock = threading.Lock()
        self._stop = threading.Event()
        self._thread = None

    def _poll_once(self):
        changed, servers = self._inner.get_servers()
        if changed:
            with self._lock:
                # first assignment is the register reply, not an update
                if self._version_seen:
                    self._is_update = True
                self._version_seen = True
                self.teacher_list = list(servers)

    def _heartbeat(self):
        while not self._stop.wait(self._heartbeat_s):
            try:
                self._poll_once()
            except EdlStoreError:
                pass  # store hiccup: next heartbeat retries

    def start(self, daemon=True, timeout=60.0):
        """Register and block until the first assignment arrives (the
        reference's _register reply); returns the teacher list."""
        self._inner.start()
        deadline = _time.monotonic() + timeout
        while _time.monotonic() < deadline:
            self._poll_once()
            if self._version_seen:
                break
            _time.sleep(0.1)
        self._thread = threading.Thread(target=self._heartbeat, daemon=daemon,
                                        name="redis-flavor-heartbeat")
        self._thread.start()
        return list(self.teacher_list)

    def get_teacher_list(self):
        """-> (is_update, servers); is_update latches until read."""
        with self._lock:
            upd, self._is_update = self._is_update, False
            return upd, list(self.teacher_list)

    def get_servers(self):
        with self._lock:
            return list(self.teacher_list)

    def stop(self):
        self._stop.set()
        if self._thread:
            self._thread.join(timeout=5)
        self._inner.stop()
