"""Teacher-pool balancing: greedy bipartite assignment + consistent-hash
sharding.

Parity: reference distill/balance_table.py Service.rebalance (139-338):
clients (students) are assigned teachers with per-server cap
ceil(clients/servers) and per-client quota max(1, servers//clients)
(bounded by the client's require_num), adjusting incrementally so existing
assignments churn as little as possible. ConsistentHash mirrors
discovery/consistent_hash.py:21-141 (300 vnodes, MD5 ring) — it shards
SERVICE NAMES over discovery servers."""
import bisect
import hashlib
import math


class ConsistentHash:
    def __init__(self, nodes=(), vnodes=300):
        self._vnodes = vnodes
        self._ring = []  # sorted list of (hash, node)
        self._keys = []
        self.nodes = set()
        for n in nodes:
            self.add_node(n)

    @staticmethod
    def _hash(key):
        return int(hashlib.md5(key.encode()).hexdigest(), 16)

    def add_node(self, node):
        if node in self.nodes:
            return
        self.nodes.add(node)
        for i in range(self._vnodes):
            h = self._hash("%s#%d" % (node, i))
            idx = bisect.bisect(self._keys, h)
            self._keys.insert(idx, h)
            self._ring.insert(idx, (h, node))

    def remove_node(self, node):
        if node not in self.nodes:
            return
        self.nodes.discard(node)
        keep = [(h, n) for h, n in self._ring if n != node]
        self._ring = keep
        self._keys = [h for h, _ in keep]

    def get_node(self, key):
        if not self._ring:
            return None
        h = self._hash(key)
        idx = bisect.bisect(self._keys, h) % len(self._ring)
        return self._ring[idx][1]


class Service:
    """One service's server/client bipartite graph + greedy rebalance."""

    def __init__(self, name):
        self.name = name
        self.servers = set()
        self.clients = {}  # client_id -> dict(require=int, assigned=[endpoints])
        self.version = 0

    def update_servers(self, servers):
        self.servers = set(servers)

    def add_client(self, client_id, require=1):
        if client_id not in self.clients:
            self.clients[client_id] = {"require": require, "assigned": []}
        else:
            self.clients[client_id]["require"] = require

    def remove_client(self, client_id):
        self.clients.pop(client_id, None)

    def update_clients(self, client_requires):
        """client_requires: {client_id: require_num}; drops absent clients."""
        for cid in list(self.clients):
            if cid not in client_requires:
                del self.clients[cid]
        for cid, req in client_requires.items():
            self.add_client(cid, req)

    def rebalance(self):
        """-> True if any assignment changed (version bumped)."""
        ns, nc = len(self.servers), len(self.clients)
        changed = False
        if nc == 0:
            return False
        if ns == 0:
            for c in self.clients.values():
                if c["assigned"]:
                    c["assigned"] = []
                    changed = True
            if changed:
                self.version += 1
            return changed

        per_client = max(1, ns // nc)
        per_server = int(math.ceil(float(nc * per_client) / ns))
        load = {s: 0 for s in self.servers}

        # keep still-valid assignments (up to quota), count load
        for c in self.clients.values():
            quota = min(per_client, c["require"]) if c["require"] else per_client
            kept = []
            for s in c["assigned"]:
                if s in self.servers and len(kept) < quota and load[s] < per_server:
                    kept.append(s)
                    load[s] += 1
            if kept != c["assigned"]:
                changed = True
            c["assigned"] = kept

        # top up under-quota clients from least-loaded servers
        for cid in sorted(self.clients):
            c = self.clients[cid]
            quota = min(per_client, c["require"]) if c["require"] else per_client
            while len(c["assigned"]) < quota:
                candidates = [s for s in self.servers
                              if s not in c["assigned"] and load[s] < per_server]
                if not candidates:
                    break
                s = min(candidates, key=lambda x: (load[x], x))
                c["assigned"].append(s)
                load[s] += 1
                changed = True
        if changed:
            self.version += 1
        return changed
