"""Balance algorithm + consistent hash unit tests (mirror reference
test_consistent_hash.py and balance_table semantics)."""
from edl_amd.distill.balance import ConsistentHash, Service


def test_ring_distribution_and_stability():
    ring = ConsistentHash(["s1", "s2", "s3"])
    keys = ["service_%d" % i for i in range(300)]
    owners = {k: ring.get_node(k) for k in keys}
    counts = {}
    for v in owners.values():
        counts[v] = counts.get(v, 0) + 1
    # roughly balanced: every node owns something substantial
    assert all(c > 30 for c in counts.values()), counts
    # removing one node must not move keys between surviving nodes
    ring.remove_node("s2")
    for k in keys:
        if owners[k] != "s2":
            assert ring.get_node(k) == owners[k]


def test_ring_single_and_empty():
    ring = ConsistentHash()
    assert ring.get_node("x") is None
    ring.add_node("only")
    assert ring.get_node("x") == "only"


def test_service_balance_even():
    s = Service("svc")
    s.update_servers(["t%d" % i for i in range(4)])
    s.update_clients({"c%d" % i: 4 for i in range(2)})
    assert s.rebalance()
    # 4 servers / 2 clients -> 2 each, disjoint (per-server cap 1)
    a0 = s.clients["c0"]["assigned"]
    a1 = s.clients["c1"]["assigned"]
    assert len(a0) == 2 and len(a1) == 2
    assert not set(a0) & set(a1)


def test_service_balance_more_clients_than_servers():
    s = Service("svc")
    s.update_servers(["t0", "t1"])
    s.update_clients({"c%d" % i: 1 for i in range(6)})
    s.rebalance()
    load = {}
    for c in s.clients.values():
        assert len(c["assigned"]) == 1
        load[c["assigned"][0]] = load.get(c["assigned"][0], 0) + 1
    assert load == {"t0": 3, "t1": 3}


def test_service_balance_minimal_churn_on_server_loss():
    s = Service("svc")
    s.update_servers(["t0", "t1", "t2", "t3"])
    s.update_clients({"c0": 4, "c1": 4})
    s.rebalance()
    before = {c: list(v["assigned"]) for c, v in s.clients.items()}
    v0 = s.version
    # lose one teacher: only assignments touching it should change
    lost = before["c0"][0]
    s.update_servers([t for t in ["t0", "t1", "t2", "t3"] if t != lost])
    s.rebalance()
    assert s.version > v0
    assert lost not in s.clients["c0"]["assigned"]
    # quota shrinks to max(1, 3//2) = 1; surviving prefix is kept (minimal
    # churn): each client retains its first still-alive previous teacher
    for cid in ("c0", "c1"):
        kept = [t for t in before[cid] if t != lost]
        assert s.clients[cid]["assigned"][0] == kept[0]


def test_service_respects_require():
    s = Service("svc")
    s.update_servers(["t%d" % i for i in range(8)])
    s.update_clients({"c0": 2})  # requires only 2 although 8 available
    s.rebalance()
    assert len(s.clients["c0"]["assigned"]) == 2


def test_service_no_servers():
    s = Service("svc")
    s.update_clients({"c0": 1})
    s.update_servers(["t0"])
    s.rebalance()
    assert s.clients["c0"]["assigned"] == ["t0"]
    s.update_servers([])
    assert s.rebalance()
    assert s.clients["c0"]["assigned"] == []
