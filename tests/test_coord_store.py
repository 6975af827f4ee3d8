"""Coordination store unit tests (substrate for everything else).

Mirrors the etcd behaviors the reference relies on: leases+TTL expiry,
put-if-absent CAS (set_server_not_exists), guarded transactions, watch."""
import threading
import time

import pytest

from edl_amd.coord.client import CoordClient
from edl_amd.utils.errors import EdlStoreError


def test_put_get_delete(coord_client):
    c = coord_client
    c.put("/a/b", "v1")
    assert c.get("/a/b") == "v1"
    c.put("/a/b", "v2")
    assert c.get("/a/b") == "v2"
    assert c.get("/missing") is None
    assert c.delete("/a/b") == 1
    assert c.get("/a/b") is None
    assert c.delete("/a/b") == 0


def test_range_and_delete_prefix(coord_client):
    c = coord_client
    for i in range(5):
        c.put("/t/nodes/%d" % i, str(i))
    c.put("/other", "x")
    kvs = c.range("/t/nodes/")
    assert [k for k, _ in kvs] == ["/t/nodes/%d" % i for i in range(5)]
    assert c.delete_prefix("/t/") == 5
    assert c.range("/t/") == []
    assert c.get("/other") == "x"


def test_lease_expiry(coord_client):
    c = coord_client
    lease = c.grant(0.6)
    c.put("/lease/k", "v", lease)
    assert c.get("/lease/k") == "v"
    time.sleep(1.0)
    assert c.get("/lease/k") is None
    assert not c.keepalive(lease)


def test_lease_keepalive(coord_client):
    c = coord_client
    lease = c.grant(0.8)
    c.put("/ka/k", "v", lease)
    for _ in range(4):
        time.sleep(0.3)
        assert c.keepalive(lease)
    assert c.get("/ka/k") == "v"


def test_cas_put_if_absent(coord_client):
    c = coord_client
    ok, val = c.put_if_absent("/rank/0", "pod_a")
    assert ok and val == "pod_a"
    ok, val = c.put_if_absent("/rank/0", "pod_b")
    assert not ok and val == "pod_a"


def test_cas_reacquire_after_lease_expiry(coord_client):
    c = coord_client
    lease = c.grant(0.5)
    ok, _ = c.put_if_absent("/rank/0", "pod_a", lease)
    assert ok
    time.sleep(0.9)
    ok, val = c.put_if_absent("/rank/0", "pod_b")
    assert ok and val == "pod_b"


def test_txn_guarded(coord_client):
    c = coord_client
    c.put("/rank/0", "leader_pod")
    assert c.txn_if("/rank/0", "leader_pod", puts=[("/cluster", "{}")])
    assert c.get("/cluster") == "{}"
    assert not c.txn_if("/rank/0", "other_pod", puts=[("/cluster", "BAD")])
    assert c.get("/cluster") == "{}"
    # txn with deletes
    assert c.txn_if("/rank/0", "leader_pod", dels=["/cluster"])
    assert c.get("/cluster") is None


def test_wait_wakes_on_change(coord_server, coord_client):
    c = coord_client
    rev = c.rev()
    result = {}

    def waiter():
        w = CoordClient(coord_server.endpoint, "test_job")
        result["changed"], result["rev"] = w.wait(rev, timeout=5.0)
        w.close()

    t = threading.Thread(target=waiter)
    t.start()
    time.sleep(0.2)
    c.put("/wake", "1")
    t.join(timeout=5)
    assert result["changed"]


def test_wait_timeout(coord_client):
    c = coord_client
    t0 = time.monotonic()
    changed, _ = c.wait(c.rev(), timeout=0.4)
    assert not changed
    assert time.monotonic() - t0 < 3.0


def test_unreachable_store_raises():
    c = CoordClient("127.0.0.1:1", "job")
    with pytest.raises(EdlStoreError):
        c.get("/x")


def test_table_key_layout(coord_client):
    assert coord_client.table_key("resource", "p1") == "/test_job/resource/nodes/p1"
    assert coord_client.table_key("resource") == "/test_job/resource/nodes/"


def test_concurrent_clients_stress(coord_server):
    """16 threads x (leases + puts + CAS races + watches): the store must
    stay consistent and exactly one CAS winner per key."""
    import threading

    winners = []
    lock = threading.Lock()
    errs = []

    def worker(i):
        try:
            c = CoordClient(coord_server.endpoint, "stress")
            lease = c.grant(5)
            for j in range(30):
                c.put("/stress/t%d/k%d" % (i, j), str(j), lease)
            ok, _ = c.put_if_absent("/stress/winner", "t%d" % i)
            if ok:
                with lock:
                    winners.append(i)
            assert c.keepalive(lease)
            kvs = c.range("/stress/t%d/" % i)
            assert len(kvs) == 30
            c.close()
        except Exception as e:  # noqa: BLE001
            errs.append(e)

    threads = [threading.Thread(target=worker, args=(i,)) for i in range(16)]
    for t in threads:
        t.start()
    for t in threads:
        t.join(30)
    assert not errs, errs
    assert len(winners) == 1
    c = CoordClient(coord_server.endpoint, "stress")
    assert len(c.range("/stress/")) == 16 * 30 + 1
    c.close()


def test_snapshot_restart_preserves_state(tmp_path):
    """Store restart with --snapshot: keys survive, leases get one TTL of
    grace (the durability the reference got from external etcd)."""
    from edl_amd.coord.client import CoordClient
    from edl_amd.coord.server import CoordServer

    snap = str(tmp_path / "coordd.json")
    srv = CoordServer(port=0, snapshot=snap).start()
    port = srv.port
    c = CoordClient(srv.endpoint, "job")
    lease = c.grant(5.0)
    c.put("/job/a/nodes/k1", "v1")
    c.put("/job/a/nodes/k2", "v2", lease=lease)
    c.close()
    srv.stop()  # final snapshot

    srv2 = CoordServer(port=port, snapshot=snap).start()
    try:
        c2 = CoordClient(srv2.endpoint, "job")
        assert c2.get("/job/a/nodes/k1") == "v1"
        assert c2.get("/job/a/nodes/k2") == "v2"  # lease in grace window
        assert c2.keepalive(lease)                # owner resumes refreshing
        c2.close()
    finally:
        srv2.stop()


def test_client_retry_window_rides_out_restart(tmp_path):
    """EDL_STORE_RETRY_S: an RPC issued while the store is down succeeds
    once a snapshot-backed replacement comes up on the same endpoint."""
    import threading
    import time

    from edl_amd.coord.client import CoordClient
    from edl_amd.coord.server import CoordServer

    snap = str(tmp_path / "coordd.json")
    srv = CoordServer(port=0, snapshot=snap).start()
    port = srv.port
    c = CoordClient(srv.endpoint, "job", retry_s=10.0)
    c.put("/job/t/nodes/x", "1")
    srv.stop()

    replacement = []

    def bring_back():
        time.sleep(1.0)
        replacement.append(CoordServer(port=port, snapshot=snap).start())

    t = threading.Thread(target=bring_back)
    t.start()
    try:
        assert c.get("/job/t/nodes/x") == "1"  # retried across the outage
    finally:
        t.join()
        c.close()
        for s in replacement:
            s.stop()


def test_snapshot_excludes_expired_leases(tmp_path):
    """Keys whose lease already lapsed must not resurrect on restart."""
    import time

    from edl_amd.coord.client import CoordClient
    from edl_amd.coord.server import CoordServer

    snap = str(tmp_path / "c.json")
    srv = CoordServer(port=0, snapshot=snap).start()
    port = srv.port
    c = CoordClient(srv.endpoint, "job")
    lease = c.grant(0.5)
    c.put("/job/x/nodes/ephemeral", "v", lease=lease)
    c.put("/job/x/nodes/durable", "v")
    time.sleep(1.2)  # lease lapses; sweeper expires + snapshots
    c.close()
    srv.stop()

    srv2 = CoordServer(port=port, snapshot=snap).start()
    try:
        c2 = CoordClient(srv2.endpoint, "job")
        assert c2.get("/job/x/nodes/ephemeral") is None
        assert c2.get("/job/x/nodes/durable") == "v"
        c2.close()
    finally:
        srv2.stop()
