"""Leader election + cluster generator + barrier integration tests.

Mirrors reference test_leader_pod.py:45-61 (seize leadership, stop leader,
assert failover after TTL) and test_cluster_generator.py:57-98 (two pods +
barrier returns cluster)."""
import time

import pytest

import edl_amd.coord.tables as tables
from edl_amd.cluster.barrier import barrier
from edl_amd.cluster.generator import ClusterGenerator
from edl_amd.cluster.leader import LeaderElector
from edl_amd.cluster.model import load_cluster
from edl_amd.cluster.resource import ResourceRegister, load_resource_pods
from edl_amd.cluster.status import Status, save_pod_status
from edl_amd.coord.register import Register
from edl_amd.utils.errors import EdlPodIDNotExistError, EdlRegisterError
from tests.test_cluster_model import make_pod


def wait_until(fn, timeout=10.0, poll=0.05):
    deadline = time.monotonic() + timeout
    while time.monotonic() < deadline:
        if fn():
            return True
        time.sleep(poll)
    return False


def test_register_ttl_and_stop(coord_client):
    r = Register(coord_client, "/k", "v", ttl=1.0).start()
    assert coord_client.get("/k") == "v"
    time.sleep(1.5)  # refresh keeps it alive past TTL
    assert coord_client.get("/k") == "v"
    r.stop()
    assert coord_client.get("/k") is None  # revoked on stop


def test_register_exclusive(coord_client):
    r1 = Register(coord_client, "/x", "a", ttl=5, exclusive=True).start()
    with pytest.raises(EdlRegisterError):
        Register(coord_client, "/x", "b", ttl=5, exclusive=True).start()
    r1.stop()
    r2 = Register(coord_client, "/x", "b", ttl=5, exclusive=True).start()
    assert coord_client.get("/x") == "b"
    r2.stop()


def test_resource_register_roundtrip(coord_client):
    pod = make_pod("pod_r")
    reg = ResourceRegister(coord_client, pod).start()
    pods = load_resource_pods(coord_client)
    assert "pod_r" in pods and pods["pod_r"] == pod
    reg.stop()
    assert load_resource_pods(coord_client) == {}


def test_leader_failover(coord_server, coord_client, monkeypatch):
    # Shrink the TTL so failover happens fast in the test.
    monkeypatch.setattr(tables, "ETCD_TTL", 1.0)
    import edl_amd.cluster.leader as leader_mod

    monkeypatch.setattr(leader_mod, "ETCD_TTL", 1.0)

    e1 = LeaderElector(coord_client, "pod_a", retry_interval=0.2).start()
    assert wait_until(lambda: e1.is_leader)
    from edl_amd.coord.client import CoordClient

    c2 = CoordClient(coord_server.endpoint, "test_job")
    e2 = LeaderElector(c2, "pod_b", retry_interval=0.2).start()
    time.sleep(0.5)
    assert not e2.is_leader
    assert e2.leader_id() == "pod_a"

    e1.stop()  # revokes lease -> key expires
    assert wait_until(lambda: e2.is_leader, timeout=10)
    assert e2.leader_id() == "pod_b"
    e2.stop()
    c2.close()


def test_generator_first_boot_and_join_and_leave(coord_server, coord_client):
    from edl_amd.coord.client import CoordClient

    pod_a, pod_b = make_pod("a"), make_pod("b")
    reg_a = ResourceRegister(coord_client, pod_a).start()
    save_pod_status(coord_client, "a", Status.INITIAL)
    # pod a is leader
    coord_client.put(coord_client.table_key(tables.ETCD_POD_RANK, "0"), "a")
    gen = ClusterGenerator(coord_client, "a", min_nodes=1, max_nodes=4, period=0.1)

    c = gen.generate_once()
    assert c is not None and c.pod_ids() == ["a"]
    assert c.world_size() == 2
    assert load_cluster(coord_client).stage == c.stage

    # join: pod b registers as INITIAL -> appended with new stage
    cb = CoordClient(coord_server.endpoint, "test_job")
    reg_b = ResourceRegister(cb, pod_b).start()
    save_pod_status(cb, "b", Status.INITIAL)
    c2 = gen.generate_once()
    assert c2.pod_ids() == ["a", "b"]
    assert c2.stage != c.stage
    assert c2.world_size() == 4
    ranks = [t.global_rank for p in c2.pods for t in p.trainers]
    assert ranks == [0, 1, 2, 3]

    # leave: pod b's registration disappears -> regenerated without it
    reg_b.stop()
    c3 = gen.generate_once()
    assert c3.pod_ids() == ["a"]
    assert c3.stage != c2.stage
    reg_a.stop()
    cb.close()


def test_generator_holds_below_min(coord_client):
    pod_a = make_pod("a")
    reg = ResourceRegister(coord_client, pod_a).start()
    save_pod_status(coord_client, "a", Status.INITIAL)
    coord_client.put(coord_client.table_key(tables.ETCD_POD_RANK, "0"), "a")
    gen = ClusterGenerator(coord_client, "a", min_nodes=2, period=0.1)
    assert gen.generate_once() is None
    assert gen.holding.is_set()
    assert load_cluster(coord_client) is None
    reg.stop()


def test_barrier_two_pods(coord_server, coord_client):
    import threading

    from edl_amd.coord.client import CoordClient

    pod_a, pod_b = make_pod("a"), make_pod("b")
    reg_a = ResourceRegister(coord_client, pod_a).start()
    cb = CoordClient(coord_server.endpoint, "test_job")
    reg_b = ResourceRegister(cb, pod_b).start()
    coord_client.put(coord_client.table_key(tables.ETCD_POD_RANK, "0"), "a")
    gen = ClusterGenerator(coord_client, "a", min_nodes=2, period=0.1)
    gen.generate_once()

    results = {}

    def arrive(name, client):
        results[name] = barrier(client, name, timeout=10)

    t1 = threading.Thread(target=arrive, args=("a", coord_client))
    t2 = threading.Thread(target=arrive, args=("b", cb))
    t1.start()
    time.sleep(0.3)
    assert "a" not in results  # a alone must not pass
    t2.start()
    t1.join(10)
    t2.join(10)
    assert results["a"].stage == results["b"].stage
    assert results["a"].pod_ids() == ["a", "b"]
    reg_a.stop()
    reg_b.stop()
    cb.close()


def test_barrier_scale_in_raises(coord_client):
    pod_a = make_pod("a")
    reg = ResourceRegister(coord_client, pod_a).start()
    coord_client.put(coord_client.table_key(tables.ETCD_POD_RANK, "0"), "a")
    gen = ClusterGenerator(coord_client, "a", min_nodes=1, period=0.1)
    gen.generate_once()
    with pytest.raises(EdlPodIDNotExistError):
        barrier(coord_client, "not_a_member", timeout=5)
    reg.stop()


def test_scale_in_request_drops_pod(coord_server, coord_client):
    """External scale-in API (reference PodServer.ScaleIn): the generator
    drops the named pod on its next pass."""
    from edl_amd.cluster.scale import read_scale_request, request_scale
    from edl_amd.coord.client import CoordClient

    pod_a, pod_b = make_pod("a"), make_pod("b")
    reg_a = ResourceRegister(coord_client, pod_a).start()
    cb = CoordClient(coord_server.endpoint, "test_job")
    reg_b = ResourceRegister(cb, pod_b).start()
    save_pod_status(coord_client, "a", Status.INITIAL)
    save_pod_status(cb, "b", Status.INITIAL)
    coord_client.put(coord_client.table_key(tables.ETCD_POD_RANK, "0"), "a")
    gen = ClusterGenerator(coord_client, "a", min_nodes=1, period=0.1)
    c = gen.generate_once()
    assert c.pod_ids() == ["a", "b"]

    save_pod_status(cb, "b", Status.RUNNING)  # running pods are not re-added
    request_scale(coord_client, remove_pods=["b"])
    c2 = gen.generate_once()
    assert c2.pod_ids() == ["a"]
    assert read_scale_request(coord_client) is None  # consumed
    c3 = gen.generate_once()
    assert c3.pod_ids() == ["a"]  # stays out (status RUNNING, not INITIAL)
    reg_a.stop()
    reg_b.stop()
    cb.close()


def test_scale_request_survives_lost_leadership(coord_server, coord_client):
    """A pending scale-in request must NOT be consumed when the guarded
    publish fails (leadership lost mid-publish) — clearing first would
    silently drop the scale-in (reference guarded-txn pattern:
    utils/cluster_generator.py:224-250)."""
    from edl_amd.cluster.scale import read_scale_request, request_scale
    from edl_amd.coord.client import CoordClient

    pod_a, pod_b = make_pod("a"), make_pod("b")
    reg_a = ResourceRegister(coord_client, pod_a).start()
    cb = CoordClient(coord_server.endpoint, "test_job")
    reg_b = ResourceRegister(cb, pod_b).start()
    save_pod_status(coord_client, "a", Status.INITIAL)
    save_pod_status(cb, "b", Status.INITIAL)
    rank_key = coord_client.table_key(tables.ETCD_POD_RANK, "0")
    coord_client.put(rank_key, "a")
    gen = ClusterGenerator(coord_client, "a", min_nodes=1, period=0.1)
    assert gen.generate_once().pod_ids() == ["a", "b"]
    save_pod_status(cb, "b", Status.RUNNING)

    # leadership flips to b BEFORE a's generator processes the request:
    # the guarded publish fails and the request must stay pending
    request_scale(coord_client, remove_pods=["b"])
    coord_client.put(rank_key, "b")
    c = gen.generate_once()
    assert c.pod_ids() == ["a", "b"]  # publish refused
    assert read_scale_request(coord_client) is not None  # NOT consumed

    # a regains leadership: the still-pending request now applies
    coord_client.put(rank_key, "a")
    c2 = gen.generate_once()
    assert c2.pod_ids() == ["a"]
    assert read_scale_request(coord_client) is None  # consumed after publish
    reg_a.stop()
    reg_b.stop()
    cb.close()


def test_scale_request_noop_is_cleared(coord_server, coord_client):
    """A request naming no known pod is a no-op and must not linger."""
    from edl_amd.cluster.scale import read_scale_request, request_scale

    pod_a = make_pod("a")
    reg_a = ResourceRegister(coord_client, pod_a).start()
    save_pod_status(coord_client, "a", Status.INITIAL)
    coord_client.put(coord_client.table_key(tables.ETCD_POD_RANK, "0"), "a")
    gen = ClusterGenerator(coord_client, "a", min_nodes=1, period=0.1)
    assert gen.generate_once().pod_ids() == ["a"]
    request_scale(coord_client, remove_pods=["ghost"])
    assert gen.generate_once().pod_ids() == ["a"]
    assert read_scale_request(coord_client) is None
    reg_a.stop()
