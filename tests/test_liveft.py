"""liveft hot-restart elasticity tests (CPU, in-process managers +
subprocess trainers)."""
import os
import sys
import threading
import time

from edl_amd.liveft.elastic import (
    ELASTIC_EXIT_CODE,
    ElasticManager,
    ElasticStatus,
    LauncherInterface,
)

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))


def _em(coord_server, **kw):
    kw.setdefault("store_endpoints", coord_server.endpoint)
    kw.setdefault("job_id", "liveft_test")
    return ElasticManager(**kw)


def test_wait_until_np(coord_server):
    m1 = _em(coord_server, np=2, host="h1@1").start()
    assert not m1.wait(timeout=1)
    m2 = _em(coord_server, np=2, host="h2@2").start()
    assert m1.wait(timeout=10) and m2.wait(timeout=10)
    assert m1.hosts == m2.hosts == ["h1@1", "h2@2"]
    assert m1.rank == 0 and m2.rank == 1
    m2.stop()
    # world shrinks -> changed
    deadline = time.monotonic() + 10
    while time.monotonic() < deadline and not m1.world_changed():
        time.sleep(0.2)
    assert m1.world_changed()
    m1.stop()


def _script(tmp_path, body):
    p = tmp_path / "t.py"
    p.write_text(body)
    return str(p)


def test_watch_completed(coord_server, tmp_path):
    m = _em(coord_server, np=1, host="h@1").start()
    assert m.wait(timeout=5)
    lf = LauncherInterface([_script(tmp_path, "print('ok')")],
                           log_dir=str(tmp_path))
    m.run(lf)
    assert m.watch(lf) == ElasticStatus.COMPLETED
    m.stop()


def test_watch_restart_on_failure(coord_server, tmp_path):
    m = _em(coord_server, np=1, host="h@1", fault_level=1).start()
    m.wait(timeout=5)
    lf = LauncherInterface([_script(tmp_path, "import sys; sys.exit(3)")],
                           log_dir=str(tmp_path))
    m.run(lf)
    assert m.watch(lf) == ElasticStatus.RESTART
    m.stop()


def test_watch_error_level0(coord_server, tmp_path):
    m = _em(coord_server, np=1, host="h@1", fault_level=0).start()
    m.wait(timeout=5)
    lf = LauncherInterface([_script(tmp_path, "import sys; sys.exit(3)")],
                           log_dir=str(tmp_path))
    m.run(lf)
    assert m.watch(lf) == ElasticStatus.ERROR
    m.stop()


def test_watch_hold_on_world_change(coord_server, tmp_path):
    m1 = _em(coord_server, np=2, host="h1@1").start()
    m2 = _em(coord_server, np=2, host="h2@2").start()
    assert m1.wait(timeout=5)
    lf = LauncherInterface([_script(tmp_path, "import time; time.sleep(30)")],
                           log_dir=str(tmp_path))
    m1.run(lf)
    result = {}

    def w():
        result["s"] = m1.watch(lf)

    t = threading.Thread(target=w)
    t.start()
    time.sleep(0.5)
    m2.stop()  # node leaves -> HOLD + procs stopped
    t.join(20)
    assert result["s"] == ElasticStatus.HOLD
    assert lf.poll() is not None  # trainers were stopped
    m1.stop()


def test_exit_code_101_means_restart(coord_server, tmp_path):
    m = _em(coord_server, np=1, host="h@1", fault_level=0).start()
    m.wait(timeout=5)
    lf = LauncherInterface(
        [_script(tmp_path, "import sys; sys.exit(%d)" % ELASTIC_EXIT_CODE)],
        log_dir=str(tmp_path))
    m.run(lf)
    assert m.watch(lf) == ElasticStatus.RESTART  # 101 restarts even at level 0
    m.stop()
