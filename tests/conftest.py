import os
import sys

import pytest

REPO_ROOT = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
if REPO_ROOT not in sys.path:
    sys.path.insert(0, REPO_ROOT)


def pytest_configure(config):
    config.addinivalue_line("markers", "gpu: test requires an MI355X GPU (run via gpurun)")


@pytest.fixture()
def coord_server():
    """A throwaway in-process coordination store (the reference's tests boot
    a throwaway etcd the same way — reference tests/unittests/etcd_test_base.py)."""
    from edl_amd.coord.server import CoordServer

    srv = CoordServer(port=0).start()
    yield srv
    srv.stop()


@pytest.fixture()
def coord_client(coord_server):
    from edl_amd.coord.client import CoordClient

    c = CoordClient(coord_server.endpoint, job_id="test_job")
    yield c
    c.close()
