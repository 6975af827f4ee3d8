"""Unit tests for the resilience primitives: retry_until_timeout (the
reference's handle_errors_until_timeout idiom, C16) and the TTL
Register's failure semantics (C6)."""
import time

import pytest

from edl_amd.utils.errors import EdlError, EdlStoreError, retry_until_timeout


class TestRetryUntilTimeout:
    def test_retries_then_succeeds(self):
        calls = []

        @retry_until_timeout(timeout=5, interval=0.01)
        def flaky():
            calls.append(1)
            if len(calls) < 3:
                raise EdlStoreError("transient")
            return "ok"

        assert flaky() == "ok"
        assert len(calls) == 3

    def test_raises_after_window(self):
        @retry_until_timeout(timeout=0.05, interval=0.01)
        def always_fails():
            raise EdlStoreError("down")

        t0 = time.monotonic()
        with pytest.raises(EdlStoreError):
            always_fails()
        assert time.monotonic() - t0 < 2.0  # bounded, not hung

    def test_per_call_window_override(self):
        calls = []

        @retry_until_timeout(timeout=60, interval=0.01)
        def fails(**kw):
            calls.append(1)
            raise EdlStoreError("x")

        t0 = time.monotonic()
        with pytest.raises(EdlStoreError):
            fails(timeout=0.05)  # kwarg overrides the 60 s default
        assert time.monotonic() - t0 < 2.0
        assert len(calls) >= 2

    def test_unlisted_exception_propagates_immediately(self):
        calls = []

        @retry_until_timeout(timeout=5, interval=0.01)
        def typo():
            calls.append(1)
            raise ValueError("bug, not outage")

        with pytest.raises(ValueError):
            typo()
        assert len(calls) == 1  # programming errors never retry


class TestRegisterTTL:
    def test_refresh_keeps_key_and_stop_releases(self):
        from edl_amd.coord.client import CoordClient
        from edl_amd.coord.register import Register
        from edl_amd.coord.server import CoordServer

        srv = CoordServer(port=0).start()
        try:
            c = CoordClient("127.0.0.1:%d" % srv.port, "jobr")
            reg = Register(c, c.table_key("resource", "podX"), "meta",
                           ttl=2).start()
            time.sleep(3.0)  # > ttl: key must survive via refresh
            assert c.get(c.table_key("resource", "podX")) == "meta"
            assert not reg.failed  # property: refresh thread healthy
            reg.stop()
            c.close()
        finally:
            srv.stop()
