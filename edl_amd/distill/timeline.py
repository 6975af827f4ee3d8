"""Distill profiling timeline (parity: reference distill/timeline.py:21-46
— per-op wall-time logger gated by DISTILL_READER_PROFILE=1)."""
import contextlib
import os
import time

from ..utils.log import get_logger

log = get_logger("edl.distill.timeline")


class _RealTimeLine:
    def __init__(self, name):
        self.name = name

    @contextlib.contextmanager
    def __call__(self, op):
        t0 = time.monotonic()
        yield
        log.info("[timeline] %s.%s %.3f ms", self.name, op,
                 (time.monotonic() - t0) * 1e3)


class _NopTimeLine:
    def __init__(self, name):
        pass

    @contextlib.contextmanager
    def __call__(self, op):
        yield


def timeline(name):
    if os.environ.get("DISTILL_READER_PROFILE") == "1":
        return _RealTimeLine(name)
    return _NopTimeLine(name)
