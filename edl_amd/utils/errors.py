"""Typed errors + the retry-until-timeout idiom.

Capability parity with reference utils/exceptions.py:20-117 (18 typed
Edl*Error classes) and utils/error_utils.py:22-39
(handle_errors_until_timeout — retry every interval until a deadline).
"""
import functools
import time


class EdlError(Exception):
    pass


class EdlStoreError(EdlError):
    """Coordination-store RPC / connectivity failure."""


class EdlBarrierError(EdlError):
    pass


class EdlLeaderError(EdlError):
    pass


class EdlGenerateClusterError(EdlError):
    pass


class EdlTableError(EdlError):
    pass


class EdlRegisterError(EdlError):
    pass


class EdlStopIteration(EdlError):
    pass


class EdlDataEndError(EdlError):
    pass


class EdlPodIDNotExistError(EdlError):
    pass


class EdlNotLeaderError(EdlError):
    pass


class EdlUnkownError(EdlError):  # name kept for paddle_edl compat
    pass


def serialize_error(exc):
    """Marshal an exception to (class_name, detail) — reference
    utils/exceptions.py:90-117 carries these through proto Status."""
    return type(exc).__name__, str(exc)


def deserialize_error(name, detail):
    cls = globals().get(name)
    if cls is not None and isinstance(cls, type) and issubclass(cls, EdlError):
        return cls(detail)
    return EdlUnkownError("%s: %s" % (name, detail))


def retry_until_timeout(timeout=60, interval=3, exceptions=(EdlError, OSError)):
    """Decorator: retry fn every `interval` s until `timeout` s elapsed.

    The universal resilience idiom of the reference
    (utils/error_utils.py:22-39 handle_errors_until_timeout). The wrapped
    function may override the window per call with kwarg ``timeout=``.
    """

    def deco(fn):
        @functools.wraps(fn)
        def wrapper(*args, **kwargs):
            tmo = kwargs.pop("timeout", timeout)
            deadline = time.monotonic() + tmo
            while True:
                try:
                    return fn(*args, **kwargs)
                except exceptions as e:
                    if time.monotonic() >= deadline:
                        raise
                    last = e
                    time.sleep(min(interval, max(0.0, deadline - time.monotonic())))

        return wrapper

    return deco
