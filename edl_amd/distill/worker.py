"""Distill pipeline workers (parity: reference distill/distill_worker.py,
the largest file of the reference — 46-847).

Processes and protocol:
  reader_worker  : regroups the user's samples into teacher_batch_size
                   tasks, flow-controlled by a semaphore of
                   2*require_num+2 (reference distill_reader.py:238),
                   ends with a _PoisonPill carrying feed_count (174-178)
  predict_worker : binds to ONE teacher endpoint; task -> teacher RPC ->
                   out_queue; on RPC failure re-queues the task and exits
                   so the manage thread can recycle the slot (496-506);
                   the LAST live worker re-queues the pill until
                   predict_count == feed_count (435-491)
  fetch          : reorders completed tasks by task_id (720-769)

The NOP teacher (reference _TestNopPaddlePredictServer, 324-333) lets the
whole 1+N-process pipeline run in CPU tests: predictions == the feed.
"""
import queue as pyqueue
import threading
import time

import numpy as np

from ..utils.log import get_logger
from .timeline import timeline

log = get_logger("edl.distill.worker")

_NOP_PREDICT_TEST = False  # set True in tests (reference distill_worker.py:36-43)


class Counter:
    """Shared counter with a lock (the pipeline runs on THREADS: forking
    after CUDA/HIP init hangs the child runtime — observed on MI355X — and
    the workers are RPC/IO-bound anyway, so threads are the right tool)."""

    def __init__(self, v=0):
        self.value = v
        self._lock = threading.Lock()

    def get_lock(self):
        return self._lock


class _PoisonPill:
    def __init__(self, feed_count):
        self.feed_count = feed_count


class ServerItem:
    def __init__(self, slot, endpoint):
        self.slot = slot
        self.endpoint = endpoint


class _NopClient:
    def __init__(self, endpoint):
        self.endpoint = endpoint

    def predict(self, feed):
        return np.asarray(feed)

    def close(self):
        pass


def _make_client(endpoint):
    if _NOP_PREDICT_TEST:
        return _NopClient(endpoint)
    from .teacher_server import TeacherClient

    return TeacherClient(endpoint)


def reader_worker(gen_factory, mode, teacher_batch_size, task_queue,
                  task_semaphore, stop_event):
    """Pulls the user generator, emits (task_id, samples) tasks."""
    tid = 0
    buf = []

    def flush():
        nonlocal tid, buf
        if not buf:
            return
        task_semaphore.acquire()
        task_queue.put((tid, buf))
        tid += 1
        buf = []

    try:
        for item in gen_factory():
            if stop_event.is_set():
                break
            if mode == "sample":
                buf.append(tuple(np.asarray(a) for a in item))
                if len(buf) >= teacher_batch_size:
                    flush()
            elif mode == "sample_list":
                for s in item:
                    buf.append(tuple(np.asarray(a) for a in s))
                    if len(buf) >= teacher_batch_size:
                        flush()
            else:  # batch: item is a tuple of batched ndarrays
                flush()  # keep ordering: don't mix partial sample buffers
                task_semaphore.acquire()
                task_queue.put((tid, tuple(np.asarray(a) for a in item)))
                tid += 1
        flush()
    finally:
        task_queue.put(_PoisonPill(tid))
        log.debug("reader done: %d tasks", tid)


def predict_worker(slot, server_queue, task_queue, out_queue, predict_count,
                   live_workers, stop_events, feed_idx):
    """One worker: take a teacher from server_queue, serve tasks until the
    teacher dies / is removed / the epoch ends."""
    item = server_queue.get()
    if item is None:
        with live_workers.get_lock():
            live_workers.value -= 1
        return
    client = None
    tl = timeline("predict_worker.%d" % slot)
    try:
        client = _make_client(item.endpoint)
        _predict_loop(slot, item, client, task_queue, out_queue, predict_count,
                      live_workers, stop_events, feed_idx, tl)
    except Exception as e:  # noqa: BLE001
        log.warning("predict worker %d (%s) died: %s", slot, item.endpoint, e)
        with live_workers.get_lock():
            live_workers.value -= 1
        out_queue.put(("worker_exit", slot, item.endpoint))
    finally:
        if client is not None:
            client.close()


def _predict_loop(slot, item, client, task_queue, out_queue, predict_count,
                  live_workers, stop_events, feed_idx, tl):
    while True:
        if stop_events[item.slot].is_set():
            with live_workers.get_lock():
                live_workers.value -= 1
            out_queue.put(("worker_exit", slot, item.endpoint))
            return
        try:
            task = task_queue.get(timeout=0.5)
        except pyqueue.Empty:
            continue
        if isinstance(task, _PoisonPill):
            with predict_count.get_lock(), live_workers.get_lock():
                done = predict_count.value >= task.feed_count
                last = live_workers.value <= 1
            if done:
                # epoch complete: pass the pill to fetch and go idle for
                # the next epoch (worker stays alive)
                out_queue.put(("pill", task.feed_count, None))
                continue
            # not everyone reported: put it back for later
            task_queue.put(task)
            if last:
                time.sleep(0.05)  # sole worker: let in-flight settle
            continue

        tid, samples = task
        try:
            with tl("predict"):
                if isinstance(samples, tuple):  # batch mode
                    feed = samples[feed_idx]
                else:
                    feed = np.stack([s[feed_idx] for s in samples])
                pred = client.predict(feed)
            with predict_count.get_lock():
                predict_count.value += 1
            out_queue.put(("done", tid, (samples, pred)))
        except Exception:
            # teacher failure: the task MUST survive (re-queue), the worker
            # dies and gets recycled (reference 496-506)
            task_queue.put(task)
            raise


def fetch_ordered(out_queue, task_semaphore, on_worker_exit=None, timeout=120.0):
    """Generator of (samples, predictions) in task_id order. Terminates
    after the pill AND all preceding tasks have been yielded."""
    store = {}
    next_tid = 0
    feed_count = None
    deadline = time.monotonic() + timeout
    while True:
        if feed_count is not None and next_tid >= feed_count:
            return
        try:
            msg = out_queue.get(timeout=1.0)
            deadline = time.monotonic() + timeout
        except pyqueue.Empty:
            if time.monotonic() > deadline:
                raise TimeoutError("distill fetch stalled at task %d" % next_tid)
            continue
        kind = msg[0]
        if kind == "pill":
            feed_count = msg[1]
        elif kind == "worker_exit":
            if on_worker_exit:
                on_worker_exit(msg[1], msg[2])
        else:
            _, tid, payload = msg
            store[tid] = payload
            while next_tid in store:
                yield store.pop(next_tid)
                next_tid += 1
                task_semaphore.release()
