"""DistillReader — the student-facing elastic distillation reader.

Parity: reference distill/distill_reader.py:85-417. Usage:

    dr = DistillReader(ins=["img", "label"], predicts=["logits"])
    dr.set_sample_list_generator(my_reader)          # or set_sample/_batch
    dr.set_fixed_teacher("host:9292,host:9293")      # or set_dynamic_teacher
    for img, label, logits in dr():
        ...

Env overrides (reference distill_reader.py:278-298):
    PADDLE_DISTILL_BALANCE_SERVER   (store endpoints for dynamic discovery)
    PADDLE_DISTILL_SERVICE_NAME
    PADDLE_DISTILL_MAX_TEACHER      (require_num)
Runs 1 reader thread + require_num predict threads + a manage thread
that diffs the discovered teacher set, stops workers of removed teachers
and fills slots for new ones (reference predict_manage_worker 58-171)."""
import os
import queue
import threading
import time

from ..utils.log import get_logger
from . import worker as W
from .discovery import DiscoveryClient

log = get_logger("edl.distill.reader")


class DistillReader:
    def __init__(self, ins, predicts, teacher_batch_size=16, require_num=None):
        self._ins = list(ins)
        self._predicts = list(predicts)
        self._teacher_batch_size = teacher_batch_size
        self._require_num = require_num or int(
            os.environ.get("PADDLE_DISTILL_MAX_TEACHER", "2"))
        self._gen_factory = None
        self._mode = None
        self._feed_idx = 0
        self._fixed_teachers = None
        self._discovery = None
        self._store_endpoints = os.environ.get("PADDLE_DISTILL_BALANCE_SERVER")
        self._service_name = os.environ.get("PADDLE_DISTILL_SERVICE_NAME")

    # ---- configuration (reference 307-353) ----
    def set_sample_generator(self, gen_factory):
        self._gen_factory, self._mode = gen_factory, "sample"
        return self

    def set_sample_list_generator(self, gen_factory):
        self._gen_factory, self._mode = gen_factory, "sample_list"
        return self

    def set_batch_generator(self, gen_factory):
        self._gen_factory, self._mode = gen_factory, "batch"
        return self

    def set_fixed_teacher(self, teachers):
        if isinstance(teachers, str):
            teachers = [t for t in teachers.split(",") if t]
        self._fixed_teachers = teachers
        return self

    def set_dynamic_teacher(self, store_endpoints=None, service_name=None):
        self._store_endpoints = store_endpoints or self._store_endpoints
        self._service_name = service_name or self._service_name
        assert self._store_endpoints and self._service_name, \
            "dynamic teacher needs store endpoints + service name"
        return self

    # ---- the pipeline ----
    def _current_teachers(self):
        if self._fixed_teachers is not None:
            return list(self._fixed_teachers)
        changed, servers = self._discovery.get_servers()
        return servers

    def _manage(self, server_queue, stop_events, live_workers, assigned, stop):
        """Diff the teacher set; stop removed, enqueue added
        (reference predict_manage_worker)."""
        while not stop.wait(1.0):
            teachers = set(self._current_teachers())
            # removed teachers -> signal their slots
            for slot, ep in list(assigned.items()):
                if ep is not None and ep not in teachers:
                    stop_events[slot].set()
                    assigned[slot] = None
            # new teachers -> hand to free slots
            active = {ep for ep in assigned.values() if ep}
            for ep in sorted(teachers - active):
                free = [s for s, cur in assigned.items() if cur is None]
                if not free:
                    break
                slot = free[0]
                stop_events[slot].clear()
                assigned[slot] = ep
                server_queue.put(W.ServerItem(slot, ep))

    def __call__(self):
        assert self._gen_factory is not None, "set a generator first"
        n = self._require_num
        task_queue = queue.Queue(maxsize=4 * n + 8)
        out_queue = queue.Queue()
        server_queue = queue.Queue()
        task_semaphore = threading.Semaphore(2 * n + 2)
        reader_stop = threading.Event()
        stop_events = [threading.Event() for _ in range(n)]
        predict_count = W.Counter()
        live_workers = W.Counter()

        if self._fixed_teachers is None:
            self._discovery = DiscoveryClient(
                self._store_endpoints, self._service_name, require=n).start()

        assigned = {s: None for s in range(n)}
        teachers = []
        t0 = time.monotonic()
        while not teachers:
            teachers = self._current_teachers()
            if teachers or W._NOP_PREDICT_TEST:
                break
            if time.monotonic() - t0 > 120:
                raise TimeoutError("no teachers discovered")
            time.sleep(0.5)

        procs = []
        reader = threading.Thread(
            target=W.reader_worker,
            args=(self._gen_factory, self._mode, self._teacher_batch_size,
                  task_queue, task_semaphore, reader_stop),
            daemon=True,
        )
        reader.start()
        with live_workers.get_lock():
            live_workers.value = n
        for slot in range(n):
            p = threading.Thread(
                target=W.predict_worker,
                args=(slot, server_queue, task_queue, out_queue, predict_count,
                      live_workers, stop_events, self._feed_idx),
                daemon=True,
            )
            p.start()
            procs.append(p)

        manage_stop = threading.Event()

        def on_worker_exit(slot, endpoint):
            # recycle: spawn a fresh worker process for the slot
            with live_workers.get_lock():
                live_workers.value += 1
            assigned[slot] = None
            p = threading.Thread(
                target=W.predict_worker,
                args=(slot, server_queue, task_queue, out_queue, predict_count,
                      live_workers, stop_events, self._feed_idx),
                daemon=True,
            )
            p.start()
            procs.append(p)

        manager = threading.Thread(
            target=self._manage,
            args=(server_queue, stop_events, live_workers, assigned, manage_stop),
            daemon=True,
        )
        manager.start()
        # seed initial assignment immediately
        for slot, ep in zip(range(n), sorted(teachers)):
            assigned[slot] = ep
            server_queue.put(W.ServerItem(slot, ep))
        if W._NOP_PREDICT_TEST and not teachers:
            for slot in range(n):
                assigned[slot] = "nop:%d" % slot
                server_queue.put(W.ServerItem(slot, "nop:%d" % slot))

        try:
            for samples, pred in W.fetch_ordered(out_queue, task_semaphore,
                                                 on_worker_exit=on_worker_exit):
                if isinstance(samples, tuple):  # batch mode
                    yield tuple(samples) + (pred,)
                else:  # per-sample modes
                    for i, s in enumerate(samples):
                        yield tuple(s) + (pred[i],)
        finally:
            reader_stop.set()
            manage_stop.set()
            for e in stop_events:
                e.set()
            # unblock a reader waiting on flow control so its thread exits
            for _ in range(4 * n + 8):
                task_semaphore.release()
            for _ in procs:
                server_queue.put(None)
            for p in [reader] + procs:
                p.join(timeout=5)
            if self._discovery:
                self._discovery.stop()
                self._discovery = None
