"""Trainer subprocess manager.

Parity: reference utils/train_process.py:25-188 — spawn one subprocess per
trainer slot with the env contract, scrub http proxies (the RCCL/torch TCP
rendezvous must not go through a proxy; reference train_process.py:37-41),
kill whole descendant trees on terminate (89-112 via psutil), poll exit
codes (130-175), tail rank-0's log (115-127)."""
import os
import signal
import subprocess
import sys
import time

import psutil

from .train.env import trainer_env_dict
from .utils.log import get_logger

log = get_logger("edl.procs")


class TrainerProc:
    def __init__(self, proc, trainer, log_path):
        self.proc = proc
        self.trainer = trainer
        self.log_path = log_path
        self.log_offset = 0


class TrainerProcs:
    def __init__(self, job_env, cluster, pod, cmd_args):
        """cmd_args: argv of the user training script, e.g.
        ['train_with_engine.py', '--epochs', '2']."""
        self._job_env = job_env
        self._cluster = cluster
        self._pod = pod
        self._cmd = list(cmd_args)
        self._procs = []

    def start(self):
        os.makedirs(self._job_env.log_dir, exist_ok=True)
        base_env = dict(os.environ)
        for k in ("http_proxy", "https_proxy", "HTTP_PROXY", "HTTPS_PROXY"):
            base_env.pop(k, None)
        for t in self._pod.trainers:
            env = dict(base_env)
            env.update(trainer_env_dict(self._job_env, self._cluster, self._pod, t))
            log_path = os.path.join(self._job_env.log_dir, "workerlog.%d" % t.rank_in_pod)
            f = open(log_path, "ab", buffering=0)
            cmd = self._cmd
            if cmd and (cmd[0].endswith(".py") or cmd[0] == "-m"):
                cmd = [sys.executable, "-u"] + cmd
            proc = subprocess.Popen(
                cmd, env=env, stdout=f, stderr=subprocess.STDOUT, start_new_session=True
            )
            f.close()
            self._procs.append(TrainerProc(proc, t, log_path))
            log.info(
                "spawned trainer rank=%d local=%d pid=%d log=%s",
                t.global_rank, t.rank_in_pod, proc.pid, log_path,
            )
        return self

    def poll(self):
        """-> (alive: bool, failed: bool). failed=True if any exited nonzero."""
        alive, failed = False, False
        for tp in self._procs:
            rc = tp.proc.poll()
            if rc is None:
                alive = True
            elif rc != 0:
                failed = True
        return alive, failed

    def exit_codes(self):
        return [tp.proc.poll() for tp in self._procs]

    def tail_rank0(self, max_bytes=8192):
        """Forward new bytes of the local rank-0 trainer log to our stdout
        (reference pull_worker_log, train_process.py:115-127)."""
        for tp in self._procs:
            if tp.trainer.rank_in_pod != 0:
                continue
            try:
                with open(tp.log_path, "rb") as f:
                    f.seek(tp.log_offset)
                    data = f.read(max_bytes)
                    tp.log_offset += len(data)
                if data:
                    sys.stdout.write(data.decode("utf-8", "replace"))
                    sys.stdout.flush()
            except OSError:
                pass

    def terminate(self, grace=3.0):
        """SIGTERM the whole descendant tree of every trainer, then SIGKILL
        stragglers (reference train_process.py:89-112)."""
        victims = []
        for tp in self._procs:
            if tp.proc.poll() is not None:
                continue
            try:
                parent = psutil.Process(tp.proc.pid)
                victims.extend(parent.children(recursive=True))
                victims.append(parent)
            except psutil.NoSuchProcess:
                continue
        for p in victims:
            try:
                p.send_signal(signal.SIGTERM)
            except psutil.NoSuchProcess:
                pass
        _, survivors = psutil.wait_procs(victims, timeout=grace)
        for p in survivors:
            try:
                p.kill()
            except psutil.NoSuchProcess:
                pass
        for tp in self._procs:
            try:
                tp.proc.wait(timeout=grace)
            except subprocess.TimeoutExpired:
                pass

    def wait(self, timeout=None):
        deadline = None if timeout is None else time.monotonic() + timeout
        while True:
            alive, failed = self.poll()
            if not alive:
                return failed
            if deadline is not None and time.monotonic() > deadline:
                raise TimeoutError("trainers still alive")
            time.sleep(0.2)
