"""liveft — hot-restart elasticity (v2 path).

Parity: reference python/edl/liveft/elastic.py:28-313 + launch.py:24-63,
the Paddle-2.x-style elastic mode: an ElasticManager per node registers its
host, waits until the desired world size `np` is reached, runs the trainer
processes, and on any world change stops them and re-enters wait — an
in-place hot restart (exit code 101) rather than the launcher's
full stop-resume machinery. Fault levels
(PADDLE_ELASTIC_FAULT_TOLERANC_LEVEL — reference spelling, elastic.py:
103-104): level 0 = any trainer failure fails the job; level 1 = nonzero
exits trigger RESTART."""
import os
import subprocess
import sys
import time

from ..coord.client import CoordClient
from ..coord.register import Register
from ..utils.log import get_logger
from ..utils.net import local_ip

log = get_logger("edl.liveft")

ELASTIC_EXIT_CODE = 101


class ElasticStatus:
    COMPLETED = "completed"
    RESTART = "restart"
    ERROR = "error"
    HOLD = "hold"
    EXIT = "exit"


class LauncherInterface:
    """Spawns/watches the local trainer processes for one world epoch."""

    def __init__(self, cmd, log_dir="./edl_logs"):
        self.cmd = list(cmd)
        self.log_dir = log_dir
        self.procs = []

    def launch(self, hosts, rank, np_total, nproc_per_node=1):
        os.makedirs(self.log_dir, exist_ok=True)
        self.procs = []
        master = hosts[0].split("@")[0]
        if master == local_ip():
            master = "127.0.0.1"
        for lr in range(nproc_per_node):
            env = dict(os.environ)
            grank = rank * nproc_per_node + lr
            env.update({
                "PADDLE_TRAINER_ID": str(grank),
                "RANK": str(grank),
                "LOCAL_RANK": str(lr),
                "WORLD_SIZE": str(np_total * nproc_per_node),
                "PADDLE_TRAINERS_NUM": str(np_total * nproc_per_node),
                "PADDLE_TRAINERS": ",".join(h.split("@")[0] for h in hosts),
                "MASTER_ADDR": master,
                "MASTER_PORT": env.get("MASTER_PORT", "29600"),
            })
            cmd = self.cmd
            if cmd and cmd[0].endswith(".py"):
                cmd = [sys.executable, "-u"] + cmd
            f = open(os.path.join(self.log_dir, "liveft.%d.log" % lr), "ab")
            self.procs.append(subprocess.Popen(
                cmd, env=env, stdout=f, stderr=subprocess.STDOUT,
                start_new_session=True))
            f.close()

    def poll(self):
        """-> None (running) | 0 (all ok) | first nonzero exit code."""
        codes = [p.poll() for p in self.procs]
        if any(c is None for c in codes):
            return None
        bad = [c for c in codes if c != 0]
        return bad[0] if bad else 0

    def stop(self):
        for p in self.procs:
            if p.poll() is None:
                try:
                    os.killpg(os.getpgid(p.pid), 15)
                except ProcessLookupError:
                    pass
        deadline = time.monotonic() + 5
        for p in self.procs:
            try:
                p.wait(timeout=max(0.1, deadline - time.monotonic()))
            except subprocess.TimeoutExpired:
                try:
                    os.killpg(os.getpgid(p.pid), 9)
                except ProcessLookupError:
                    pass


class ElasticManager:
    def __init__(self, job_id=None, np=None, store_endpoints=None, host=None,
                 fault_level=None):
        env = os.environ
        self.job_id = job_id or env.get("PADDLE_ELASTIC_JOB_ID", "liveft_job")
        self.np = int(np or env.get("PADDLE_ELASTIC_NP", "1"))
        self.fault_level = int(fault_level if fault_level is not None else
                               env.get("PADDLE_ELASTIC_FAULT_TOLERANC_LEVEL", "1"))
        endpoints = store_endpoints or env.get("PADDLE_ELASTIC_SERVER") or \
            env.get("EDL_STORE_ENDPOINTS", "127.0.0.1:2379")
        self._client = CoordClient(endpoints, self.job_id)
        # host id = ip@timestamp (reference registers under nodes/<ts>)
        self.host = host or "%s@%d" % (local_ip(), time.time_ns() % 10**9)
        self._reg = None
        self.hosts = []
        self.rank = -1
        self.enabled = True

    def _host_key(self):
        return self._client.table_key("liveft_nodes", self.host)

    def start(self):
        self._reg = Register(self._client, self._host_key(), "1").start()
        return self

    def _load_hosts(self):
        pfx = self._client.table_key("liveft_nodes")
        return sorted(k[len(pfx):] for k, _ in self._client.range(pfx))

    def wait(self, timeout=600):
        """Block until exactly np hosts are present; assign rank-preserving
        ranks (reference _update_hosts 238-261: a returning host keeps its
        slot where possible — here: sorted order is stable because host ids
        are stable for a process lifetime)."""
        deadline = time.monotonic() + timeout
        while time.monotonic() < deadline:
            hosts = self._load_hosts()
            if len(hosts) == self.np:
                self.hosts = hosts
                self.rank = hosts.index(self.host)
                return True
            time.sleep(0.5)
        return False

    def world_changed(self):
        return self._load_hosts() != self.hosts

    def run(self, launcher, nproc_per_node=1):
        launcher.launch(self.hosts, self.rank, self.np, nproc_per_node)

    def watch(self, launcher, poll=0.5):
        """-> ElasticStatus (reference watch() 284-307)."""
        while True:
            code = launcher.poll()
            if code == 0:
                return ElasticStatus.COMPLETED
            if code is not None:
                if code == ELASTIC_EXIT_CODE or self.fault_level >= 1:
                    return ElasticStatus.RESTART
                return ElasticStatus.ERROR
            if self.world_changed():
                launcher.stop()
                return ElasticStatus.HOLD
            time.sleep(poll)

    def stop(self):
        if self._reg:
            self._reg.stop()
        self._client.close()


def launch(cmd, job_id=None, np=None, store_endpoints=None, nproc_per_node=1,
           max_restarts=10, log_dir="./edl_logs"):
    """The wait->run->watch loop (reference liveft/launch.py:24-59)."""
    em = ElasticManager(job_id=job_id, np=np, store_endpoints=store_endpoints)
    em.start()
    restarts = 0
    try:
        while True:
            if not em.wait():
                log.error("liveft: world never reached np=%d", em.np)
                return 1
            launcher = LauncherInterface(cmd, log_dir=log_dir)
            em.run(launcher, nproc_per_node)
            status = em.watch(launcher)
            log.info("liveft epoch done: %s (world=%s)", status, em.hosts)
            if status == ElasticStatus.COMPLETED:
                return 0
            if status == ElasticStatus.ERROR:
                return 1
            launcher.stop()
            restarts += 1
            if restarts > max_restarts:
                log.error("liveft: too many restarts")
                return 1
    finally:
        em.stop()
