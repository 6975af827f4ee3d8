"""Job / trainer environment contract.

Parity: reference utils/env.py (JobEnv 40-176, TrainerEnv 179-229). Keeps
the paddle_edl env names (PADDLE_*) for API compatibility and adds EDL_*
aliases plus the torch-standard RANK/WORLD_SIZE/MASTER_* variables so a
stock `torch.distributed.init_process_group(init_method="env://")` works
unchanged inside a spawned trainer.
"""
import os

from ..utils.net import find_free_port, local_ip


def _get(env, *names, default=None):
    for n in names:
        v = env.get(n)
        if v not in (None, ""):
            return v
    return default


def get_visible_gpus(env=None):
    """GPU ids this agent may use. Reference env.get_gpus (env.py:22-30):
    CUDA_VISIBLE_DEVICES order defines the local slots."""
    env = env if env is not None else os.environ
    for name in ("CUDA_VISIBLE_DEVICES", "HIP_VISIBLE_DEVICES"):
        v = env.get(name)
        if v is not None:  # NB: "" is a valid value meaning ZERO gpus
            return [g for g in v.split(",") if g != ""]
    try:
        import torch

        if torch.cuda.is_available():
            return [str(i) for i in range(torch.cuda.device_count())]
    except Exception:  # noqa: BLE001
        pass
    return []


class JobEnv:
    """Merged CLI args + environment for the per-node agent."""

    def __init__(self, args=None, env=None):
        env = env if env is not None else os.environ
        a = vars(args) if args is not None and not isinstance(args, dict) else (args or {})

        self.job_id = a.get("job_id") or _get(env, "PADDLE_JOB_ID", "EDL_JOB_ID", default="edl_job")
        self.store_endpoints = (
            a.get("store_endpoints")
            or _get(env, "PADDLE_ETCD_ENDPOINTS", "EDL_STORE_ENDPOINTS", default="127.0.0.1:2379")
        )
        nodes_range = a.get("nodes_range") or _get(
            # NB: "RANAGE" is the reference's own spelling (env.py) — kept for compat.
            env, "PADDLE_EDLNODES_RANAGE", "EDL_NODES_RANGE", default="1:8"
        )
        lo, _, hi = str(nodes_range).partition(":")
        self.min_nodes = int(lo)
        self.max_nodes = int(hi or lo)

        self.pod_ip = a.get("pod_ip") or _get(env, "POD_IP", default=None) or local_ip()
        self.gpus = get_visible_gpus(env)
        # nproc_per_node: CPU-only runs (tests, fit_a_line config) override
        self.nproc_per_node = a.get("nproc_per_node") or (
            int(env["EDL_NPROC_PER_NODE"]) if env.get("EDL_NPROC_PER_NODE") else None
        )
        ports = _get(env, "PADDLE_TRAINER_PORTS", "EDL_TRAINER_PORTS")
        if ports:
            self.trainer_ports = [int(p) for p in ports.split(",")]
        else:
            n = self.nproc_per_node or max(1, len(self.gpus))
            got = find_free_port(n)
            self.trainer_ports = got if isinstance(got, list) else [got]

        self.log_dir = a.get("log_dir") or env.get("EDL_LOG_DIR") or "./edl_logs"
        self.ce_test = bool(int(env.get("PADDLE_EDL_ONLY_FOR_CE_TEST", "0")))
        # checkpoint root (reference: fleet checkpoint to LocalFS/HDFS)
        self.checkpoint_dir = (
            a.get("checkpoint_dir")
            or _get(env, "EDL_CHECKPOINT_DIR", "PADDLE_EDL_HDFS_CHECKPOINT_PATH")
            or os.path.join(self.log_dir, "checkpoints")
        )


class TrainerEnv:
    """What a spawned trainer process reads back (reference env.py:179-229)."""

    def __init__(self, env=None):
        env = env if env is not None else os.environ
        self.job_id = _get(env, "PADDLE_JOB_ID", "EDL_JOB_ID", default="edl_job")
        self.global_rank = int(_get(env, "PADDLE_TRAINER_ID", "RANK", default="0"))
        self.rank_in_pod = int(_get(env, "PADDLE_TRAINER_RANK_IN_POD", "LOCAL_RANK", default="0"))
        self.world_size = int(_get(env, "PADDLE_TRAINERS_NUM", "WORLD_SIZE", default="1"))
        eps = _get(env, "PADDLE_TRAINER_ENDPOINTS", default="")
        self.trainer_endpoints = [e for e in eps.split(",") if e]
        self.current_endpoint = _get(env, "PADDLE_CURRENT_ENDPOINT", default="")
        gpus = _get(env, "FLAGS_selected_gpus", default="")
        self.gpus = [g for g in str(gpus).split(",") if g != ""]
        self.store_endpoints = _get(env, "PADDLE_ETCD_ENDPOINTS", "EDL_STORE_ENDPOINTS", default="")
        self.cluster_stage = _get(env, "EDL_CLUSTER_STAGE", default="")
        self.job_stage = int(_get(env, "EDL_JOB_STAGE", default="0"))
        self.master_addr = _get(env, "MASTER_ADDR", default="127.0.0.1")
        self.master_port = int(_get(env, "MASTER_PORT", default="29500"))
        self.checkpoint_dir = _get(env, "EDL_CHECKPOINT_DIR", default="./edl_logs/checkpoints")

    @property
    def is_rank0(self):
        return self.global_rank == 0


def trainer_env_dict(job_env, cluster, pod, trainer):
    """Build the env-var dict for ONE trainer subprocess.

    Parity: reference utils/train_process.py:46-73 (PADDLE_TRAINER_ID,
    FLAGS_selected_gpus, PADDLE_TRAINER_ENDPOINTS, ...) plus torch env://
    rendezvous vars. MASTER_* point at global rank 0's endpoint; on every
    elastic restart a new cluster stage yields a fresh world."""
    endpoints = cluster.trainer_endpoints()
    master_host, master_port = endpoints[0].rsplit(":", 1)
    if master_host not in ("127.0.0.1", "localhost") and master_host == local_ip():
        # single-node job: always rendezvous on loopback (container hostnames
        # may not resolve; BASELINE is single-node 1-8 GPU)
        master_host = "127.0.0.1"
    e = {
        "PADDLE_JOB_ID": job_env.job_id,
        "EDL_JOB_ID": job_env.job_id,
        "PADDLE_TRAINER_ID": str(trainer.global_rank),
        "PADDLE_TRAINER_RANK_IN_POD": str(trainer.rank_in_pod),
        "PADDLE_TRAINERS_NUM": str(cluster.world_size()),
        "PADDLE_TRAINER_ENDPOINTS": ",".join(endpoints),
        "PADDLE_CURRENT_ENDPOINT": trainer.endpoint,
        "FLAGS_selected_gpus": ",".join(trainer.gpus),
        "PADDLE_ETCD_ENDPOINTS": job_env.store_endpoints,
        "EDL_STORE_ENDPOINTS": job_env.store_endpoints,
        "EDL_CLUSTER_STAGE": cluster.stage,
        "EDL_JOB_STAGE": str(cluster.job_stage),
        "EDL_POD_ID": pod.pod_id,
        "EDL_CHECKPOINT_DIR": job_env.checkpoint_dir,
        # torch-standard rendezvous
        "RANK": str(trainer.global_rank),
        "LOCAL_RANK": str(trainer.rank_in_pod),
        "WORLD_SIZE": str(cluster.world_size()),
        "LOCAL_WORLD_SIZE": str(len(pod.trainers)),
        "MASTER_ADDR": master_host,
        "MASTER_PORT": master_port,
    }
    return e
