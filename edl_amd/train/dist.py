"""torch.distributed process-group setup for the elastic world.

On GPU the backend string "nccl" IS RCCL on ROCm — one process per GPU
over xGMI. On CPU (tests, fit_a_line config) we use gloo. The rendezvous
is torch's env:// TCPStore on MASTER_ADDR/MASTER_PORT, which the launcher
points at global rank 0's trainer endpoint — this replaces the reference's
NCCL-uniqueId-over-socket bootstrap (utils/train_process.py:37-41): on
every elastic restart the launcher hands out a fresh world and the new
processes rebuild the communicator from scratch (stop-resume semantics,
SURVEY.md §0.2)."""
import datetime
import os

import torch
import torch.distributed as dist

from .env import TrainerEnv


def init_from_env(env=None, timeout_s=120):
    """Initialise the default process group from the launcher's env
    contract. Returns (TrainerEnv, device)."""
    tenv = env or TrainerEnv()
    use_cuda = torch.cuda.is_available()
    if use_cuda:
        # FLAGS_selected_gpus indexes into CUDA_VISIBLE_DEVICES order;
        # each trainer owns one visible device slot (= its local rank).
        device = torch.device(
            "cuda", tenv.rank_in_pod % max(1, torch.cuda.device_count()))
        torch.cuda.set_device(device)
        backend = "nccl"
    else:
        device = torch.device("cpu")
        backend = "gloo"

    # test hook: EDL_FORCE_BACKEND=gloo lets N ranks share one GPU (RCCL
    # forbids duplicate devices) so the full multi-rank path is testable
    # on a 1-GPU box
    forced = os.environ.get("EDL_FORCE_BACKEND")
    if forced:
        backend = forced
    if tenv.world_size <= 1:
        return tenv, device  # single process: no communicator needed
    if not dist.is_initialized():
        os.environ.setdefault("MASTER_ADDR", tenv.master_addr)
        os.environ.setdefault("MASTER_PORT", str(tenv.master_port))
        dist.init_process_group(
            backend=backend,
            rank=tenv.global_rank,
            world_size=tenv.world_size,
            timeout=datetime.timedelta(seconds=timeout_s),
        )
    return tenv, device


def cleanup():
    if dist.is_initialized():
        dist.destroy_process_group()


def world_size():
    return dist.get_world_size() if dist.is_initialized() else 1


def rank():
    return dist.get_rank() if dist.is_initialized() else 0


def barrier(device=None):
    if not dist.is_initialized():
        return
    if (device is not None and device.type == "cuda"
            and dist.get_backend() == "nccl"):
        dist.barrier(device_ids=[device.index])
    else:
        dist.barrier()
