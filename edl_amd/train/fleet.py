"""paddle_edl-compatible training API facade.

The reference's user scripts drive Paddle Fleet
(train_with_fleet.py:367-381: fleet.init(role), DistributedStrategy,
fleet.distributed_optimizer(...).minimize, fleet.save/load_check_point with
TrainStatus). This module keeps those NAMES working on top of the MI355X
engine so a paddle_edl user can port a script mechanically:

    from edl_amd.train import fleet
    fleet.init()
    engine = fleet.distributed_engine(model, lr=0.1)   # torch-native core
    ts = fleet.load_check_point(engine, path)
    for epoch in range(ts.next(), epochs): ...
    fleet.save_check_point(engine, TrainStatus(epoch), path)
"""
from dataclasses import dataclass, field

import torch

from . import dist as edist
from .checkpoint import CheckpointManager
from .env import TrainerEnv


@dataclass
class DistributedStrategy:
    """Knob parity with the reference's strategy/flags:
    values are recorded; the engine applies the equivalents."""

    nccl_comm_num: int = 1                   # RCCL communicators
    use_hierarchical_allreduce: bool = False  # moot intra-node over xGMI
    fuse_all_reduce_ops: bool = True          # always on: flat bucket views
    fuse_all_reduce_ops_mb: int = 25          # bucket size
    use_amp: bool = True                      # bf16 autocast
    use_dgc: bool = False                     # optional sparse allreduce
    use_recompute: bool = False
    extra: dict = field(default_factory=dict)


class TrainStatus:
    """Epoch cursor (reference doc/fault_tolerance.md:55-62)."""

    def __init__(self, epoch_no=-1):
        self.epoch_no = epoch_no

    def next(self):
        return self.epoch_no + 1

    def __eq__(self, o):
        return isinstance(o, TrainStatus) and o.epoch_no == self.epoch_no


_state = {"env": None, "device": None}


def init(role=None, strategy=None):
    """fleet.init: join the RCCL/gloo world from the launcher env."""
    env, device = edist.init_from_env()
    _state["env"], _state["device"] = env, device
    return env


def worker_index():
    return (_state["env"] or TrainerEnv()).global_rank


def worker_num():
    return (_state["env"] or TrainerEnv()).world_size


def is_first_worker():
    return worker_index() == 0


def distributed_engine(model_name="resnet50_vd", strategy=None, **kwargs):
    """Build the TrainerEngine (our fleet.distributed_optimizer analog)."""
    from .engine import TrainerEngine

    strategy = strategy or DistributedStrategy()
    kwargs.setdefault("bucket_mb", strategy.fuse_all_reduce_ops_mb)
    kwargs.setdefault("dtype", "bf16" if strategy.use_amp and
                      torch.cuda.is_available() else "fp32")
    kwargs.setdefault("dgc", strategy.use_dgc)
    kwargs.setdefault("recompute", strategy.use_recompute)
    eng = TrainerEngine(model=model_name, **kwargs)
    eng.setup(_state["env"])
    return eng


def save_check_point(engine, train_status, path, fs=None):
    """Rank-0 versioned checkpoint (reference train_with_fleet.py:562-570)."""
    if engine.env is not None and not engine.env.is_rank0:
        return None
    cm = engine.ckpt if (engine.ckpt and engine.ckpt.path == path) else \
        CheckpointManager(path)
    return cm.save(engine.model.state_dict(),
                   {"epoch_no": train_status.epoch_no,
                    "global_step": engine.global_step},
                   optimizer_state=engine.opt.state_dict())


def load_check_point(engine, path, fs=None, trainer_id=None):
    """-> TrainStatus or None (reference train_with_fleet.py:427-431)."""
    cm = CheckpointManager(path)
    got = cm.load()
    if got is None:
        return None
    model_state, opt_state, ts = got
    engine.model.load_state_dict(model_state)
    if opt_state is not None:
        try:
            engine.opt.load_state_dict(opt_state)
        except (KeyError, ValueError):
            pass
    engine.global_step = int(ts.get("global_step", 0))
    return TrainStatus(int(ts.get("epoch_no", -1)))
