"""Flagship collective trainer — the `train_with_fleet.py` equivalent.

Spawned by the edlrun launcher (one process per GPU); reads the env
contract (TrainerEnv), builds the RCCL world, trains ResNet50_vd (or any
model in the zoo) on synthetic ImageNet-shaped data with elastic
stop-resume: per-epoch (and optional per-N-steps) checkpoints, LR rescaled
by the linear-scaling rule on every world change, train status reported to
the coordination store so the generator can veto near-end scaling.

CLI parity subset of reference example/collective/resnet50/train_with_fleet.py.
"""
import argparse
import os
import sys
import time

import torch

from ..cluster.status import TrainStatus, save_train_status
from ..coord.client import CoordClient
from ..data.synthetic import SyntheticImageNet
from ..utils.log import get_logger
from . import dist as edist
from .engine import TrainerEngine
from .env import TrainerEnv

log = get_logger("edl.train")


def parse_args(argv=None):
    p = argparse.ArgumentParser("edl_amd resnet trainer")
    p.add_argument("--model", default="resnet50_vd")
    p.add_argument("--batch_size", type=int, default=32, help="per-GPU batch")
    p.add_argument("--num_epochs", type=int, default=2)
    p.add_argument("--steps_per_epoch", type=int, default=100)
    p.add_argument("--lr", type=float, default=0.1)
    p.add_argument("--momentum", type=float, default=0.9)
    p.add_argument("--weight_decay", type=float, default=1e-4)
    p.add_argument("--checkpoint", default=None, help="checkpoint dir (resume + save)")
    p.add_argument("--checkpoint_steps", type=int, default=0,
                   help="also checkpoint every N steps (0 = per-epoch only)")
    p.add_argument("--data_dir", default=None,
                   help="txt record files: train through the elastic "
                        "data plane (leader-balanced Reader) instead of "
                        "the synthetic loader; records deterministically "
                        "seed synthetic images (no dataset network)")
    p.add_argument("--image_hw", type=int, default=224)
    p.add_argument("--use_hip_ops", type=int, default=1)
    p.add_argument("--graph_capture", type=int, default=None)
    p.add_argument("--bucket_mb", type=int, default=25)
    p.add_argument("--dtype", default="bf16", choices=["bf16", "fp16", "fp32"])
    p.add_argument("--profile", action="store_true")
    return p.parse_args(argv)


def main(argv=None):
    args = parse_args(argv)
    tenv = TrainerEnv()
    ckpt_dir = args.checkpoint or tenv.checkpoint_dir

    engine = TrainerEngine(
        model=args.model,
        per_device_batch=args.batch_size,
        base_lr=args.lr,
        momentum=args.momentum,
        weight_decay=args.weight_decay,
        dtype=args.dtype if torch.cuda.is_available() else "fp32",
        channels_last=torch.cuda.is_available(),
        bucket_mb=args.bucket_mb,
        checkpoint_dir=ckpt_dir,
        use_hip_ops=bool(args.use_hip_ops) and torch.cuda.is_available(),
        graph_capture=None if args.graph_capture is None else bool(args.graph_capture),
    ).setup(tenv)

    # status reporting back to the control plane (reference TrainStatus flow,
    # utils/train_status.py; generator reads it to veto near-end scale-out)
    store = None
    if tenv.store_endpoints:
        try:
            store = CoordClient(tenv.store_endpoints, tenv.job_id)
        except Exception as e:  # noqa: BLE001
            log.warning("no coordination store (%s); status reporting off", e)
    pod_id = os.environ.get("EDL_POD_ID", "pod")

    def report(status):
        if store is not None and tenv.is_rank0:
            try:
                save_train_status(store, pod_id, status)
            except Exception:  # noqa: BLE001
                pass

    report(TrainStatus.RUNNING)
    shape = (3, args.image_hw, args.image_hw)
    cl = engine.channels_last and engine.device.type == "cuda"
    loader = None
    if not args.data_dir:
        loader = SyntheticImageNet(
            args.batch_size, engine.device, image_shape=shape,
            channels_last=cl, seed=1234 + engine.env.global_rank,
        )

    def plane_epoch_loader():
        """Elastic data plane: leader-balanced record slices -> loader.
        Ranks can receive slightly unequal record counts, so all ranks
        agree on the MIN step count before training (collectives must
        stay matched across the world)."""
        import torch.distributed as dist

        from ..data.plane import RecordImageSet, fetch_epoch_records

        recs = fetch_epoch_records(tenv, args.data_dir, args.batch_size)
        ds = RecordImageSet(recs, args.batch_size, engine.device,
                            image_shape=shape, channels_last=cl)
        steps = ds.steps()
        if dist.is_initialized():
            t = torch.tensor([steps], dtype=torch.long,
                             device=engine.device
                             if dist.get_backend() == "nccl" else "cpu")
            dist.all_reduce(t, op=dist.ReduceOp.MIN)
            steps = int(t.item())
        if args.steps_per_epoch:
            steps = min(steps, args.steps_per_epoch)
        return ds, steps

    def on_step(epoch, it):
        if args.checkpoint_steps and (engine.global_step % args.checkpoint_steps == 0):
            engine.save_checkpoint(
                epoch, extra={"mid_epoch": True, "step_in_epoch": it + 1})

    t_start = time.monotonic()
    for epoch in range(engine.start_epoch, args.num_epochs):
        if epoch == args.num_epochs - 1:
            report(TrainStatus.NEARTHEEND)
        if args.data_dir:
            ep_loader, ep_steps = plane_epoch_loader()
            if ep_steps == 0:
                log.warning("epoch %d: no full batch from the data plane", epoch)
                continue
        else:
            ep_loader, ep_steps = loader, args.steps_per_epoch
        stats = engine.train_epoch(
            epoch, ep_loader, ep_steps, on_step=on_step,
            start_step=engine.start_step if epoch == engine.start_epoch else 0)
        if engine.env.is_rank0:
            log.info("epoch %d done: %.1f img/s (world=%d, global_batch=%d)",
                     epoch, stats["img_per_s"], engine.world_size, engine.global_batch)
        engine.save_checkpoint(epoch)
    if engine.ckpt is not None:
        engine.ckpt.wait()
    report(TrainStatus.SUCCEED)
    edist.barrier(engine.device)
    if engine.env.is_rank0:
        log.info("training done in %.1fs", time.monotonic() - t_start)
    edist.cleanup()
    return 0


if __name__ == "__main__":
    sys.exit(main())
