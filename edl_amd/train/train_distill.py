"""Distill student trainer — BASELINE config 4: ResNeXt101_32x16d_wsl
teachers serving ResNet50_vd students (reference
example/distill/resnet/train_with_fleet.py).

Each student rank trains ResNet50_vd with the KD soft-label CE loss
against teacher logits fetched from the elastic teacher pool over the
DistillReader pipeline (fixed teacher list or dynamic discovery).
Synthetic ImageNet-shaped data (BASELINE.json).

    # teacher side (per teacher GPU):
    python -m edl_amd.distill.teacher_server --port 9292 [--service_name S...]
    # student side (spawned by edlrun, one per GPU):
    python -m edl_amd.train.train_distill --teachers host:9292,host:9293 ...
"""
import argparse
import sys
import time

import numpy as np
import torch

from ..data.synthetic import SyntheticImageNet
from ..distill.reader import DistillReader
from ..utils.log import get_logger
from . import dist as edist
from .engine import TrainerEngine
from .env import TrainerEnv

log = get_logger("edl.distill.train")


def parse_args(argv=None):
    p = argparse.ArgumentParser("edl_amd distill student trainer")
    p.add_argument("--model", default="resnet50_vd")
    p.add_argument("--batch_size", type=int, default=32)
    p.add_argument("--teacher_batch_size", type=int, default=16)
    p.add_argument("--num_epochs", type=int, default=1)
    p.add_argument("--steps_per_epoch", type=int, default=50)
    p.add_argument("--lr", type=float, default=0.1)
    p.add_argument("--kd_alpha", type=float, default=1.0)
    p.add_argument("--teachers", default=None, help="fixed host:port list")
    p.add_argument("--service_name", default=None)
    p.add_argument("--balance_server", default=None, help="store endpoints")
    p.add_argument("--require_num", type=int, default=2)
    p.add_argument("--checkpoint", default=None)
    p.add_argument("--dtype", default="bf16")
    p.add_argument("--image_shape", default="3x224x224",
                   help="CxHxW of the synthetic student input (e.g. "
                        "1x28x28 for the mnist_* students)")
    p.add_argument("--num_classes", type=int, default=1000)
    return p.parse_args(argv)


def main(argv=None):
    args = parse_args(argv)
    tenv = TrainerEnv()
    engine = TrainerEngine(
        model=args.model,
        per_device_batch=args.batch_size,
        base_lr=args.lr,
        dtype=args.dtype if torch.cuda.is_available() else "fp32",
        channels_last=torch.cuda.is_available(),
        checkpoint_dir=args.checkpoint,
        use_hip_ops=torch.cuda.is_available(),
        kd_alpha=args.kd_alpha,
        num_classes=args.num_classes,
    ).setup(tenv)
    engine.model.train()

    shape = tuple(int(d) for d in args.image_shape.split("x"))
    loader = SyntheticImageNet(args.batch_size, torch.device("cpu"),
                               seed=1234 + engine.env.global_rank, pool=8,
                               image_shape=shape,
                               num_classes=args.num_classes)

    def batch_gen():
        for _ in range(args.steps_per_epoch):
            x, y = loader._host[loader._i % len(loader._host)]
            loader._i += 1
            yield (x.numpy(), y.numpy())

    dr = DistillReader(ins=["img", "label"], predicts=["logits"],
                       teacher_batch_size=args.batch_size,
                       require_num=args.require_num)
    dr.set_batch_generator(batch_gen)
    if args.teachers:
        dr.set_fixed_teacher(args.teachers)
    else:
        dr.set_dynamic_teacher(args.balance_server, args.service_name)

    for epoch in range(engine.start_epoch, args.num_epochs):
        t0 = time.monotonic()
        imgs = 0
        steps = 0
        for img, label, logits in dr():
            x = torch.from_numpy(np.ascontiguousarray(img)).to(engine.device)
            y = torch.from_numpy(np.ascontiguousarray(label)).to(engine.device)
            t = torch.from_numpy(np.ascontiguousarray(logits)).to(engine.device)
            if engine.channels_last and x.dim() == 4:
                x = x.contiguous(memory_format=torch.channels_last)
            engine.set_lr(engine.scaled_lr(epoch))
            loss = engine.train_step(x, y, teacher_logits=t)
            imgs += x.shape[0] * engine.world_size
            steps += 1
        if engine.device.type == "cuda":
            torch.cuda.synchronize()
        dt = time.monotonic() - t0
        if engine.env.is_rank0:
            log.info("distill epoch %d: %d steps, %.1f img/s (whole job), loss=%.4f",
                     epoch, steps, imgs / dt, float(loss.detach()))
        engine.save_checkpoint(epoch)
    if engine.ckpt:
        engine.ckpt.wait()
    edist.barrier(engine.device)
    edist.cleanup()
    return 0


if __name__ == "__main__":
    sys.exit(main())
