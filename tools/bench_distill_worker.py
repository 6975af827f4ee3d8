"""One student rank of the service-distill bench (spawned by
bench_distill.py --teacher_gpus/--student_gpus via torchrun). Trains the
student against the shared teacher pool through the DistillReader and
reports whole-job img/s (MAX step time over ranks, rank 0 prints)."""
import argparse
import json
import os
import sys
import time

import numpy as np

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
sys.path.insert(0, REPO)


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--steps", type=int, default=20)
    ap.add_argument("--warmup", type=int, default=5)
    ap.add_argument("--batch_size", type=int, default=32)
    ap.add_argument("--teacher_batch_size", type=int, default=16)
    ap.add_argument("--require_num", type=int, default=1)
    ap.add_argument("--student_model", default="resnet50_vd")
    ap.add_argument("--teachers", required=True)
    args = ap.parse_args()

    import torch
    import torch.distributed as dist

    from edl_amd.distill.reader import DistillReader
    from edl_amd.train.engine import TrainerEngine

    engine = TrainerEngine(
        model=args.student_model, per_device_batch=args.batch_size,
        base_lr=0.01, use_hip_ops=torch.cuda.is_available(),
        dtype="bf16" if torch.cuda.is_available() else "fp32",
        channels_last=torch.cuda.is_available(),
        graph_capture=False, kd_alpha=1.0).setup()
    engine.model.train()
    rank = engine.env.global_rank
    total = args.warmup + args.steps
    rng = np.random.RandomState(rank)
    batches = [(rng.randn(args.batch_size, 3, 224, 224).astype(np.float32),
                rng.randint(0, 1000, (args.batch_size,)).astype(np.int64))
               for _ in range(4)]

    def batch_gen():
        for i in range(total):
            yield batches[i % len(batches)]

    dr = DistillReader(["img", "label"], ["logits"],
                       teacher_batch_size=args.teacher_batch_size,
                       require_num=args.require_num)
    dr.set_batch_generator(batch_gen)
    dr.set_fixed_teacher(args.teachers.split(","))

    n, t0 = 0, None
    for img, label, logits in dr():
        x = torch.from_numpy(img).to(engine.device)
        if engine.device.type == "cuda":
            x = x.to(torch.bfloat16).contiguous(
                memory_format=torch.channels_last)
        y = torch.from_numpy(label).to(engine.device)
        lg = np.ascontiguousarray(logits)
        if not lg.flags.writeable:
            lg = lg.copy()
        t = torch.from_numpy(lg).to(engine.device)
        engine.train_step(x, y, teacher_logits=t)
        n += 1
        if n == args.warmup:
            if engine.device.type == "cuda":
                torch.cuda.synchronize()
            if dist.is_initialized():
                dist.barrier()
            t0 = time.monotonic()
    if engine.device.type == "cuda":
        torch.cuda.synchronize()
    dt = time.monotonic() - t0
    timed = n - args.warmup
    if dist.is_initialized():
        tmax = torch.tensor([dt])
        dist.all_reduce(tmax, op=dist.ReduceOp.MAX)
        dt = float(tmax)
    world = engine.world_size
    if rank == 0:
        print(json.dumps({
            "metric": "img/s",
            "value": round(timed * args.batch_size * world / dt, 1),
            "unit": "img/s", "mode": "distill_service",
            "n_students": world, "steps": timed,
            "ms_per_step": round(dt / timed * 1e3, 2),
            "dtype": "bf16" if torch.cuda.is_available() else "fp32",
            "data": "synthetic",
            "vs_baseline_service_1514": round(
                timed * args.batch_size * world / dt / 1514.0, 3),
        }), flush=True)
    from edl_amd.train import dist as edist

    edist.cleanup()
    return 0


if __name__ == "__main__":
    sys.exit(main())
