#!/usr/bin/env python3
"""BASELINE config 5 bench point: wide&deep CTR on the dense-embedding
all-reduce path.

    python tools/bench_ctr.py [--steps 200] [--batch 512]        # world 1
    python -m torch.distributed.run --nproc-per-node 2 ... \
        tools/bench_ctr.py   (EDL_FORCE_BACKEND=gloo on a 1-GPU box)

Prints one JSON line (rank 0): samples/s whole job, MAX-over-ranks.
"""
import argparse
import json
import os
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--steps", type=int, default=200)
    ap.add_argument("--warmup", type=int, default=30)
    ap.add_argument("--batch", type=int, default=512)
    ap.add_argument("--lr", type=float, default=0.02)
    args = ap.parse_args()

    import torch
    import torch.nn.functional as F

    from edl_amd.data.synthetic import SyntheticCTR
    from edl_amd.models import WideAndDeep
    from edl_amd.train import dist as edist
    from edl_amd.train.bucketed_ddp import BucketedAllReducer

    env, device = edist.init_from_env()
    world = env.world_size
    rank = env.global_rank
    torch.manual_seed(7)
    m = WideAndDeep().to(device)
    reducer = BucketedAllReducer(m.parameters(), bucket_cap_mb=8)
    reducer.broadcast_params(src=0)
    opt = torch.optim.SGD(m.parameters(), lr=args.lr)
    data = SyntheticCTR(args.batch, device, seed=13 + rank)

    def step():
        dense, sparse, label = data.next()
        loss = F.binary_cross_entropy_with_logits(
            m(dense, sparse).view(-1), label.view(-1))
        reducer.zero_grad()
        loss.backward()
        reducer.finalize()
        if reducer.grad_scale != 1.0:
            for b in reducer._buckets:
                b.buffer.mul_(reducer.grad_scale)
        opt.step()
        return loss

    for _ in range(args.warmup):
        loss = step()
    edist.barrier(device)
    if device.type == "cuda":
        torch.cuda.synchronize()
    t0 = time.monotonic()
    for _ in range(args.steps):
        loss = step()
    if device.type == "cuda":
        torch.cuda.synchronize()
    edist.barrier(device)
    el = time.monotonic() - t0
    if world > 1:
        import torch.distributed as dist

        t = torch.tensor([el], dtype=torch.float64)
        dist.all_reduce(t, op=dist.ReduceOp.MAX)
        el = float(t.item())
    if rank == 0:
        print(json.dumps({
            "metric": "samples/s", "value": round(args.steps * args.batch *
                                                  world / el, 1),
            "unit": "samples/s", "n_gpus": world, "steps": args.steps,
            "warmup": args.warmup, "ms_per_step": round(el / args.steps * 1e3, 3),
            "higher_is_better": True, "scaling": "weak",
            "vs_baseline": None, "dtype": "fp32", "data": "synthetic",
            "config": {"model": "wide_and_deep_ctr",
                       "global_batch": args.batch * world,
                       "parallelism": "dp%d" % world,
                       "loss": round(float(loss.item()), 4)},
        }))
    edist.cleanup()


if __name__ == "__main__":
    main()
