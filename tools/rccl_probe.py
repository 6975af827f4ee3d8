#!/usr/bin/env python3
"""Multi-rank RCCL de-risk probe.

Launched as N ranks on ONE MI355X (all ranks map to cuda:0 — dist.py picks
rank_in_pod % device_count). Exercises, on real silicon, the exact code
the driver's 8-GPU SCALE run will execute for the first time:

  1. RCCL (backend "nccl" on ROCm) process-group init with >1 rank
  2. correctness of a raw all_reduce
  3. the full TrainerEngine step at world>1: bucketed async all-reduce
     overlap (post-accumulate-grad hooks), finalize(), grad_scale fold,
     broadcast_params, momentum broadcast, MAX-over-ranks timing
  4. reducer.rebuild() (elastic re-bucket) mid-run, then more steps
  5. clean destroy

Prints one JSON line per rank; rank 0's line is the verdict.

Usage: python -m torch.distributed.run --nnodes=1 --nproc-per-node 2 \
           --master-addr 127.0.0.1 --master-port 29531 tools/rccl_probe.py
"""
import json
import os
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))


def main():
    import torch
    import torch.distributed as dist

    rank = int(os.environ.get("RANK", "0"))
    world = int(os.environ.get("WORLD_SIZE", "1"))
    out = {"probe": "rccl_multirank", "rank": rank, "world": world,
           "stages": {}}

    def stage(name, ok, detail=""):
        out["stages"][name] = {"ok": bool(ok), "detail": str(detail)[:300]}
        if rank == 0:
            print("[stage] %s: %s %s" % (name, "OK" if ok else "FAIL", detail),
                  flush=True)
        if not ok:
            print(json.dumps(out), flush=True)
            sys.exit(1)

    # 1. RCCL init, every rank on cuda:0
    try:
        torch.cuda.set_device(0)
        dist.init_process_group(backend="nccl", rank=rank, world_size=world)
        stage("rccl_init", True, "world=%d device=cuda:0(all)" % world)
    except Exception as e:  # noqa: BLE001
        stage("rccl_init", False, repr(e))

    # 2. raw allreduce correctness
    try:
        t = torch.full((1 << 20,), float(rank + 1), device="cuda")
        dist.all_reduce(t)
        expect = sum(range(1, world + 1))
        ok = bool((t == expect).all().item())
        stage("allreduce_value", ok, "sum=%s expect=%d" % (t[0].item(), expect))
    except Exception as e:  # noqa: BLE001
        stage("allreduce_value", False, repr(e))

    # 3. full engine step at world>1 (bucketed overlap path)
    try:
        from edl_amd.data.synthetic import SyntheticImageNet
        from edl_amd.train.engine import TrainerEngine

        eng = TrainerEngine(model="resnet50_vd", per_device_batch=8,
                            dtype="bf16", checkpoint_dir=None).setup()
        loader = SyntheticImageNet(8, eng.device, channels_last=True,
                                   seed=100 + rank)
        for _ in range(3):
            x, y = loader.next()
            loss = eng.train_step(x, y)
        torch.cuda.synchronize()
        # params must be identical across ranks after synced steps
        p0 = next(eng.model.parameters()).detach().float()
        pmax = p0.clone()
        dist.all_reduce(pmax, op=dist.ReduceOp.MAX)
        pmin = p0.clone()
        dist.all_reduce(pmin, op=dist.ReduceOp.MIN)
        drift = float((pmax - pmin).abs().max().item())
        stage("engine_world%d_step" % world, drift == 0.0,
              "loss=%.4f param_drift=%g" % (loss.item(), drift))
    except Exception as e:  # noqa: BLE001
        stage("engine_world%d_step" % world, False, repr(e))

    # 4. elastic re-bucket mid-run (rebuild + optimizer snapshot/restore)
    try:
        eng.reducer.rebuild(bucket_cap_mb=50)
        for _ in range(2):
            x, y = loader.next()
            loss = eng.train_step(x, y)
        torch.cuda.synchronize()
        p0 = next(eng.model.parameters()).detach().float()
        pmax = p0.clone()
        dist.all_reduce(pmax, op=dist.ReduceOp.MAX)
        pmin = p0.clone()
        dist.all_reduce(pmin, op=dist.ReduceOp.MIN)
        drift = float((pmax - pmin).abs().max().item())
        stage("rebuild_then_step", drift == 0.0,
              "loss=%.4f drift=%g buckets=%s" %
              (loss.item(), drift,
               [round(m, 1) for m in eng.reducer.bucket_sizes_mb()[:4]]))
    except Exception as e:  # noqa: BLE001
        stage("rebuild_then_step", False, repr(e))

    # 5. MAX-over-ranks timing flow (the bench contract) + teardown
    try:
        t0 = time.monotonic()
        for _ in range(2):
            x, y = loader.next()
            eng.train_step(x, y)
        torch.cuda.synchronize()
        el = torch.tensor([time.monotonic() - t0], dtype=torch.float64,
                          device="cuda")
        dist.all_reduce(el, op=dist.ReduceOp.MAX)
        stage("max_over_ranks", True, "max_elapsed=%.3fs" % el.item())
        dist.barrier(device_ids=[0])
        dist.destroy_process_group()
        stage("teardown", True)
    except Exception as e:  # noqa: BLE001
        stage("teardown", False, repr(e))

    out["ok"] = all(s["ok"] for s in out["stages"].values())
    print(json.dumps(out), flush=True)
    return 0


if __name__ == "__main__":
    sys.exit(main())
