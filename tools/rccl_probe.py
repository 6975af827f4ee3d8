#!/usr/bin/env python3
"""Multi-rank de-risk probe for the 8-GPU SCALE path.

Measured fact (gpurun_out/derisk, r2): RCCL 2.26.6 REFUSES two ranks on
one device — `Duplicate GPU detected : rank 0 and rank 1 both on CUDA
device` at the first collective — so the literal "N RCCL ranks pinned to
cuda:0" experiment is impossible on a 1-GPU box. What a 1-GPU box CAN
exercise, and what this probe covers:

  * world=1, backend nccl: real RCCL communicator init on MI355X, eager
    allreduce/broadcast/barrier, and an allreduce captured INSIDE a
    hipGraph + replayed (evidence for the RCCL-in-graph interaction the
    engine gates at world>1).
  * world=2, backend nccl: documents the Duplicate-GPU refusal (exit 0
    with mode=rccl_refuses_dup — evidence, not failure).
  * world=2, EDL_FORCE_BACKEND=gloo, tensors on cuda:0: the ENTIRE
    engine world>1 path on silicon — bucketed overlap hooks, finalize,
    grad_scale fold, broadcast_params + momentum broadcast, elastic
    rebuild(), MAX-over-ranks timing, teardown. Collectives run gloo;
    every compute kernel is the real HIP path.

Usage:
  python -m torch.distributed.run --nnodes=1 --nproc-per-node 1 \
      --master-addr 127.0.0.1 --master-port P tools/rccl_probe.py   # world-1 RCCL
  EDL_FORCE_BACKEND=gloo python -m torch.distributed.run ... --nproc-per-node 2 \
      ... tools/rccl_probe.py                                       # engine path
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
    backend = os.environ.get("EDL_FORCE_BACKEND", "nccl")
    out = {"probe": "rccl_multirank", "rank": rank, "world": world,
           "backend": backend, "stages": {}}

    def stage(name, ok, detail="", fatal=True):
        out["stages"][name] = {"ok": bool(ok), "detail": str(detail)[:300]}
        print("[stage %d] %s: %s %s" % (rank, name, "OK" if ok else "FAIL",
                                        detail), flush=True)
        if not ok and fatal:
            print(json.dumps(out), flush=True)
            sys.exit(1)

    torch.cuda.set_device(0)
    try:
        dist.init_process_group(backend=backend, rank=rank, world_size=world)
        stage("init", True, "backend=%s world=%d device=cuda:0" % (backend, world))
    except Exception as e:  # noqa: BLE001
        stage("init", False, repr(e))

    # raw allreduce correctness (detects the RCCL duplicate-GPU refusal)
    try:
        t = torch.full((1 << 20,), float(rank + 1), device="cuda")
        dist.all_reduce(t)
        expect = sum(range(1, world + 1))
        stage("allreduce_value", bool((t == expect).all().item()),
              "sum=%s expect=%d" % (t[0].item(), expect))
    except Exception as e:  # noqa: BLE001
        if "Duplicate GPU" in repr(e) and backend == "nccl" and world > 1:
            out["mode"] = "rccl_refuses_dup"
            out["ok"] = True
            print("[stage %d] allreduce_value: RCCL refuses multi-rank-per-GPU"
                  " (expected on a 1-GPU box) — use EDL_FORCE_BACKEND=gloo for"
                  " the engine path" % rank, flush=True)
            print(json.dumps(out), flush=True)
            return 0
        stage("allreduce_value", False, repr(e))

    if world == 1 and backend == "nccl":
        # world-1 RCCL: broadcast + barrier + allreduce inside a hipGraph
        try:
            b = torch.full((1024,), 7.0, device="cuda")
            dist.broadcast(b, src=0)
            dist.barrier(device_ids=[0])
            stage("rccl_world1_eager", True)
        except Exception as e:  # noqa: BLE001
            stage("rccl_world1_eager", False, repr(e))
        try:
            g = torch.cuda.CUDAGraph()
            x = torch.ones(1 << 16, device="cuda")
            # warmup on a side stream, then capture an allreduce
            s = torch.cuda.Stream()
            s.wait_stream(torch.cuda.current_stream())
            with torch.cuda.stream(s):
                dist.all_reduce(x)
            torch.cuda.current_stream().wait_stream(s)
            torch.cuda.synchronize()
            with torch.cuda.graph(g):
                dist.all_reduce(x)
            x.fill_(3.0)
            g.replay()
            torch.cuda.synchronize()
            stage("rccl_allreduce_in_hipgraph", bool((x == 3.0).all().item()),
                  "captured+replayed", fatal=False)
        except Exception as e:  # noqa: BLE001
            stage("rccl_allreduce_in_hipgraph", False, repr(e), fatal=False)
        dist.destroy_process_group()
        out["ok"] = all(s["ok"] for n, s in out["stages"].items()
                        if n != "rccl_allreduce_in_hipgraph")
        print(json.dumps(out), flush=True)
        return 0 if out["ok"] else 1

    # ---- world>1 engine path (gloo collectives, HIP compute) ----
    from edl_amd.data.synthetic import SyntheticImageNet
    from edl_amd.train.engine import TrainerEngine

    def drift_of(eng):
        p0 = next(eng.model.parameters()).detach().float()
        pmax = p0.clone()
        dist.all_reduce(pmax, op=dist.ReduceOp.MAX)
        pmin = p0.clone()
        dist.all_reduce(pmin, op=dist.ReduceOp.MIN)
        return float((pmax - pmin).abs().max().item())

    try:
        eng = TrainerEngine(model="resnet50_vd", per_device_batch=8,
                            dtype="bf16", checkpoint_dir=None).setup()
        loader = SyntheticImageNet(8, eng.device, channels_last=True,
                                   seed=100 + rank)
        for _ in range(3):
            x, y = loader.next()
            loss = eng.train_step(x, y)
        torch.cuda.synchronize()
        d = drift_of(eng)
        stage("engine_world%d_step" % world, d == 0.0,
              "loss=%.4f param_drift=%g" % (loss.item(), d))
    except Exception as e:  # noqa: BLE001
        stage("engine_world%d_step" % world, False, repr(e))

    try:
        eng.reducer.rebuild(bucket_cap_mb=50)
        for _ in range(2):
            x, y = loader.next()
            loss = eng.train_step(x, y)
        torch.cuda.synchronize()
        d = drift_of(eng)
        stage("rebuild_then_step", d == 0.0,
              "loss=%.4f drift=%g buckets=%s" %
              (loss.item(), d,
               [round(m, 1) for m in eng.reducer.bucket_sizes_mb()[:4]]))
    except Exception as e:  # noqa: BLE001
        stage("rebuild_then_step", False, repr(e))

    try:
        t0 = time.monotonic()
        for _ in range(2):
            x, y = loader.next()
            eng.train_step(x, y)
        torch.cuda.synchronize()
        el = torch.tensor([time.monotonic() - t0], dtype=torch.float64)
        dist.all_reduce(el, op=dist.ReduceOp.MAX)
        stage("max_over_ranks", True, "max_elapsed=%.3fs" % el.item())
        dist.barrier()
        dist.destroy_process_group()
        stage("teardown", True)
    except Exception as e:  # noqa: BLE001
        stage("teardown", False, repr(e))

    out["ok"] = all(s["ok"] for s in out["stages"].values())
    print(json.dumps(out), flush=True)
    return 0


if __name__ == "__main__":
    sys.exit(main())
