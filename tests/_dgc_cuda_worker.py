"""Worker for test_gpu_extras.test_dgc_two_rank_cuda: 2 gloo ranks, CUDA
tensors on cuda:0 — DGC compressed exchange + error feedback on GPU."""
import json
import os
import sys

import torch
import torch.distributed as dist

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

from edl_amd.train.bucketed_ddp import BucketedAllReducer  # noqa: E402
from edl_amd.train.dgc import DGCCompressor  # noqa: E402


def main():
    rank = int(os.environ["RANK"])
    world = int(os.environ["WORLD_SIZE"])
    torch.cuda.set_device(0)
    dist.init_process_group("gloo", rank=rank, world_size=world)
    torch.manual_seed(7)  # same params on both ranks
    model = torch.nn.Sequential(
        torch.nn.Linear(256, 256), torch.nn.ReLU(), torch.nn.Linear(256, 8)
    ).cuda()
    reducer = BucketedAllReducer(model.parameters(), bucket_cap_mb=1)
    dgc = DGCCompressor(reducer, compress_ratio=0.05, rampup_begin_step=0)

    ok = True
    detail = ""
    for step in range(3):
        torch.manual_seed(100 + step * world + rank)  # per-rank data
        x = torch.randn(16, 256, device="cuda")
        y = model(x).sum()
        reducer.zero_grad()
        y.backward()
        dgc.step()
        # after the compressed exchange both ranks must hold IDENTICAL
        # bucket gradients
        for b in reducer._buckets:
            flat = b.buffer.detach().clone()
            gathered = [torch.empty_like(flat) for _ in range(world)]
            dist.all_gather(gathered, flat)
            if not all(torch.equal(g.cpu(), gathered[0].cpu()) for g in gathered):
                ok = False
                detail = "rank grads diverge at step %d" % step
    # error feedback: residuals exist on CUDA and are finite
    for r in dgc._residuals:
        if not torch.isfinite(r).all():
            ok = False
            detail = "non-finite residual"
    dist.barrier()
    dist.destroy_process_group()
    print(json.dumps({"dgc_cuda": True, "rank": rank, "ok": ok,
                      "detail": detail}), flush=True)
    return 0 if ok else 1


if __name__ == "__main__":
    sys.exit(main())
