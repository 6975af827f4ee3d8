"""Small-model elastic trainers: fit_a_line (BASELINE config 1 — linear
regression plumbing check on CPU/gloo) and ctr (config 5 — wide&deep on
the dense-embedding all-reduce path, elastic 1..8 GPUs).

    python -m edl_amd.train.train_simple --model fit_a_line --steps 200
    python -m edl_amd.train.train_simple --model ctr --steps 200

Spawned by edlrun; reads the launcher env contract."""
import argparse
import sys
import time

import torch
import torch.nn.functional as F

from ..data.synthetic import SyntheticCTR
from ..models import FitALine, WideAndDeep
from ..utils.log import get_logger
from . import dist as edist
from .bucketed_ddp import BucketedAllReducer
from .checkpoint import CheckpointManager
from .env import TrainerEnv
from ..ops.sgd import FusedSGD

log = get_logger("edl.train.simple")


def main(argv=None):
    p = argparse.ArgumentParser()
    p.add_argument("--model", default="fit_a_line", choices=["fit_a_line", "ctr"])
    p.add_argument("--batch_size", type=int, default=64)
    p.add_argument("--num_epochs", type=int, default=2)
    p.add_argument("--steps_per_epoch", type=int, default=100)
    p.add_argument("--lr", type=float, default=0.01)
    p.add_argument("--checkpoint", default=None)
    p.add_argument("--data_dir", default=None,
                   help="fit_a_line only: directory of text files "
                        "('f1 ... f13 y' per line) read through the ELASTIC "
                        "data plane (leader-balanced Reader) instead of "
                        "synthetic data")
    args = p.parse_args(argv)

    tenv = TrainerEnv()
    tenv, device = edist.init_from_env(tenv)
    torch.manual_seed(42)
    if args.model == "fit_a_line":
        model = FitALine().to(device)
        data = None
    else:
        model = WideAndDeep().to(device)
        data = SyntheticCTR(args.batch_size, device,
                            seed=1234 + tenv.global_rank)
    reducer = BucketedAllReducer(model.parameters(), bucket_cap_mb=4)
    opt = FusedSGD(model.parameters(), lr=args.lr, momentum=0.9,
                   weight_decay=0.0, grad_scale=reducer.grad_scale,
                   reducer=reducer)
    ckpt = CheckpointManager(args.checkpoint) if args.checkpoint else None
    start_epoch = 0
    if ckpt:
        got = ckpt.load()
        if got:
            model.load_state_dict(got[0])
            start_epoch = int(got[2]["epoch_no"]) + 1
    reducer.broadcast_params(0)

    def data_plane_epoch():
        """One pass over the shared file set via the elastic data plane:
        trainer global rank 0 runs the leader DataServer; endpoints
        rendezvous through the coordination store (dist_reader table)."""
        import glob
        import os as _os

        from ..coord.client import CoordClient
        from ..coord.tables import ETCD_DIST_READER
        from ..data.data_server import DataServer
        from ..data.reader import Reader

        files = sorted(glob.glob(_os.path.join(args.data_dir, "*")))
        pod_ids = [str(r) for r in range(tenv.world_size)]
        me = str(tenv.global_rank)
        srv = DataServer(file_list=files, pod_ids=pod_ids).start()
        store = CoordClient(tenv.store_endpoints, tenv.job_id)
        lease = store.grant(60)
        store.put(store.table_key(ETCD_DIST_READER, me),
                  "127.0.0.1:%d" % srv.port, lease)
        eps = {}
        deadline = time.monotonic() + 60
        while len(eps) < tenv.world_size and time.monotonic() < deadline:
            pfx = store.table_key(ETCD_DIST_READER)
            eps = {k[len(pfx):]: v for k, v in store.range(pfx)}
            time.sleep(0.1)
        assert len(eps) == tenv.world_size, "data-plane rendezvous failed"
        reader = Reader(me, eps["0"], srv, eps, batch_size=args.batch_size)
        batch = []
        for item in reader:
            for rec in item["data"]:
                vals = [float(v) for v in rec.split()]
                batch.append(vals)
                if len(batch) >= args.batch_size:
                    yield torch.tensor(batch)
                    batch = []
        if batch:
            yield torch.tensor(batch)
        reader.close()
        store.revoke(lease)
        store.close()
        srv.stop()

    g = torch.Generator().manual_seed(100 + tenv.global_rank)
    t0 = time.monotonic()
    for epoch in range(start_epoch, args.num_epochs):
        if args.model == "fit_a_line" and args.data_dir:
            # ranks may get UNEQUAL batch counts from the balancer: every
            # step first agrees (all-reduce) whether anyone still has data;
            # ranks without a batch contribute zero gradients (finalize
            # all-reduces their zeroed buckets) so collectives stay matched
            import torch.distributed as dist

            n_recs = 0
            it = iter(data_plane_epoch())
            while True:
                xy = next(it, None)
                if dist.is_initialized():
                    have = torch.tensor(
                        [0.0 if xy is None else 1.0], device=device)
                    dist.all_reduce(have)
                    if have.item() == 0:
                        break
                elif xy is None:
                    break
                reducer.zero_grad()
                if xy is not None:
                    x, y = xy[:, :13].to(device), xy[:, 13:14].to(device)
                    n_recs += x.shape[0]
                    loss = F.mse_loss(model(x), y)
                    loss.backward()
                reducer.finalize()
                opt.step()
            log.info("rank %d consumed %d records this epoch",
                     tenv.global_rank, n_recs)
            if ckpt and tenv.is_rank0:
                ckpt.save(model.state_dict(), {"epoch_no": epoch}, blocking=True)
            continue
        for _ in range(args.steps_per_epoch):
            reducer.zero_grad()
            if args.model == "fit_a_line":
                x = torch.randn(args.batch_size, 13, generator=g).to(device)
                y = (x.sum(1, keepdim=True) * 0.5 + 1.0)
                loss = F.mse_loss(model(x), y)
            else:
                dense, sparse, label = data.next()
                loss = F.binary_cross_entropy_with_logits(
                    model(dense, sparse), label)
            loss.backward()
            reducer.finalize()
            opt.step()
        if ckpt and tenv.is_rank0:
            ckpt.save(model.state_dict(), {"epoch_no": epoch}, blocking=True)
        if tenv.is_rank0:
            log.info("%s epoch %d loss=%.5f (world=%d)",
                     args.model, epoch, loss.item(), tenv.world_size)
    edist.barrier(device)
    if tenv.is_rank0:
        steps = (args.num_epochs - start_epoch) * args.steps_per_epoch
        log.info("done: %d steps in %.2fs", steps, time.monotonic() - t0)
    edist.cleanup()
    return 0


if __name__ == "__main__":
    sys.exit(main())
