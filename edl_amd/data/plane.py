"""Elastic data-plane glue for the image trainer.

`fetch_epoch_records` runs one pass of the leader-balanced reader
(data_server.py / reader.py — reference utils/data_server.py +
collective/distribute_reader.py) and returns this rank's
record lines for the epoch: rank 0's DataServer is the leader that slices
the file list round-robin over pods and levels the produced-batch queues;
endpoints rendezvous through the coordination store's dist_reader table.

`RecordImageSet` adapts those text records to the TrainerEngine loader
protocol (.next() -> (images, labels)): each record deterministically
seeds a synthetic image (there is no dataset network access — BASELINE
"synthetic data" note), with the label parsed from a leading integer when
present, else derived from the record hash. Records with equal content
produce equal tensors on every rank/world size, so elastic resizes resume
on identical data.
"""
import glob
import hashlib
import os
import time

import torch

from ..coord.client import CoordClient
from ..coord.tables import ETCD_DIST_READER
from ..utils.log import get_logger
from .data_server import DataServer
from .reader import Reader

log = get_logger("edl.data_plane")


def fetch_epoch_records(tenv, data_dir, batch_size, timeout=60):
    """One data-plane epoch -> list[str] records for this rank."""
    files = sorted(glob.glob(os.path.join(data_dir, "*")))
    pod_ids = [str(r) for r in range(tenv.world_size)]
    me = str(tenv.global_rank)
    srv = DataServer(file_list=files, pod_ids=pod_ids).start()
    store = CoordClient(tenv.store_endpoints, tenv.job_id)
    lease = store.grant(timeout)
    store.put(store.table_key(ETCD_DIST_READER, me),
              "127.0.0.1:%d" % srv.port, lease)
    eps = {}
    deadline = time.monotonic() + timeout
    while len(eps) < tenv.world_size and time.monotonic() < deadline:
        pfx = store.table_key(ETCD_DIST_READER)
        eps = {k[len(pfx):]: v for k, v in store.range(pfx)}
        time.sleep(0.1)
    if len(eps) != tenv.world_size:
        raise RuntimeError("data-plane rendezvous failed: %s" % eps)
    reader = Reader(me, eps["0"], srv, eps, batch_size=batch_size)
    records = []
    for item in reader:
        records.extend(item["data"])
    reader.close()
    # every pod must finish FETCHING before any pod stops its server —
    # assignments are balanced across pods, so a fast pod tearing down
    # early resets connections of peers still pulling its batches
    import torch.distributed as dist

    if dist.is_initialized():
        dist.barrier()
    store.revoke(lease)
    store.close()
    srv.stop()
    log.info("rank %s: %d records this epoch", me, len(records))
    return records


class RecordImageSet:
    """TrainerEngine loader over data-plane records (deterministic
    record -> (image, label) synthesis)."""

    def __init__(self, records, batch_size, device, image_shape=(3, 224, 224),
                 num_classes=1000, channels_last=False):
        self.records = records
        self.batch_size = batch_size
        self.device = device
        self.image_shape = image_shape
        self.num_classes = num_classes
        self.channels_last = channels_last
        self._i = 0

    def steps(self):
        return len(self.records) // self.batch_size

    @staticmethod
    def _rec_seed(rec):
        return int.from_bytes(
            hashlib.md5(rec.encode("utf-8", "replace")).digest()[:4], "little")

    def _rec_label(self, rec):
        head = rec.split()
        if head and head[0].lstrip("-").isdigit():
            return int(head[0]) % self.num_classes
        return self._rec_seed(rec) % self.num_classes

    def next(self):
        xs, ys = [], []
        for _ in range(self.batch_size):
            rec = self.records[self._i % len(self.records)]
            self._i += 1
            g = torch.Generator().manual_seed(self._rec_seed(rec))
            xs.append(torch.randn(self.image_shape, generator=g))
            ys.append(self._rec_label(rec))
        x = torch.stack(xs)
        if self.channels_last:
            x = x.contiguous(memory_format=torch.channels_last)
        y = torch.tensor(ys, dtype=torch.long)
        return x.to(self.device), y.to(self.device)
