"""Synthetic input pipelines (BASELINE.json: "synthetic 3x224x224 data,
random-init weights — there is no network for datasets").

The ImageNet-shaped loader keeps a pool of pinned host batches and
prefetches the next batch to the device on a side stream while the current
step computes — the H2D copy is real work on every step (an honest stand-in
for the reference's DALI GPU pipeline, example/collective/resnet50/dali.py),
it is just overlapped, as it would be in production."""
import torch


class SyntheticImageNet:
    def __init__(self, batch_size, device, dtype=torch.float32, num_classes=1000,
                 pool=8, image_shape=(3, 224, 224), channels_last=False, seed=1234):
        g = torch.Generator().manual_seed(seed)
        self.device = device
        self._host = []
        pin = device.type == "cuda"
        for _ in range(pool):
            x = torch.randn((batch_size, *image_shape), generator=g, dtype=dtype)
            y = torch.randint(0, num_classes, (batch_size,), generator=g)
            if channels_last:
                x = x.contiguous(memory_format=torch.channels_last)
            if pin:
                x, y = x.pin_memory(), y.pin_memory()
            self._host.append((x, y))
        self._i = 0
        self._stream = torch.cuda.Stream() if device.type == "cuda" else None
        self._next = None
        self._prefetch()

    def _copy(self, hx, hy):
        x = hx.to(self.device, non_blocking=True)
        y = hy.to(self.device, non_blocking=True)
        return x, y

    def _prefetch(self):
        hx, hy = self._host[self._i % len(self._host)]
        self._i += 1
        if self._stream is not None:
            with torch.cuda.stream(self._stream):
                self._next = self._copy(hx, hy)
        else:
            self._next = (hx, hy)

    def next(self):
        """-> (images, labels) on device; prefetches the following batch."""
        if self._stream is not None:
            torch.cuda.current_stream().wait_stream(self._stream)
        batch = self._next
        if self._stream is not None:
            # the tensors are consumed on the compute stream
            batch[0].record_stream(torch.cuda.current_stream())
            batch[1].record_stream(torch.cuda.current_stream())
        self._prefetch()
        return batch


class SyntheticCTR:
    """Criteo-shaped batches for the wide&deep config (reference example/ctr)."""

    def __init__(self, batch_size, device, num_dense=13, num_sparse=26,
                 vocab_size=100000, pool=8, seed=1234):
        g = torch.Generator().manual_seed(seed)
        self.device = device
        self._host = []
        for _ in range(pool):
            dense = torch.randn(batch_size, num_dense, generator=g)
            sparse = torch.randint(0, vocab_size, (batch_size, num_sparse), generator=g)
            label = torch.randint(0, 2, (batch_size, 1), generator=g).float()
            self._host.append((dense, sparse, label))
        self._i = 0

    def next(self):
        b = self._host[self._i % len(self._host)]
        self._i += 1
        return tuple(t.to(self.device, non_blocking=True) for t in b)
