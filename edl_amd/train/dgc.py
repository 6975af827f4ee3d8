"""DGC — deep gradient compression (optional sparse all-reduce path).

Parity: the reference exposes Paddle's DGCMomentumOptimizer
(train_with_fleet.py:98,106-111: rampup_begin_step, momentum correction)
as an optional strategy. This is the standard DGC scheme (Lin et al.,
public algorithm) on our bucket layout:

  * per bucket, keep local ERROR FEEDBACK: residual += grad
  * pick the top-k% of |residual|; zero them out of the residual
  * exchange (indices, values) with all_gather; every rank scatter-adds
    all ranks' sparse contributions into the bucket buffer (÷ world)
  * before rampup_begin_step, fall back to dense all-reduce

Enabled via DistributedStrategy(use_dgc=True) -> TrainerEngine(dgc=...) or
directly: reducer = BucketedAllReducer(..., async_reduce=False);
dgc = DGCCompressor(reducer, ...); dgc.step() replaces reducer.finalize().
"""
import torch
import torch.distributed as dist


class DGCCompressor:
    def __init__(self, reducer, compress_ratio=0.01, rampup_begin_step=0,
                 process_group=None):
        self._reducer = reducer
        self.compress_ratio = compress_ratio
        self.rampup_begin_step = rampup_begin_step
        self._pg = process_group
        self._step = 0
        self._residuals = [torch.zeros_like(b.buffer)
                           for b in reducer._buckets]
        # DGC exchanges instead of the reducer's own async all-reduce
        reducer._async = False

    @property
    def enabled(self):
        return dist.is_initialized() and dist.get_world_size() > 1

    @torch.no_grad()
    def step(self):
        """Call instead of reducer.finalize() after backward."""
        self._step += 1
        if not self.enabled:
            return
        if self._step <= self.rampup_begin_step:
            self._reducer.finalize()
            return
        world = dist.get_world_size()
        for b, res in zip(self._reducer._buckets, self._residuals):
            res.add_(b.buffer)  # error feedback accumulates the raw grad
            n = res.numel()
            k = max(1, int(n * self.compress_ratio))
            _, idx = torch.topk(res.abs(), k, sorted=False)
            vals = res[idx].clone()
            res[idx] = 0  # transmitted mass leaves the residual

            # exchange sparse (idx, vals) with all ranks
            idx_list = [torch.empty_like(idx) for _ in range(world)]
            val_list = [torch.empty_like(vals) for _ in range(world)]
            dist.all_gather(idx_list, idx, group=self._pg)
            dist.all_gather(val_list, vals, group=self._pg)

            b.buffer.zero_()
            for i, v in zip(idx_list, val_list):
                b.buffer.scatter_add_(0, i, v)
            # NOTE: the reducer's grad_scale (1/world) is applied by the
            # optimizer, matching the dense path's averaging.
