"""FusedSGD — momentum SGD over the reducer's flat buckets.

Replaces the reference's Paddle Momentum optimizer + fused-allreduce combo
(train_with_fleet.py:106-111 SGD momentum + L2 decay). Because
BucketedAllReducer re-homes both params and grads into flat per-bucket
buffers, the whole model update is ONE HIP kernel launch per bucket
(edl_amd._C.fused_sgd) instead of one per parameter tensor — and the DP
gradient average (1/world) is folded in as grad_scale, saving a full
read+write pass over all gradients.

State-dict format matches torch.optim.SGD (per-param momentum_buffer), so
checkpoints survive re-bucketing across elastic resizes."""
import torch

from . import available, ext
from ..train.bucketed_ddp import _view_like


class FusedSGD:
    handles_grad_scale = True

    def __init__(self, params, lr=0.1, momentum=0.9, weight_decay=1e-4, grad_scale=1.0,
                 reducer=None):
        # params is accepted for API parity; the flat buffers come from the
        # reducer (set via attach() or inferred from param .grad views).
        self.defaults = dict(lr=lr, momentum=momentum, weight_decay=weight_decay)
        self.param_groups = [dict(self.defaults, params=[p for p in params if p.requires_grad])]
        self.grad_scale = grad_scale
        self._reducer = reducer
        self._buckets = None  # list of (pflat, gflat, mflat, params, offsets)
        self._use_ext = available() and torch.cuda.is_available()
        # LR lives in a device scalar on GPU: the update kernel reads it at
        # run time, so a hipGraph-captured step tracks set_lr() on replay
        # instead of freezing the capture-time value (warmup/decay schedules
        # kept working under graph capture).
        self._lr_dev = None
        if reducer is not None:
            self.attach(reducer)

    def attach(self, reducer):
        self._reducer = reducer
        self._buckets = None
        reducer._attached_opt = self  # rebuild() snapshots/restores momentum
        return self

    def set_lr(self, lr):
        """Set the learning rate (updates the device scalar when present)."""
        for g in self.param_groups:
            g["lr"] = float(lr)
        if self._lr_dev is not None:
            self._lr_dev.fill_(float(lr))

    def _materialize(self):
        if self._buckets is not None:
            return self._buckets
        r = self._reducer
        if r is None:
            raise RuntimeError("FusedSGD needs a BucketedAllReducer (call attach())")
        out = []
        for b in r._buckets:
            if b.param_flat is None:
                raise RuntimeError("FusedSGD requires flatten_params=True buckets")
            m = torch.zeros_like(b.param_flat)
            offsets = []
            off = 0
            for p in b.params:
                offsets.append((off, p.numel()))
                off += p.numel()
            out.append({"p": b.param_flat, "g": b.buffer, "m": m,
                        "params": b.params, "offsets": offsets})
        self._buckets = out
        if self._use_ext and out and out[0]["p"].is_cuda and self._lr_dev is None:
            self._lr_dev = torch.full(
                (1,), self.param_groups[0]["lr"], dtype=torch.float32,
                device=out[0]["p"].device)
        return out

    @torch.no_grad()
    def step(self):
        from .conv import bump_weight_epoch

        bump_weight_epoch()  # invalidate per-step weight-derived caches
        g0 = self.param_groups[0]
        lr, mu, wd = g0["lr"], g0["momentum"], g0["weight_decay"]
        for bk in self._materialize():
            if self._use_ext:
                ext().fused_sgd(bk["p"], bk["g"], bk["m"], lr, mu, wd,
                                self.grad_scale, self._lr_dev)
            else:
                # torch fallback (CPU tests / bring-up): same math
                g = bk["g"]
                if self.grad_scale != 1.0:
                    g = g.mul(self.grad_scale)
                g = g.add(bk["p"], alpha=wd)
                bk["m"].mul_(mu).add_(g)
                bk["p"].add_(bk["m"], alpha=-lr)

    def zero_grad(self, set_to_none=False):
        if self._reducer is not None:
            self._reducer.zero_grad()

    # ---- torch.optim.SGD-compatible state dict ----
    def state_dict(self):
        state = {}
        idx = 0
        for bk in self._materialize():
            for (off, n), p in zip(bk["offsets"], bk["params"]):
                state[idx] = {"momentum_buffer": _view_like(bk["m"][off:off + n], p).clone()}
                idx += 1
        groups = [{k: v for k, v in self.param_groups[0].items() if k != "params"}]
        groups[0]["params"] = list(range(idx))
        return {"state": state, "param_groups": groups}

    def load_state_dict(self, sd):
        state = sd.get("state", {})
        idx = 0
        for bk in self._materialize():
            for (off, n), p in zip(bk["offsets"], bk["params"]):
                ent = state.get(idx, state.get(str(idx)))
                if ent is not None and "momentum_buffer" in ent and ent["momentum_buffer"] is not None:
                    _view_like(bk["m"][off:off + n], p).copy_(
                        ent["momentum_buffer"].to(bk["m"].device)
                    )
                idx += 1
        if sd.get("param_groups"):
            for k in ("lr", "momentum", "weight_decay"):
                if k in sd["param_groups"][0]:
                    self.param_groups[0][k] = sd["param_groups"][0][k]
            if self._lr_dev is not None:
                self._lr_dev.fill_(float(self.param_groups[0]["lr"]))
