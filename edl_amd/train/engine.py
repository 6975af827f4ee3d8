"""The MI355X training engine — what Paddle Fleet was to the reference.

One process per GPU (RCCL over xGMI), bf16 autocast + channels_last NHWC,
gradient-bucketed overlap all-reduce (bucketed_ddp), fused HIP ops where
the extension is available (edl_amd.ops), optional whole-step hipGraph
capture (launch-bound CNN steps replay as ONE graph), versioned async
checkpointing, and elastic resume driven by the launcher's stop-resume.

Replaces: fleet.distributed_optimizer + DistributedStrategy
(train_with_fleet.py:367-381), FLAGS_sync_nccl_allreduce, fuse_all_reduce_ops,
fleet save/load_check_point."""
import os
import time

import torch
import torch.nn.functional as F

from ..models import build_model
from ..utils.log import get_logger
from . import dist as edist
from .bucketed_ddp import BucketedAllReducer
from .checkpoint import CheckpointManager

log = get_logger("edl.engine")

# EDL_PROFILE_MARKERS=1 emits roctx ranges (torch.cuda.nvtx IS roctx on
# ROCm) around the step phases so `rocprofv3 --marker-trace` can attribute
# kernels to forward/backward/allreduce/optimizer (SURVEY.md §5.1 — the
# reference's env-gated _TimeLine, but rocprof-native)
_MARKERS = os.environ.get("EDL_PROFILE_MARKERS") == "1"


class _mark:
    __slots__ = ("name",)

    def __init__(self, name):
        self.name = name

    def __enter__(self):
        if _MARKERS:
            torch.cuda.nvtx.range_push(self.name)

    def __exit__(self, *exc):
        if _MARKERS:
            torch.cuda.nvtx.range_pop()


def piecewise_lr(base_lr, epoch, boundaries=(30, 60, 90), decay=0.1, warmup_epochs=5,
                 step_in_epoch=0.0):
    """Reference LR policy (train_with_fleet.py piecewise_decay 30/60/90
    x0.1) + linear warmup."""
    e = epoch + step_in_epoch
    if e < warmup_epochs:
        return base_lr * (e + 1e-9) / warmup_epochs
    mult = 1.0
    for b in boundaries:
        if epoch >= b:
            mult *= decay
    return base_lr * mult


class DynamicLossScaler:
    """Dynamic loss scaling for fp16 training (reference
    train_with_fleet.py:318-321: mixed_precision.decorate with
    scale_loss=128 + use_dynamic_loss_scaling). bf16 never needs it —
    this engages only for dtype=fp16: the loss is multiplied by `value`
    before backward, the optimizer folds 1/value into its grad scale,
    and a step whose gradients overflowed is SKIPPED while the scale
    backs off; `growth_interval` clean steps double it again."""

    def __init__(self, init_scale=2.0 ** 15, growth_factor=2.0,
                 backoff_factor=0.5, growth_interval=2000):
        self.value = float(init_scale)
        self.growth_factor = growth_factor
        self.backoff_factor = backoff_factor
        self.growth_interval = growth_interval
        self._good_steps = 0

    def update(self, found_inf):
        """-> True if the step should be applied, False to skip it."""
        if found_inf:
            self.value = max(1.0, self.value * self.backoff_factor)
            self._good_steps = 0
            return False
        self._good_steps += 1
        if self._good_steps >= self.growth_interval:
            self.value *= self.growth_factor
            self._good_steps = 0
        return True


class TrainerEngine:
    def __init__(
        self,
        model="resnet50_vd",
        per_device_batch=32,
        num_classes=1000,
        base_lr=0.1,
        momentum=0.9,
        weight_decay=1e-4,
        label_smoothing=0.1,
        dtype="bf16",
        channels_last=True,
        bucket_mb=25,
        checkpoint_dir=None,
        use_hip_ops=True,
        graph_capture=None,
        kd_teacher=None,
        kd_alpha=1.0,
        recompute=False,
        dgc=False,
        dgc_ratio=0.01,
        dgc_rampup=0,
    ):
        self.model_name = model
        self.per_device_batch = per_device_batch
        self.num_classes = num_classes
        self.base_lr = base_lr
        self.momentum = momentum
        self.weight_decay = weight_decay
        self.label_smoothing = label_smoothing
        self.dtype = {"bf16": torch.bfloat16, "fp16": torch.float16,
                      "fp32": torch.float32}[dtype]
        # fp16 needs dynamic loss scaling (bf16/fp32 do not)
        self.scaler = DynamicLossScaler() if self.dtype == torch.float16 else None
        self.channels_last = channels_last
        self.bucket_mb = bucket_mb
        self.checkpoint_dir = checkpoint_dir
        self.use_hip_ops = use_hip_ops
        self.graph_capture = graph_capture
        self.kd_teacher = kd_teacher  # callable(images)->soft logits, or None
        self.kd_alpha = kd_alpha
        self.recompute = recompute  # activation checkpointing (reference
        # dist_strategy.forward_recompute, train_with_fleet.py:322-325)
        self.dgc_cfg = (dgc, dgc_ratio, dgc_rampup)  # optional sparse allreduce
        self.dgc = None

        self.env = None
        self.device = None
        self.model = None
        self.reducer = None
        self.opt = None
        self.ckpt = None
        self.start_epoch = 0
        self.start_step = 0  # step within start_epoch (mid-epoch resume)
        self.global_step = 0
        self._graph = None
        self._static = {}
        # side stream for the per-epoch weight-repack prefill
        # (EDL_PREFILL_DERIVED=1 opt-in: measured -5% under hipGraph
        # capture on MI355X — the two-stream graph replays with less
        # launch pipelining than the single-stream one; see NOTES)
        self._prefill_stream = None
        self._prefill_mods = None

    # ---- setup ----
    def setup(self, env=None):
        self.env, self.device = edist.init_from_env(env)
        torch.manual_seed(42 + self.env.global_rank)
        self.model = build_model(self.model_name, num_classes=self.num_classes)
        self.model.to(self.device)
        if (self.device.type == "cuda"
                and os.environ.get("EDL_PREFILL_DERIVED", "0") == "1"):
            self._prefill_stream = torch.cuda.Stream()
        if self.channels_last and self.device.type == "cuda":
            self.model.to(memory_format=torch.channels_last)
        if self.use_hip_ops and self.device.type == "cuda":
            from .. import ops

            ops.assert_hip_ops_available(self.model)
            self._mark_channels_last_conv_weights()
        if self.recompute and hasattr(self.model, "stages"):
            import torch.utils.checkpoint as ckpt_mod

            class _Recompute(torch.nn.Module):
                def __init__(self, inner):
                    super().__init__()
                    self.inner = inner

                def forward(self, x):
                    if self.training and torch.is_grad_enabled():
                        return ckpt_mod.checkpoint(self.inner, x, use_reentrant=False)
                    return self.inner(x)

            self.model.stages = torch.nn.Sequential(
                *[_Recompute(s) for s in self.model.stages]
            )

        self.reducer = BucketedAllReducer(
            self.model.parameters(), bucket_cap_mb=self.bucket_mb
        )
        if self.dgc_cfg[0]:
            from .dgc import DGCCompressor

            self.dgc = DGCCompressor(self.reducer, self.dgc_cfg[1],
                                     self.dgc_cfg[2])
        self.opt = self._build_optimizer()
        if self.checkpoint_dir:
            self.ckpt = CheckpointManager(self.checkpoint_dir)
            self._resume()
        # everyone starts from rank-0's weights (fresh or resumed)
        self.reducer.broadcast_params(src=0)
        # ... and rank-0's momentum: with non-shared checkpoint dirs the
        # per-rank local checkpoints may disagree, and divergent momentum
        # makes parameters drift apart permanently (params alone are synced)
        self._broadcast_opt_state(src=0)
        from ..ops.conv import bump_weight_epoch

        bump_weight_epoch()  # weights changed outside an optimizer step
        self._init_bf16_mirrors()
        # elastic re-bucketing re-homes params into new flats; the bf16
        # compute mirrors must follow or conv kernels read dead storage
        self.reducer.register_rebuild_callback(self._init_bf16_mirrors)
        edist.barrier(self.device)
        return self

    def _mark_channels_last_conv_weights(self):
        """Flag dense 3x3 conv weights for channels-last BUCKET storage
        ([Cout, 3, 3, Cin] physical, logical view unchanged): the s-major
        w3 repack becomes a zero-cost view of the bf16 mirror (one less
        full-weight copy per conv per step) and the 3x3 wgrad's s-major
        epilogue can accumulate the grad bucket COALESCED (round-1's
        stride-9 scatter measured-negative disappears)."""
        from ..ops.conv import Conv2dFast

        for m in self.model.modules():
            if (isinstance(m, Conv2dFast) and m.kernel_size == (3, 3)
                    and m.groups == 1 and m.bias is None
                    and m.in_channels % 64 == 0 and m.out_channels % 64 == 0
                    and m.padding == (1, 1)):
                w = m.weight
                w._edl_phys_shape = (w.shape[0], 3, 3, w.shape[1])
                w._edl_phys_perm = (0, 3, 1, 2)

    def _broadcast_opt_state(self, src=0):
        import torch.distributed as dist

        if not (dist.is_initialized() and dist.get_world_size() > 1):
            return
        if hasattr(self.opt, "_materialize"):
            for bk in self.opt._materialize():
                dist.broadcast(bk["m"], src=src)

    def _init_bf16_mirrors(self):
        """Per-bucket bf16 mirror of the flat fp32 params: each conv's
        plain bf16 compute copy becomes a VIEW into the mirror, and one
        copy_ per bucket after every optimizer step replaces ~100
        individual per-weight cast kernels per step (the repacked
        layouts — transposed/s-major/rotated — stay per-weight)."""
        self._bf16_mirrors = []
        if not (self.use_hip_ops and self.device.type == "cuda"):
            return
        for b in self.reducer._buckets:
            if b.param_flat is None or b.param_flat.dtype != torch.float32:
                continue
            mirror = torch.empty_like(b.param_flat, dtype=torch.bfloat16)
            self._bf16_mirrors.append((mirror, b.param_flat))
            off = 0
            from .bucketed_ddp import _view_like
            for p in b.params:
                p._edl_bf16 = _view_like(mirror[off:off + p.numel()], p)
                off += p.numel()
        self._refresh_bf16_mirrors()

    def _refresh_bf16_mirrors(self):
        for mirror, flat in getattr(self, "_bf16_mirrors", ()):
            mirror.copy_(flat, non_blocking=True)

    def _build_optimizer(self):
        from ..ops.sgd import FusedSGD

        return FusedSGD(
            self.model.parameters(), lr=self.base_lr, momentum=self.momentum,
            weight_decay=self.weight_decay, grad_scale=self.reducer.grad_scale,
            reducer=self.reducer,
        )

    @property
    def world_size(self):
        return edist.world_size()

    @property
    def global_batch(self):
        return self.per_device_batch * self.world_size

    def scaled_lr(self, epoch, step_in_epoch=0.0):
        """Linear-scaling rule: base_lr is quoted at total batch 256
        (reference train_parameters, models/resnet_vd.py) and rescales on
        every elastic world change."""
        lr = self.base_lr * self.global_batch / 256.0
        return piecewise_lr(lr, epoch, step_in_epoch=step_in_epoch)

    def set_lr(self, lr):
        if hasattr(self.opt, "set_lr"):
            # FusedSGD keeps LR in a device scalar the update kernel reads,
            # so a captured graph tracks schedule changes on replay
            self.opt.set_lr(lr)
        else:
            for g in self.opt.param_groups:
                g["lr"] = lr

    # ---- checkpoint ----
    def _resume(self):
        got = self.ckpt.load(map_location="cpu")
        if got is None:
            if self.env.is_rank0:
                log.info("no checkpoint in %s; fresh start", self.checkpoint_dir)
            return
        model_state, opt_state, ts = got
        self.model.load_state_dict(model_state)
        if opt_state is not None:
            try:
                self.opt.load_state_dict(opt_state)
            except (ValueError, KeyError) as e:
                log.warning("optimizer state mismatch (%s); reset", e)
        if ts.get("mid_epoch"):
            # step-level checkpoint: resume INSIDE the epoch (beyond the
            # reference's per-epoch granularity, doc/fault_tolerance.md
            # "step level ... next version")
            self.start_epoch = int(ts.get("epoch_no", 0))
            self.start_step = int(ts.get("step_in_epoch", 0))
        else:
            self.start_epoch = int(ts.get("epoch_no", -1)) + 1
            self.start_step = 0
        self.global_step = int(ts.get("global_step", 0))
        if self.env.is_rank0:
            log.info(
                "resumed from checkpoint v%s: epoch %d step %d, global_step %d (world=%d)",
                ts.get("_version"), self.start_epoch, self.start_step,
                self.global_step, self.world_size,
            )

    def save_checkpoint(self, epoch, extra=None, blocking=False):
        if self.ckpt is None or not self.env.is_rank0:
            return None
        ts = {"epoch_no": epoch, "global_step": self.global_step,
              "world_size": self.world_size, "total_batch_size": self.global_batch}
        ts.update(extra or {})
        return self.ckpt.save(
            self.model.state_dict(), ts, optimizer_state=self.opt.state_dict(),
            blocking=blocking,
        )

    # ---- the step ----
    def _loss(self, logits, labels, teacher_logits=None):
        if teacher_logits is not None:
            # KD soft-label cross-entropy (reference
            # example/distill/resnet/train_with_fleet.py:254-259: student
            # trains against teacher soft labels) — fused HIP kernel on GPU
            from ..ops.functional import kd_soft_cross_entropy

            kd = kd_soft_cross_entropy(logits, teacher_logits)
            if self.kd_alpha >= 1.0:
                return kd
            ce = F.cross_entropy(logits.float(), labels,
                                 label_smoothing=self.label_smoothing)
            return self.kd_alpha * kd + (1 - self.kd_alpha) * ce
        return F.cross_entropy(logits.float(), labels,
                               label_smoothing=self.label_smoothing)

    def train_step(self, images, labels, teacher_logits=None):
        self.reducer.zero_grad()
        self._prefill_derived_async()
        with _mark("edl.forward"):
            with torch.autocast(device_type=self.device.type, dtype=self.dtype,
                                enabled=self.dtype != torch.float32):
                logits = self.model(images)
            loss = self._loss(logits, labels, teacher_logits)
        with _mark("edl.backward"):
            if self._prefill_stream is not None:
                # dgrad kernels read the prefilled repacks (see
                # _prefill_derived_async) — join the side stream here,
                # after forward has run in parallel with the repacks
                torch.cuda.current_stream().wait_stream(self._prefill_stream)
            if self.scaler is not None:
                (loss * self.scaler.value).backward()
            else:
                loss.backward()
        with _mark("edl.allreduce_finalize"):
            if self.dgc is not None:
                self.dgc.step()
            else:
                self.reducer.finalize()
        with _mark("edl.optimizer"):
            inv_scale = 1.0
            if self.scaler is not None:
                # buckets already hold the all-reduced (scaled) grads, so
                # every rank sees the same values and makes the same call.
                # One device-side reduction over all buckets, ONE host sync
                # (a per-bucket .all().item() loop would serialize the step
                # with N GPU->CPU round-trips).
                flags = torch.stack(
                    [torch.isfinite(b.buffer).all()
                     for b in self.reducer._buckets])
                found_inf = not bool(flags.all().item())
                if not self.scaler.update(found_inf):
                    self.global_step += 1
                    return loss  # overflow: skip the update, scale backed off
                inv_scale = 1.0 / self.scaler.value
            if getattr(self.opt, "handles_grad_scale", False):
                if inv_scale != 1.0:
                    base = self.opt.grad_scale
                    self.opt.grad_scale = base * inv_scale
                    try:
                        self.opt.step()
                    finally:
                        self.opt.grad_scale = base
                else:
                    self.opt.step()
            else:
                scale = self.reducer.grad_scale * inv_scale
                if scale != 1.0:
                    for b in self.reducer._buckets:
                        b.buffer.mul_(scale)
                self.opt.step()
            self._refresh_bf16_mirrors()
        self.global_step += 1
        return loss

    def _prefill_derived_async(self):
        """Launch the per-epoch weight repacks (wt_t, dgrad w3 repacks)
        on a side stream at step start: they only depend on the bf16
        mirrors (written at the END of the previous step), so they run in
        parallel with forward; backward joins the stream before its first
        dgrad. Capture-safe: fork/join via wait_stream edges."""
        if self._prefill_stream is None:
            return
        mods = self._prefill_mods
        if mods is None:
            from ..ops.conv import Conv2dFast

            mods = [m for m in self.model.modules()
                    if isinstance(m, Conv2dFast)]
            self._prefill_mods = mods
        cur = torch.cuda.current_stream()
        self._prefill_stream.wait_stream(cur)
        with torch.cuda.stream(self._prefill_stream):
            for m in mods:
                m.prefill_derived()

    # ---- hipGraph capture ----
    def maybe_capture(self, images, labels):
        """Capture the full train step (fwd+bwd+allreduce+opt) as ONE hip
        graph. CNN steps at bs 32/GPU are launch-bound; replaying a single
        graph removes per-kernel launch gaps (guide §launches-baseline).
        Returns True if capture succeeded."""
        if self.device.type != "cuda" or self._graph is not None:
            return False
        if self.scaler is not None:
            # fp16 loss scaling is data-dependent control flow (inf check,
            # step skip) — not capturable as one static graph
            return False
        want = self.graph_capture
        if want is None:
            env = os.environ.get("EDL_GRAPH_CAPTURE")
            if env is not None:
                want = env == "1"
            else:
                # default: capture only single-GPU steps — capturing RCCL
                # collectives inside a graph is riskier (a hang on one rank
                # stalls the job); opt in with EDL_GRAPH_CAPTURE=1
                want = self.world_size == 1
        if not want:
            return False
        try:
            self._static["x"] = images.clone()
            self._static["y"] = labels.clone()
            torch.cuda.synchronize()
            # warmup on a side stream (required before capture)
            s = torch.cuda.Stream()
            s.wait_stream(torch.cuda.current_stream())
            with torch.cuda.stream(s):
                for _ in range(3):
                    self.train_step(self._static["x"], self._static["y"])
            torch.cuda.current_stream().wait_stream(s)
            torch.cuda.synchronize()
            g = torch.cuda.CUDAGraph()
            step_before = self.global_step
            with torch.cuda.graph(g):
                self._static["loss"] = self.train_step(self._static["x"], self._static["y"])
            self.global_step = step_before  # capture itself is not a step
            self._graph = g
            log.info("captured train step as hipGraph")
            return True
        except Exception as e:  # noqa: BLE001 - fall back to eager
            log.warning("hipGraph capture failed (%s); running eager", e)
            self._graph = None
            self._static = {}
            return False

    def replay_step(self, images, labels):
        """Run one training step; uses the captured graph when available."""
        if self._graph is not None:
            self._static["x"].copy_(images, non_blocking=True)
            self._static["y"].copy_(labels, non_blocking=True)
            self._graph.replay()
            self.global_step += 1
            return self._static["loss"]
        return self.train_step(images, labels)

    # ---- epoch driver ----
    def train_epoch(self, epoch, loader, steps, log_every=50, on_step=None,
                    start_step=0):
        self.model.train()
        t0 = time.monotonic()
        imgs_done = 0
        loss = None
        for it in range(start_step, steps):
            self.set_lr(self.scaled_lr(epoch, it / max(1, steps)))
            x, y = loader.next()
            loss = self.replay_step(x, y)
            imgs_done += self.global_batch
            if on_step:
                on_step(epoch, it)
            if log_every and (it + 1) % log_every == 0 and self.env.is_rank0:
                if self.device.type == "cuda":
                    torch.cuda.synchronize()
                dt = time.monotonic() - t0
                log.info(
                    "epoch %d step %d/%d loss=%.4f %.1f img/s (world=%d)",
                    epoch, it + 1, steps, loss.item(), imgs_done / dt, self.world_size,
                )
        if self.device.type == "cuda":
            torch.cuda.synchronize()
        dt = time.monotonic() - t0
        return {"epoch": epoch, "steps": steps, "time_s": dt,
                "img_per_s": imgs_done / dt,
                "loss": float(loss.item()) if loss is not None else None}
