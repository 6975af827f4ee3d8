"""Versioned, atomic, async-staged checkpointing.

Parity: the reference's fleet save/load_check_point semantics
(doc/fault_tolerance.md:20-36; train_with_fleet.py:427-431,562-570):
rank 0 saves, incrementing version directories, write-temp-then-rename
atomicity, TrainStatus carries epoch_no and resume is next().

MI355X-native addition (BASELINE north star): the GPU->host copy is staged
through PINNED host buffers on a side HIP stream (hipMemcpyAsync
underneath torch's non_blocking copy), so a per-step checkpoint costs the
training stream almost nothing; a background thread serialises to disk
after the copy event fires."""
import json
import os
import re
import shutil
import threading

import torch

_CKPT_RE = re.compile(r"^checkpoint\.(\d+)$")


class CheckpointManager:
    def __init__(self, path, keep=2, use_pinned=None):
        self.path = path
        self.keep = keep
        self._use_pinned = torch.cuda.is_available() if use_pinned is None else use_pinned
        self._stream = torch.cuda.Stream() if self._use_pinned else None
        self._pinned = {}  # name -> pinned host tensor cache
        self._thread = None
        os.makedirs(path, exist_ok=True)

    # ---- versions ----
    def versions(self):
        out = []
        try:
            for d in os.listdir(self.path):
                m = _CKPT_RE.match(d)
                if m and os.path.isdir(os.path.join(self.path, d)):
                    out.append(int(m.group(1)))
        except FileNotFoundError:
            pass
        return sorted(out)

    def latest_version(self):
        v = self.versions()
        return v[-1] if v else None

    # ---- save ----
    def _stage_to_host(self, state_dict):
        """Async copy of every tensor to pinned host memory on the side
        stream. Returns (host_state_dict, event-or-None)."""
        if not self._use_pinned:
            return {k: v.detach().cpu().clone() if torch.is_tensor(v) else v
                    for k, v in state_dict.items()}, None
        ev = torch.cuda.Event()
        host = {}
        with torch.cuda.stream(self._stream):
            # the side stream must see finished values from the compute stream
            self._stream.wait_stream(torch.cuda.current_stream())
            for k, v in state_dict.items():
                if torch.is_tensor(v) and v.is_cuda:
                    buf = self._pinned.get(k)
                    if buf is None or buf.shape != v.shape or buf.dtype != v.dtype:
                        buf = torch.empty_like(v, device="cpu", pin_memory=True)
                        self._pinned[k] = buf
                    buf.copy_(v.detach(), non_blocking=True)
                    host[k] = buf
                elif torch.is_tensor(v):
                    host[k] = v.detach().clone()
                else:
                    host[k] = v
            ev.record(self._stream)
        return host, ev

    def save(self, model_state, train_status, optimizer_state=None, version=None,
             blocking=False):
        """Write checkpoint.<version> atomically. train_status: dict with at
        least epoch_no (and anything else resumable)."""
        if version is None:
            latest = self.latest_version()
            version = 0 if latest is None else latest + 1
        host_model, ev = self._stage_to_host(model_state)
        host_opt = None
        if optimizer_state is not None:
            flat = _flatten_opt(optimizer_state)
            host_flat, ev2 = self._stage_to_host(flat)
            host_opt = (optimizer_state, host_flat)
            ev = ev2 or ev
        if ev is not None:
            # the NEXT optimizer step (compute stream) must not overwrite
            # params/momentum while the D2H copies are still in flight —
            # without this the non-blocking save can snapshot a torn state
            torch.cuda.current_stream().wait_event(ev)

        def _write():
            if ev is not None:
                ev.synchronize()
            tmp = os.path.join(self.path, ".tmp.checkpoint.%d" % version)
            final = os.path.join(self.path, "checkpoint.%d" % version)
            shutil.rmtree(tmp, ignore_errors=True)
            os.makedirs(tmp)
            # clone pinned buffers so the next save can reuse them
            torch.save({k: (v.clone() if torch.is_tensor(v) else v)
                        for k, v in host_model.items()},
                       os.path.join(tmp, "model.pt"))
            if host_opt is not None:
                skeleton, flat_host = host_opt
                torch.save(_unflatten_opt(skeleton, {
                    k: (v.clone() if torch.is_tensor(v) else v)
                    for k, v in flat_host.items()
                }), os.path.join(tmp, "optimizer.pt"))
            with open(os.path.join(tmp, "train_status.json"), "w") as f:
                json.dump(train_status, f)
            # layout manifest (docs/fault_tolerance.md "paddle_edl
            # checkpoint-format mapping"): lets tooling verify integrity
            # without unpickling tensors
            files = ["model.pt", "train_status.json"]
            if host_opt is not None:
                files.insert(1, "optimizer.pt")
            with open(os.path.join(tmp, "checkpoint_meta.json"), "w") as f:
                json.dump({"format": "edl_amd.v1", "version": version,
                           "files": files, "saved_by_rank": 0}, f)
            shutil.rmtree(final, ignore_errors=True)
            os.rename(tmp, final)
            self._gc()

        self.wait()  # one in-flight save at a time (pinned buffers are reused)
        if blocking:
            _write()
        else:
            self._thread = threading.Thread(target=_write, daemon=True, name="ckpt-write")
            self._thread.start()
        return version

    def wait(self):
        if self._thread is not None:
            self._thread.join()
            self._thread = None

    def _gc(self):
        vs = self.versions()
        for v in vs[: max(0, len(vs) - self.keep)]:
            shutil.rmtree(os.path.join(self.path, "checkpoint.%d" % v), ignore_errors=True)

    # ---- load ----
    def load(self, map_location="cpu"):
        """-> (model_state, optimizer_state_or_None, train_status) of the
        newest COMPLETE checkpoint, or None."""
        for v in reversed(self.versions()):
            d = os.path.join(self.path, "checkpoint.%d" % v)
            try:
                model = torch.load(os.path.join(d, "model.pt"),
                                   map_location=map_location, weights_only=True)
                with open(os.path.join(d, "train_status.json")) as f:
                    ts = json.load(f)
                opt_path = os.path.join(d, "optimizer.pt")
                opt = None
                if os.path.exists(opt_path):
                    opt = torch.load(opt_path, map_location=map_location, weights_only=True)
                ts["_version"] = v
                return model, opt, ts
            except Exception:  # noqa: BLE001 - torn/corrupt dir (any unpickling
                continue       # or IO error): fall back to the previous version
        return None


def _flatten_opt(opt_state):
    """Flatten an optimizer state_dict's tensors into a {path: tensor} map."""
    flat = {}

    def rec(prefix, obj):
        if torch.is_tensor(obj):
            flat[prefix] = obj
        elif isinstance(obj, dict):
            for k, v in obj.items():
                rec("%s/%s" % (prefix, k), v)
        elif isinstance(obj, (list, tuple)):
            for i, v in enumerate(obj):
                rec("%s/%d" % (prefix, i), v)

    rec("", opt_state)
    return flat


def _unflatten_opt(skeleton, flat):
    def rec(prefix, obj):
        if torch.is_tensor(obj):
            return flat[prefix]
        if isinstance(obj, dict):
            return {k: rec("%s/%s" % (prefix, k), v) for k, v in obj.items()}
        if isinstance(obj, list):
            return [rec("%s/%d" % (prefix, i), v) for i, v in enumerate(obj)]
        if isinstance(obj, tuple):
            return tuple(rec("%s/%d" % (prefix, i), v) for i, v in enumerate(obj))
        return obj

    return rec("", skeleton)
