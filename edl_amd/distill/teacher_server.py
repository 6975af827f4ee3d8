"""Teacher inference server — our Paddle Serving replacement.

Serves a model's forward (default: ResNeXt101_32x16d_wsl, the reference's
distill teacher, README.md:51-75) over the proto.py TCP protocol. On GPU
the forward runs eval-mode bf16 channels_last — BN through the fused CDNA4
kernels, 1x1 convs on the MFMA GEMM. Registers itself in the coordination
store under /<job>/service/nodes/<name>/<endpoint> with a TTL lease
(reference edl.discovery.register, register.py:40-77).

    python -m edl_amd.distill.teacher_server --port 9292 \
        --model resnext101_32x16d_wsl [--service_name S --store host:port]
"""
import argparse
import socket
import socketserver
import threading

import numpy as np
import torch

from ..models import build_model
from ..utils.log import get_logger
from . import proto
from .registry import ServerRegister

log = get_logger("edl.teacher")


class TeacherService:
    def __init__(self, model_name="resnext101_32x16d_wsl", num_classes=1000,
                 device=None, model=None):
        self.device = device or (
            torch.device("cuda", 0) if torch.cuda.is_available() else torch.device("cpu")
        )
        self.model = model or build_model(model_name, num_classes=num_classes)
        self.model.eval().to(self.device)
        for p in self.model.parameters():
            # frozen: lets the conv layer cache its repacked weights
            # forever instead of per (student) weight epoch
            p.requires_grad_(False)
        self.use_bf16 = self.device.type == "cuda"
        if self.use_bf16:
            self.model.to(memory_format=torch.channels_last)
        self._lock = threading.Lock()  # one forward at a time per GPU

    @torch.no_grad()
    def predict(self, images):
        """images: numpy [B, 3, H, W] float32 -> logits numpy [B, classes]."""
        arr = np.ascontiguousarray(images)
        if not arr.flags.writeable:  # proto recv buffers are read-only views
            arr = arr.copy()
        x = torch.from_numpy(arr).to(self.device)
        with self._lock:
            if self.use_bf16:
                # channels_last only applies to 4-D image batches; text
                # teachers feed integer token ids [B, T]
                if x.dim() == 4:
                    x = x.contiguous(memory_format=torch.channels_last)
                with torch.autocast("cuda", torch.bfloat16):
                    y = self.model(x)
            else:
                y = self.model(x)
        return y.float().cpu().numpy()


class _Handler(socketserver.BaseRequestHandler):
    def handle(self):
        self.request.setsockopt(socket.IPPROTO_TCP, socket.TCP_NODELAY, 1)
        svc = self.server.service
        try:
            while True:
                header, arrays = proto.recv_arrays(self.request)
                op = header.get("op")
                if op == "predict":
                    try:
                        out = svc.predict(arrays[0])
                        proto.send_arrays(self.request, "ok", [out])
                    except Exception as e:  # noqa: BLE001
                        proto.send_arrays(self.request, "err", [],
                                          meta={"err": str(e)})
                elif op == "ping":
                    proto.send_arrays(self.request, "ok", [])
                else:
                    proto.send_arrays(self.request, "err", [],
                                      meta={"err": "bad op %r" % op})
        except (ConnectionError, OSError):
            pass


class _TCP(socketserver.ThreadingTCPServer):
    allow_reuse_address = True
    daemon_threads = True


class TeacherServer:
    def __init__(self, service, host="0.0.0.0", port=0):
        self._srv = _TCP((host, port), _Handler)
        self._srv.service = service
        self.port = self._srv.server_address[1]
        self._thread = None

    def start(self):
        self._thread = threading.Thread(target=self._srv.serve_forever, daemon=True,
                                        name="teacher-srv")
        self._thread.start()
        return self

    def stop(self):
        self._srv.shutdown()
        self._srv.server_close()


class TeacherClient:
    """Client used by the distill predict workers."""

    def __init__(self, endpoint, timeout=60.0):
        self.endpoint = endpoint
        self._sock = proto.connect(endpoint, timeout)

    def predict(self, images):
        proto.send_arrays(self._sock, "predict", [images])
        header, arrays = proto.recv_arrays(self._sock)
        if header.get("op") != "ok":
            raise RuntimeError("teacher error: %s" % header.get("meta", {}).get("err"))
        return arrays[0]

    def close(self):
        try:
            self._sock.close()
        except OSError:
            pass


def main(argv=None):
    ap = argparse.ArgumentParser("edl_amd teacher server")
    ap.add_argument("--model", default="resnext101_32x16d_wsl")
    ap.add_argument("--num_classes", type=int, default=1000)
    ap.add_argument("--host", default="0.0.0.0")
    ap.add_argument("--port", type=int, default=9292)
    ap.add_argument("--service_name", default=None)
    ap.add_argument("--store_endpoints", default=None)
    ap.add_argument("--job_id", default="distill")
    ap.add_argument("--advertise_ip", default=None)
    args = ap.parse_args(argv)

    svc = TeacherService(args.model, args.num_classes)
    srv = TeacherServer(svc, args.host, args.port).start()
    log.info("teacher %s serving on port %d (device %s)", args.model, srv.port,
             svc.device)
    reg = None
    if args.service_name and args.store_endpoints:
        from ..coord.client import CoordClient
        from ..utils.net import local_ip

        client = CoordClient(args.store_endpoints, args.job_id)
        ep = "%s:%d" % (args.advertise_ip or local_ip(), srv.port)
        reg = ServerRegister(client, args.service_name, ep).start()
        log.info("registered %s under service %s", ep, args.service_name)
    try:
        threading.Event().wait()
    except KeyboardInterrupt:
        if reg:
            reg.stop()
        srv.stop()


if __name__ == "__main__":
    main()
