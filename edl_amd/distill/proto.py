"""Teacher-serving wire protocol: length-prefixed JSON header + raw
tensor payloads.

Replaces the reference's Paddle Serving RPC (distill_worker.py:197-321
PaddlePredictServer with feed-shape negotiation). Tensors travel as raw
bytes (dtype/shape in the header) — no pickling, no base64."""
import json
import socket
import struct

import numpy as np

_HEAD = struct.Struct("!I")
MAX_FRAME = 1 << 30


def _send(sock, header, payloads):
    body = json.dumps(header, separators=(",", ":")).encode()
    sock.sendall(_HEAD.pack(len(body)))
    sock.sendall(body)
    for p in payloads:
        sock.sendall(_HEAD.pack(len(p)))
        sock.sendall(p)


def _recv_exact(sock, n):
    # bytearray + recv_into: no per-chunk concatenation, and the buffer
    # stays WRITABLE so np.frombuffer views are writable (torch refuses
    # read-only arrays without a copy)
    buf = bytearray(n)
    view = memoryview(buf)
    got = 0
    while got < n:
        r = sock.recv_into(view[got:], min(1 << 20, n - got))
        if r == 0:
            raise ConnectionError("peer closed")
        got += r
    return buf


def _recv(sock):
    (n,) = _HEAD.unpack(_recv_exact(sock, 4))
    if n > MAX_FRAME:
        raise ValueError("frame too large")
    header = json.loads(bytes(_recv_exact(sock, n)).decode())
    payloads = []
    for _ in range(header.get("n_payloads", 0)):
        (m,) = _HEAD.unpack(_recv_exact(sock, 4))
        payloads.append(_recv_exact(sock, m))
    return header, payloads


def send_arrays(sock, op, arrays, meta=None):
    header = {
        "op": op,
        "meta": meta or {},
        "n_payloads": len(arrays),
        "tensors": [{"dtype": str(a.dtype), "shape": list(a.shape)} for a in arrays],
    }
    _send(sock, header, [np.ascontiguousarray(a).tobytes() for a in arrays])


def recv_arrays(sock):
    header, payloads = _recv(sock)
    arrays = [
        np.frombuffer(p, dtype=t["dtype"]).reshape(t["shape"])
        for p, t in zip(payloads, header.get("tensors", []))
    ]
    return header, arrays


def connect(endpoint, timeout=30.0):
    host, port = endpoint.rsplit(":", 1)
    s = socket.create_connection((host, int(port)), timeout=timeout)
    s.setsockopt(socket.IPPROTO_TCP, socket.TCP_NODELAY, 1)
    return s
