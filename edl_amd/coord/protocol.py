"""Wire protocol for the coordination store: 4-byte length + JSON.

The reference used an external etcd (discovery/etcd_client.py) plus a
hand-rolled epoll TCP server with length-prefixed JSON frames for the redis
balance tier (distill/redis/balance_server.py:38-211, HEAD_FORMAT "!4si").
We keep the length-prefixed-JSON idea but make ONE in-repo store the single
coordination substrate — there is no etcd and no redis in the MI355X image.
"""
import json
import socket
import struct

_HEAD = struct.Struct("!I")
MAX_FRAME = 64 * 1024 * 1024


class ProtocolError(Exception):
    pass


def send_msg(sock, obj):
    body = json.dumps(obj, separators=(",", ":")).encode("utf-8")
    sock.sendall(_HEAD.pack(len(body)) + body)


def _recv_exact(sock, n):
    buf = bytearray()
    while len(buf) < n:
        chunk = sock.recv(n - len(buf))
        if not chunk:
            raise ConnectionError("peer closed")
        buf.extend(chunk)
    return bytes(buf)


def recv_msg(sock):
    (n,) = _HEAD.unpack(_recv_exact(sock, 4))
    if n > MAX_FRAME:
        raise ProtocolError("frame too large: %d" % n)
    return json.loads(_recv_exact(sock, n).decode("utf-8"))


def connect(endpoint, timeout=6.0):
    """endpoint: 'host:port' string."""
    host, port = endpoint.rsplit(":", 1)
    s = socket.create_connection((host, int(port)), timeout=timeout)
    s.setsockopt(socket.IPPROTO_TCP, socket.TCP_NODELAY, 1)
    return s
