"""Network helpers (parity: reference utils/network_utils.py:21-54)."""
import socket


def find_free_port(n=1):
    """Return n distinct free TCP ports (briefly bound then released)."""
    socks, ports = [], []
    try:
        for _ in range(n):
            s = socket.socket(socket.AF_INET, socket.SOCK_STREAM)
            s.setsockopt(socket.SOL_SOCKET, socket.SO_REUSEADDR, 1)
            s.bind(("127.0.0.1", 0))
            socks.append(s)
            ports.append(s.getsockname()[1])
    finally:
        for s in socks:
            s.close()
    return ports[0] if n == 1 else ports


def local_ip():
    """Best-effort local IP; single-node jobs use 127.0.0.1 anyway."""
    try:
        s = socket.socket(socket.AF_INET, socket.SOCK_DGRAM)
        try:
            s.connect(("10.255.255.255", 1))
            return s.getsockname()[0]
        finally:
            s.close()
    except OSError:
        return "127.0.0.1"
