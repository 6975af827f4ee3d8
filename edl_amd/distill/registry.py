"""Teacher registry over the coordination store.

Parity: reference edl.discovery.register.ServerRegister (register.py:40-77,
TCP-alive-checked TTL heartbeat) + server_alive.py:19-33. Teachers appear
under /<job>/service/nodes/<service_name>/<endpoint>; their key expires
with the lease when they die."""
import socket
import time

from ..coord.register import Register
from ..coord.tables import ETCD_SERVICE
from ..utils.log import get_logger

log = get_logger("edl.registry")


def is_server_alive(endpoint, timeout=3.0):
    """TCP connect probe (reference discovery/server_alive.py:19-33)."""
    host, port = endpoint.rsplit(":", 1)
    try:
        s = socket.create_connection((host, int(port)), timeout=timeout)
        s.close()
        return True
    except OSError:
        return False


def service_key(client, service_name, endpoint=""):
    base = client.table_key(ETCD_SERVICE, service_name)
    return "%s/%s" % (base, endpoint) if endpoint else base + "/"


class ServerRegister:
    """Registers a teacher endpoint once the server answers TCP, then
    keeps the lease refreshed; deregisters on stop."""

    def __init__(self, client, service_name, endpoint, ttl=10, wait_alive=True,
                 alive_timeout=60.0):
        self._client = client
        self._service = service_name
        self._endpoint = endpoint
        self._ttl = ttl
        self._wait_alive = wait_alive
        self._alive_timeout = alive_timeout
        self._reg = None

    def start(self):
        if self._wait_alive:
            deadline = time.monotonic() + self._alive_timeout
            while not is_server_alive(self._endpoint):
                if time.monotonic() > deadline:
                    raise TimeoutError("server %s never came alive" % self._endpoint)
                time.sleep(0.5)
        self._reg = Register(
            self._client, service_key(self._client, self._service, self._endpoint),
            "1", ttl=self._ttl,
        ).start()
        return self

    @property
    def failed(self):
        return self._reg is not None and self._reg.failed

    def stop(self):
        if self._reg:
            self._reg.stop()


def list_servers(client, service_name):
    pfx = service_key(client, service_name)
    return sorted(k[len(pfx):] for k, _ in client.range(pfx))
