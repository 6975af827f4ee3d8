"""Resource pod registry (parity: reference utils/resource_pods.py:25-71).

Each agent registers its Pod JSON under resource/nodes/<pod_id> bound to a
TTL lease; the leader's generator reconciles the cluster from this table."""
import time

from ..coord.register import Register
from ..coord.tables import ETCD_POD_RESOURCE
from .model import Pod


class ResourceRegister(Register):
    def __init__(self, client, pod):
        super().__init__(client, client.table_key(ETCD_POD_RESOURCE, pod.pod_id), pod.to_json())


def load_resource_pods(client):
    pfx = client.table_key(ETCD_POD_RESOURCE)
    return {k[len(pfx):]: Pod.from_json(v) for k, v in client.range(pfx)}


def wait_resource_empty(client, my_pod_id, timeout=300):
    """Leader-at-exit waits for followers to deregister
    (reference resource_pods.wait_resource, 58-68)."""
    deadline = time.monotonic() + timeout
    while time.monotonic() < deadline:
        pods = load_resource_pods(client)
        pods.pop(my_pod_id, None)
        if not pods:
            return True
        time.sleep(1.0)
    return False
