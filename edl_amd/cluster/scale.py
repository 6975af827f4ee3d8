"""Scale-in / scale-out API (parity: reference PodServer.ScaleOut/ScaleIn
gRPC stubs, utils/pod_server.py:47-67 + protos/pod_server.proto:40-45 —
leader-checked entry points for an external scheduler/controller).

Store-backed: the controller writes a desired-world hint; the leader's
generator reads it and the k8s controller (edl_amd.k8s.controller) or a
human adds/removes agent processes to satisfy it. Scale-in of a specific
pod marks it for removal — the generator drops it on the next pass."""
import json

SCALE_KEY = "scale"


def request_scale(client, desired_nodes=None, remove_pods=()):
    """Controller-side: publish the scaling intent."""
    client.put(
        client.table_key(SCALE_KEY, "request"),
        json.dumps({"desired_nodes": desired_nodes, "remove_pods": list(remove_pods)}),
    )


def read_scale_request(client):
    v = client.get(client.table_key(SCALE_KEY, "request"))
    if not v:
        return None
    try:
        return json.loads(v)
    except ValueError:
        return None


def clear_scale_request(client):
    client.delete(client.table_key(SCALE_KEY, "request"))
