"""Stage barrier over the coordination store.

Parity: reference PodServer.Barrier gRPC (utils/pod_server.py:69-116 +
pod_server_client.py:25-60): arrivals are grouped by cluster stage and the
barrier completes when the arrival set equals the cluster's pod-id set.

MI355X-native simplification: instead of a per-pod gRPC server whose only
live method is the leader's barrier, arrivals are lease-bound keys under
barrier/nodes/<stage>/<pod_id>; every pod independently observes
completion. Stale arrivals expire with their lease."""
import time

from ..coord.tables import ETCD_BARRIER
from ..utils.errors import EdlBarrierError, EdlPodIDNotExistError
from .model import load_cluster

BARRIER_ARRIVAL_TTL = 60


def barrier(client, pod_id, timeout=60.0, expect_stage=None, poll=0.2, allow_join=False):
    """Arrive at the barrier for the current published cluster stage and
    wait for everyone. Returns the Cluster whose stage completed.

    If the published cluster changes stage while waiting, we re-arrive at
    the new stage (the reference client retries the RPC the same way,
    pod_server_client.py:37-60)."""
    deadline = time.monotonic() + timeout
    lease = client.grant(BARRIER_ARRIVAL_TTL)
    arrived_stage = None
    last_keepalive = time.monotonic()
    # NOTE: the lease is deliberately NOT revoked on return — peers may
    # still be polling for our arrival key; it expires by TTL instead.
    while time.monotonic() < deadline:
        if time.monotonic() - last_keepalive > BARRIER_ARRIVAL_TTL / 3:
            client.keepalive(lease)
            last_keepalive = time.monotonic()
        cluster = load_cluster(client)
        if cluster is None or (expect_stage and cluster.stage != expect_stage):
            time.sleep(poll)
            continue
        if pod_id not in cluster.pod_ids():
            if allow_join:
                # initial barrier of a joining pod: the leader's generator
                # will append us (INITIAL + registered) with a new stage —
                # keep waiting (reference: launcher re-barriers, the 3 s
                # generator loop appends INITIAL pods).
                time.sleep(poll)
                continue
            # re-barrier of a running pod: scale-in — we are out
            raise EdlPodIDNotExistError(
                "pod %s not in cluster stage %s" % (pod_id, cluster.stage)
            )
        if arrived_stage != cluster.stage:
            client.put(
                client.table_key(ETCD_BARRIER, "%s/%s" % (cluster.stage, pod_id)),
                "1",
                lease,
            )
            arrived_stage = cluster.stage
        pfx = client.table_key(ETCD_BARRIER, cluster.stage + "/")
        arrived = {k[len(pfx):] for k, _ in client.range(pfx)}
        if arrived >= set(cluster.pod_ids()):
            return cluster
        time.sleep(poll)
    raise EdlBarrierError("barrier timeout after %.0fs (pod %s)" % (timeout, pod_id))
