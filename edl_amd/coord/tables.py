"""Key-space table names + TTLs (parity: reference utils/constants.py:15-39).

Layout: /<job_id>/<table>/nodes/<key>
"""

ETCD_POD_RESOURCE = "resource"      # pod JSON under its own id, lease-bound
ETCD_POD_RANK = "rank"              # leader election: server key "0"
ETCD_POD_STATUS = "pod_status"      # per-pod Status
ETCD_JOB_STATUS = "job_status"      # whole-job Status flag
ETCD_TRAIN_STATUS = "train_status"  # trainer-reported TrainStatus
ETCD_CLUSTER = "cluster"            # published Cluster JSON (key "cluster")
ETCD_READER = "reader"              # data reader metadata
ETCD_STATE = "state"                # train State (checkpoint metadata)
ETCD_BARRIER = "barrier"            # barrier arrivals per stage
ETCD_DIST_READER = "dist_reader"    # distributed reader leader endpoint
ETCD_SERVICE = "service"            # distill teacher registry (per service name)
ETCD_SERVICE_CLIENTS = "service_clients"  # distill student registrations
ETCD_SERVICE_ASSIGN = "service_assign"    # balance output: client -> teachers
ETCD_BALANCE = "balance"            # discovery-server self-registry (__balance__)

import os

# Lease TTL (seconds). Reference constants.py:26 fixes 15 s; we keep that
# default but allow env override — failure-detection latency is part of the
# resize-recovery headline metric and tests shrink it.
ETCD_TTL = float(os.environ.get("EDL_LEASE_TTL", "15"))
ETCD_CONN_TIMEOUT = 6  # seconds

LEADER_KEY = "0"       # rank table server name for the leader
CLUSTER_KEY = "cluster"
