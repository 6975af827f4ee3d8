"""Pod/job/train status tables (parity: reference utils/status.py:22-110,
utils/train_status.py:20-45)."""
import json

from ..coord.tables import ETCD_JOB_STATUS, ETCD_POD_STATUS, ETCD_TRAIN_STATUS


class Status:
    INITIAL = "INITIAL"
    RUNNING = "RUNNING"
    PENDING = "PENDING"
    SUCCEED = "SUCCEED"
    FAILED = "FAILED"


class TrainStatus:
    """Reported by trainers so the generator can veto meaningless scaling
    near job end (reference cluster_generator.py:206-215)."""

    INITIAL = "INITIAL"
    RUNNING = "RUNNING"
    NEARTHEEND = "NEARTHEEND"
    SUCCEED = "SUCCEED"
    FAILED = "FAILED"


def save_pod_status(client, pod_id, status):
    client.put(client.table_key(ETCD_POD_STATUS, pod_id), status)


def load_pod_status(client, pod_id):
    return client.get(client.table_key(ETCD_POD_STATUS, pod_id))


def load_pods_status(client):
    pfx = client.table_key(ETCD_POD_STATUS)
    return {k[len(pfx):]: v for k, v in client.range(pfx)}


def save_job_status(client, status):
    client.put(client.table_key(ETCD_JOB_STATUS, "job_status"), status)


def load_job_status(client):
    return client.get(client.table_key(ETCD_JOB_STATUS, "job_status"))


def save_train_status(client, pod_id, status, meta=None):
    v = json.dumps({"status": status, "meta": meta or {}})
    client.put(client.table_key(ETCD_TRAIN_STATUS, pod_id), v)


def load_train_statuses(client):
    pfx = client.table_key(ETCD_TRAIN_STATUS)
    out = {}
    for k, v in client.range(pfx):
        try:
            out[k[len(pfx):]] = json.loads(v)["status"]
        except (ValueError, KeyError):
            pass
    return out
