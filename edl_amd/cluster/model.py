"""Pod / Trainer / Cluster model.

Parity: reference utils/pod.py:26-182, utils/trainer.py:19-55,
utils/cluster.py:29-175. A Pod is one per-node agent owning N trainer
process slots (one per GPU); a Cluster is the rank-ordered pod list with a
`stage` uuid bumped on every membership change (cluster.py:137-138) — the
stage is what barriers and watchers key on."""
import json
import uuid

from ..coord.tables import CLUSTER_KEY, ETCD_CLUSTER


def new_id():
    return uuid.uuid4().hex[:12]


class Trainer:
    """One trainer process slot: a GPU slice + ranks."""

    def __init__(self, endpoint="", gpus=None, rank_in_pod=0, global_rank=-1):
        self.endpoint = endpoint
        self.gpus = gpus or []
        self.rank_in_pod = rank_in_pod
        self.global_rank = global_rank

    def to_dict(self):
        return dict(self.__dict__)

    @classmethod
    def from_dict(cls, d):
        t = cls()
        t.__dict__.update(d)
        return t

    def __eq__(self, o):
        return isinstance(o, Trainer) and self.__dict__ == o.__dict__

    def __repr__(self):
        return "Trainer(rank=%s, gpus=%s, ep=%s)" % (self.global_rank, self.gpus, self.endpoint)


class Pod:
    def __init__(self, pod_id=None, addr="127.0.0.1", port=0, gpus=None, trainers=None):
        self.pod_id = pod_id or new_id()
        self.addr = addr
        self.port = port  # agent control port (barrier etc. go via the store)
        self.gpus = gpus or []
        self.trainers = trainers or []
        self.rank = -1

    @classmethod
    def from_env(cls, job_env):
        """Build the local pod: one Trainer per GPU (or per proc slot on CPU).
        Reference Pod.from_env pod.py:72-103."""
        pod = cls(addr=job_env.pod_ip, gpus=list(job_env.gpus))
        n = max(1, len(job_env.gpus)) if job_env.nproc_per_node is None else job_env.nproc_per_node
        ports = job_env.trainer_ports[:n] if job_env.trainer_ports else [0] * n
        for i in range(n):
            gpus = [job_env.gpus[i]] if i < len(job_env.gpus) else []
            pod.trainers.append(
                Trainer(endpoint="%s:%d" % (pod.addr, ports[i]), gpus=gpus, rank_in_pod=i)
            )
        return pod

    def set_rank(self, rank, rank_offset):
        """Assign pod rank + global trainer ranks (reference pod.py:145-150)."""
        self.rank = rank
        for i, t in enumerate(self.trainers):
            t.global_rank = rank_offset + i

    def to_dict(self):
        d = dict(self.__dict__)
        d["trainers"] = [t.to_dict() for t in self.trainers]
        return d

    @classmethod
    def from_dict(cls, d):
        p = cls()
        p.__dict__.update({k: v for k, v in d.items() if k != "trainers"})
        p.trainers = [Trainer.from_dict(t) for t in d.get("trainers", [])]
        return p

    def to_json(self):
        return json.dumps(self.to_dict())

    @classmethod
    def from_json(cls, s):
        return cls.from_dict(json.loads(s))

    def __eq__(self, o):
        return isinstance(o, Pod) and self.to_dict() == o.to_dict()

    def __repr__(self):
        return "Pod(%s rank=%s %dtrainers)" % (self.pod_id, self.rank, len(self.trainers))


class Cluster:
    """Rank-ordered pod list + stage uuid (reference utils/cluster.py:29-175)."""

    def __init__(self, pods=None, stage=None, job_stage=0):
        self.pods = pods or []
        self.stage = stage or new_id()
        self.job_stage = job_stage  # monotonically increasing resize count

    def new_stage(self):
        self.stage = new_id()
        self.job_stage += 1

    def assign_ranks(self):
        offset = 0
        for rank, pod in enumerate(self.pods):
            pod.set_rank(rank, offset)
            offset += len(pod.trainers)

    def pod_ids(self):
        return [p.pod_id for p in self.pods]

    def get_pod(self, pod_id):
        for p in self.pods:
            if p.pod_id == pod_id:
                return p
        return None

    def world_size(self):
        return sum(len(p.trainers) for p in self.pods)

    def trainer_endpoints(self):
        return [t.endpoint for p in self.pods for t in p.trainers]

    def to_json(self):
        return json.dumps(
            {
                "stage": self.stage,
                "job_stage": self.job_stage,
                "pods": [p.to_dict() for p in self.pods],
            }
        )

    @classmethod
    def from_json(cls, s):
        d = json.loads(s)
        return cls(
            pods=[Pod.from_dict(p) for p in d["pods"]],
            stage=d["stage"],
            job_stage=d.get("job_stage", 0),
        )

    def same_members(self, other):
        """Membership (rank-ordered pod-id list) equality — the watcher's
        change test (reference cluster_watcher.py:71-95)."""
        return other is not None and self.pod_ids() == other.pod_ids()

    def __eq__(self, o):
        return isinstance(o, Cluster) and self.to_json() == o.to_json()

    def __repr__(self):
        return "Cluster(stage=%s pods=%s)" % (self.stage, self.pod_ids())


def save_cluster(client, cluster, leader_guard=None):
    """Publish cluster JSON; when leader_guard=(key, val) is given, write via
    a still-being-leader transaction (reference cluster_generator.py:224-250)."""
    key = client.table_key(ETCD_CLUSTER, CLUSTER_KEY)
    if leader_guard is None:
        client.put(key, cluster.to_json())
        return True
    gk, gv = leader_guard
    return client.txn_if(gk, gv, puts=[(key, cluster.to_json())])


def load_cluster(client):
    s = client.get(client.table_key(ETCD_CLUSTER, CLUSTER_KEY))
    return Cluster.from_json(s) if s else None
