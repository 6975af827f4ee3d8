"""Pod/Trainer/Cluster model tests (mirror reference test_pod.py/test_cluster.py)."""
import json

from edl_amd.cluster.model import Cluster, Pod, Trainer, load_cluster, save_cluster
from edl_amd.train.env import JobEnv


def make_pod(pod_id, n_trainers=2):
    p = Pod(pod_id=pod_id, addr="127.0.0.1")
    for i in range(n_trainers):
        p.trainers.append(Trainer(endpoint="127.0.0.1:%d" % (9000 + i), gpus=[str(i)],
                                  rank_in_pod=i))
    return p


def test_pod_json_roundtrip():
    p = make_pod("p1")
    p.set_rank(3, 6)
    q = Pod.from_json(p.to_json())
    assert q == p
    assert q.trainers[1].global_rank == 7


def test_pod_from_env():
    env = JobEnv(
        {"nproc_per_node": 2},
        env={"CUDA_VISIBLE_DEVICES": "", "PADDLE_TRAINER_PORTS": "7001,7002"},
    )
    pod = Pod.from_env(env)
    assert len(pod.trainers) == 2
    assert pod.trainers[0].endpoint.endswith(":7001")


def test_cluster_rank_assignment_and_stage():
    c = Cluster(pods=[make_pod("a"), make_pod("b"), make_pod("c")])
    c.assign_ranks()
    assert c.world_size() == 6
    ranks = [t.global_rank for p in c.pods for t in p.trainers]
    assert ranks == list(range(6))
    s0 = c.stage
    c.new_stage()
    assert c.stage != s0
    assert c.job_stage == 1


def test_cluster_json_and_members():
    c = Cluster(pods=[make_pod("a"), make_pod("b")])
    c.assign_ranks()
    d = Cluster.from_json(c.to_json())
    assert d == c
    assert d.same_members(c)
    e = Cluster(pods=[make_pod("a")])
    assert not e.same_members(c)
    # stage survives serialization
    assert json.loads(c.to_json())["stage"] == c.stage


def test_save_load_cluster_guarded(coord_client):
    c = Cluster(pods=[make_pod("a")])
    c.assign_ranks()
    # unguarded write
    assert save_cluster(coord_client, c)
    got = load_cluster(coord_client)
    assert got == c
    # guarded write only applies when the guard matches
    coord_client.put("/test_job/rank/nodes/0", "a")
    c.new_stage()
    assert save_cluster(coord_client, c, leader_guard=("/test_job/rank/nodes/0", "a"))
    assert load_cluster(coord_client).stage == c.stage
    c2 = Cluster(pods=[make_pod("z")])
    assert not save_cluster(coord_client, c2, leader_guard=("/test_job/rank/nodes/0", "zzz"))
    assert load_cluster(coord_client).stage == c.stage


def test_trainer_env_contract():
    """The full env contract a spawned trainer sees (reference
    utils/train_process.py:46-73 PADDLE_* names + torch env:// vars), and
    that TrainerEnv reads it back to the same identity."""
    from edl_amd.train.env import TrainerEnv, trainer_env_dict

    job = JobEnv({"job_id": "jctr", "store_endpoints": "127.0.0.1:2379",
                  "nproc_per_node": 2},
                 env={"CUDA_VISIBLE_DEVICES": ""})
    c = Cluster(pods=[make_pod("a"), make_pod("b")])
    c.assign_ranks()
    pod = c.pods[1]
    tr = pod.trainers[1]  # global rank 3
    e = trainer_env_dict(job, c, pod, tr)

    assert e["PADDLE_JOB_ID"] == "jctr"
    assert e["PADDLE_TRAINER_ID"] == e["RANK"] == "3"
    assert e["PADDLE_TRAINER_RANK_IN_POD"] == e["LOCAL_RANK"] == "1"
    assert e["PADDLE_TRAINERS_NUM"] == e["WORLD_SIZE"] == "4"
    assert e["PADDLE_TRAINER_ENDPOINTS"].count(",") == 3
    assert e["PADDLE_CURRENT_ENDPOINT"] == tr.endpoint
    assert e["FLAGS_selected_gpus"] == "1"
    assert e["PADDLE_ETCD_ENDPOINTS"] == "127.0.0.1:2379"
    assert e["EDL_CLUSTER_STAGE"] == c.stage
    assert e["EDL_POD_ID"] == "b"
    # MASTER_* = global rank 0's endpoint
    host, port = c.trainer_endpoints()[0].rsplit(":", 1)
    assert e["MASTER_PORT"] == port

    tenv = TrainerEnv(env=e)
    assert tenv.global_rank == 3 and tenv.rank_in_pod == 1
    assert tenv.world_size == 4
    assert tenv.current_endpoint == tr.endpoint
    assert tenv.cluster_stage == c.stage
    assert tenv.master_port == int(port)


def test_master_addr_multinode_vs_local():
    """MASTER_ADDR selection: a remote rank-0 endpoint is kept verbatim
    (multi-node); an endpoint on THIS host's IP rendezvous on loopback
    (container hostnames may not resolve — single-node default)."""
    from edl_amd.train.env import trainer_env_dict
    from edl_amd.utils.net import local_ip

    job = JobEnv({"job_id": "j", "nproc_per_node": 1},
                 env={"CUDA_VISIBLE_DEVICES": ""})

    def cluster_with(host):
        p = Pod(pod_id="a", addr=host)
        p.trainers.append(Trainer(endpoint="%s:9100" % host, gpus=["0"],
                                  rank_in_pod=0))
        c = Cluster(pods=[p])
        c.assign_ranks()
        return c

    c = cluster_with("10.9.8.7")  # not this host
    e = trainer_env_dict(job, c, c.pods[0], c.pods[0].trainers[0])
    assert e["MASTER_ADDR"] == "10.9.8.7"

    c = cluster_with(local_ip())
    e = trainer_env_dict(job, c, c.pods[0], c.pods[0].trainers[0])
    assert e["MASTER_ADDR"] in ("127.0.0.1", "localhost")
