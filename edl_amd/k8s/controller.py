"""TrainingJob controller — the reference's out-of-repo Go autoscaler,
in-repo (reference k8s/edl_controller.yaml runs `edl -max_load_desired
0.9`; the in-repo glue was k8s/k8s_tools.py). Reconciles the number of
agent pods of each TrainingJob between min/max replicas, honoring
scale requests from the coordination store (cluster/scale.py).

The cluster API is pluggable: KubectlClient shells out to kubectl (real
clusters); tests inject a FakeClusterClient. Run:

    python -m edl_amd.k8s.controller --namespace default --period 10
"""
import argparse
import json
import subprocess
import time

from ..utils.log import get_logger

log = get_logger("edl.k8s")


class ClusterClient:
    """What the controller needs from the cluster."""

    def list_training_jobs(self):
        """-> [{name, min_instance, max_instance, desired, running}]"""
        raise NotImplementedError

    def scale_job(self, name, replicas):
        raise NotImplementedError


class KubectlClient(ClusterClient):
    def __init__(self, namespace="default"):
        self.namespace = namespace

    def _run(self, *args):
        return subprocess.run(["kubectl", "-n", self.namespace] + list(args),
                              capture_output=True, text=True, check=True).stdout

    def list_training_jobs(self):
        out = self._run("get", "trainingjobs", "-o", "json")
        jobs = []
        for item in json.loads(out).get("items", []):
            spec = item.get("spec", {})
            trainer = spec.get("trainer", {})
            jobs.append({
                "name": item["metadata"]["name"],
                "min_instance": trainer.get("minInstance", 1),
                "max_instance": trainer.get("maxInstance", 1),
                "desired": trainer.get("desired"),
                "running": item.get("status", {}).get("replicas", 0),
            })
        return jobs

    def scale_job(self, name, replicas):
        # agents run as a StatefulSet named <job>-trainer
        self._run("scale", "statefulset", "%s-trainer" % name,
                  "--replicas", str(replicas))


def reconcile(client, free_slots=None, total_slots=None,
              max_load_desired=0.9):
    """One reconcile pass. Policy (reference doc/usage.md autoscaler,
    edl_controller.yaml `-max_load_desired 0.9`): give every job its min;
    distribute remaining capacity toward max, fair-share; when the
    cluster capacity is known (total_slots), allocate at most
    floor(total_slots * max_load_desired) so the cluster keeps headroom
    for non-training workloads. free_slots/total_slots=None means
    capacity-unconstrained."""
    jobs = client.list_training_jobs()
    if total_slots is not None:
        free_slots = int(total_slots * max_load_desired)
    actions = []
    want = {}
    for j in jobs:
        desired = j["desired"] if j.get("desired") else j["max_instance"]
        want[j["name"]] = max(j["min_instance"], min(desired, j["max_instance"]))
    if free_slots is not None:
        total_min = sum(j["min_instance"] for j in jobs)
        budget = max(free_slots, total_min)
        # start from mins, hand out the rest round-robin up to max
        want = {j["name"]: j["min_instance"] for j in jobs}
        budget -= total_min
        progress = True
        while budget > 0 and progress:
            progress = False
            for j in jobs:
                cap = min(j["max_instance"],
                          (j["desired"] or j["max_instance"]))
                if budget > 0 and want[j["name"]] < cap:
                    want[j["name"]] += 1
                    budget -= 1
                    progress = True
    for j in jobs:
        if want[j["name"]] != j["running"]:
            client.scale_job(j["name"], want[j["name"]])
            actions.append((j["name"], j["running"], want[j["name"]]))
            log.info("scale %s: %d -> %d", j["name"], j["running"], want[j["name"]])
    return actions


def main(argv=None):
    ap = argparse.ArgumentParser("edl_amd k8s controller")
    ap.add_argument("--namespace", default="default")
    ap.add_argument("--period", type=float, default=10.0)
    ap.add_argument("--free_slots", type=int, default=None)
    ap.add_argument("--total_slots", type=int, default=None,
                    help="cluster trainer-slot capacity; allocation is "
                         "capped at total_slots * max_load_desired")
    ap.add_argument("--max_load_desired", type=float, default=0.9)
    args = ap.parse_args(argv)
    client = KubectlClient(args.namespace)
    while True:
        try:
            reconcile(client, args.free_slots, args.total_slots,
                      args.max_load_desired)
        except Exception as e:  # noqa: BLE001
            log.warning("reconcile error: %s", e)
        time.sleep(args.period)


if __name__ == "__main__":
    main()
