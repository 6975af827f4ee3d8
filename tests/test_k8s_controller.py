"""Controller reconcile-policy tests with a fake cluster client."""
from edl_amd.k8s.controller import ClusterClient, reconcile


class FakeCluster(ClusterClient):
    def __init__(self, jobs):
        self.jobs = jobs
        self.scaled = []

    def list_training_jobs(self):
        return [dict(j) for j in self.jobs]

    def scale_job(self, name, replicas):
        self.scaled.append((name, replicas))
        for j in self.jobs:
            if j["name"] == name:
                j["running"] = replicas


def job(name, mn, mx, running, desired=None):
    return {"name": name, "min_instance": mn, "max_instance": mx,
            "running": running, "desired": desired}


def test_scale_up_to_max_when_unconstrained():
    fc = FakeCluster([job("a", 1, 4, 1)])
    actions = reconcile(fc)
    assert actions == [("a", 1, 4)]


def test_desired_caps_at_bounds():
    fc = FakeCluster([job("a", 2, 8, 8, desired=1)])
    reconcile(fc)
    assert fc.jobs[0]["running"] == 2  # desired below min -> min


def test_fair_share_under_capacity():
    fc = FakeCluster([job("a", 1, 4, 4), job("b", 1, 4, 1)])
    reconcile(fc, free_slots=4)
    assert fc.jobs[0]["running"] + fc.jobs[1]["running"] == 4
    assert fc.jobs[0]["running"] >= 1 and fc.jobs[1]["running"] >= 1


def test_noop_when_at_target():
    fc = FakeCluster([job("a", 1, 2, 2)])
    assert reconcile(fc) == []


def test_max_load_desired_caps_allocation():
    """Reference `-max_load_desired 0.9`: with capacity known, never
    allocate past the load target (headroom for other workloads)."""
    c = FakeCluster([job("a", 1, 8, 1), job("b", 1, 8, 1)])
    reconcile(c, total_slots=10, max_load_desired=0.9)  # budget = 9
    total = sum(j["running"] for j in c.jobs)
    assert total == 9, c.jobs
    assert all(j["running"] >= 1 for j in c.jobs)
