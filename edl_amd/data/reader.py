"""Distributed elastic reader (parity: reference
collective/distribute_reader.py — which has fatal typos and is treated as
spec).

Per pod: a generator thread splits this pod's file slice into batches,
caches them in the local DataServer and reports ids to the leader; the
accesser side pulls BALANCED assignments from the leader and fetches each
batch from whichever pod produced it. `Reader(...)` yields
{"meta": (owner, batch_id), "data": batch}."""
import threading

from ..utils.log import get_logger
from .data_server import DataClient, DataServer
from .dataset import TxtFileSplitter

log = get_logger("edl.data_reader")


class Reader:
    def __init__(self, pod_id, leader_endpoint, local_server, pod_endpoints,
                 splitter=None, batch_size=2):
        """pod_endpoints: {pod_id: data-server endpoint} for remote fetch."""
        self.pod_id = pod_id
        self.leader = DataClient(leader_endpoint)
        self.local_server = local_server
        self.pod_endpoints = dict(pod_endpoints)
        self.splitter = splitter or TxtFileSplitter()
        self.batch_size = batch_size
        self._clients = {}
        self._gen_thread = None

    def _client_for(self, owner):
        if owner == self.pod_id:
            return None
        if owner not in self._clients:
            self._clients[owner] = DataClient(self.pod_endpoints[owner])
        return self._clients[owner]

    def _generate(self):
        """Produce batches from this pod's file slice; cache + report."""
        files = self.leader.get_file_list(self.pod_id)
        batch_id = 0
        batch = []
        produced = []

        def emit():
            nonlocal batch_id, batch
            if not batch:
                return
            bid = "%s-%d" % (self.pod_id, batch_id)
            self.local_server.put_batch(bid, __import__("pickle").dumps(batch))
            produced.append(bid)
            batch_id += 1
            batch = []

        for f in files:
            for _idx, rec in self.splitter(f):
                batch.append(rec)
                if len(batch) >= self.batch_size:
                    emit()
                    if len(produced) >= 4:
                        self.leader.report(self.pod_id, produced)
                        produced = []
        emit()
        self.leader.report(self.pod_id, produced, finished=True)
        log.debug("pod %s produced %d batches", self.pod_id, batch_id)

    def __iter__(self):
        self._gen_thread = threading.Thread(target=self._generate, daemon=True)
        self._gen_thread.start()
        import pickle
        import time

        while True:
            items, done = self.leader.get_meta(self.pod_id)
            for owner, bid in items:
                if owner == self.pod_id:
                    data = pickle.loads(self.local_server.get_batch(bid))
                else:
                    data = self._client_for(owner).get_batch(bid)
                yield {"meta": (owner, bid), "data": data}
            if done:
                return
            if not items:
                time.sleep(0.05)

    def close(self):
        for c in self._clients.values():
            c.close()
        self.leader.close()


def start_pod_data_server(file_list=None, pod_ids=None, port=0):
    """Helper: start this pod's DataServer (leader passes the file list)."""
    return DataServer(port=port, file_list=file_list or [],
                      pod_ids=pod_ids or []).start()
