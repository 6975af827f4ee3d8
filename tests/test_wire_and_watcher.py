"""Direct unit tests for the wire protocols (C17) and the cluster
watcher (C13) — previously covered only through the integration suites."""
import socket
import struct
import threading

import numpy as np
import pytest

from edl_amd.coord import protocol
from edl_amd.distill import proto


class TestCoordProtocol:
    def _pair(self):
        a, b = socket.socketpair()
        return a, b

    def test_roundtrip_and_multiframe(self):
        a, b = self._pair()
        try:
            msgs = [{"op": "put", "k": "x" * 1000, "v": 1},
                    {"op": "get"}, {"nested": {"a": [1, 2, 3]}}]
            for m in msgs:
                protocol.send_msg(a, m)
            for m in msgs:
                assert protocol.recv_msg(b) == m
        finally:
            a.close(); b.close()

    def test_oversize_frame_rejected(self):
        a, b = self._pair()
        try:
            a.sendall(struct.pack("!I", protocol.MAX_FRAME + 1))
            with pytest.raises(protocol.ProtocolError):
                protocol.recv_msg(b)
        finally:
            a.close(); b.close()

    def test_peer_close_raises(self):
        a, b = self._pair()
        a.close()
        try:
            with pytest.raises(ConnectionError):
                protocol.recv_msg(b)
        finally:
            b.close()


class TestDistillProto:
    def test_header_and_payload_roundtrip(self):
        a, b = socket.socketpair()
        try:
            x = np.arange(4096, dtype=np.float32)
            y = np.ones((3, 5), dtype=np.uint8)
            proto._send(a, {"op": "predict", "n_payloads": 2,
                            "shapes": [list(x.shape), list(y.shape)]},
                        [x.tobytes(), y.tobytes()])
            hdr, payloads = proto._recv(b)
            assert hdr["op"] == "predict" and len(payloads) == 2
            xr = np.frombuffer(payloads[0], dtype=np.float32)
            assert np.array_equal(xr, x)
            # buffers must be WRITABLE (torch refuses read-only arrays)
            xr2 = np.frombuffer(payloads[0], dtype=np.float32)
            assert xr2.flags.writeable or bytearray is type(payloads[0])
        finally:
            a.close(); b.close()

    def test_oversize_header_rejected(self):
        a, b = socket.socketpair()
        try:
            a.sendall(struct.pack("!I", proto.MAX_FRAME + 1))
            with pytest.raises(ValueError):
                proto._recv(b)
        finally:
            a.close(); b.close()


class TestClusterWatcher:
    def test_detects_stage_change_and_ignores_noise(self):
        from edl_amd.cluster.model import Cluster, Pod, save_cluster
        from edl_amd.cluster.watcher import ClusterWatcher
        from edl_amd.coord.client import CoordClient
        from edl_amd.coord.server import CoordServer

        srv = CoordServer(port=0).start()
        try:
            ep = "127.0.0.1:%d" % srv.port
            client = CoordClient(ep, "jobw")
            base = Cluster(pods=[Pod(pod_id="p0"), Pod(pod_id="p1")])
            save_cluster(client, base)

            w = ClusterWatcher(ep, "jobw", base).start()
            # unrelated key writes must not trigger it
            client.put(client.table_key("misc", "noise"), "1")
            assert not w.wait_changed(0.5)

            grown = Cluster(pods=[Pod(pod_id="p0"), Pod(pod_id="p1"),
                                  Pod(pod_id="p2")])
            grown.new_stage()
            save_cluster(client, grown)
            assert w.wait_changed(5.0), "watcher missed the stage change"
            assert w.new_cluster().stage == grown.stage
            w.stop()
            client.close()
        finally:
            srv.stop()
