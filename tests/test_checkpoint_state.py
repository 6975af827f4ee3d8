"""Checkpoint manager + elastic train state tests."""
import os

import torch

from edl_amd.train.checkpoint import CheckpointManager
from edl_amd.train.state import DataCheckpoint, EpochAttr, State, load_state, save_state


def test_checkpoint_save_load_versioned(tmp_path):
    cm = CheckpointManager(str(tmp_path), keep=2)
    model = {"w": torch.randn(4, 4), "b": torch.randn(4)}
    v0 = cm.save(model, {"epoch_no": 0}, blocking=True)
    assert v0 == 0
    v1 = cm.save({"w": model["w"] + 1, "b": model["b"]}, {"epoch_no": 1}, blocking=True)
    assert v1 == 1
    got = cm.load()
    assert got is not None
    m, opt, ts = got
    assert ts["epoch_no"] == 1 and ts["_version"] == 1
    assert torch.allclose(m["w"], model["w"] + 1)


def test_checkpoint_gc_keeps_last(tmp_path):
    cm = CheckpointManager(str(tmp_path), keep=2)
    for e in range(5):
        cm.save({"w": torch.tensor([float(e)])}, {"epoch_no": e}, blocking=True)
    assert cm.versions() == [3, 4]


def test_checkpoint_skips_corrupt_latest(tmp_path):
    cm = CheckpointManager(str(tmp_path), keep=5)
    cm.save({"w": torch.tensor([1.0])}, {"epoch_no": 0}, blocking=True)
    cm.save({"w": torch.tensor([2.0])}, {"epoch_no": 1}, blocking=True)
    # corrupt the newest
    with open(os.path.join(str(tmp_path), "checkpoint.1", "model.pt"), "wb") as f:
        f.write(b"garbage")
    m, _, ts = cm.load()
    assert ts["epoch_no"] == 0
    assert torch.allclose(m["w"], torch.tensor([1.0]))


def test_checkpoint_optimizer_roundtrip(tmp_path):
    cm = CheckpointManager(str(tmp_path))
    opt_state = {"state": {0: {"momentum_buffer": torch.randn(3)}},
                 "param_groups": [{"lr": 0.1, "params": [0]}]}
    cm.save({"w": torch.randn(2)}, {"epoch_no": 0}, optimizer_state=opt_state,
            blocking=True)
    _, opt, _ = cm.load()
    assert torch.allclose(opt["state"][0]["momentum_buffer"],
                          opt_state["state"][0]["momentum_buffer"])
    assert opt["param_groups"][0]["lr"] == 0.1


def test_state_roundtrip_and_adjust(coord_client):
    st = State(total_batch_size=256, user_defined={"note": "x"})
    st.data_checkpoint = DataCheckpoint(["a.txt", "b.txt"], {"a.txt": [[0, 100]]})
    st.train_status.update_epoch(EpochAttr(3, 8, 100, 0.1))
    st.train_status.global_step_no = 400

    calls = []
    st.register_adjust_function(lambda s, ow, nw: calls.append((ow, nw)))
    st.adjust(8, 4)
    assert calls == [(8, 4)]

    assert save_state(coord_client, st)
    got = load_state(coord_client)
    assert got.total_batch_size == 256
    assert got.train_status.next_epoch() == 4
    assert got.train_status.global_step_no == 400
    assert got.data_checkpoint.file_list == ["a.txt", "b.txt"]
    assert got.user_defined == {"note": "x"}

    # leader-guarded save
    coord_client.put("/test_job/rank/nodes/0", "leader")
    assert save_state(coord_client, st, guard=("/test_job/rank/nodes/0", "leader"))
    assert not save_state(coord_client, st, guard=("/test_job/rank/nodes/0", "bogus"))
