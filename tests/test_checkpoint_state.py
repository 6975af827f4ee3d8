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


def test_paddle_edl_layout_contract(tmp_path):
    """Asserts the paddle_edl checkpoint-format mapping documented in
    docs/fault_tolerance.md (reference doc/fault_tolerance.md:20-62):
    incrementing `checkpoint.<N>` version dirs, atomic temp-then-rename
    (no .tmp residue), train_status.json with epoch_no, a manifest, and
    TrainStatus(pass_id).next() resume semantics via the fleet facade."""
    import json

    from edl_amd.train.fleet import TrainStatus

    cm = CheckpointManager(str(tmp_path), keep=3)
    for epoch in range(2):
        v = cm.save({"w": torch.ones(2)}, {"epoch_no": epoch}, blocking=True)
        assert v == epoch  # incrementing version numbers
    names = sorted(os.listdir(tmp_path))
    assert names == ["checkpoint.0", "checkpoint.1"]  # no .tmp residue
    d = tmp_path / "checkpoint.1"
    assert sorted(os.listdir(d)) == [
        "checkpoint_meta.json", "model.pt", "train_status.json"]
    meta = json.load(open(d / "checkpoint_meta.json"))
    assert meta["format"] == "edl_amd.v1" and meta["version"] == 1
    assert meta["saved_by_rank"] == 0
    ts = json.load(open(d / "train_status.json"))
    assert ts["epoch_no"] == 1
    # resume flow: TrainStatus(pass_id).next() is the first epoch to run
    status = TrainStatus(ts["epoch_no"])
    assert status.next() == 2
    # fresh-start sentinel matches the reference (-1 -> start at 0)
    assert TrainStatus().next() == 0
