"""Elastic train state in the coordination store.

Parity: reference utils/state.py:25-217 — DataCheckpoint (file list +
processed record ranges), EpochAttr (per-epoch stats), TrainStatus
(epoch map + global step), State (total batch size + user-defined substate
+ register_adjust_function hook for LR/batch re-adjust on world resize),
leader-guarded save via store transaction (state.py:186-200)."""
import json

from ..coord.tables import ETCD_STATE


class DataCheckpoint:
    def __init__(self, file_list=None, processed=None):
        self.file_list = file_list or []
        self.processed = processed or {}  # file -> [[begin, end], ...]

    def to_dict(self):
        return {"file_list": self.file_list, "processed": self.processed}

    @classmethod
    def from_dict(cls, d):
        return cls(d.get("file_list"), d.get("processed"))


class EpochAttr:
    def __init__(self, epoch_no=0, world_size=0, step_num=0, avg_step_time=0.0):
        self.epoch_no = epoch_no
        self.world_size = world_size
        self.step_num = step_num
        self.avg_step_time = avg_step_time

    def to_dict(self):
        return dict(self.__dict__)

    @classmethod
    def from_dict(cls, d):
        e = cls()
        e.__dict__.update(d)
        return e


class TrainStatusState:
    """Epoch map + global step (reference state.py:61-111). Named to avoid
    colliding with cluster.status.TrainStatus (the enum)."""

    def __init__(self):
        self.epochs = {}  # epoch_no(str) -> EpochAttr
        self.global_step_no = 0
        self.epoch_no = -1

    def update_epoch(self, attr):
        self.epochs[str(attr.epoch_no)] = attr
        self.epoch_no = max(self.epoch_no, attr.epoch_no)

    def next_epoch(self):
        """Resume point (reference TrainStatus.next(), doc/fault_tolerance.md:55-62)."""
        return self.epoch_no + 1

    def to_dict(self):
        return {
            "epochs": {k: v.to_dict() for k, v in self.epochs.items()},
            "global_step_no": self.global_step_no,
            "epoch_no": self.epoch_no,
        }

    @classmethod
    def from_dict(cls, d):
        t = cls()
        t.global_step_no = d.get("global_step_no", 0)
        t.epoch_no = d.get("epoch_no", -1)
        t.epochs = {k: EpochAttr.from_dict(v) for k, v in d.get("epochs", {}).items()}
        return t


class State:
    """The resumable train state + world-resize adjustment hooks."""

    def __init__(self, total_batch_size=0, user_defined=None):
        self.total_batch_size = total_batch_size
        self.user_defined = user_defined or {}
        self.data_checkpoint = DataCheckpoint()
        self.train_status = TrainStatusState()
        self._adjust_fns = []

    def register_adjust_function(self, fn):
        """fn(state, old_world, new_world) — called when the world size
        changes (LR rescale etc.; reference state.py:142-143,
        doc/edl_collective_design_doc.md:14-17)."""
        self._adjust_fns.append(fn)

    def adjust(self, old_world, new_world):
        for fn in self._adjust_fns:
            fn(self, old_world, new_world)

    def to_json(self):
        return json.dumps(
            {
                "total_batch_size": self.total_batch_size,
                "user_defined": self.user_defined,
                "data_checkpoint": self.data_checkpoint.to_dict(),
                "train_status": self.train_status.to_dict(),
            }
        )

    @classmethod
    def from_json(cls, s):
        d = json.loads(s)
        st = cls(d.get("total_batch_size", 0), d.get("user_defined"))
        st.data_checkpoint = DataCheckpoint.from_dict(d.get("data_checkpoint", {}))
        st.train_status = TrainStatusState.from_dict(d.get("train_status", {}))
        return st


def save_state(client, state, name="train", guard=None):
    key = client.table_key(ETCD_STATE, name)
    if guard is None:
        client.put(key, state.to_json())
        return True
    return client.txn_if(guard[0], guard[1], puts=[(key, state.to_json())])


def load_state(client, name="train"):
    s = client.get(client.table_key(ETCD_STATE, name))
    return State.from_json(s) if s else None
