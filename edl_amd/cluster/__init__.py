from .model import Pod, Trainer, Cluster
from .status import Status, TrainStatus
