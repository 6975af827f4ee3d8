from .resnet_vd import (
    ResNetVd,
    resnet18_vd,
    resnet34_vd,
    resnet50_vd,
    resnet101_vd,
    resnet152_vd,
    resnet200_vd,
)
from .resnext_wsl import resnext101_32x16d_wsl
from .linear import FitALine
from .ctr import WideAndDeep


def build_model(name, num_classes=1000):
    name = name.lower()
    table = {
        "resnet18_vd": resnet18_vd,
        "resnet34_vd": resnet34_vd,
        "resnet50_vd": resnet50_vd,
        "resnet101_vd": resnet101_vd,
        "resnet152_vd": resnet152_vd,
        "resnet200_vd": resnet200_vd,
        "resnext101_32x16d_wsl": resnext101_32x16d_wsl,
    }
    if name not in table:
        raise ValueError("unknown model %r (have %s)" % (name, sorted(table)))
    return table[name](num_classes=num_classes)
