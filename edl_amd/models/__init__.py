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
from .mnist import MnistCNN, MnistMLP, MnistSoftmaxRegression
from .text import TextBOW, TextCNN


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
        # distill example families (reference example/distill/mnist_distill
        # nn_type selector + train_with_fleet.py:55-95)
        "mnist_cnn": MnistCNN,
        "mnist_mlp": MnistMLP,
        "mnist_softmax": MnistSoftmaxRegression,
    }
    if name not in table:
        raise ValueError("unknown model %r (have %s)" % (name, sorted(table)))
    kwargs = {"num_classes": num_classes}
    if name.startswith("mnist") and num_classes == 1000:
        kwargs["num_classes"] = 10
    return table[name](**kwargs)
