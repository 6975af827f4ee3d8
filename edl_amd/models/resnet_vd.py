"""ResNet-vd family (the reference's student model zoo).

Architecture parity with the reference's Paddle model zoo
(example/distill/resnet/models/resnet_vd.py: deep 3×3 stem 32-32-64, vd
shortcut = 2×2 avg-pool + 1×1 conv on stride-2 stages, bottleneck with
stride on the 3×3), re-implemented natively in PyTorch — this is the "bag
of tricks" ResNet-D of He et al., public architecture.

MI355X-first structure: every BN is an edl_amd.ops.bnrelu fused module —
BN+ReLU after convs, BN+Add+ReLU at block tails — so on GPU the whole
non-conv side of the network runs as hand-written NHWC bf16 CDNA4 kernels
(plain-torch fallback keeps CPU tests runnable)."""
import torch
import torch.nn as nn

from ..ops.bnrelu import BNAddReLU2d, BNReLU2d
from ..ops.conv import Conv2dFast
from ..ops.pool import AvgPool2x2, MaxPool3x3s2


class ConvBN(nn.Module):
    def __init__(self, cin, cout, k, stride=1, act=True):
        super().__init__()
        self.conv = Conv2dFast(cin, cout, k, stride=stride, padding=(k - 1) // 2,
                              bias=False)
        self.bn = BNReLU2d(cout, act=act)

    def forward(self, x):
        # conv folds BN stats partials into its epilogue when training
        y = self.conv(x, bn_stats=self.bn.training)
        return self.bn(y, partials=self.conv.pop_bn_part())


class VdShortcut(nn.Module):
    """Identity, or (vd) avgpool+1x1conv+BN projection."""

    def __init__(self, cin, cout, stride, if_first):
        super().__init__()
        self.pool = None
        self.proj = None
        if cin != cout or stride != 1:
            if stride != 1 and not if_first:
                # the "vd" trick: downsample by avg-pool, then 1x1 stride 1
                self.pool = AvgPool2x2()
                conv_stride = 1
            else:
                conv_stride = stride
            self.proj = ConvBN(cin, cout, 1, stride=conv_stride, act=False)

    def forward(self, x):
        if self.pool is not None:
            x = self.pool(x)
        if self.proj is not None:
            x = self.proj(x)
        return x


class BottleneckVd(nn.Module):
    expansion = 4

    def __init__(self, cin, planes, stride=1, if_first=False):
        super().__init__()
        cout = planes * self.expansion
        self.conv0 = ConvBN(cin, planes, 1)
        self.conv1 = ConvBN(planes, planes, 3, stride=stride)
        self.conv2 = Conv2dFast(planes, cout, 1, bias=False)
        self.bn_add = BNAddReLU2d(cout)  # relu(bn(conv2) + shortcut), fused
        self.shortcut = VdShortcut(cin, cout, stride, if_first)

    def forward(self, x):
        s = self.shortcut(x)
        y = self.conv2(self.conv1(self.conv0(x)),
                       bn_stats=self.bn_add.training)
        return self.bn_add(y, s, partials=self.conv2.pop_bn_part())


class BasicBlockVd(nn.Module):
    expansion = 1

    def __init__(self, cin, planes, stride=1, if_first=False):
        super().__init__()
        cout = planes * self.expansion
        self.conv0 = ConvBN(cin, planes, 3, stride=stride)
        self.conv1 = Conv2dFast(planes, cout, 3, padding=1, bias=False)
        self.bn_add = BNAddReLU2d(cout)
        self.shortcut = VdShortcut(cin, cout, stride, if_first)

    def forward(self, x):
        s = self.shortcut(x)
        y = self.conv1(self.conv0(x), bn_stats=self.bn_add.training)
        return self.bn_add(y, s, partials=self.conv1.pop_bn_part())


_DEPTHS = {
    18: (BasicBlockVd, [2, 2, 2, 2]),
    34: (BasicBlockVd, [3, 4, 6, 3]),
    50: (BottleneckVd, [3, 4, 6, 3]),
    101: (BottleneckVd, [3, 4, 23, 3]),
    152: (BottleneckVd, [3, 8, 36, 3]),
    200: (BottleneckVd, [3, 12, 48, 3]),
}


class ResNetVd(nn.Module):
    def __init__(self, layers=50, num_classes=1000):
        super().__init__()
        block, depths = _DEPTHS[layers]
        self.stem = nn.Sequential(
            ConvBN(3, 32, 3, stride=2),
            ConvBN(32, 32, 3),
            ConvBN(32, 64, 3),
            MaxPool3x3s2(),
        )
        planes = [64, 128, 256, 512]
        cin = 64
        stages = []
        for bi, (p, d) in enumerate(zip(planes, depths)):
            blocks = []
            for i in range(d):
                stride = 2 if i == 0 and bi != 0 else 1
                blocks.append(block(cin, p, stride=stride, if_first=(bi == 0 and i == 0)))
                cin = p * block.expansion
            stages.append(nn.Sequential(*blocks))
        self.stages = nn.Sequential(*stages)
        self.avgpool = nn.AdaptiveAvgPool2d(1)
        self.fc = nn.Linear(cin, num_classes)
        self._init_weights()

    def _init_weights(self):
        for m in self.modules():
            if isinstance(m, nn.Conv2d):
                nn.init.kaiming_normal_(m.weight, mode="fan_out", nonlinearity="relu")

    def forward(self, x):
        x = self.stages(self.stem(x))
        x = self.avgpool(x).flatten(1)
        return self.fc(x)


def resnet18_vd(num_classes=1000):
    return ResNetVd(18, num_classes)


def resnet34_vd(num_classes=1000):
    return ResNetVd(34, num_classes)


def resnet50_vd(num_classes=1000):
    return ResNetVd(50, num_classes)


def resnet101_vd(num_classes=1000):
    return ResNetVd(101, num_classes)


def resnet152_vd(num_classes=1000):
    return ResNetVd(152, num_classes)


def resnet200_vd(num_classes=1000):
    return ResNetVd(200, num_classes)
