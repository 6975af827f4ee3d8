"""ResNet-vd family (the reference's student model zoo).

Architecture parity with the reference's Paddle model zoo
(example/distill/resnet/models/resnet_vd.py: deep 3×3 stem 32-32-64, vd
shortcut = 2×2 avg-pool + 1×1 conv on stride-2 stages, bottleneck with
stride on the 3×3), re-implemented natively in PyTorch — this is the "bag
of tricks" ResNet-D of He et al., public architecture.

MI355X notes: intended to run channels_last (NHWC) + bf16 autocast; the
hot fused ops (BN+ReLU epilogues, KD loss, SGD) are swapped in by
edl_amd.ops at the engine level, keeping this definition plain torch so it
also runs on CPU for tests.
"""
import torch
import torch.nn as nn


def _conv_bn(cin, cout, k, stride=1, act=True):
    layers = [
        nn.Conv2d(cin, cout, k, stride=stride, padding=(k - 1) // 2, bias=False),
        nn.BatchNorm2d(cout),
    ]
    if act:
        layers.append(nn.ReLU(inplace=True))
    return nn.Sequential(*layers)


class BottleneckVd(nn.Module):
    expansion = 4

    def __init__(self, cin, planes, stride=1, if_first=False):
        super().__init__()
        cout = planes * self.expansion
        self.conv0 = _conv_bn(cin, planes, 1)
        self.conv1 = _conv_bn(planes, planes, 3, stride=stride)
        self.conv2 = _conv_bn(planes, cout, 1, act=False)
        self.shortcut = None
        if cin != cout or stride != 1:
            sc = []
            if stride != 1 and not if_first:
                # the "vd" trick: downsample by avg-pool, then 1x1 stride 1
                sc.append(nn.AvgPool2d(2, 2, ceil_mode=True))
            sc.append(nn.Conv2d(cin, cout, 1, stride=1 if not if_first else stride, bias=False))
            sc.append(nn.BatchNorm2d(cout))
            self.shortcut = nn.Sequential(*sc)
        self.relu = nn.ReLU(inplace=True)

    def forward(self, x):
        s = x if self.shortcut is None else self.shortcut(x)
        y = self.conv2(self.conv1(self.conv0(x)))
        return self.relu(y + s)


class BasicBlockVd(nn.Module):
    expansion = 1

    def __init__(self, cin, planes, stride=1, if_first=False):
        super().__init__()
        cout = planes * self.expansion
        self.conv0 = _conv_bn(cin, planes, 3, stride=stride)
        self.conv1 = _conv_bn(planes, cout, 3, act=False)
        self.shortcut = None
        if cin != cout or stride != 1:
            sc = []
            if stride != 1 and not if_first:
                sc.append(nn.AvgPool2d(2, 2, ceil_mode=True))
            sc.append(nn.Conv2d(cin, cout, 1, stride=1 if not if_first else stride, bias=False))
            sc.append(nn.BatchNorm2d(cout))
            self.shortcut = nn.Sequential(*sc)
        self.relu = nn.ReLU(inplace=True)

    def forward(self, x):
        s = x if self.shortcut is None else self.shortcut(x)
        return self.relu(self.conv1(self.conv0(x)) + s)


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
            _conv_bn(3, 32, 3, stride=2),
            _conv_bn(32, 32, 3),
            _conv_bn(32, 64, 3),
            nn.MaxPool2d(3, 2, padding=1),
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
            elif isinstance(m, nn.BatchNorm2d):
                nn.init.ones_(m.weight)
                nn.init.zeros_(m.bias)

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
