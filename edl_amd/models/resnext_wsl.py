"""ResNeXt101_32x16d_wsl — the reference's distill TEACHER model
(README.md:51-75: remote ResNeXt101_32x16d_wsl teachers served to
ResNet50_vd students). Standard ResNeXt architecture (grouped 3×3,
width = planes·(base_width/64)·groups), 7×7 stem.

Served by edl_amd.distill.teacher_server; BN(+Add)+ReLU run as the fused
CDNA4 kernels (edl_amd.ops.bnrelu) on GPU."""
import torch.nn as nn

from ..ops.bnrelu import BNAddReLU2d, BNReLU2d
from ..ops.conv import Conv2dFast
from ..ops.pool import MaxPool3x3s2


class ResNeXtBottleneck(nn.Module):
    expansion = 4

    def __init__(self, cin, planes, stride=1, groups=32, base_width=16):
        super().__init__()
        width = int(planes * (base_width / 64.0)) * groups
        cout = planes * self.expansion
        self.conv1 = Conv2dFast(cin, width, 1, bias=False)
        self.bn1 = BNReLU2d(width)
        self.conv2 = Conv2dFast(width, width, 3, stride=stride, padding=1,
                               groups=groups, bias=False)
        self.bn2 = BNReLU2d(width)
        self.conv3 = Conv2dFast(width, cout, 1, bias=False)
        self.bn_add = BNAddReLU2d(cout)
        self.downsample = None
        if stride != 1 or cin != cout:
            self.downsample = nn.Sequential(
                Conv2dFast(cin, cout, 1, stride=stride, bias=False),
                BNReLU2d(cout, act=False),
            )

    def forward(self, x):
        s = x if self.downsample is None else self.downsample(x)
        y = self.bn1(self.conv1(x))
        y = self.bn2(self.conv2(y))
        return self.bn_add(self.conv3(y), s)


class ResNeXtWSL(nn.Module):
    def __init__(self, depths=(3, 4, 23, 3), groups=32, base_width=16, num_classes=1000):
        super().__init__()
        self.conv1 = Conv2dFast(3, 64, 7, stride=2, padding=3, bias=False)
        self.bn1 = BNReLU2d(64)
        self.maxpool = MaxPool3x3s2()
        cin = 64
        stages = []
        for bi, (p, d) in enumerate(zip([64, 128, 256, 512], depths)):
            blocks = []
            for i in range(d):
                stride = 2 if i == 0 and bi != 0 else 1
                blocks.append(ResNeXtBottleneck(cin, p, stride, groups, base_width))
                cin = p * ResNeXtBottleneck.expansion
            stages.append(nn.Sequential(*blocks))
        self.stages = nn.Sequential(*stages)
        self.avgpool = nn.AdaptiveAvgPool2d(1)
        self.fc = nn.Linear(cin, num_classes)
        for m in self.modules():
            if isinstance(m, nn.Conv2d):
                nn.init.kaiming_normal_(m.weight, mode="fan_out", nonlinearity="relu")

    def forward(self, x):
        x = self.maxpool(self.bn1(self.conv1(x)))
        x = self.stages(x)
        return self.fc(self.avgpool(x).flatten(1))


def resnext101_32x16d_wsl(num_classes=1000):
    return ResNeXtWSL((3, 4, 23, 3), groups=32, base_width=16, num_classes=num_classes)
