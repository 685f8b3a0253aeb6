"""ResNet-18/50 (He et al. 2015), written for this framework.

The reference trains unspecified single-device torch models (README.md:6);
BASELINE configs 2-3 name ResNet-18/50 as the PS benchmark models.  Convs run
through PyTorch-ROCm (MIOpen); BatchNorm (+residual add +ReLU) runs on this
framework's fused CDNA4 kernels (ops/bn.py) — on MI355X the stock torch
channels_last BN kernels were 57% of the step time.
"""

from __future__ import annotations

import torch.nn as nn

from ..ops.bn import FusedBatchNorm2d


class BasicBlock(nn.Module):
    expansion = 1

    def __init__(self, cin, cout, stride=1):
        super().__init__()
        self.conv1 = nn.Conv2d(cin, cout, 3, stride, 1, bias=False)
        self.bn1 = FusedBatchNorm2d(cout, relu=True)
        self.conv2 = nn.Conv2d(cout, cout, 3, 1, 1, bias=False)
        self.bn2 = FusedBatchNorm2d(cout, relu=True)  # fused y=relu(bn(x)+z)
        self.down_conv = None
        if stride != 1 or cin != cout * self.expansion:
            self.down_conv = nn.Conv2d(cin, cout * self.expansion, 1, stride,
                                       bias=False)
            self.down_bn = FusedBatchNorm2d(cout * self.expansion)

    def forward(self, x):
        idt = x if self.down_conv is None else self.down_bn(self.down_conv(x))
        y = self.bn1(self.conv1(x))
        return self.bn2(self.conv2(y), z=idt)


class Bottleneck(nn.Module):
    expansion = 4

    def __init__(self, cin, cout, stride=1):
        super().__init__()
        self.conv1 = nn.Conv2d(cin, cout, 1, bias=False)
        self.bn1 = FusedBatchNorm2d(cout, relu=True)
        self.conv2 = nn.Conv2d(cout, cout, 3, stride, 1, bias=False)
        self.bn2 = FusedBatchNorm2d(cout, relu=True)
        self.conv3 = nn.Conv2d(cout, cout * 4, 1, bias=False)
        self.bn3 = FusedBatchNorm2d(cout * 4, relu=True)  # y=relu(bn(x)+z)
        self.down_conv = None
        if stride != 1 or cin != cout * 4:
            self.down_conv = nn.Conv2d(cin, cout * 4, 1, stride, bias=False)
            self.down_bn = FusedBatchNorm2d(cout * 4)

    def forward(self, x):
        idt = x if self.down_conv is None else self.down_bn(self.down_conv(x))
        y = self.bn1(self.conv1(x))
        y = self.bn2(self.conv2(y))
        return self.bn3(self.conv3(y), z=idt)


class ResNet(nn.Module):
    def __init__(self, block, layers, num_classes=1000):
        super().__init__()
        self.cin = 64
        self.conv1 = nn.Conv2d(3, 64, 7, 2, 3, bias=False)
        self.bn1 = FusedBatchNorm2d(64, relu=True)
        self.maxpool = nn.MaxPool2d(3, 2, 1)
        self.layer1 = self._make(block, 64, layers[0], 1)
        self.layer2 = self._make(block, 128, layers[1], 2)
        self.layer3 = self._make(block, 256, layers[2], 2)
        self.layer4 = self._make(block, 512, layers[3], 2)
        self.pool = nn.AdaptiveAvgPool2d(1)
        self.fc = nn.Linear(512 * block.expansion, num_classes)
        for m in self.modules():
            if isinstance(m, nn.Conv2d):
                nn.init.kaiming_normal_(m.weight, mode="fan_out",
                                        nonlinearity="relu")
            elif isinstance(m, nn.BatchNorm2d):
                nn.init.ones_(m.weight)
                nn.init.zeros_(m.bias)

    def _make(self, block, cout, n, stride):
        blocks = []
        for i in range(n):
            blocks.append(block(self.cin, cout, stride if i == 0 else 1))
            self.cin = cout * block.expansion
        return nn.Sequential(*blocks)

    def forward(self, x):
        x = self.maxpool(self.bn1(self.conv1(x)))
        x = self.layer4(self.layer3(self.layer2(self.layer1(x))))
        return self.fc(self.pool(x).flatten(1))


def resnet18(num_classes=1000):
    return ResNet(BasicBlock, [2, 2, 2, 2], num_classes)


def resnet50(num_classes=1000):
    return ResNet(Bottleneck, [3, 4, 6, 3], num_classes)
