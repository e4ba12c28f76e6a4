"""ResNet family for the sharing benchmarks (BASELINE.json: ResNet50 on
synthetic data / random init; the reference's e2e workloads are
torchelastic resnet18/50 + vgg16 pods, test/distribute/**).

Written for MI355X training throughput: channels-last memory format and
bf16 autocast are applied by the caller (bench/worker); the module keeps
to stock conv/bn so MIOpen picks its gfx950 kernels, with the fused
BN+ReLU HIP op from kubeshare_amd.ops swapped in where profitable.
"""
from __future__ import annotations

import torch
import torch.nn as nn


class Bottleneck(nn.Module):
    expansion = 4
    fused_ops = False  # set by kubeshare_amd.ops.fuse_model

    def __init__(self, in_ch: int, width: int, stride: int = 1):
        super().__init__()
        out_ch = width * self.expansion
        self.conv1 = nn.Conv2d(in_ch, width, 1, bias=False)
        self.bn1 = nn.BatchNorm2d(width)
        self.conv2 = nn.Conv2d(width, width, 3, stride=stride, padding=1,
                               bias=False)
        self.bn2 = nn.BatchNorm2d(width)
        self.conv3 = nn.Conv2d(width, out_ch, 1, bias=False)
        self.bn3 = nn.BatchNorm2d(out_ch)
        self.relu = nn.ReLU(inplace=True)
        self.down = None
        if stride != 1 or in_ch != out_ch:
            self.down = nn.Sequential(
                nn.Conv2d(in_ch, out_ch, 1, stride=stride, bias=False),
                nn.BatchNorm2d(out_ch),
            )

    def forward(self, x):
        if self.fused_ops:
            from .. import ops
            # downsample BN has no activation: fused no-ReLU variant
            # (otherwise it falls back to MIOpen's fp32 spatial BN)
            identity = x if self.down is None else \
                ops.bn_relu(self.down[0](x), self.down[1], relu=False)
            out = ops.bn_relu(self.conv1(x), self.bn1)
            out = ops.bn_relu(self.conv2(out), self.bn2)
            return ops.bn_relu(self.conv3(out), self.bn3, res=identity)
        identity = x if self.down is None else self.down(x)
        out = self.relu(self.bn1(self.conv1(x)))
        out = self.relu(self.bn2(self.conv2(out)))
        out = self.bn3(self.conv3(out))
        return self.relu(out + identity)


class BasicBlock(nn.Module):
    expansion = 1
    fused_ops = False  # set by kubeshare_amd.ops.fuse_model

    def __init__(self, in_ch: int, width: int, stride: int = 1):
        super().__init__()
        self.conv1 = nn.Conv2d(in_ch, width, 3, stride=stride, padding=1,
                               bias=False)
        self.bn1 = nn.BatchNorm2d(width)
        self.conv2 = nn.Conv2d(width, width, 3, padding=1, bias=False)
        self.bn2 = nn.BatchNorm2d(width)
        self.relu = nn.ReLU(inplace=True)
        self.down = None
        if stride != 1 or in_ch != width:
            self.down = nn.Sequential(
                nn.Conv2d(in_ch, width, 1, stride=stride, bias=False),
                nn.BatchNorm2d(width),
            )

    def forward(self, x):
        if self.fused_ops:
            from .. import ops
            identity = x if self.down is None else \
                ops.bn_relu(self.down[0](x), self.down[1], relu=False)
            out = ops.bn_relu(self.conv1(x), self.bn1)
            return ops.bn_relu(self.conv2(out), self.bn2, res=identity)
        identity = x if self.down is None else self.down(x)
        out = self.relu(self.bn1(self.conv1(x)))
        out = self.bn2(self.conv2(out))
        return self.relu(out + identity)


class ResNet(nn.Module):
    fused_ops = False  # stem BN+ReLU fusion (set by ops.fuse_model)

    def __init__(self, block, layers, num_classes: int = 1000):
        super().__init__()
        self.in_ch = 64
        self.conv1 = nn.Conv2d(3, 64, 7, stride=2, padding=3, bias=False)
        self.bn1 = nn.BatchNorm2d(64)
        self.relu = nn.ReLU(inplace=True)
        self.maxpool = nn.MaxPool2d(3, stride=2, padding=1)
        self.layer1 = self._make_layer(block, 64, layers[0])
        self.layer2 = self._make_layer(block, 128, layers[1], stride=2)
        self.layer3 = self._make_layer(block, 256, layers[2], stride=2)
        self.layer4 = self._make_layer(block, 512, layers[3], stride=2)
        self.avgpool = nn.AdaptiveAvgPool2d(1)
        self.fc = nn.Linear(512 * block.expansion, num_classes)
        for m in self.modules():
            if isinstance(m, nn.Conv2d):
                nn.init.kaiming_normal_(m.weight, mode="fan_out",
                                        nonlinearity="relu")
            elif isinstance(m, nn.BatchNorm2d):
                nn.init.ones_(m.weight)
                nn.init.zeros_(m.bias)

    def _make_layer(self, block, width, n, stride=1):
        layers = [block(self.in_ch, width, stride)]
        self.in_ch = width * block.expansion
        layers += [block(self.in_ch, width) for _ in range(n - 1)]
        return nn.Sequential(*layers)

    def forward(self, x):
        if self.fused_ops:
            from .. import ops
            x = self.maxpool(ops.bn_relu(self.conv1(x), self.bn1))
        else:
            x = self.maxpool(self.relu(self.bn1(self.conv1(x))))
        x = self.layer4(self.layer3(self.layer2(self.layer1(x))))
        x = torch.flatten(self.avgpool(x), 1)
        return self.fc(x)


def resnet18(num_classes: int = 1000) -> ResNet:
    return ResNet(BasicBlock, [2, 2, 2, 2], num_classes)


def resnet50(num_classes: int = 1000) -> ResNet:
    return ResNet(Bottleneck, [3, 4, 6, 3], num_classes)
