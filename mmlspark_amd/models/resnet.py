"""ResNet family implemented in plain PyTorch (torchvision is not available).

Backbone for ImageFeaturizer — the MI355X re-expression of the reference's
CNTK ResNet50 featurizer (deep-learning/.../ImageFeaturizer.scala:41,
ModelDownloader schemas). Conv/GEMM run through MIOpen/rocBLAS via
PyTorch-ROCm; DP scaling uses DDP over RCCL.
"""
from __future__ import annotations

from typing import List

import torch.nn as nn


class Bottleneck(nn.Module):
    expansion = 4

    def __init__(self, in_ch, ch, stride=1, downsample=None):
        super().__init__()
        self.conv1 = nn.Conv2d(in_ch, ch, 1, bias=False)
        self.bn1 = nn.BatchNorm2d(ch)
        self.conv2 = nn.Conv2d(ch, ch, 3, stride=stride, padding=1, bias=False)
        self.bn2 = nn.BatchNorm2d(ch)
        self.conv3 = nn.Conv2d(ch, ch * 4, 1, bias=False)
        self.bn3 = nn.BatchNorm2d(ch * 4)
        self.relu = nn.ReLU(inplace=True)
        self.downsample = downsample

    def forward(self, x):
        idt = x
        out = self.relu(self.bn1(self.conv1(x)))
        out = self.relu(self.bn2(self.conv2(out)))
        out = self.bn3(self.conv3(out))
        if self.downsample is not None:
            idt = self.downsample(x)
        return self.relu(out + idt)


class BasicBlock(nn.Module):
    expansion = 1

    def __init__(self, in_ch, ch, stride=1, downsample=None):
        super().__init__()
        self.conv1 = nn.Conv2d(in_ch, ch, 3, stride=stride, padding=1, bias=False)
        self.bn1 = nn.BatchNorm2d(ch)
        self.conv2 = nn.Conv2d(ch, ch, 3, padding=1, bias=False)
        self.bn2 = nn.BatchNorm2d(ch)
        self.relu = nn.ReLU(inplace=True)
        self.downsample = downsample

    def forward(self, x):
        idt = x
        out = self.relu(self.bn1(self.conv1(x)))
        out = self.bn2(self.conv2(out))
        if self.downsample is not None:
            idt = self.downsample(x)
        return self.relu(out + idt)


class ResNet(nn.Module):
    def __init__(self, block, layers: List[int], num_classes=1000):
        super().__init__()
        self.in_ch = 64
        self.stem = nn.Sequential(
            nn.Conv2d(3, 64, 7, stride=2, padding=3, bias=False),
            nn.BatchNorm2d(64), nn.ReLU(inplace=True),
            nn.MaxPool2d(3, stride=2, padding=1))
        self.layer1 = self._make(block, 64, layers[0])
        self.layer2 = self._make(block, 128, layers[1], 2)
        self.layer3 = self._make(block, 256, layers[2], 2)
        self.layer4 = self._make(block, 512, layers[3], 2)
        self.avgpool = nn.AdaptiveAvgPool2d(1)
        self.fc = nn.Linear(512 * block.expansion, num_classes)
        self.feature_dim = 512 * block.expansion
        for m in self.modules():
            if isinstance(m, nn.Conv2d):
                nn.init.kaiming_normal_(m.weight, mode="fan_out",
                                        nonlinearity="relu")
            elif isinstance(m, nn.BatchNorm2d):
                nn.init.ones_(m.weight)
                nn.init.zeros_(m.bias)

    def _make(self, block, ch, n, stride=1):
        down = None
        if stride != 1 or self.in_ch != ch * block.expansion:
            down = nn.Sequential(
                nn.Conv2d(self.in_ch, ch * block.expansion, 1, stride=stride,
                          bias=False),
                nn.BatchNorm2d(ch * block.expansion))
        blocks = [block(self.in_ch, ch, stride, down)]
        self.in_ch = ch * block.expansion
        for _ in range(1, n):
            blocks.append(block(self.in_ch, ch))
        return nn.Sequential(*blocks)

    def forward(self, x, cut_output_layers: int = 0):
        """cut_output_layers: 0 = logits; 1 = pooled features (2048-d);
        2 = pre-pool conv featuremap — the reference's layer-cut semantics
        (ImageFeaturizer cutOutputLayers)."""
        x = self.stem(x)
        x = self.layer1(x)
        x = self.layer2(x)
        x = self.layer3(x)
        x = self.layer4(x)
        if cut_output_layers >= 2:
            return x
        x = self.avgpool(x).flatten(1)
        if cut_output_layers >= 1:
            return x
        return self.fc(x)


def resnet18(num_classes=1000):
    return ResNet(BasicBlock, [2, 2, 2, 2], num_classes)


def resnet34(num_classes=1000):
    return ResNet(BasicBlock, [3, 4, 6, 3], num_classes)


def resnet50(num_classes=1000):
    return ResNet(Bottleneck, [3, 4, 6, 3], num_classes)


def resnet101(num_classes=1000):
    return ResNet(Bottleneck, [3, 4, 23, 3], num_classes)


MODELS = {"ResNet18": resnet18, "ResNet34": resnet34, "ResNet50": resnet50,
          "ResNet101": resnet101}
