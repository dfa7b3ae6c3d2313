"""ResNet-50 (bottleneck v1, textbook architecture) for the DP-SGD benchmark
(BASELINE.md config #3 — the reference benchmarked data-parallel convnet
training in examples/)."""
from typing import List

import torch
import torch.nn as nn


class Bottleneck(nn.Module):
    expansion = 4

    def __init__(self, in_ch, ch, stride=1, downsample=None):
        super().__init__()
        self.conv1 = nn.Conv2d(in_ch, ch, 1, bias=False)
        self.bn1 = nn.BatchNorm2d(ch)
        self.conv2 = nn.Conv2d(ch, ch, 3, stride=stride, padding=1, bias=False)
        self.bn2 = nn.BatchNorm2d(ch)
        self.conv3 = nn.Conv2d(ch, ch * self.expansion, 1, bias=False)
        self.bn3 = nn.BatchNorm2d(ch * self.expansion)
        self.relu = nn.ReLU(inplace=True)
        self.downsample = downsample
        self.stride = stride

    def forward(self, x):
        identity = x
        if getattr(self.bn1, "supports_fusion", False) and x.is_cuda:
            # FusedBatchNorm2d path: relu and the residual add run inside
            # the BN epilogue kernels (no separate elementwise passes).
            out = self.bn1(self.conv1(x), relu=True)
            out = self.bn2(self.conv2(out), relu=True)
            if self.downsample is not None:
                identity = self.downsample(x)
            return self.bn3(self.conv3(out), relu=True, res=identity)
        out = self.relu(self.bn1(self.conv1(x)))
        out = self.relu(self.bn2(self.conv2(out)))
        out = self.bn3(self.conv3(out))
        if self.downsample is not None:
            identity = self.downsample(x)
        return self.relu(out + identity)


class ResNet(nn.Module):
    def __init__(self, layers: List[int], num_classes: int = 1000):
        super().__init__()
        self.in_ch = 64
        self.conv1 = nn.Conv2d(3, 64, 7, stride=2, padding=3, bias=False)
        self.bn1 = nn.BatchNorm2d(64)
        self.relu = nn.ReLU(inplace=True)
        self.maxpool = nn.MaxPool2d(3, stride=2, padding=1)
        self.layer1 = self._make_layer(64, layers[0])
        self.layer2 = self._make_layer(128, layers[1], 2)
        self.layer3 = self._make_layer(256, layers[2], 2)
        self.layer4 = self._make_layer(512, layers[3], 2)
        self.avgpool = nn.AdaptiveAvgPool2d(1)
        self.fc = nn.Linear(512 * Bottleneck.expansion, num_classes)
        for m in self.modules():
            if isinstance(m, nn.Conv2d):
                nn.init.kaiming_normal_(m.weight, mode="fan_out",
                                        nonlinearity="relu")
            elif isinstance(m, nn.BatchNorm2d):
                nn.init.ones_(m.weight)
                nn.init.zeros_(m.bias)

    def _make_layer(self, ch, blocks, stride=1):
        downsample = None
        if stride != 1 or self.in_ch != ch * Bottleneck.expansion:
            downsample = nn.Sequential(
                nn.Conv2d(self.in_ch, ch * Bottleneck.expansion, 1, stride=stride,
                          bias=False),
                nn.BatchNorm2d(ch * Bottleneck.expansion),
            )
        layers = [Bottleneck(self.in_ch, ch, stride, downsample)]
        self.in_ch = ch * Bottleneck.expansion
        for _ in range(1, blocks):
            layers.append(Bottleneck(self.in_ch, ch))
        return nn.Sequential(*layers)

    def forward(self, x):
        if getattr(self.bn1, "supports_fusion", False) and x.is_cuda:
            x = self.maxpool(self.bn1(self.conv1(x), relu=True))
        else:
            x = self.maxpool(self.relu(self.bn1(self.conv1(x))))
        x = self.layer4(self.layer3(self.layer2(self.layer1(x))))
        x = self.avgpool(x).flatten(1)
        return self.fc(x)


def resnet50(num_classes: int = 1000) -> ResNet:
    return ResNet([3, 4, 6, 3], num_classes)
