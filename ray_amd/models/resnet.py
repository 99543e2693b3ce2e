"""ResNet-50 for the Train DDP bench (north-star config 2: ResNet-50
DDP bf16, synthetic ImageNet). Standard bottleneck architecture,
written here because torchvision is not in the image; GEMM/conv via
MIOpen through torch, input normalize via the HIP img_normalize kernel.
"""
from __future__ import annotations

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
        self.conv3 = nn.Conv2d(ch, ch * 4, 1, bias=False)
        self.bn3 = nn.BatchNorm2d(ch * 4)
        self.relu = nn.ReLU(inplace=True)
        self.downsample = downsample

    def forward(self, x):
        idn = x
        out = self.relu(self.bn1(self.conv1(x)))
        out = self.relu(self.bn2(self.conv2(out)))
        out = self.bn3(self.conv3(out))
        if self.downsample is not None:
            idn = self.downsample(x)
        return self.relu(out + idn)


class ResNet50(nn.Module):
    def __init__(self, num_classes=1000):
        super().__init__()
        self.in_ch = 64
        self.conv1 = nn.Conv2d(3, 64, 7, stride=2, padding=3, bias=False)
        self.bn1 = nn.BatchNorm2d(64)
        self.relu = nn.ReLU(inplace=True)
        self.maxpool = nn.MaxPool2d(3, stride=2, padding=1)
        self.layer1 = self._make_layer(64, 3)
        self.layer2 = self._make_layer(128, 4, stride=2)
        self.layer3 = self._make_layer(256, 6, stride=2)
        self.layer4 = self._make_layer(512, 3, stride=2)
        self.avgpool = nn.AdaptiveAvgPool2d(1)
        self.fc = nn.Linear(2048, num_classes)

    def _make_layer(self, ch, blocks, stride=1):
        downsample = None
        if stride != 1 or self.in_ch != ch * 4:
            downsample = nn.Sequential(
                nn.Conv2d(self.in_ch, ch * 4, 1, stride=stride, bias=False),
                nn.BatchNorm2d(ch * 4),
            )
        layers = [Bottleneck(self.in_ch, ch, stride, downsample)]
        self.in_ch = ch * 4
        layers += [Bottleneck(self.in_ch, ch) for _ in range(blocks - 1)]
        return nn.Sequential(*layers)

    def forward(self, x):
        x = self.maxpool(self.relu(self.bn1(self.conv1(x))))
        x = self.layer4(self.layer3(self.layer2(self.layer1(x))))
        x = self.avgpool(x).flatten(1)
        return self.fc(x)

    def num_params(self):
        return sum(p.numel() for p in self.parameters())
