"""(Wide)ResNet for the reference's wresnet benchmark family.

reference: benchmark/torch/model/wresnet.py (WideResNet-50/101 via
torchvision). torchvision is not in this image, so the bottleneck
architecture is implemented directly; `wresnet50()` matches the
torchvision `wide_resnet50_2` geometry (width_per_group=128).
"""
from __future__ import annotations

import torch
import torch.nn as nn
import torch.nn.functional as F


class Bottleneck(nn.Module):
    expansion = 4

    def __init__(self, inplanes, planes, stride=1, downsample=None,
                 base_width=64):
        super().__init__()
        width = int(planes * (base_width / 64.0))
        self.conv1 = nn.Conv2d(inplanes, width, 1, bias=False)
        self.bn1 = nn.BatchNorm2d(width)
        self.conv2 = nn.Conv2d(width, width, 3, stride=stride, padding=1,
                               bias=False)
        self.bn2 = nn.BatchNorm2d(width)
        self.conv3 = nn.Conv2d(width, planes * self.expansion, 1,
                               bias=False)
        self.bn3 = nn.BatchNorm2d(planes * self.expansion)
        self.downsample = downsample
        self.relu = nn.ReLU(inplace=False)

    def forward(self, x):
        idt = x
        out = self.relu(self.bn1(self.conv1(x)))
        out = self.relu(self.bn2(self.conv2(out)))
        out = self.bn3(self.conv3(out))
        if self.downsample is not None:
            idt = self.downsample(x)
        return self.relu(out + idt)


class ResNet(nn.Module):
    def __init__(self, layers, base_width=64, n_classes=1000):
        super().__init__()
        self.inplanes = 64
        self.base_width = base_width
        self.conv1 = nn.Conv2d(3, 64, 7, stride=2, padding=3, bias=False)
        self.bn1 = nn.BatchNorm2d(64)
        self.relu = nn.ReLU(inplace=False)
        self.maxpool = nn.MaxPool2d(3, stride=2, padding=1)
        self.layer1 = self._make_layer(64, layers[0])
        self.layer2 = self._make_layer(128, layers[1], 2)
        self.layer3 = self._make_layer(256, layers[2], 2)
        self.layer4 = self._make_layer(512, layers[3], 2)
        self.avgpool = nn.AdaptiveAvgPool2d(1)
        self.fc = nn.Linear(512 * Bottleneck.expansion, n_classes)

    def _make_layer(self, planes, blocks, stride=1):
        downsample = None
        out_planes = planes * Bottleneck.expansion
        if stride != 1 or self.inplanes != out_planes:
            downsample = nn.Sequential(
                nn.Conv2d(self.inplanes, out_planes, 1, stride=stride,
                          bias=False),
                nn.BatchNorm2d(out_planes))
        layers = [Bottleneck(self.inplanes, planes, stride, downsample,
                             self.base_width)]
        self.inplanes = out_planes
        for _ in range(1, blocks):
            layers.append(Bottleneck(self.inplanes, planes,
                                     base_width=self.base_width))
        return nn.Sequential(*layers)

    def forward(self, x):
        x = self.maxpool(self.relu(self.bn1(self.conv1(x))))
        x = self.layer4(self.layer3(self.layer2(self.layer1(x))))
        return self.fc(self.avgpool(x).flatten(1))


def resnet18_shape():
    """Small variant for CPU plumbing tests (BASELINE config #1)."""
    return ResNet([1, 1, 1, 1], n_classes=10)


def resnet50():
    return ResNet([3, 4, 6, 3])


def resnet101():
    return ResNet([3, 4, 23, 3])


def wresnet50():
    """wide_resnet50_2 geometry (reference bench_case.py:17-20)."""
    return ResNet([3, 4, 6, 3], base_width=128)


def wresnet101():
    return ResNet([3, 4, 23, 3], base_width=128)


def resnet_train_step(model, opt, x, y):
    dev = "cuda" if x.is_cuda else "cpu"
    with torch.autocast(device_type=dev, dtype=torch.bfloat16,
                        enabled=x.is_cuda):
        logits = model(x)
    loss = F.cross_entropy(logits.float(), y)
    loss.backward()
    opt.step()
    opt.zero_grad(True)
    return loss
