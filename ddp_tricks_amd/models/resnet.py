"""ResNet-18/34/50 on the framework's HIP-backed modules (BASELINE.json
configs 4-5).

MI355X-first block design: every BN+ReLU pair runs as one fused HIP kernel,
and each block's closing ``relu(bn(out) + identity)`` is a single kernel via
``BatchNorm2d.forward(x, residual=...)`` — the skip add never materialises a
separate elementwise pass, and its gradient (the relu-masked dy) comes out of
the same backward kernel.  Activations are bf16 channels_last throughout.

state_dict keys follow the torchvision naming (conv1, bn1, layer{1-4}.{i}.*,
fc) so reference-style ``torch.save(module.state_dict())`` checkpoints are
interchangeable with stock definitions of the same architecture.

``cifar_stem=True`` (ResNet-18 CIFAR-10 config) replaces the 7x7/2 stem +
3x3/2 maxpool with a 3x3/1 conv, the standard CIFAR variant.
"""
from __future__ import annotations

import torch
import torch.nn as nn

from ..ops.functional import conv_bn
from ..ops.modules import (
    AdaptiveAvgPool2d, BatchNorm2d, Conv2d, Identity, Linear, MaxPool2d,
)


def conv3x3(cin, cout, stride=1):
    return Conv2d(cin, cout, 3, stride=stride, padding=1, bias=False)


def conv1x1(cin, cout, stride=1):
    return Conv2d(cin, cout, 1, stride=stride, bias=False)


class BasicBlock(nn.Module):
    expansion = 1

    def __init__(self, inplanes, planes, stride=1, downsample=None):
        super().__init__()
        self.conv1 = conv3x3(inplanes, planes, stride)
        self.bn1 = BatchNorm2d(planes, fuse_relu=True)
        self.relu = Identity()  # fused into bn1/bn2 epilogues
        self.conv2 = conv3x3(planes, planes)
        self.bn2 = BatchNorm2d(planes, fuse_relu=True)
        self.downsample = downsample
        self.stride = stride

    def forward(self, x):
        identity = x if self.downsample is None else self.downsample(x)
        out = conv_bn(self.conv1, self.bn1, x)
        return conv_bn(self.conv2, self.bn2, out, residual=identity)


class Bottleneck(nn.Module):
    expansion = 4

    def __init__(self, inplanes, planes, stride=1, downsample=None):
        super().__init__()
        self.conv1 = conv1x1(inplanes, planes)
        self.bn1 = BatchNorm2d(planes, fuse_relu=True)
        self.conv2 = conv3x3(planes, planes, stride)
        self.bn2 = BatchNorm2d(planes, fuse_relu=True)
        self.conv3 = conv1x1(planes, planes * self.expansion)
        self.bn3 = BatchNorm2d(planes * self.expansion, fuse_relu=True)
        self.relu = Identity()
        self.downsample = downsample
        self.stride = stride

    def forward(self, x):
        identity = x if self.downsample is None else self.downsample(x)
        out = conv_bn(self.conv1, self.bn1, x)
        out = conv_bn(self.conv2, self.bn2, out)
        return conv_bn(self.conv3, self.bn3, out, residual=identity)


class ResNet(nn.Module):
    def __init__(self, block, layers, num_classes=1000, cifar_stem=False):
        super().__init__()
        self.inplanes = 64
        if cifar_stem:
            self.conv1 = Conv2d(3, 64, 3, stride=1, padding=1, bias=False)
            self.maxpool = Identity()
        else:
            self.conv1 = Conv2d(3, 64, 7, stride=2, padding=3, bias=False)
            self.maxpool = MaxPool2d(kernel_size=3, stride=2, padding=1)
        self.bn1 = BatchNorm2d(64, fuse_relu=True)
        self.relu = Identity()
        self.layer1 = self._make_layer(block, 64, layers[0])
        self.layer2 = self._make_layer(block, 128, layers[1], stride=2)
        self.layer3 = self._make_layer(block, 256, layers[2], stride=2)
        self.layer4 = self._make_layer(block, 512, layers[3], stride=2)
        self.avgpool = AdaptiveAvgPool2d((1, 1))
        self.fc = Linear(512 * block.expansion, num_classes)

        for m in self.modules():
            if isinstance(m, Conv2d):
                nn.init.kaiming_normal_(m.weight, mode="fan_out",
                                        nonlinearity="relu")
            elif isinstance(m, BatchNorm2d):
                nn.init.constant_(m.weight, 1)
                nn.init.constant_(m.bias, 0)

    def _make_layer(self, block, planes, blocks, stride=1):
        downsample = None
        if stride != 1 or self.inplanes != planes * block.expansion:
            downsample = nn.Sequential(
                conv1x1(self.inplanes, planes * block.expansion, stride),
                BatchNorm2d(planes * block.expansion),
            )
        strides = [stride] + [1] * (blocks - 1)
        mods = []
        for s in strides:
            mods.append(block(self.inplanes, planes, s, downsample))
            downsample = None
            self.inplanes = planes * block.expansion
        return nn.Sequential(*mods)

    def forward(self, x: torch.Tensor) -> torch.Tensor:
        x = self.maxpool(conv_bn(self.conv1, self.bn1, x))
        x = self.layer4(self.layer3(self.layer2(self.layer1(x))))
        x = self.avgpool(x)
        return self.fc(torch.flatten(x, 1))


def resnet18(num_classes=1000, cifar_stem=False):
    return ResNet(BasicBlock, [2, 2, 2, 2], num_classes, cifar_stem)


def resnet34(num_classes=1000, cifar_stem=False):
    return ResNet(BasicBlock, [3, 4, 6, 3], num_classes, cifar_stem)


def resnet50(num_classes=1000, cifar_stem=False):
    return ResNet(Bottleneck, [3, 4, 6, 3], num_classes, cifar_stem)
