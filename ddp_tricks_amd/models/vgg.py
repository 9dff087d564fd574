"""VGG-16 with BatchNorm, built on the framework's HIP-backed modules.

Zoo extension beyond the reference's Toy_Net (reference utils/model.py is
the only model there): the classic 13-conv/3-linear stack exercises the
conv kernels at deep same-C 3×3 chains and the GEMM ladder at the wide
classifier shapes ([B, 512·7·7] × [25088, 4096] at 224 input).  Same
conv→BN fused-stats path as ResNet (conv_bn), ReLUs fused into BN
epilogues, 2×2 maxpool on the dedicated kernel.

``cifar_head=True`` shrinks the classifier for 32×32 inputs
(512·1·1 → 512 hidden), mirroring common CIFAR-VGG practice.

State-dict layout follows torchvision's vgg16_bn naming
(features.N.weight / classifier.N.weight) so torchvision-trained
checkpoints map by key.
"""
from __future__ import annotations

import torch
import torch.nn as nn

from ..ops.functional import conv_bn
from ..ops.modules import (BatchNorm2d, Conv2d, Flatten, Identity,
                           Linear, MaxPool2d)

# vgg16 plan: channel widths with 'M' = 2x2/2 maxpool
_CFG16 = [64, 64, "M", 128, 128, "M", 256, 256, 256, "M",
          512, 512, 512, "M", 512, 512, 512, "M"]


class VGG16BN(nn.Module):
    def __init__(self, num_classes: int = 1000, cifar_head: bool = False,
                 dropout: float = 0.0):
        super().__init__()
        layers = []
        cin = 3
        for v in _CFG16:
            if v == "M":
                layers.append(MaxPool2d(kernel_size=2, stride=2))
            else:
                layers.append(Conv2d(cin, v, 3, padding=1))
                layers.append(BatchNorm2d(v, fuse_relu=True))
                layers.append(Identity())   # ReLU slot (fused; keeps
                cin = v                     # torchvision index layout)
        self.features = nn.Sequential(*layers)
        feat = 512 if cifar_head else 512 * 7 * 7
        hidden = 512 if cifar_head else 4096
        self.classifier = nn.Sequential(
            Linear(feat, hidden),
            Identity(),                     # ReLU (applied in forward)
            nn.Dropout(p=dropout),
            Linear(hidden, hidden),
            Identity(),
            nn.Dropout(p=dropout),
            Linear(hidden, num_classes),
        )
        self.flatten = Flatten()   # NHWC-aware at the conv->dense junction
        self.cifar_head = cifar_head
        for m in self.modules():
            if isinstance(m, Conv2d):
                nn.init.kaiming_normal_(m.weight, mode="fan_out",
                                        nonlinearity="relu")
                if m.bias is not None:
                    nn.init.constant_(m.bias, 0)
            elif isinstance(m, BatchNorm2d):
                nn.init.constant_(m.weight, 1)
                nn.init.constant_(m.bias, 0)

    def forward(self, x: torch.Tensor) -> torch.Tensor:
        # walk features explicitly so each conv→BN pair takes the fused
        # stats path (indices — and state_dict keys — are unchanged)
        f = self.features
        i = 0
        while i < len(f):
            if isinstance(f[i], Conv2d):
                x = conv_bn(f[i], f[i + 1], x)
                i += 3
            else:
                x = f[i](x)
                i += 1
        x = self.flatten(x)
        c = self.classifier
        x = torch.relu(c[0](x))
        x = c[2](x)
        x = torch.relu(c[3](x))
        x = c[5](x)
        return c[6](x)


def vgg16_bn(num_classes=1000, cifar_head=False, dropout=0.0):
    return VGG16BN(num_classes, cifar_head, dropout)
