"""Toy_Net — the reference's MNIST CNN, built on the framework's HIP-backed
modules.

Architecture and state_dict layout match the reference exactly
(reference utils/model.py:4-34; 37-key fp32 state_dict):
conv.{0,3,7,10} Conv2d / conv.{1,4,8,11} BatchNorm2d / conv.{6,13} MaxPool2d,
dense.1 Linear(8192,512) / dense.2 BatchNorm1d / dense.4 Linear(512,10).

ReLUs are fused into the preceding BatchNorm's HIP epilogue
(fuse_relu=True); the Sequential keeps an Identity at each ReLU index so
indices — and therefore parameter/buffer keys — are unchanged (ReLU has no
state).
"""
from __future__ import annotations

import torch
import torch.nn as nn

from ..ops.modules import (
    BatchNorm1d, BatchNorm2d, Conv2d, Flatten, Identity, Linear, MaxPool2d,
)


class Toy_Net(nn.Module):
    def __init__(self):
        super().__init__()
        self.conv = nn.Sequential(
            Conv2d(1, 64, 3),
            BatchNorm2d(64, fuse_relu=True),
            Identity(),
            Conv2d(64, 128, 3),
            BatchNorm2d(128, fuse_relu=True),
            Identity(),
            MaxPool2d(2),
            Conv2d(128, 256, 3),
            BatchNorm2d(256, fuse_relu=True),
            Identity(),
            Conv2d(256, 512, 3),
            BatchNorm2d(512, fuse_relu=True),
            Identity(),
            MaxPool2d(2),
        )
        self.dense = nn.Sequential(
            Flatten(),
            Linear(8192, 512),
            BatchNorm1d(512, fuse_relu=True),
            Identity(),
            Linear(512, 10),
        )

    def forward(self, x: torch.Tensor) -> torch.Tensor:
        # walk the Sequentials explicitly so each conv->BN pair runs the
        # fused stats path (parameter keys are unchanged — same children)
        from ..ops.functional import conv_bn
        c = self.conv
        out = conv_bn(c[0], c[1], x)
        out = conv_bn(c[3], c[4], out)
        out = c[6](out)
        out = conv_bn(c[7], c[8], out)
        out = conv_bn(c[10], c[11], out)
        out = c[13](out)
        out = self.dense(out)
        return out
