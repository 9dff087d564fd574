"""nn.Module wrappers over the framework ops.

These subclass the torch modules (identical parameter/buffer registration →
identical state_dict keys and init) but route forward through
ops/functional, which dispatches to the HIP kernels on GPU.  BN+ReLU pairs
are fused by Toy_Net using ``BatchNorm2d(..., fuse_relu=True)`` style flags
at the model level (see models/toy_net.py); the standalone modules keep
reference semantics.
"""
from __future__ import annotations

import os

import torch
import torch.nn as nn

from . import functional as F_ops


class Conv2d(nn.Conv2d):
    def forward(self, x):
        return F_ops.conv2d(x, self.weight, self.bias, self.stride, self.padding)


class Linear(nn.Linear):
    def forward(self, x):
        return F_ops.linear(x, self.weight, self.bias)


class _BatchNormBase:
    def _bn_forward(self, x, residual=None, pre_stats=None):
        self._check_input_dim(x)
        if self.training:
            if self.num_batches_tracked is not None:
                self.num_batches_tracked.add_(1)
        momentum = self.momentum if self.momentum is not None else 0.1
        fuse_relu = getattr(self, "fuse_relu", False)
        return F_ops.batch_norm(
            x, self.running_mean, self.running_var, self.weight, self.bias,
            self.training or not self.track_running_stats, momentum, self.eps,
            fuse_relu=fuse_relu, residual=residual, pre_stats=pre_stats)


class BatchNorm2d(nn.BatchNorm2d, _BatchNormBase):
    def __init__(self, *args, fuse_relu: bool = False, **kwargs):
        super().__init__(*args, **kwargs)
        self.fuse_relu = fuse_relu

    def forward(self, x, residual=None, pre_stats=None):
        # residual: optional skip tensor added before the (fused) ReLU —
        # relu(bn(x) + residual) in one kernel on GPU (ResNet blocks);
        # pre_stats: partial statistics from the producing conv's epilogue
        return self._bn_forward(x, residual, pre_stats)


class BatchNorm1d(nn.BatchNorm1d, _BatchNormBase):
    def __init__(self, *args, fuse_relu: bool = False, **kwargs):
        super().__init__(*args, **kwargs)
        self.fuse_relu = fuse_relu

    def forward(self, x):
        return self._bn_forward(x)


class ReLU(nn.ReLU):
    def forward(self, x):
        return F_ops.relu(x, inplace=self.inplace)


class MaxPool2d(nn.MaxPool2d):
    def forward(self, x):
        return F_ops.max_pool2d(x, self.kernel_size, self.stride, self.padding)


class AdaptiveAvgPool2d(nn.AdaptiveAvgPool2d):
    """Only the (1,1) global case is supported (the ResNet head)."""

    def forward(self, x):
        assert self.output_size in (1, (1, 1))
        return F_ops.global_avg_pool2d(x).reshape(x.shape[0], x.shape[1], 1, 1)


class Flatten(nn.Flatten):
    """flatten(start_dim=1) with a dedicated kernel at the NHWC conv→dense
    junction: ``nn.Flatten`` on a channels_last tensor materialises the
    NCHW semantic order through ATen's strided permute-copy (~23 µs each
    way per Toy_Net step); the HIP pair does the same transpose with
    coalesced access both directions (reference utils/model.py:23)."""

    def forward(self, x):
        if (self.start_dim == 1 and self.end_dim == -1 and x.dim() == 4
                and x.is_cuda and x.dtype == torch.bfloat16
                and 1 < x.size(2) * x.size(3) <= 64 and x.size(1) > 1
                and not x.is_contiguous()  # NCHW-contig flatten is a view
                and x.is_contiguous(memory_format=torch.channels_last)
                and os.environ.get("DDPX_NHWC_FLATTEN", "1") == "1"):
            return F_ops.nhwc_flatten(x)
        return super().forward(x)


class Identity(nn.Identity):
    pass
