"""Autograd-integrated ops: HIP/CDNA4 kernels on GPU, torch reference on CPU.

Each op is a torch.autograd.Function whose forward/backward call the
_hip_ops extension (bf16 compute, fp32 accumulate — the framework's amp-O1
policy: conv/linear whitelisted to bf16, BN stats/CE in fp32; reference
behavior via apex amp.initialize at utils/train.py:58).  On CPU (the
no-GPU test tier) the same Python entry points run the equivalent torch
ops in fp32 so they double as the numerics oracle.

Weight bf16 casts are cached per parameter version (one cast per optimizer
step, like apex's per-iteration cast cache).
"""
from __future__ import annotations

from typing import Optional, Tuple

import torch
import torch.nn.functional as F

from . import require_ext_for
from .. import amp as amp_mod


# --------------------------------------------------------------- casting ---

_BF16_CACHE = {}  # param -> (version, bf16 tensor)


def bf16_weight(w: torch.Tensor) -> torch.Tensor:
    """Cached fp32->bf16 cast of a master weight (re-cast after each step)."""
    if w.dtype == torch.bfloat16:
        return w
    ent = _BF16_CACHE.get(w)
    if ent is not None and ent[0] == w._version:
        return ent[1]
    wb = w.detach().to(torch.bfloat16)
    _BF16_CACHE[w] = (w._version, wb)
    return wb


def clear_weight_cache() -> None:
    _BF16_CACHE.clear()


def _to_bf16(x: torch.Tensor) -> torch.Tensor:
    return x if x.dtype == torch.bfloat16 else x.to(torch.bfloat16)


# ----------------------------------------------------------------- conv2d ---

class _HIPConv2d(torch.autograd.Function):
    @staticmethod
    def forward(ctx, x, weight, bias, stride, padding):
        ext = require_ext_for(x)
        xb = _to_bf16(x.contiguous())
        wb = bf16_weight(weight)
        y = ext.conv2d_fwd(xb, wb, stride, padding)
        if bias is not None:
            y += bias.to(y.dtype).view(1, -1, 1, 1)
        ctx.save_for_backward(xb, wb)
        ctx.meta = (stride, padding, x.dtype, weight.shape, bias is not None)
        return y

    @staticmethod
    def backward(ctx, dy):
        xb, wb = ctx.saved_tensors
        stride, padding, x_dtype, w_shape, has_bias = ctx.meta
        ext = require_ext_for(dy)
        dyb = _to_bf16(dy.contiguous())
        dx = dw = db = None
        if ctx.needs_input_grad[0]:
            dx = ext.conv2d_dgrad(dyb, wb, xb.shape[2], xb.shape[3], stride, padding)
            if x_dtype != dx.dtype:
                dx = dx.to(x_dtype)
        if ctx.needs_input_grad[1]:
            dw = ext.conv2d_wgrad(dyb, xb, w_shape[2], w_shape[3], stride, padding)
        if has_bias and ctx.needs_input_grad[2]:
            db = dyb.float().sum(dim=(0, 2, 3))
        return dx, dw, db, None, None


def conv2d(x, weight, bias=None, stride=1, padding=0):
    stride = stride[0] if isinstance(stride, (tuple, list)) else stride
    padding = padding[0] if isinstance(padding, (tuple, list)) else padding
    if x.is_cuda and require_ext_for(x) is not None:
        return _HIPConv2d.apply(x, weight, bias, stride, padding)
    if x.is_cuda and amp_mod.is_enabled():  # explicit torch-fallback bring-up path
        return F.conv2d(_to_bf16(x), bf16_weight(weight),
                        bias.to(torch.bfloat16) if bias is not None else None,
                        stride, padding)
    return F.conv2d(x, weight, bias, stride, padding)


# ----------------------------------------------------------------- linear ---

class _HIPLinear(torch.autograd.Function):
    @staticmethod
    def forward(ctx, x, weight, bias):
        ext = require_ext_for(x)
        xb = _to_bf16(x.contiguous())
        wb = bf16_weight(weight)
        y = ext.linear_fwd(xb, wb, bias if bias is not None else None)
        ctx.save_for_backward(xb, wb)
        ctx.meta = (x.dtype, bias is not None)
        return y

    @staticmethod
    def backward(ctx, dy):
        xb, wb = ctx.saved_tensors
        x_dtype, has_bias = ctx.meta
        ext = require_ext_for(dy)
        dyb = _to_bf16(dy.contiguous())
        dx = dw = db = None
        if ctx.needs_input_grad[0]:
            dx = ext.linear_dgrad(dyb, wb)          # dy [M,N] @ W [N,K] -> [M,K]
            if dx.dtype != x_dtype and x_dtype == torch.float32:
                pass  # keep bf16 grads flowing between bf16 layers
        if ctx.needs_input_grad[1]:
            dw = ext.linear_wgrad(dyb, xb)          # dy^T [N,M] @ X [M,K] -> fp32
        if has_bias and ctx.needs_input_grad[2]:
            db = dyb.float().sum(dim=0)
        return dx, dw, db


def linear(x, weight, bias=None):
    if x.is_cuda and require_ext_for(x) is not None:
        return _HIPLinear.apply(x, weight, bias)
    if x.is_cuda and amp_mod.is_enabled():
        return F.linear(_to_bf16(x), bf16_weight(weight),
                        bias.to(torch.bfloat16) if bias is not None else None)
    return F.linear(x, weight, bias)


# ------------------------------------------------------------- batch norm ---

class _HIPBatchNorm(torch.autograd.Function):
    """BatchNorm (2d NCHW or 1d NC) with optional fused ReLU.

    Stats in fp32 over bf16 activations; running stats updated in-place
    (rank-local, broadcast from rank 0 each forward by the DDP wrapper —
    reference DDP broadcast_buffers semantics).
    """

    @staticmethod
    def forward(ctx, x, weight, bias, running_mean, running_var,
                training, momentum, eps, fuse_relu):
        ext = require_ext_for(x)
        xb = _to_bf16(x.contiguous())
        if training:
            y, save_mean, save_invstd = ext.bn_fwd_train(
                xb, weight, bias, running_mean, running_var,
                momentum, eps, fuse_relu)
            ctx.save_for_backward(xb, weight, save_mean, save_invstd, y)
        else:
            y = ext.bn_fwd_eval(xb, weight, bias, running_mean, running_var,
                                eps, fuse_relu)
        ctx.fuse_relu = fuse_relu
        ctx.training = training
        ctx.x_dtype = x.dtype
        return y

    @staticmethod
    def backward(ctx, dy):
        assert ctx.training, "backward through eval-mode BN is unsupported"
        xb, weight, save_mean, save_invstd, y = ctx.saved_tensors
        ext = require_ext_for(dy)
        dyb = _to_bf16(dy.contiguous())
        dx, dweight, dbias = ext.bn_bwd(xb, dyb, weight, save_mean,
                                        save_invstd, y, ctx.fuse_relu)
        if dx.dtype != ctx.x_dtype and ctx.x_dtype == torch.float32:
            dx = dx.float()
        return dx, dweight, dbias, None, None, None, None, None, None


def batch_norm(x, running_mean, running_var, weight, bias,
               training, momentum, eps, fuse_relu=False):
    if x.is_cuda and require_ext_for(x) is not None:
        return _HIPBatchNorm.apply(x, weight, bias, running_mean, running_var,
                                   training, momentum, eps, fuse_relu)
    xf = x.float() if x.dtype != torch.float32 else x
    y = F.batch_norm(xf, running_mean, running_var, weight, bias,
                     training, momentum, eps)
    if fuse_relu:
        y = F.relu(y)
    return y.to(x.dtype) if x.is_cuda and amp_mod.is_enabled() else y


# ---------------------------------------------------------------- pooling ---

class _HIPMaxPool2x2(torch.autograd.Function):
    @staticmethod
    def forward(ctx, x):
        ext = require_ext_for(x)
        xb = _to_bf16(x.contiguous())
        y, idx = ext.maxpool2x2_fwd(xb)
        ctx.save_for_backward(idx)
        ctx.x_shape = xb.shape
        return y

    @staticmethod
    def backward(ctx, dy):
        (idx,) = ctx.saved_tensors
        ext = require_ext_for(dy)
        dx = ext.maxpool2x2_bwd(_to_bf16(dy.contiguous()), idx,
                                ctx.x_shape[2], ctx.x_shape[3])
        return dx


def max_pool2d(x, kernel_size=2, stride=None):
    ks = kernel_size[0] if isinstance(kernel_size, (tuple, list)) else kernel_size
    st = stride or ks
    st = st[0] if isinstance(st, (tuple, list)) else st
    if x.is_cuda and require_ext_for(x) is not None and ks == 2 and st == 2:
        return _HIPMaxPool2x2.apply(x)
    return F.max_pool2d(x, kernel_size, stride)


def relu(x, inplace=False):
    # standalone ReLU (non-fused path); fused into BN epilogue on GPU
    return F.relu(x, inplace=inplace)


# ------------------------------------------------------------ cross entropy ---

class _HIPCrossEntropy(torch.autograd.Function):
    @staticmethod
    def forward(ctx, logits, target):
        ext = require_ext_for(logits)
        lb = logits.contiguous()
        loss, lse = ext.ce_fwd(lb, target)
        ctx.save_for_backward(lb, target, lse)
        return loss

    @staticmethod
    def backward(ctx, dloss):
        logits, target, lse = ctx.saved_tensors
        ext = require_ext_for(logits)
        dlogits = ext.ce_bwd(logits, target, lse, dloss)
        return dlogits, None


def cross_entropy_loss(logits, target):
    """Mean-reduced CE (the engine applies the reference's extra /B on top —
    reference utils/process.py:22-23)."""
    if logits.is_cuda and require_ext_for(logits) is not None:
        return _HIPCrossEntropy.apply(logits, target)
    return F.cross_entropy(logits.float(), target)


def argmax_correct(logits: torch.Tensor, target: torch.Tensor) -> torch.Tensor:
    """Number of rows whose argmax equals target — device-side i64 scalar."""
    if logits.is_cuda and require_ext_for(logits) is not None:
        ext = require_ext_for(logits)
        return ext.argmax_correct(logits.contiguous(), target)
    return (torch.argmax(logits, dim=1) == target).sum()
