"""Autograd-integrated ops: HIP/CDNA4 kernels on GPU, torch reference on CPU.

GPU data layout: activations are bf16 **channels_last** (NHWC — channels
fastest, the MFMA implicit-GEMM-friendly layout on CDNA4); weights are
cached per-parameter-version as bf16 channels_last (KRSC) plus the
transformed WT2[C, R·S·K] operand for backward-data.  Statistics, biases,
losses and weight gradients are fp32 (the framework's amp-O1 policy —
reference behavior via apex at utils/train.py:58).

On CPU (the no-GPU test tier) the same entry points run equivalent torch
ops in fp32, doubling as the numerics oracle the GPU tests compare against.
"""
from __future__ import annotations

import os

import torch
import torch.nn.functional as F

from . import require_ext_for
from .. import amp as amp_mod

CL = torch.channels_last


# --------------------------------------------------------------- casting ---

_WCACHE = {}  # param -> {"version": v, variant: tensor}


def weight_variant(w: torch.Tensor, variant: str) -> torch.Tensor:
    """Per-step cached transforms of fp32 master weights.

    variants: "flat"     — bf16, native contiguous (linear [N,K])
              "nhwc"     — bf16 channels_last (conv KRSC memory)
              "wt2"      — bf16 [C, R*S*K] (conv dgrad operand)
              "nhwc_p8"  — bf16 channels_last with Cin zero-padded to 8
                           (MFMA path for C<8 stems, e.g. ResNet RGB)
              "wt2_p8"   — dgrad operand of the padded weights
    """
    ent = _WCACHE.get(w)
    if ent is None or ent["version"] != w._version:
        ent = {"version": w._version}
        _WCACHE[w] = ent
    if variant not in ent:
        ext = require_ext_for(w) if w.is_cuda else None
        if (ext is not None and w.dtype == torch.float32
                and variant in ("nhwc", "wt2", "nhwc_p8", "wt2_p8")):
            # one fused kernel fills BOTH conv operand layouts per step
            pad8 = variant.endswith("_p8")
            sfx = "_p8" if pad8 else ""
            nhwc, wt2 = ext.pack_conv_weight(w.detach().contiguous(), pad8,
                                             True)
            ent["nhwc" + sfx] = nhwc
            ent["wt2" + sfx] = wt2
            return ent[variant]
        wb = w.detach().to(torch.bfloat16)
        if variant == "flat":
            ent[variant] = wb.contiguous()
        elif variant == "nhwc":
            ent[variant] = wb.contiguous(memory_format=CL)
        elif variant == "wt2":
            K, C, R, S = w.shape
            ent[variant] = wb.permute(1, 2, 3, 0).reshape(C, R * S * K).contiguous()
        elif variant in ("nhwc_p8", "wt2_p8"):
            K, C, R, S = w.shape
            wp = torch.zeros(K, 8, R, S, dtype=torch.bfloat16, device=w.device)
            wp[:, :C] = wb
            if variant == "nhwc_p8":
                ent[variant] = wp.contiguous(memory_format=CL)
            else:
                ent[variant] = wp.permute(1, 2, 3, 0).reshape(8, R * S * K).contiguous()
        else:
            raise KeyError(variant)
    return ent[variant]


def clear_weight_cache() -> None:
    _WCACHE.clear()


# ------------------------------------------------------- skip-grad fusion ---

def _skipfuse_on() -> bool:
    """ResNet skip-gradient fusion: at a residual
    junction the block input is consumed twice — by the first conv and as
    the BN epilogue's residual — so autograd sums conv-dgrad dx with the
    BN's dresid in a separate ATen add pass (4.3% of the ResNet-50 step).
    With fusion on, the BN stashes dresid in a box attached to the shared
    tensor object and returns None; the conv's dgrad (which runs strictly
    later — it is upstream in the backward order) streams it into its
    epilogue (dx += dresid in-flight)."""
    return os.environ.get("DDPX_SKIPFUSE", "1") == "1"


def _attach_skipbox(ctx, x):
    """Conv-side: ensure the input tensor carries a junction box and
    remember it on the ctx (object attribute — allocator-reuse immune)."""
    box = getattr(x, "_ddpx_skipbox", None)
    if box is None:
        box = {}
        x._ddpx_skipbox = box
    ctx.skip_box = box


def _pop_skip_grad(ctx):
    box = getattr(ctx, "skip_box", None)
    if box is None:
        return None
    return box.pop("g", None)


def bf16_weight(w: torch.Tensor) -> torch.Tensor:
    if w.dtype == torch.bfloat16:
        return w
    return weight_variant(w, "flat")


def _to_bf16(x: torch.Tensor) -> torch.Tensor:
    return x if x.dtype == torch.bfloat16 else x.to(torch.bfloat16)


def _chlast(x: torch.Tensor) -> torch.Tensor:
    if x.dim() == 4:
        return x.contiguous(memory_format=CL)
    return x.contiguous()


def _nhwc_2d(x: torch.Tensor) -> torch.Tensor:
    """Zero-copy [N*H*W, C] view of a channels_last 4D tensor."""
    return x.permute(0, 2, 3, 1).reshape(-1, x.shape[1])


# ----------------------------------------------------------------- conv2d ---

def _pad8(xb: torch.Tensor) -> torch.Tensor:
    """Zero-pad channels to 8 (channels_last) so C<8 inputs can take the
    MFMA implicit-GEMM path — one fused kernel on GPU."""
    ext = require_ext_for(xb)
    if ext is not None:
        return ext.pad8_channels(xb)
    N, C, H, W = xb.shape
    xp = torch.zeros(N, 8, H, W, dtype=xb.dtype, device=xb.device)
    xp = xp.contiguous(memory_format=CL)
    xp[:, :C] = xb
    return xp


def _use_pad8(C: int, H: int, W: int) -> bool:
    return C < 8 and H * W >= 1024


def _conv_fwd_prep(ctx, x, weight, bias, stride, padding):
    """Shared operand prep for the conv autograd functions; returns
    (ext, xb_for_kernel, wb, bias_f) and fills ctx bookkeeping."""
    ext = require_ext_for(x)
    xb = _chlast(_to_bf16(x))
    C_in = weight.shape[1]
    if C_in < 8:
        pad8_bwd = _use_pad8(C_in, xb.shape[2], xb.shape[3])
        xp = _pad8(xb)
        wb = weight_variant(weight, "nhwc_p8")
        ctx.save_for_backward(xp if pad8_bwd else xb, weight)
        ctx.meta = (stride, padding, x.dtype, weight.shape,
                    bias is not None, pad8_bwd)
        return ext, xp, wb, (bias if bias is None else bias.detach().float())
    wb = weight_variant(weight, "nhwc")
    ctx.save_for_backward(xb, weight)
    ctx.meta = (stride, padding, x.dtype, weight.shape, bias is not None,
                False)
    if stride == 1 and x.is_cuda and _skipfuse_on():
        _attach_skipbox(ctx, x)
    if stride == 1 and xb is x \
            and os.environ.get("DDPX_BNFUSE", "0") == "1":
        # producer-side BN-backward fusion (attr set by batch_norm).
        # OPT-IN: measured net-negative at ResNet-50 (8.38k vs 8.72k
        # img/s) — the epilogue's +32 VGPR (partial accumulators +
        # coefficient cache) pushes the dgrad to the 512-reg cliff while
        # the skipped BN partial pass only saves ~1 activation read.
        ctx.bn_box = getattr(x, "_ddpx_bnbwd", None)
    return ext, xb, wb, (bias if bias is None else bias.detach().float())


def _conv_bwd_impl(ctx, dy):
    xb, weight = ctx.saved_tensors
    stride, padding, x_dtype, w_shape, has_bias, pad8 = ctx.meta
    ext = require_ext_for(dy)
    dyb = _chlast(_to_bf16(dy))
    K, C, R, S = w_shape
    dx = dw = db = None
    if ctx.needs_input_grad[0]:
        wt2 = weight_variant(weight, "wt2_p8" if pad8 else "wt2")
        box = getattr(ctx, "bn_box", None)
        # stashed junction skip-grad (only ever set for stride-1 non-pad8
        # convs — see _attach_skipbox call sites)
        g = _pop_skip_grad(ctx) if (stride == 1 and not pad8) else None
        if (box is not None and not pad8 and stride == 1
                and x_dtype != torch.float32 and g is None):
            dx, slab = ext.conv2d_dgrad_bn(
                dyb, wt2, xb.shape[0], xb.shape[1], xb.shape[2], xb.shape[3],
                R, S, padding, box["x"], box["mask"], box["mean"],
                box["invstd"])
            box["slab"] = slab
            box["dx_ref"] = dx
        else:
            dx = ext.conv2d_dgrad(dyb, wt2, xb.shape[0], xb.shape[1],
                                  xb.shape[2], xb.shape[3], R, S,
                                  stride, padding, g)
            if pad8:
                dx = dx[:, :C]
            if x_dtype == torch.float32:
                dx = dx.float()
    if ctx.needs_input_grad[1]:
        dw = ext.conv2d_wgrad(dyb, xb, R, S, stride, padding)
        if pad8:
            dw = dw[:, :C].contiguous()
    if has_bias and ctx.needs_input_grad[2]:
        db = ext.col_sum(_nhwc_2d(dyb))
    return dx, dw, db


class _HIPConv2dStats(torch.autograd.Function):
    """Conv forward that also emits the consuming BN's partial statistics
    from the epilogue (stats output is non-differentiable)."""

    @staticmethod
    def forward(ctx, x, weight, bias, stride, padding):
        ext, xk, wb, bias_f = _conv_fwd_prep(ctx, x, weight, bias, stride,
                                             padding)
        y, stats = ext.conv2d_fwd_stats(xk, wb, bias_f, stride, padding)
        ctx.mark_non_differentiable(stats)
        return y, stats

    @staticmethod
    def backward(ctx, dy, dstats):
        dx, dw, db = _conv_bwd_impl(ctx, dy)
        return dx, dw, db, None, None


class _HIPConv2d(torch.autograd.Function):
    @staticmethod
    def forward(ctx, x, weight, bias, stride, padding):
        ext = require_ext_for(x)
        xb = _chlast(_to_bf16(x))
        C_in = weight.shape[1]
        if C_in < 8:
            # fwd always takes the MFMA path on zero-padded channels
            # (measured faster than the direct VALU kernel even at MNIST
            # size: 48 vs 64 us).  wgrad for SMALL spatial inputs stays on
            # the specialized C<8 kernels (measured: MFMA wgrad at Kgemm=72
            # is 362 us vs 132 us for the sliding-window kernel), so the
            # UNPADDED x is saved in that case.
            pad8_bwd = _use_pad8(C_in, xb.shape[2], xb.shape[3])
            xp = _pad8(xb)
            wb = weight_variant(weight, "nhwc_p8")
            y = ext.conv2d_fwd(xp, wb, bias if bias is None
                               else bias.detach().float(), stride, padding)
            ctx.save_for_backward(xp if pad8_bwd else xb, weight)
            ctx.meta = (stride, padding, x.dtype, weight.shape,
                        bias is not None, pad8_bwd)
            return y
        wb = weight_variant(weight, "nhwc")
        bias_f = bias if bias is None else bias.detach().float()
        y = ext.conv2d_fwd(xb, wb, bias_f, stride, padding)
        ctx.save_for_backward(xb, weight)
        ctx.meta = (stride, padding, x.dtype, weight.shape, bias is not None,
                    False)
        if stride == 1 and x.is_cuda and _skipfuse_on():
            _attach_skipbox(ctx, x)
        if stride == 1 and xb is x \
                and os.environ.get("DDPX_BNFUSE", "0") == "1":
            # producer-side BN-backward fusion (opt-in; see _conv_fwd_prep)
            ctx.bn_box = getattr(x, "_ddpx_bnbwd", None)
        return y

    @staticmethod
    def backward(ctx, dy):
        dx, dw, db = _conv_bwd_impl(ctx, dy)
        return dx, dw, db, None, None


def conv2d(x, weight, bias=None, stride=1, padding=0):
    stride = stride[0] if isinstance(stride, (tuple, list)) else stride
    padding = padding[0] if isinstance(padding, (tuple, list)) else padding
    if x.is_cuda and require_ext_for(x) is not None:
        # NOTE: a 1x1-as-GEMM route through the linear kernels was tried and
        # REVERTED — measured 2.6 ms vs 0.27 ms for the conv path at
        # ResNet-50's [M~800k, 64x64] shapes (the GEMM ladder's split-K is
        # tuned for classifier-head sizes, the conv wgrad splits for deep M).
        return _HIPConv2d.apply(x, weight, bias, stride, padding)
    if x.is_cuda and amp_mod.is_enabled():  # explicit torch-fallback bring-up path
        return F.conv2d(_to_bf16(x), bf16_weight(weight).view_as(weight),
                        bias.to(torch.bfloat16) if bias is not None else None,
                        stride, padding)
    return F.conv2d(x, weight, bias, stride, padding)


# ----------------------------------------------------------------- linear ---

class _HIPLinear(torch.autograd.Function):
    @staticmethod
    def forward(ctx, x, weight, bias):
        ext = require_ext_for(x)
        xb = _to_bf16(x.contiguous())
        wb = weight_variant(weight, "flat")
        bias_f = bias if bias is None else bias.detach().float()
        y = ext.linear_fwd(xb, wb, bias_f)
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
            dx = ext.linear_dgrad(dyb, wb)
            if x_dtype == torch.float32:
                dx = dx.float()
        if ctx.needs_input_grad[1]:
            dw = ext.linear_wgrad(dyb, xb)
        if has_bias and ctx.needs_input_grad[2]:
            db = ext.col_sum(dyb)
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
    """BatchNorm (2d channels_last or 1d) with optional fused ReLU.

    fp32 stats over bf16 activations; running stats updated in-place
    (rank-local; broadcast from rank 0 each forward by the DDP wrapper —
    reference broadcast_buffers semantics)."""

    @staticmethod
    def forward(ctx, x, weight, bias, running_mean, running_var,
                training, momentum, eps, fuse_relu, residual, pre_stats=None):
        ext = require_ext_for(x)
        xb = _chlast(_to_bf16(x))
        rb = None if residual is None else _chlast(_to_bf16(residual))
        if training:
            y, save_mean, save_invstd, mask = ext.bn_fwd_train(
                xb, weight.detach(), bias.detach(), running_mean, running_var,
                momentum, eps, fuse_relu, rb, pre_stats)
            ctx.save_for_backward(xb, weight, save_mean, save_invstd, mask)
            # producer-side bwd fusion: the conv that consumes y can emit
            # this BN's backward partials from its dgrad epilogue; the box
            # carries what it needs and brings the result back (used only
            # when dy's identity proves a single consumer — see backward)
            ctx.bnbwd_box = {"x": xb, "mask": mask, "mean": save_mean,
                             "invstd": save_invstd, "slab": None,
                             "dx_ref": None}
        elif (ctx.needs_input_grad[0] or ctx.needs_input_grad[1]
              or ctx.needs_input_grad[2]):
            # (grad mode is force-disabled inside Function.forward, so the
            # needs_input_grad flags — not torch.is_grad_enabled() — tell
            # whether a backward will run)
            # eval-mode BN on the grad path (fine-tuning with frozen stats;
            # VERDICT r01 weak #8): keep the relu mask + frozen stats for
            # backward.  The C-length stat vectors are cloned — the DDP
            # buffer broadcast rewrites the originals in place each forward.
            y, mask = ext.bn_fwd_eval_mask(xb, weight.detach(), bias.detach(),
                                           running_mean, running_var, eps,
                                           fuse_relu, rb)
            ctx.save_for_backward(xb, weight, running_mean.detach().clone(),
                                  running_var.detach().clone(), mask)
            ctx.eps = eps
        else:
            y = ext.bn_fwd_eval(xb, weight.detach(), bias.detach(),
                                running_mean, running_var, eps, fuse_relu, rb)
        ctx.fuse_relu = fuse_relu
        ctx.training = training
        ctx.x_dtype = x.dtype
        ctx.has_residual = residual is not None
        # skip-grad fusion arming: the residual tensor is ALSO the junction
        # conv's input (its box was attached by that conv's forward, which
        # ran earlier).  Stash dresid there in backward instead of
        # returning it; the conv's dgrad (strictly later in backward
        # order) streams it into its epilogue.
        ctx.skip_armed = False
        if (training and residual is not None and residual.is_cuda
                and residual.dtype == torch.bfloat16 and _skipfuse_on()
                and residual.requires_grad):
            sb = getattr(residual, "_ddpx_skipbox", None)
            if sb is not None:
                ctx.skip_armed = True
                ctx.skip_stash = sb
        return y

    @staticmethod
    def backward(ctx, dy):
        if not ctx.training:
            # frozen-stats backward: mean/var are constants, so
            # dx = gamma*invstd*dy_eff (eval_stats=True zeroes the batch
            # correction terms); dgamma/dbeta reduce against frozen xhat
            xb, weight, rm, rv, mask = ctx.saved_tensors
            ext = require_ext_for(dy)
            dyb = _chlast(_to_bf16(dy))
            si = (rv + ctx.eps).rsqrt()
            out = ext.bn_bwd(xb, dyb, weight.detach(), rm, si, mask,
                             ctx.fuse_relu, ctx.has_residual, None, True)
            dx, dweight, dbias = out[0], out[1], out[2]
            dresid = out[3] if ctx.has_residual else None
            if ctx.x_dtype == torch.float32:
                dx = dx.float()
            return (dx, dweight, dbias, None, None, None, None, None, None,
                    dresid, None)
        xb, weight, save_mean, save_invstd, mask = ctx.saved_tensors
        ext = require_ext_for(dy)
        box = getattr(ctx, "bnbwd_box", None)
        pre_slab = None
        if (box is not None and box["slab"] is not None
                and dy is box["dx_ref"]):
            # dy IS the consuming conv's dgrad output (single consumer —
            # a multi-consumer sum would be a different tensor object), so
            # the partials it emitted are exactly this BN's
            pre_slab = box["slab"]
        dyb = _chlast(_to_bf16(dy))
        out = ext.bn_bwd(xb, dyb, weight.detach(), save_mean,
                         save_invstd, mask, ctx.fuse_relu, ctx.has_residual,
                         pre_slab)
        dx, dweight, dbias = out[0], out[1], out[2]
        dresid = out[3] if ctx.has_residual else None
        if ctx.x_dtype == torch.float32:
            dx = dx.float()
        if dresid is not None and getattr(ctx, "skip_armed", False):
            stash = ctx.skip_stash
            prev = stash.get("g")
            stash["g"] = dresid if prev is None else prev + dresid
            dresid = None   # delivered through the junction conv's dgrad
        return (dx, dweight, dbias, None, None, None, None, None, None,
                dresid, None)


def batch_norm(x, running_mean, running_var, weight, bias,
               training, momentum, eps, fuse_relu=False, residual=None,
               pre_stats=None):
    """BN with optionally fused ReLU and residual add: relu(bn(x) + residual)
    — the ResNet block epilogue in one kernel (skip grad returned in bwd).
    pre_stats: [2,C,S] partial sums emitted by the producing conv's epilogue
    (conv2d_stats) — skips the BN statistics pass entirely."""
    if x.is_cuda and require_ext_for(x) is not None:
        y = _HIPBatchNorm.apply(x, weight, bias, running_mean, running_var,
                                training, momentum, eps, fuse_relu,
                                residual, pre_stats)
        # surface the producer-fusion box on the tensor object so the
        # consuming conv can find it (object-attribute, not data_ptr —
        # immune to allocator reuse); ctx attributes are visible on grad_fn
        if training and y.grad_fn is not None:
            box = getattr(y.grad_fn, "bnbwd_box", None)
            if box is not None:
                y._ddpx_bnbwd = box
        return y
    xf = x.float() if x.dtype != torch.float32 else x
    y = F.batch_norm(xf, running_mean, running_var, weight, bias,
                     training, momentum, eps)
    if residual is not None:
        y = y + residual.float()
    if fuse_relu:
        y = F.relu(y)
    return y.to(x.dtype) if x.is_cuda and amp_mod.is_enabled() else y


# ---------------------------------------------------------------- pooling ---

class _HIPMaxPool2x2(torch.autograd.Function):
    @staticmethod
    def forward(ctx, x):
        ext = require_ext_for(x)
        xb = _chlast(_to_bf16(x))
        y, idx = ext.maxpool2x2_fwd(xb)
        ctx.save_for_backward(idx)
        ctx.x_shape = xb.shape
        return y

    @staticmethod
    def backward(ctx, dy):
        (idx,) = ctx.saved_tensors
        ext = require_ext_for(dy)
        dx = ext.maxpool2x2_bwd(_chlast(_to_bf16(dy)), idx,
                                ctx.x_shape[2], ctx.x_shape[3])
        return dx


class _HIPMaxPool(torch.autograd.Function):
    """General max pool (kernel ks, stride st, padding pad) — ResNet stem."""

    @staticmethod
    def forward(ctx, x, ks, st, pad):
        ext = require_ext_for(x)
        xb = _chlast(_to_bf16(x))
        y, idx = ext.maxpool_fwd(xb, ks, st, pad)
        ctx.save_for_backward(idx)
        ctx.meta = (xb.shape[2], xb.shape[3], ks, st, pad)
        return y

    @staticmethod
    def backward(ctx, dy):
        (idx,) = ctx.saved_tensors
        H, W, ks, st, pad = ctx.meta
        ext = require_ext_for(dy)
        dx = ext.maxpool_bwd(_chlast(_to_bf16(dy)), idx, H, W, ks, st, pad)
        return dx, None, None, None


def max_pool2d(x, kernel_size=2, stride=None, padding=0):
    ks = kernel_size[0] if isinstance(kernel_size, (tuple, list)) else kernel_size
    st = stride or ks
    st = st[0] if isinstance(st, (tuple, list)) else st
    pad = padding[0] if isinstance(padding, (tuple, list)) else padding
    if x.is_cuda and require_ext_for(x) is not None:
        if ks == 2 and st == 2 and pad == 0:
            return _HIPMaxPool2x2.apply(x)
        return _HIPMaxPool.apply(x, ks, st, pad)
    return F.max_pool2d(x, kernel_size, stride, padding)


class _HIPGlobalAvgPool(torch.autograd.Function):
    @staticmethod
    def forward(ctx, x):
        ext = require_ext_for(x)
        xb = _chlast(_to_bf16(x))
        ctx.hw = (xb.shape[2], xb.shape[3])
        return ext.global_avgpool_fwd(xb)

    @staticmethod
    def backward(ctx, dy):
        ext = require_ext_for(dy)
        return ext.global_avgpool_bwd(_to_bf16(dy.contiguous()), *ctx.hw)


def global_avg_pool2d(x):
    """[N,C,H,W] -> [N,C] mean over H,W (ResNet head: adaptive avgpool 1x1
    + flatten in one op)."""
    if x.is_cuda and require_ext_for(x) is not None:
        return _HIPGlobalAvgPool.apply(x)
    return x.float().mean(dim=(2, 3)).to(x.dtype) \
        if x.is_cuda and amp_mod.is_enabled() else x.mean(dim=(2, 3))


def relu(x, inplace=False):
    # standalone ReLU (non-fused path); fused into BN epilogue on GPU
    return F.relu(x, inplace=inplace)


class _HIPFlattenNHWC(torch.autograd.Function):
    """[N,C,H,W] channels_last bf16 -> [N, C*H*W] in NCHW semantic order
    (the reference fc1 weight layout, reference utils/model.py:23-25) with
    coalesced HIP transposes both ways — replaces ATen's strided
    permute-copies at the conv->dense junction."""

    @staticmethod
    def forward(ctx, x):
        ext = require_ext_for(x)
        ctx.chw = (x.shape[1], x.shape[2], x.shape[3])
        return ext.nhwc_flatten(x)

    @staticmethod
    def backward(ctx, dy):
        ext = require_ext_for(dy)
        return ext.nhwc_unflatten(dy.contiguous(), *ctx.chw)


def nhwc_flatten(x):
    if require_ext_for(x) is not None:
        return _HIPFlattenNHWC.apply(x)
    return torch.flatten(x, 1)


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
        dlogits = ext.ce_bwd(logits, target, lse, dloss.contiguous())
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


def conv_bn(conv_mod, bn_mod, x, residual=None):
    """Fused producer/consumer: on the GPU training path the conv epilogue
    emits the BN's per-channel partial statistics, and the BN forward skips
    its stats pass (SURVEY §7 M5 fusion note).  Falls back to the plain
    composition on CPU, in eval mode, or off the MFMA path."""
    use_stats = (x.is_cuda and require_ext_for(x) is not None
                 and bn_mod.training and getattr(conv_mod, "dilation",
                                                 (1, 1)) in ((1, 1), 1))
    if use_stats:
        stride = conv_mod.stride[0] if isinstance(conv_mod.stride, tuple) \
            else conv_mod.stride
        padding = conv_mod.padding[0] if isinstance(conv_mod.padding, tuple) \
            else conv_mod.padding
        y, stats = _HIPConv2dStats.apply(x, conv_mod.weight, conv_mod.bias,
                                         stride, padding)
        return bn_mod(y, residual=residual, pre_stats=stats)
    return bn_mod(conv_mod(x), residual=residual)
