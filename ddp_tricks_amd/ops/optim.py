"""FusedSGD — nesterov-momentum SGD with a single multi-tensor step.

Replaces torch.optim.SGD in the reference recipe (reference
utils/train.py:41: SGD(lr, momentum=0.9, nesterov=True), stepped through
Lookahead).  On GPU the step is one fused multi-tensor HIP kernel over the
flattened param/grad/momentum lists; on CPU (and as the
reference oracle) the identical math runs via torch._foreach ops:

    buf = mu * buf + g            (no dampening, matches reference cfg)
    d   = g + mu * buf            (nesterov)
    p  -= lr * d
"""
from __future__ import annotations

from typing import List

import torch
from torch.optim import Optimizer


class FusedSGD(Optimizer):
    def __init__(self, params, lr: float, momentum: float = 0.0,
                 weight_decay: float = 0.0, nesterov: bool = False):
        if lr < 0.0:
            raise ValueError(f"invalid lr {lr}")
        if nesterov and momentum <= 0:
            raise ValueError("nesterov momentum requires momentum > 0")
        defaults = dict(lr=lr, momentum=momentum, weight_decay=weight_decay,
                        nesterov=nesterov)
        super().__init__(params, defaults)

    @torch.no_grad()
    def step(self, closure=None):
        loss = None
        if closure is not None:
            with torch.enable_grad():
                loss = closure()
        for group in self.param_groups:
            params: List[torch.Tensor] = []
            grads: List[torch.Tensor] = []
            bufs: List[torch.Tensor] = []
            momentum = group["momentum"]
            for p in group["params"]:
                if p.grad is None:
                    continue
                params.append(p)
                grads.append(p.grad)
                if momentum != 0:
                    state = self.state[p]
                    if "momentum_buffer" not in state:
                        state["momentum_buffer"] = torch.zeros_like(p)
                    bufs.append(state["momentum_buffer"])
            if not params:
                continue
            self._step_group(params, grads, bufs, group)
        return loss

    def _step_group(self, params, grads, bufs, group):
        lr, momentum = group["lr"], group["momentum"]
        wd, nesterov = group["weight_decay"], group["nesterov"]
        if params[0].is_cuda:
            from . import load_extension
            ext = load_extension(required=False)
            if ext is not None:
                from .. import amp as amp_mod
                ext.fused_sgd(params, grads, bufs, lr, momentum, wd,
                              1.0 if nesterov else 0.0,
                              amp_mod.pending_found_inf())
                # raw-pointer writes don't bump Tensor._version — drop the
                # per-version bf16 weight cast cache so the next forward
                # re-casts the updated masters
                from .functional import clear_weight_cache
                clear_weight_cache()
                return
        if wd != 0:
            grads = torch._foreach_add(grads, params, alpha=wd)
        if momentum != 0:
            torch._foreach_mul_(bufs, momentum)
            torch._foreach_add_(bufs, grads)
            if nesterov:
                d = torch._foreach_add(grads, bufs, alpha=momentum)
            else:
                d = bufs
            torch._foreach_add_(params, d, alpha=-lr)
        else:
            torch._foreach_add_(params, grads, alpha=-lr)

    def zero_grad(self, set_to_none: bool = False):
        """Default zeros in place (NOT set-to-none): gradients are views
        into the DDP bucket flats and must keep their storage."""
        grads = [p.grad for g in self.param_groups for p in g["params"]
                 if p.grad is not None]
        if not grads:
            return
        if set_to_none:
            for g in self.param_groups:
                for p in g["params"]:
                    p.grad = None
        else:
            torch._foreach_zero_(grads)
