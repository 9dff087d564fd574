"""Lookahead optimizer wrapper (k fast steps, then slow-weight interpolation).

API- and semantics-compatible with the reference's vendored wrapper
(reference utils/lookahead.py:8-72), re-implemented for MI355X:

  * interpolation math: ``slow += alpha * (fast - slow); fast <- slow``
    executed on the *whole group at once* via ``torch._foreach_`` multi-tensor
    ops (one fused HIP kernel sequence on ROCm instead of a Python per-param
    loop), or by the framework's fused SGD+Lookahead HIP kernel when the inner
    optimizer is ``ddp_tricks_amd.ops.FusedSGD`` (see ops/optim.py).
  * trigger schedule matches the reference exactly: the group counter starts
    at 0 and the update fires when ``counter == 0`` *after* a fast step, so
    interpolations happen on steps 1, k+1, 2k+1, ... (reference
    utils/lookahead.py:33-41); the first update initializes the slow weights
    to the fast weights (a numeric no-op).
  * ``state_dict``/``load_state_dict`` keep the reference's
    {fast_state, slow_state, param_groups} split (utils/lookahead.py:43-68).
"""
from __future__ import annotations

from collections import OrderedDict, defaultdict

import torch
from torch.optim import Optimizer


class Lookahead(Optimizer):
    def __init__(self, optimizer: Optimizer, k: int = 5, alpha: float = 0.5):
        self.optimizer = optimizer
        self.k = k
        self.alpha = alpha
        # We intentionally do NOT run Optimizer.__init__ (it would deep-copy
        # param groups; the whole point is SHARING the inner optimizer's
        # groups) — but torch>=2 Optimizer methods expect these attributes:
        self.defaults = {"k": k, "alpha": alpha}
        self._optimizer_step_pre_hooks = OrderedDict()
        self._optimizer_step_post_hooks = OrderedDict()
        self._optimizer_state_dict_pre_hooks = OrderedDict()
        self._optimizer_state_dict_post_hooks = OrderedDict()
        self._optimizer_load_state_dict_pre_hooks = OrderedDict()
        self._optimizer_load_state_dict_post_hooks = OrderedDict()
        # Shared param_groups object keeps LR schedulers pointed at the inner
        # optimizer coherent with stepping through the wrapper
        # (SURVEY Appendix A.4).
        self.param_groups = self.optimizer.param_groups
        self.state = defaultdict(dict)
        self.fast_state = self.optimizer.state
        for group in self.param_groups:
            group["counter"] = 0

    @torch.no_grad()
    def update(self, group) -> None:
        fasts, slows = [], []
        for fast in group["params"]:
            param_state = self.state[fast]
            if "slow_param" not in param_state:
                param_state["slow_param"] = fast.detach().clone()
            fasts.append(fast.data)
            slows.append(param_state["slow_param"])
        if not fasts:
            return
        if fasts[0].is_cuda:
            from ..ops import load_extension
            ext = load_extension(required=False)
            if ext is not None:
                from .. import amp as amp_mod
                ext.fused_lookahead(fasts, slows, self.alpha,
                                    amp_mod.pending_found_inf())
                from ..ops.functional import clear_weight_cache
                clear_weight_cache()  # raw writes don't bump _version
                return
        # slow = slow + alpha*(fast - slow) == lerp(slow, fast, alpha)
        torch._foreach_lerp_(slows, fasts, self.alpha)
        torch._foreach_copy_(fasts, slows)

    def update_lookahead(self) -> None:
        for group in self.param_groups:
            self.update(group)

    def step(self, closure=None):
        loss = self.optimizer.step(closure)
        for group in self.param_groups:
            if group["counter"] == 0:
                self.update(group)
            group["counter"] += 1
            if group["counter"] >= self.k:
                group["counter"] = 0
        return loss

    def zero_grad(self, set_to_none: bool = False):
        # Default False, NOT torch's True: gradients are views into the DDP
        # bucket flats — set_to_none would drop the views, autograd would
        # accumulate into fresh tensors, and the bucket all-reduce would
        # reduce zeros (silent rank desync at world>1).  The DDP hook also
        # self-heals re-pointed grads, but keeping the views is the fast
        # path.
        return self.optimizer.zero_grad(set_to_none=set_to_none)

    def state_dict(self):
        fast_state_dict = self.optimizer.state_dict()
        # Same {fast_state, slow_state, param_groups} split as the reference
        # (utils/lookahead.py:43-55), but slow entries are keyed by PARAM
        # INDEX (torch's convention) instead of the reference's id(k):
        # raw Python ids cannot be remapped onto a new process's params, so
        # the reference's format silently drops the slow weights on load —
        # harmless there (its training path never loads a Lookahead state,
        # SURVEY §5.4) but fatal to our --resume extension, where resumed
        # slow weights must continue the straight-run trajectory exactly
        # (tests/test_resume.py::test_resume_bitwise_equals_straight_run).
        index = {p: i for i, p in enumerate(
            p for g in self.param_groups for p in g["params"])}
        slow_state = {
            (index[k] if isinstance(k, torch.Tensor) else k): v
            for k, v in self.state.items()
        }
        return {
            "fast_state": fast_state_dict["state"],
            "slow_state": slow_state,
            "param_groups": fast_state_dict["param_groups"],
        }

    def load_state_dict(self, state_dict):
        fast_state_dict = {
            "state": state_dict["fast_state"],
            "param_groups": state_dict["param_groups"],
        }
        self.optimizer.load_state_dict(fast_state_dict)
        # the inner load REPLACES its param_groups list — re-share it (and
        # the per-group counters it restored) through the wrapper
        self.param_groups = self.optimizer.param_groups
        for group in self.param_groups:
            group.setdefault("counter", 0)
        params = [p for g in self.param_groups for p in g["params"]]
        self.state = defaultdict(dict)
        for k, v in state_dict["slow_state"].items():
            if isinstance(k, int) and 0 <= k < len(params):
                self.state[params[k]] = v
            # id(k)-keyed entries (reference-format checkpoints) cannot be
            # mapped to live params; they are dropped, matching what the
            # reference's own load achieves
        self.fast_state = self.optimizer.state

    def add_param_group(self, param_group):
        param_group["counter"] = 0
        self.optimizer.add_param_group(param_group)
