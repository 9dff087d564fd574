"""LR schedulers — self-contained implementations of the two scheduler
semantics the reference wires up (reference utils/train.py:48-53,104-106):

  * WarmupLambdaLR — torch.optim.lr_scheduler.LambdaLR semantics: on
    construction applies ``lr = base_lr * lr_lambda(0)`` (so epoch 0 trains at
    LR 0 with the reference's warm-up lambda — SURVEY Appendix A.2), and each
    ``step()`` advances the epoch counter and re-applies the lambda.  The
    caller gates stepping (reference steps it only while
    ``epoch <= warmup_epochs``, which freezes the cosine branch — Appendix A.3).

  * ReduceLROnPlateau — torch semantics with the defaults the reference uses:
    mode='min', factor, patience, rel threshold 1e-4, eps 1e-8.

Both schedulers mutate ``optimizer.param_groups[i]['lr']`` directly, so they
compose with Lookahead's shared param_groups object exactly as in the
reference (a plateau reduction during the warmup
window would be overwritten by the next warmup step — preserved here).
"""
from __future__ import annotations

from typing import Callable, List


class WarmupLambdaLR:
    def __init__(self, optimizer, lr_lambda: Callable[[int], float]):
        self.optimizer = optimizer
        self.lr_lambda = lr_lambda
        self.base_lrs: List[float] = [g["lr"] for g in optimizer.param_groups]
        self.last_epoch = -1
        self.step()  # applies lambda(0) at construction, like torch LambdaLR

    def get_lr(self) -> List[float]:
        return [base * self.lr_lambda(self.last_epoch) for base in self.base_lrs]

    def step(self) -> None:
        self.last_epoch += 1
        for group, lr in zip(self.optimizer.param_groups, self.get_lr()):
            group["lr"] = lr

    def state_dict(self):
        return {"base_lrs": self.base_lrs, "last_epoch": self.last_epoch}

    def load_state_dict(self, state):
        self.base_lrs = list(state["base_lrs"])
        self.last_epoch = state["last_epoch"]


class ReduceLROnPlateau:
    def __init__(
        self,
        optimizer,
        mode: str = "min",
        factor: float = 0.1,
        patience: int = 10,
        threshold: float = 1e-4,
        threshold_mode: str = "rel",
        cooldown: int = 0,
        min_lr: float = 0.0,
        eps: float = 1e-8,
        verbose: bool = False,
    ):
        if factor >= 1.0:
            raise ValueError("factor should be < 1.0")
        if mode not in ("min", "max"):
            raise ValueError(f"mode {mode!r} is unknown")
        if threshold_mode not in ("rel", "abs"):
            raise ValueError(f"threshold_mode {threshold_mode!r} is unknown")
        self.optimizer = optimizer
        self.mode = mode
        self.factor = factor
        self.patience = patience
        self.threshold = threshold
        self.threshold_mode = threshold_mode
        self.cooldown = cooldown
        self.cooldown_counter = 0
        self.min_lrs = [min_lr] * len(optimizer.param_groups)
        self.eps = eps
        self.verbose = verbose
        self.best = float("inf") if mode == "min" else float("-inf")
        self.num_bad_epochs = 0
        self.last_epoch = 0

    def _is_better(self, current: float) -> bool:
        if self.mode == "min":
            if self.threshold_mode == "rel":
                return current < self.best * (1.0 - self.threshold)
            return current < self.best - self.threshold
        if self.threshold_mode == "rel":
            return current > self.best * (1.0 + self.threshold)
        return current > self.best + self.threshold

    def step(self, metrics: float) -> None:
        current = float(metrics)
        self.last_epoch += 1
        if self._is_better(current):
            self.best = current
            self.num_bad_epochs = 0
        else:
            self.num_bad_epochs += 1
        if self.cooldown_counter > 0:
            self.cooldown_counter -= 1
            self.num_bad_epochs = 0
        if self.num_bad_epochs > self.patience:
            self._reduce_lr()
            self.cooldown_counter = self.cooldown
            self.num_bad_epochs = 0

    def _reduce_lr(self) -> None:
        for i, group in enumerate(self.optimizer.param_groups):
            old_lr = float(group["lr"])
            new_lr = max(old_lr * self.factor, self.min_lrs[i])
            if old_lr - new_lr > self.eps:
                group["lr"] = new_lr
                if self.verbose:
                    print(f"ReduceLROnPlateau: reducing learning rate of group {i} to {new_lr:.4e}.")

    def state_dict(self):
        return {
            k: v for k, v in self.__dict__.items() if k != "optimizer"
        }

    def load_state_dict(self, state):
        self.__dict__.update(state)
