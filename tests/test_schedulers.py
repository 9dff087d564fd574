"""Scheduler semantics vs torch equivalents (reference utils/train.py:48-53)."""
import math

import torch

from ddp_tricks_amd.ops.optim import FusedSGD
from ddp_tricks_amd.utils.schedulers import ReduceLROnPlateau, WarmupLambdaLR


def _opt(lr=0.1):
    p = torch.nn.Parameter(torch.zeros(1))
    return FusedSGD([p], lr=lr)


def _torch_opt(lr=0.1):
    p = torch.nn.Parameter(torch.zeros(1))
    return torch.optim.SGD([p], lr=lr)


def test_warmup_lambda_matches_torch_lambdalr():
    warmup_epochs = 10
    lam = lambda e: e / warmup_epochs if e <= warmup_epochs else 1  # noqa: E731
    ours, theirs = _opt(), _torch_opt()
    s1 = WarmupLambdaLR(ours, lam)
    s2 = torch.optim.lr_scheduler.LambdaLR(theirs, lam)
    for _ in range(15):
        assert abs(ours.param_groups[0]["lr"] - theirs.param_groups[0]["lr"]) < 1e-12
        s1.step()
        s2.step()


def test_epoch0_lr_is_zero():
    ours = _opt(0.1)
    WarmupLambdaLR(ours, lambda e: e / 10 if e <= 10 else 1)
    assert ours.param_groups[0]["lr"] == 0.0


def test_dead_cosine_freeze():
    """Reference quirk A.3: warmup only stepped while epoch <= warmup_epochs,
    so the cosine branch evaluates once at epoch warmup+1 and freezes."""
    epochs, warmup_epochs = 500, 10
    lam = lambda e: e / warmup_epochs if e <= warmup_epochs else 0.5 * (  # noqa: E731
        math.cos((e - warmup_epochs) / (epochs - warmup_epochs) * math.pi) + 1)
    ours = _opt(0.1)
    s = WarmupLambdaLR(ours, lam)
    for epoch in range(50):
        if epoch <= warmup_epochs:
            s.step()
    assert abs(ours.param_groups[0]["lr"] - 0.0999989726) < 1e-7


def test_plateau_matches_torch():
    seq = [1.0, 0.9, 0.9, 0.9, 0.9, 0.9, 0.9, 0.9, 0.9, 0.85, 0.85, 0.85,
           0.85, 0.85, 0.85, 0.85, 0.85, 0.85, 0.84]
    ours, theirs = _opt(), _torch_opt()
    s1 = ReduceLROnPlateau(ours, mode="min", factor=0.1, patience=6)
    s2 = torch.optim.lr_scheduler.ReduceLROnPlateau(theirs, mode="min",
                                                    factor=0.1, patience=6)
    for v in seq:
        s1.step(v)
        s2.step(v)
        assert abs(ours.param_groups[0]["lr"] - theirs.param_groups[0]["lr"]) < 1e-12
