"""iterate_loader — the per-epoch batch loop (the hot loop).

Behavior contract from the reference (utils/process.py:6-42):
zero_grad → H2D → forward → argmax → CE loss → extra ``/= len(outputs)``
double normalization → amp.scale_loss backward → optimizer.step → loss/acc
accumulation; returns (loss, acc, curr_lr) when training, (loss, acc)
otherwise, where loss/acc are sample-weighted epoch means and curr_lr reads
``param_groups[0]["lr"]``.

MI355X implementation notes:
  * loss and correct-count accumulate in DEVICE-side tensors — the
    reference's three per-batch ``.item()`` syncs become one sync per
    epoch; reported numbers are identical.
  * the accuracy reduction is the fused rowwise-argmax+compare HIP kernel
    on GPU (ops.functional.argmax_correct).
  * tqdm progress on every rank like the reference (utils/process.py:15);
    disable with DDPX_NO_TQDM=1 (benchmarks).
"""
from __future__ import annotations

import os
from typing import Optional

import torch

from .. import amp
from ..ops import functional as F_ops


def iterate_loader(
    loader,
    model: torch.nn.Module,
    loss_function,
    local_rank,
    apex_optimizer: Optional[torch.optim.Optimizer],
    training: bool = False,
    timers=None,
):
    device = next(model.parameters()).device
    loss_sum = torch.zeros((), dtype=torch.float64, device=device)
    correct_sum = torch.zeros((), dtype=torch.long, device=device)
    num = 0

    if timers is None and os.environ.get("DDPX_PHASE_TIMERS", "0") == "1":
        from .timers import PhaseTimers
        # cache keyed by device: events/streams are device-bound
        cache = getattr(iterate_loader, "_timers", None)
        if cache is None:
            cache = iterate_loader._timers = {}
        timers = cache.get(device)
        if timers is None:
            timers = cache[device] = PhaseTimers(device)
    from contextlib import nullcontext
    ph = timers.phase if timers is not None else (lambda name: nullcontext())

    iterator = loader
    if os.environ.get("DDPX_NO_TQDM", "0") != "1":
        from tqdm import tqdm
        iterator = tqdm(loader, total=len(loader))

    for image, target in iterator:
        if training:
            apex_optimizer.zero_grad()
        with ph("h2d"):
            image = image.to(device, non_blocking=True)
            target = target.to(device, dtype=torch.long, non_blocking=True)
        with ph("fwd"):
            outputs = model(image)
            batch_loss = loss_function(outputs, target)
            batch_loss = batch_loss / outputs.shape[0]  # reference's double normalization
        if training:
            with ph("bwd"):
                with amp.scale_loss(batch_loss, apex_optimizer) as scaled_loss:
                    scaled_loss.backward()
            with ph("opt"):
                apex_optimizer.step()

        # device-side metric accumulation (one host sync per epoch)
        loss_sum += batch_loss.detach().double() * image.shape[0]
        correct_sum += F_ops.argmax_correct(outputs.detach(), target)
        num += image.shape[0]

    if training:
        amp.maybe_sync_scaler()  # async-amp: fold deferred overflow count
    loss = (loss_sum / num).item()
    acc = (correct_sum.double() / num).item()

    if training:
        curr_lr = apex_optimizer.param_groups[0]["lr"]
        return loss, acc, curr_lr
    return loss, acc
