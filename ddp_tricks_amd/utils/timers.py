"""Per-phase HIP-event timers (SURVEY §5.1 — the reference has no tracing;
this is the framework's phase-level observability layer).

Phases: h2d (host→device copies), fwd, bwd (backward incl. overlapped bucket
all-reduces on the comm stream), opt (optimizer step).  GPU timing uses HIP
events recorded on the current stream — zero host syncs during the epoch;
``summary()`` synchronizes once and returns accumulated milliseconds.

Enable with env ``DDPX_PHASE_TIMERS=1``; the trainer appends the phase
breakdown to the per-epoch stdout line.  Also usable directly::

    t = PhaseTimers(device)
    with t.phase("fwd"):
        out = model(x)
    print(t.summary())
"""
from __future__ import annotations

import time
from contextlib import contextmanager
from typing import Dict, List, Tuple

import torch

PHASES = ("h2d", "fwd", "bwd", "opt")


class PhaseTimers:
    def __init__(self, device):
        self.device = torch.device(device) if device is not None else None
        self.use_cuda = (self.device is not None
                         and self.device.type == "cuda"
                         and torch.cuda.is_available())
        self._pairs: Dict[str, List[Tuple]] = {p: [] for p in PHASES}
        self._cpu_acc: Dict[str, float] = {p: 0.0 for p in PHASES}

    @contextmanager
    def phase(self, name: str):
        if name not in self._pairs:
            self._pairs[name] = []
            self._cpu_acc[name] = 0.0
        if self.use_cuda:
            start = torch.cuda.Event(enable_timing=True)
            end = torch.cuda.Event(enable_timing=True)
            start.record()
            try:
                yield
            finally:
                end.record()
                self._pairs[name].append((start, end))
        else:
            t0 = time.perf_counter()
            try:
                yield
            finally:
                self._cpu_acc[name] += (time.perf_counter() - t0) * 1000.0

    def summary(self) -> Dict[str, float]:
        """Accumulated milliseconds per phase (syncs once on GPU)."""
        out = {}
        if self.use_cuda:
            torch.cuda.synchronize(self.device)
        for name in self._pairs:
            ms = self._cpu_acc.get(name, 0.0)
            for s, e in self._pairs[name]:
                ms += s.elapsed_time(e)
            if ms > 0:
                out[name] = ms
        return out

    def reset(self) -> None:
        for p in self._pairs:
            self._pairs[p] = []
            self._cpu_acc[p] = 0.0

    def format(self) -> str:
        s = self.summary()
        return " ".join(f"{k}={v:.1f}ms" for k, v in s.items())
