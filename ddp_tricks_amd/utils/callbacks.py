"""Seeding + EarlyStopping.

Behavioral spec from the reference (re-implemented, not copied):
  - same_seeds: reference utils/customized.py:5-13 — seed numpy/random/torch/
    all CUDA devices and force deterministic kernel selection.
  - EarlyStopping: reference utils/customized.py:16-50 — loss-mode sign flip,
    no-improvement iff score < best + delta (an equal score with delta=0
    counts as improvement), `.early_stop` set once counter reaches patience.
"""
from __future__ import annotations

import random

import numpy as np
import torch


def same_seeds(seed: int = 18) -> None:
    """Seed every RNG and force deterministic execution.

    MI355X note: our HIP kernels are deterministic by construction (fixed-order
    slab reductions, no atomics on the gradient path), so unlike the CUDA
    reference (cudnn.benchmark/deterministic, utils/customized.py:12-13) there
    is no autotuner to disable; the torch flags are still set so any torch-op
    fallback path is deterministic too.
    """
    np.random.seed(seed)
    random.seed(seed)
    torch.manual_seed(seed)
    if torch.cuda.is_available():
        torch.cuda.manual_seed(seed)
        torch.cuda.manual_seed_all(seed)
    torch.backends.cudnn.benchmark = False
    torch.backends.cudnn.deterministic = True


class EarlyStopping:
    """Stop training after `patience` epochs without metric improvement.

    Matches reference utils/customized.py:16-50 semantics exactly:
    ``score = -metrics`` when ``loss=True``; a step with
    ``score < best_score + delta`` increments the counter, anything else
    (including equality at delta=0) resets it and takes the new best.
    """

    def __init__(self, patience: int = 100, verbose: bool = True, delta: float = 0.0):
        self.patience = patience
        self.verbose = verbose
        self.counter = 0
        self.best_score = None
        self.early_stop = False
        self.delta = delta

    def __call__(self, metrics: float, loss: bool = True) -> None:
        score = -metrics if loss else metrics
        if self.best_score is None:
            self.best_score = score
        elif score < self.best_score + self.delta:
            self.counter += 1
            if self.counter >= self.patience:
                self.early_stop = True
                if self.verbose:
                    print("EarlyStopping")
        else:
            self.best_score = score
            self.counter = 0

    def state_dict(self):
        return {"counter": self.counter, "best_score": self.best_score,
                "early_stop": self.early_stop}

    def load_state_dict(self, state):
        self.counter = state["counter"]
        self.best_score = state["best_score"]
        self.early_stop = state["early_stop"]
