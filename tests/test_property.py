"""Property-based conformance tests (hypothesis): our re-implemented
plumbing must match the reference/torch semantics over RANDOM configs,
not just the hand-picked cases in the unit tests."""
import math

import torch
from hypothesis import given, settings, strategies as st

from ddp_tricks_amd.utils.callbacks import EarlyStopping
from ddp_tricks_amd.utils.data import DistributedSampler


class _RefEarlyStopping:
    """Literal transcription of the reference algorithm (reference
    utils/customized.py:16-50) used as the oracle."""

    def __init__(self, patience, delta):
        self.patience = patience
        self.counter = 0
        self.best_score = None
        self.early_stop = False
        self.delta = delta

    def __call__(self, metrics, loss=True):
        score = -metrics if loss else metrics
        if self.best_score is None:
            self.best_score = score
        elif score < self.best_score + self.delta:
            self.counter += 1
            if self.counter >= self.patience:
                self.early_stop = True
        else:
            self.best_score = score
            self.counter = 0


@settings(max_examples=200, deadline=None)
@given(
    seq=st.lists(st.floats(min_value=-10, max_value=10,
                           allow_nan=False), min_size=1, max_size=60),
    patience=st.integers(min_value=1, max_value=10),
    delta=st.sampled_from([0.0, 1e-3, 0.1]),
    loss_mode=st.booleans(),
)
def test_early_stopping_matches_reference_algorithm(seq, patience, delta,
                                                    loss_mode):
    ours = EarlyStopping(patience=patience, verbose=False, delta=delta)
    ref = _RefEarlyStopping(patience, delta)
    for v in seq:
        ours(v, loss=loss_mode)
        ref(v, loss=loss_mode)
        assert ours.counter == ref.counter
        assert ours.best_score == ref.best_score
        assert ours.early_stop == ref.early_stop
        if ours.early_stop:
            break


@settings(max_examples=100, deadline=None)
@given(
    n=st.integers(min_value=1, max_value=300),
    world=st.integers(min_value=1, max_value=8),
    epoch=st.integers(min_value=0, max_value=5),
    shuffle=st.booleans(),
)
def test_sampler_matches_torch_distributed_sampler(n, world, epoch, shuffle):
    class _DS:
        def __len__(self):
            return n

    ds = _DS()
    for rank in range(world):
        ours = DistributedSampler(ds, num_replicas=world, rank=rank,
                                  shuffle=shuffle, seed=0)
        ref = torch.utils.data.distributed.DistributedSampler(
            ds, num_replicas=world, rank=rank, shuffle=shuffle, seed=0)
        ours.set_epoch(epoch)
        ref.set_epoch(epoch)
        assert list(ours) == list(ref), (n, world, rank, epoch, shuffle)
        assert ours.num_samples == ref.num_samples == math.ceil(n / world)


@settings(max_examples=60, deadline=None)
@given(
    warmup=st.integers(min_value=1, max_value=20),
    epochs=st.integers(min_value=25, max_value=200),
    base_lr=st.sampled_from([0.1, 0.01, 1.0]),
)
def test_warmup_schedule_matches_reference_formula(warmup, epochs, base_lr):
    """Stepping gated at epoch<=warmup (reference utils/train.py:104-105):
    LR is base*e/warmup for e<=warmup then frozen at base — including the
    epoch-0-at-LR-0 quirk."""
    from ddp_tricks_amd.utils.schedulers import WarmupLambdaLR
    p = torch.nn.Parameter(torch.zeros(1))
    opt = torch.optim.SGD([p], lr=base_lr)

    def warm_up(epoch):
        return epoch / warmup if epoch <= warmup else 1

    sch = WarmupLambdaLR(optimizer=opt, lr_lambda=warm_up)
    seen = []
    for epoch in range(min(epochs, warmup + 5)):
        seen.append(opt.param_groups[0]["lr"])
        if epoch <= warmup:
            sch.step()
    for epoch, lr in enumerate(seen):
        expect = base_lr * min(epoch / warmup, 1.0)
        assert abs(lr - expect) < 1e-12, (epoch, lr, expect)
