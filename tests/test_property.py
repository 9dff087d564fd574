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


@settings(max_examples=60, deadline=None)
@given(
    k=st.integers(min_value=1, max_value=7),
    alpha=st.floats(min_value=0.1, max_value=0.9, allow_nan=False),
    steps=st.integers(min_value=1, max_value=20),
    seed=st.integers(min_value=0, max_value=1000),
)
def test_lookahead_matches_reference_algorithm(k, alpha, steps, seed):
    """slow += alpha*(fast-slow); fast <- slow every k inner steps
    (reference utils/lookahead.py:19-41), over random (k, alpha, steps)."""
    from ddp_tricks_amd.utils.lookahead import Lookahead
    torch.manual_seed(seed)
    p_ours = torch.nn.Parameter(torch.randn(6))
    p_ref = torch.nn.Parameter(p_ours.detach().clone())
    grads = [torch.randn(6) for _ in range(steps)]

    la = Lookahead(torch.optim.SGD([p_ours], lr=0.1), k=k, alpha=alpha)
    # literal transcription of the reference step (utils/lookahead.py:33-41):
    # inner step first; slow-update while counter==0 (slow lazily snapshot
    # at its first use, i.e. AFTER the first inner step); then counter++
    # and reset at k
    opt_ref = torch.optim.SGD([p_ref], lr=0.1)
    slow = None
    counter = 0
    for g in grads:
        p_ours.grad = g.clone()
        la.step()
        p_ref.grad = g.clone()
        opt_ref.step()
        if counter == 0:
            with torch.no_grad():
                if slow is None:
                    slow = p_ref.detach().clone()
                slow += alpha * (p_ref.detach() - slow)
                p_ref.copy_(slow)
        counter += 1
        if counter >= k:
            counter = 0
    assert torch.allclose(p_ours.detach(), p_ref.detach(), atol=1e-6), \
        (k, alpha, steps)


@settings(max_examples=60, deadline=None)
@given(
    seq=st.lists(st.floats(min_value=0.0, max_value=5.0, allow_nan=False),
                 min_size=1, max_size=40),
    patience=st.integers(min_value=0, max_value=5),
    factor=st.sampled_from([0.1, 0.5]),
)
def test_plateau_matches_torch(seq, patience, factor):
    """ReduceLROnPlateau re-implementation vs torch over random valid-loss
    sequences (reference wires torch's at utils/train.py:53)."""
    from ddp_tricks_amd.utils.schedulers import ReduceLROnPlateau
    p1 = torch.nn.Parameter(torch.zeros(1))
    o1 = torch.optim.SGD([p1], lr=0.1)
    p2 = torch.nn.Parameter(torch.zeros(1))
    o2 = torch.optim.SGD([p2], lr=0.1)
    ours = ReduceLROnPlateau(optimizer=o1, mode="min", factor=factor,
                             patience=patience, verbose=False)
    ref = torch.optim.lr_scheduler.ReduceLROnPlateau(
        o2, mode="min", factor=factor, patience=patience)
    for v in seq:
        ours.step(v)
        ref.step(v)
        assert abs(o1.param_groups[0]["lr"]
                   - o2.param_groups[0]["lr"]) < 1e-12


@settings(max_examples=80, deadline=None)
@given(
    momentum=st.sampled_from([0.0, 0.5, 0.9]),
    wd=st.sampled_from([0.0, 1e-4, 1e-2]),
    nesterov=st.booleans(),
    lr=st.sampled_from([0.1, 0.01]),
    steps=st.integers(min_value=1, max_value=8),
    seed=st.integers(min_value=0, max_value=500),
)
def test_fused_sgd_cpu_matches_torch(momentum, wd, nesterov, lr, steps, seed):
    """FusedSGD's CPU math vs torch.optim.SGD across the config grid
    (reference wires SGD(momentum=0.9, nesterov=True) at utils/train.py:41)."""
    if nesterov and momentum == 0.0:
        return
    from ddp_tricks_amd.ops.optim import FusedSGD
    torch.manual_seed(seed)
    p1 = torch.nn.Parameter(torch.randn(10))
    p2 = torch.nn.Parameter(p1.detach().clone())
    o1 = FusedSGD([p1], lr=lr, momentum=momentum, weight_decay=wd,
                  nesterov=nesterov)
    o2 = torch.optim.SGD([p2], lr=lr, momentum=momentum, weight_decay=wd,
                         nesterov=nesterov)
    for _ in range(steps):
        g = torch.randn(10)
        p1.grad = g.clone()
        p2.grad = g.clone()
        o1.step()
        o2.step()
    assert torch.allclose(p1.detach(), p2.detach(), atol=1e-6), \
        (momentum, wd, nesterov, lr, steps)
