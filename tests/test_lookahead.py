"""Lookahead wrapper semantics (reference utils/lookahead.py)."""
import torch

from ddp_tricks_amd import Lookahead
from ddp_tricks_amd.ops.optim import FusedSGD


def _make(k=3, alpha=0.5, lr=0.1):
    p = torch.nn.Parameter(torch.ones(4))
    opt = FusedSGD([p], lr=lr)
    la = Lookahead(opt, k=k, alpha=alpha)
    return p, opt, la


def test_interpolation_schedule():
    # update fires on steps 1, k+1, 2k+1 ... (counter==0 check after step)
    p, opt, la = _make(k=3)
    for step in range(1, 8):
        p.grad = torch.ones(4)
        la.step()
        counters = [g["counter"] for g in la.param_groups]
        assert counters == [step % 3]


def test_lookahead_math_matches_manual():
    torch.manual_seed(0)
    p, opt, la = _make(k=2, alpha=0.5, lr=0.1)
    # manual replica
    fast = torch.ones(4)
    slow = None
    counter = 0
    for i in range(6):
        g = torch.full((4,), float(i + 1))
        p.grad = g.clone()
        la.step()
        # manual plain SGD (no momentum)
        fast = fast - 0.1 * g
        if counter == 0:
            if slow is None:
                slow = fast.clone()
            slow = slow + 0.5 * (fast - slow)
            fast = slow.clone()
        counter = (counter + 1) % 2
        assert torch.allclose(p.data, fast, atol=1e-6), (i, p.data, fast)


def test_state_dict_split_roundtrip():
    p, opt, la = _make(k=3)
    for _ in range(4):
        p.grad = torch.randn(4)
        la.step()
    sd = la.state_dict()
    assert set(sd) == {"fast_state", "slow_state", "param_groups"}
    p2, opt2, la2 = _make(k=3)
    for _ in range(1):
        p2.grad = torch.randn(4)
        la2.step()
    la2.load_state_dict(sd)
    assert la2.fast_state is la2.optimizer.state


def test_shared_param_groups_object():
    p, opt, la = _make()
    assert la.param_groups is opt.param_groups
    la.param_groups[0]["lr"] = 0.05
    assert opt.param_groups[0]["lr"] == 0.05
