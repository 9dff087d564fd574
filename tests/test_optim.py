"""FusedSGD trajectory equivalence with torch.optim.SGD (nesterov momentum,
the reference's configuration at utils/train.py:41)."""
import torch

from ddp_tricks_amd.ops.optim import FusedSGD


def test_matches_torch_sgd_nesterov():
    torch.manual_seed(0)
    shapes = [(8, 4), (16,), (3, 3, 2)]
    pa = [torch.nn.Parameter(torch.randn(s)) for s in shapes]
    pb = [torch.nn.Parameter(p.detach().clone()) for p in pa]
    oa = FusedSGD(pa, lr=0.1, momentum=0.9, nesterov=True)
    ob = torch.optim.SGD(pb, lr=0.1, momentum=0.9, nesterov=True)
    for step in range(10):
        grads = [torch.randn_like(p) for p in pa]
        for p, q, g in zip(pa, pb, grads):
            p.grad = g.clone()
            q.grad = g.clone()
        oa.step()
        ob.step()
        for p, q in zip(pa, pb):
            assert torch.allclose(p, q, atol=1e-6), step


def test_weight_decay_path():
    p = torch.nn.Parameter(torch.ones(4))
    q = torch.nn.Parameter(torch.ones(4))
    oa = FusedSGD([p], lr=0.1, momentum=0.9, nesterov=True, weight_decay=0.01)
    ob = torch.optim.SGD([q], lr=0.1, momentum=0.9, nesterov=True, weight_decay=0.01)
    for _ in range(5):
        p.grad = torch.full((4,), 0.5)
        q.grad = torch.full((4,), 0.5)
        oa.step()
        ob.step()
    assert torch.allclose(p, q, atol=1e-6)


def test_zero_grad_keeps_storage():
    p = torch.nn.Parameter(torch.ones(4))
    opt = FusedSGD([p], lr=0.1)
    buf = torch.zeros(4)
    p.grad = buf.narrow(0, 0, 4)
    ptr = p.grad.data_ptr()
    opt.zero_grad()
    assert p.grad is not None and p.grad.data_ptr() == ptr
    assert torch.equal(p.grad, torch.zeros(4))
