"""amp runtime: dynamic scaler policy, scale_loss unscale, step skip
(apex O1 semantics — SURVEY N4)."""
import torch

from ddp_tricks_amd import amp
from ddp_tricks_amd.amp import DynamicLossScaler
from ddp_tricks_amd.ops.optim import FusedSGD


def setup_function(_):
    # reset global amp state between tests
    amp._state.__init__()


def test_scaler_policy():
    s = DynamicLossScaler(init_scale=1024.0, growth_interval=4)
    s.update(found_inf=True)
    assert s.scale == 512.0
    for _ in range(4):
        s.update(found_inf=False)
    assert s.scale == 1024.0
    s.update(found_inf=True)
    assert s.scale == 512.0


def test_scale_loss_unscales_grads():
    p = torch.nn.Parameter(torch.ones(4))
    opt = FusedSGD([p], lr=0.1)
    model = torch.nn.Module()
    model, opt = amp.initialize(model, opt, opt_level="O1")
    x = torch.ones(4, requires_grad=False)
    loss = (p * x).sum()
    with amp.scale_loss(loss, opt) as scaled:
        scaled.backward()
    # d(loss)/dp = 1; after unscale the grad must be 1 regardless of scale
    assert torch.allclose(p.grad, torch.ones(4), atol=1e-6)


def test_overflow_skips_step_and_backoff():
    p = torch.nn.Parameter(torch.ones(2))
    opt = FusedSGD([p], lr=0.1)
    model = torch.nn.Module()
    model, opt = amp.initialize(model, opt, opt_level="O1")
    scale0 = amp.state().scaler.scale
    loss = (p * torch.tensor([float("inf"), 1.0])).sum()
    with amp.scale_loss(loss, opt) as scaled:
        scaled.backward()
    assert amp.state().scaler.scale == scale0 * 0.5
    before = p.detach().clone()
    opt.step()  # must be skipped
    assert torch.equal(p.detach(), before)
    # next good step is not skipped
    p.grad = None
    loss = p.sum()
    with amp.scale_loss(loss, opt) as scaled:
        scaled.backward()
    opt.step()
    assert not torch.equal(p.detach(), before)


def test_o0_passthrough():
    p = torch.nn.Parameter(torch.ones(2))
    opt = FusedSGD([p], lr=0.1)
    model = torch.nn.Module()
    model, opt = amp.initialize(model, opt, opt_level="O0")
    assert not amp.is_enabled()
    loss = p.sum()
    with amp.scale_loss(loss, opt) as scaled:
        scaled.backward()
    assert torch.allclose(p.grad, torch.ones(2))


def test_register_float_function_casts_up():
    """apex-API parity: registered functions see fp32 inputs while amp is
    enabled (reference README 'Apex' §4, commented L11)."""
    import types
    from ddp_tricks_amd import amp

    seen = {}

    def probe(x):
        seen["dtype"] = x.dtype
        return x

    ns = types.SimpleNamespace(probe=probe)
    amp.register_float_function(ns, "probe")
    amp._state.enabled = True
    try:
        ns.probe(torch.zeros(2, dtype=torch.bfloat16))
        assert seen["dtype"] == torch.float32
    finally:
        amp._state.enabled = False
    ns.probe(torch.zeros(2, dtype=torch.bfloat16))
    assert seen["dtype"] == torch.bfloat16   # passthrough when disabled
    # idempotent registration
    amp.register_float_function(ns, "probe")
    assert ns.probe._amp_registered
