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


def _async_record(fi_val: float):
    """Drive the async-mode device-side recording exactly as scale_loss
    does (amp/__init__.py async branch), with CPU buffers."""
    amp._state.async_mode = True
    amp._state.async_steps += 1
    fi = amp._device_buffers(torch.device("cpu"))
    fi.zero_()
    fi += fi_val
    amp._state.overflow_count += fi
    of = fi > 0
    step = float(amp._state.async_steps)
    amp._state.first_of_step.masked_fill_(of & (amp._state.first_of_step < 0), step)
    amp._state.last_of_step.masked_fill_(of, step)
    if amp._state.async_steps >= amp._state.scaler.growth_interval:
        amp.maybe_sync_scaler()


def test_async_replay_matches_apex_sequencing():
    """Windowed overflow replay must land on the same scale as apex's
    per-step policy, for arbitrary overflow patterns and window breaks
    (VERDICT r01 weak #6 / next-round #9)."""
    import random
    rng = random.Random(7)
    for trial in range(50):
        interval = rng.choice([3, 5, 8])
        n_steps = rng.randint(1, 60)
        pattern = [rng.random() < 0.25 for _ in range(n_steps)]
        # windows break at arbitrary points (epoch ends) no longer than
        # growth_interval (enforced inside the recorder)
        amp._state.__init__()
        amp._state.scaler = amp.DynamicLossScaler(init_scale=2.0 ** 16,
                                                  growth_interval=interval)
        ref = amp.DynamicLossScaler(init_scale=2.0 ** 16,
                                    growth_interval=interval)
        for of in pattern:
            ref.update(found_inf=of)
            _async_record(1.0 if of else 0.0)
            if rng.random() < 0.15:
                amp.maybe_sync_scaler()   # epoch boundary
        amp.maybe_sync_scaler()
        assert amp._state.scaler.scale == ref.scale, (
            f"trial {trial}: pattern={pattern} interval={interval} "
            f"got {amp._state.scaler.scale} want {ref.scale}")
        assert amp._state.scaler._good_steps == ref._good_steps
    amp._state.__init__()


def test_register_half_function_casts_kwargs():
    """Floating tensor kwargs are cast like positional args."""
    class NS:
        @staticmethod
        def f(a, b=None):
            return a, b

    ns = NS()
    amp.register_half_function(ns, "f")
    model, opt = amp.initialize(torch.nn.Module(), None, "O1")
    a, b = ns.f(torch.ones(2, dtype=torch.float32),
                b=torch.ones(2, dtype=torch.float32))
    assert a.dtype == torch.bfloat16
    assert b.dtype == torch.bfloat16
    amp._state.__init__()
