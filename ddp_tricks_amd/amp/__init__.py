"""amp — the framework's mixed-precision runtime (apex-amp-O1 replacement).

The reference initializes apex amp O1 (utils/train.py:58) and wraps every
backward in ``amp.scale_loss`` (utils/process.py:26-27).  This module keeps
that exact API shape with MI355X-native mechanics:

  * compute dtype bf16 (MFMA-native on CDNA4) with fp32 master weights —
    the casts happen inside ops/functional at the conv/linear boundaries
    (whitelist), BN stats / CE / reductions stay fp32 (blacklist);
  * dynamic loss scaling with apex's policy: init 2**16, halve on
    overflow, double after 2000 clean steps (bf16 shares fp32's exponent
    range so the scaler is parity machinery, kept for API and behavior
    compatibility);
  * ``scale_loss.__exit__`` = the reducer/unscale rendezvous: waits the
    DDP bucket all-reduces (launched on the side stream during backward),
    then runs one fused multi-tensor unscale+inf-check over the flat
    gradient buckets, folding in the 1/world_size gradient average
    (all-reduce is SUM) — one kernel pass over the 23 MB gradient payload
    (SURVEY N4 disposition);
  * on overflow the wrapped optimizer's next step is skipped, like apex.
"""
from __future__ import annotations

import contextlib
import os
from typing import Optional

import torch


class DynamicLossScaler:
    def __init__(self, init_scale: float = 2.0 ** 16, growth_factor: float = 2.0,
                 backoff_factor: float = 0.5, growth_interval: int = 2000,
                 min_scale: float = 1.0):
        self.scale = init_scale
        self.growth_factor = growth_factor
        self.backoff_factor = backoff_factor
        self.growth_interval = growth_interval
        self.min_scale = min_scale
        self._good_steps = 0

    def update(self, found_inf: bool) -> None:
        if found_inf:
            self.scale = max(self.scale * self.backoff_factor, self.min_scale)
            self._good_steps = 0
        else:
            self._good_steps += 1
            if self._good_steps >= self.growth_interval:
                self.scale *= self.growth_factor
                self._good_steps = 0


class _AmpState:
    def __init__(self):
        self.enabled = False
        self.scaler: Optional[DynamicLossScaler] = None
        self.optimizer = None
        self.ddp_model = None
        self.last_overflow = False
        # async (device-side) overflow tracking — GPU HIP path only
        self.found_inf = None        # per-step flag, read by the fused
        #                              optimizer kernels (device-side skip)
        self.overflow_count = None   # running count, host-synced per epoch
        self.first_of_step = None    # 1-based step idx of first overflow in
        self.last_of_step = None     #   the window / of the last one — lets
        #                              the window replay reconstruct apex's
        #                              exact backoff/growth sequencing
        self.async_steps = 0         # backward count since last scaler sync
        self.async_mode = False


_state = _AmpState()


def _device_buffers(device):
    if _state.found_inf is None or _state.found_inf.device != device:
        _state.found_inf = torch.zeros(1, dtype=torch.float32, device=device)
        _state.overflow_count = torch.zeros(1, dtype=torch.float32,
                                            device=device)
        _state.first_of_step = torch.full((1,), -1.0, dtype=torch.float32,
                                          device=device)
        _state.last_of_step = torch.zeros(1, dtype=torch.float32,
                                          device=device)
    return _state.found_inf


def pending_found_inf():
    """Device-side overflow flag for the fused optimizer kernels (or None)."""
    return _state.found_inf if _state.async_mode else None


def maybe_sync_scaler() -> None:
    """Window-level host sync of the deferred overflow tracking (async mode):
    folds overflows seen since the last sync into the dynamic scaler with
    apex's exact sequencing.  Replay = (first_overflow_step - 1) good
    updates, then n overflow updates, then (steps - last_overflow_step)
    good updates.  Good steps *between* overflows are dropped: with the
    window length capped at growth_interval (scale_loss syncs whenever
    async_steps reaches it) an inter-overflow streak can never reach the
    growth threshold, so dropping them cannot change the scaler state —
    the replay is exactly apex-equivalent.

    Remaining (documented) divergence from apex's synchronous policy: the
    *unscale multiplier* within the window uses the window-entry scale, and
    Lookahead's k-counter still advances on device-skipped overflow steps.
    Both are moot for bf16 compute (same exponent range as fp32: overflow
    means genuinely non-finite loss)."""
    if not _state.async_mode or _state.overflow_count is None:
        return
    n = int(_state.overflow_count.item())
    if n == 0:
        for _ in range(_state.async_steps):
            _state.scaler.update(found_inf=False)
    else:
        first = int(_state.first_of_step.item())
        last = int(_state.last_of_step.item())
        for _ in range(max(0, first - 1)):
            _state.scaler.update(found_inf=False)
        for _ in range(n):
            _state.scaler.update(found_inf=True)
        for _ in range(max(0, _state.async_steps - last)):
            _state.scaler.update(found_inf=False)
        _state.overflow_count.zero_()
        _state.first_of_step.fill_(-1.0)
        _state.last_of_step.zero_()
    _state.async_steps = 0


def state_dict() -> dict:
    """Serializable scaler state (checkpoint/resume — SURVEY §5.4 ext)."""
    if _state.scaler is None:
        return {}
    sc = _state.scaler
    return {"scale": sc.scale, "good_steps": sc._good_steps}


def load_state_dict(sd: dict) -> None:
    if _state.scaler is not None and sd:
        _state.scaler.scale = sd["scale"]
        _state.scaler._good_steps = sd["good_steps"]


def is_enabled() -> bool:
    return _state.enabled


def state() -> _AmpState:
    return _state


def register_ddp(ddp_model) -> None:
    _state.ddp_model = ddp_model


def initialize(model, optimizers, opt_level: str = "O1", loss_scale="dynamic"):
    """apex-compatible entry point (model, optimizer [, ...]) -> same tuple."""
    if opt_level not in ("O0", "O1", "O2"):
        raise ValueError(f"unsupported opt_level {opt_level!r}")
    _state.enabled = opt_level != "O0"
    if loss_scale == "dynamic" or loss_scale is None:
        _state.scaler = DynamicLossScaler()
    else:
        _state.scaler = DynamicLossScaler(init_scale=float(loss_scale),
                                          growth_interval=10 ** 12)
    _state.optimizer = optimizers
    if optimizers is not None:
        _patch_step(optimizers)
    return model, optimizers


def _patch_step(optimizer) -> None:
    """Skip optimizer.step() after an overflow backward, like apex amp."""
    if getattr(optimizer, "_amp_patched", False):
        return
    raw_step = optimizer.step

    def step(closure=None):
        if _state.last_overflow:
            _state.last_overflow = False
            return None
        return raw_step(closure) if closure is not None else raw_step()

    optimizer.step = step
    optimizer._amp_patched = True


def _collect_grads(optimizer):
    grads = []
    for group in optimizer.param_groups:
        for p in group["params"]:
            if p.grad is not None:
                grads.append(p.grad)
    return grads


def _unscale_and_check(tensors, inv_scale: float) -> bool:
    """Multiply tensors by inv_scale in place; return True if any inf/nan."""
    if not tensors:
        return False
    dev = tensors[0].device
    found_inf = torch.zeros(1, dtype=torch.float32, device=dev)
    inv = torch.full((1,), inv_scale, dtype=torch.float32, device=dev)
    if dev.type == "cuda":
        from ..ops import load_extension
        ext = load_extension(required=False)
        if ext is not None:
            ext.multi_tensor_unscale(tensors, found_inf, inv_scale)
            return bool(found_inf.item())
    torch._amp_foreach_non_finite_check_and_unscale_(tensors, found_inf, inv)
    return bool(found_inf.item())


@contextlib.contextmanager
def scale_loss(loss, optimizer, delay_unscale: bool = False):
    """`with amp.scale_loss(loss, opt) as scaled: scaled.backward()`.

    On exit (backward done): finalize DDP bucket all-reduces, then fused
    unscale(+1/world average)+inf-check, then scaler update / step-skip.
    """
    if not _state.enabled or _state.scaler is None:
        yield loss
        if _state.ddp_model is not None:
            _state.ddp_model.finalize_backward(average=True)
        return

    scaler = _state.scaler
    yield loss * scaler.scale

    world = 1
    tensors = None
    if _state.ddp_model is not None:
        _state.ddp_model.finalize_backward(average=False)
        world = _state.ddp_model.world_size
        tensors = _state.ddp_model.bucket_flats()
    if tensors is None:
        tensors = _collect_grads(optimizer)
    inv = 1.0 / (scaler.scale * world)

    # Async path (GPU + HIP ext): unscale+check leaves the flag on device;
    # the fused SGD/Lookahead kernels skip themselves when it is set and
    # the scaler syncs once per epoch (maybe_sync_scaler).
    if tensors and tensors[0].is_cuda and os.environ.get(
            "DDPX_SYNC_AMP", "0") != "1":
        from ..ops import load_extension
        ext = load_extension(required=False)
        if ext is not None:
            _state.async_mode = True
            _state.async_steps += 1
            fi = _device_buffers(tensors[0].device)
            fi.zero_()  # stream-ordered: prior step's kernels already read it
            ext.multi_tensor_unscale(tensors, fi, inv)
            _state.overflow_count += fi
            # device-side (no host sync) record of first/last overflow step
            of = fi > 0
            step = float(_state.async_steps)
            _state.first_of_step.masked_fill_(
                of & (_state.first_of_step < 0), step)
            _state.last_of_step.masked_fill_(of, step)
            # cap the window at growth_interval so the replay in
            # maybe_sync_scaler stays exactly apex-equivalent
            if _state.async_steps >= _state.scaler.growth_interval:
                maybe_sync_scaler()
            return

    found_inf = _unscale_and_check(tensors, inv)
    scaler.update(found_inf)
    _state.last_overflow = found_inf


def master_params(optimizer):
    for group in optimizer.param_groups:
        for p in group["params"]:
            yield p


def register_float_function(mod, name: str) -> None:
    """apex-API parity (reference README.md L11 'Apex' §4: registered
    functions run in fp32 under amp).  Wraps ``mod.name`` to cast floating
    tensor args up to fp32 while amp is enabled.  Unlike apex this may be
    called before OR after ``initialize`` (the wrapper checks at call
    time)."""
    _register_cast(mod, name, torch.float32)


def register_half_function(mod, name: str) -> None:
    """apex-API parity: run ``mod.name`` in the compute dtype (bf16 here —
    MFMA-native on CDNA4, where apex used fp16)."""
    _register_cast(mod, name, torch.bfloat16)


def _register_cast(mod, name: str, dtype) -> None:
    fn = getattr(mod, name)
    if getattr(fn, "_amp_registered", False):
        return

    def wrapped(*args, **kwargs):
        if _state.enabled:
            cast = (lambda a: a.to(dtype)
                    if torch.is_tensor(a) and a.is_floating_point() else a)
            args = tuple(cast(a) for a in args)
            kwargs = {k: cast(v) for k, v in kwargs.items()}  # apex casts kwargs too
        return fn(*args, **kwargs)

    wrapped._amp_registered = True
    setattr(mod, name, wrapped)
