"""Phase-timer subsystem: CPU fallback accumulation, reset,
and trainer integration via DDPX_PHASE_TIMERS."""
import time

from ddp_tricks_amd.utils.timers import PhaseTimers


def test_cpu_phase_accumulation():
    t = PhaseTimers(None)
    with t.phase("fwd"):
        time.sleep(0.01)
    with t.phase("fwd"):
        time.sleep(0.01)
    with t.phase("opt"):
        time.sleep(0.005)
    s = t.summary()
    assert s["fwd"] >= 15.0 and s["opt"] >= 4.0
    assert "bwd" not in s  # empty phases omitted
    t.reset()
    assert t.summary() == {}


def test_format():
    t = PhaseTimers(None)
    with t.phase("h2d"):
        time.sleep(0.002)
    assert "h2d=" in t.format()
