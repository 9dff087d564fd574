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


def test_phase_timers_epoch_line(tmp_path, capsys, monkeypatch):
    """DDPX_PHASE_TIMERS=1 end-to-end: the epoch line carries the phase
    breakdown (regression: the per-device timer cache broke the trainer's
    consumer in round 2)."""
    import argparse

    from ddp_tricks_amd.utils.train import train
    monkeypatch.setenv("DDPX_PHASE_TIMERS", "1")
    monkeypatch.setenv("DDPX_SYNTH_SAMPLES", "512")
    monkeypatch.setenv("DDPX_NO_TQDM", "1")
    args = argparse.Namespace(
        exp_name="timers", learning_rate=0.1, batch_size=128, epochs=1,
        warmup_epochs=1, warmup_type="linear", seed_num=42,
        data_path=str(tmp_path / "d"), model_path=str(tmp_path / "m"),
        local_rank=0)
    train(args)
    out = capsys.readouterr().out
    assert "phases[" in out and "fwd=" in out and "opt=" in out
