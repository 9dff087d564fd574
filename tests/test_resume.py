"""Checkpoint/resume extension: sidecar checkpoint restores
model/optimizer/scheduler/early-stop/scaler state and continues the epoch
count; the reference best-weights .pt layout is untouched."""
import os
import types

import pytest
import torch
import torch.distributed as dist

from ddp_tricks_amd import amp


def _args(tmp_path, epochs, resume):
    return types.SimpleNamespace(
        exp_name="RES", learning_rate=0.05, batch_size=64, epochs=epochs,
        warmup_epochs=2, warmup_type="linear", seed_num=42,
        data_path="/nonexistent", model_path=str(tmp_path), local_rank=0,
        resume=resume)


@pytest.fixture
def dist_env():
    if not dist.is_initialized():
        os.environ.setdefault("MASTER_ADDR", "127.0.0.1")
        os.environ.setdefault("MASTER_PORT", "29681")
        os.environ.setdefault("RANK", "0")
        os.environ.setdefault("WORLD_SIZE", "1")
        dist.init_process_group("gloo", rank=0, world_size=1)
    yield


def test_resume_continues(tmp_path, monkeypatch, dist_env, capsys):
    monkeypatch.setenv("DDPX_SYNTH_SAMPLES", "256")
    monkeypatch.setenv("DDPX_NO_TQDM", "1")
    from ddp_tricks_amd.utils.train import train

    amp._state.__init__()
    train(_args(tmp_path, epochs=2, resume=True))
    ck = os.path.join(tmp_path, "RES.resume.pt")
    assert os.path.exists(ck)
    saved = torch.load(ck, weights_only=False)
    assert saved["epoch"] == 1
    assert "optimizer" in saved and "scheduler_wu" in saved

    amp._state.__init__()
    capsys.readouterr()
    train(_args(tmp_path, epochs=4, resume=True))
    out = capsys.readouterr().out
    assert "resumed RES at epoch 2" in out
    assert "epoch: 002/4" in out and "epoch: 000/4" not in out
    assert torch.load(ck, weights_only=False)["epoch"] == 3


def test_no_resume_flag_ignores_sidecar(tmp_path, monkeypatch, dist_env, capsys):
    monkeypatch.setenv("DDPX_SYNTH_SAMPLES", "256")
    monkeypatch.setenv("DDPX_NO_TQDM", "1")
    from ddp_tricks_amd.utils.train import train
    amp._state.__init__()
    train(_args(tmp_path, epochs=1, resume=False))
    assert not os.path.exists(os.path.join(tmp_path, "RES.resume.pt"))


def test_resume_bitwise_equals_straight_run(tmp_path, monkeypatch, dist_env):
    """Strong conformance: 2 epochs + resume for 2 more must produce the
    SAME final weights as 4 straight epochs — every piece of state the
    sidecar carries (model/momentum/schedulers/early-stop/scaler) and the
    per-epoch determinism (sampler set_epoch seeding) must line up."""
    from ddp_tricks_amd.utils.train import train
    monkeypatch.setenv("DDPX_SYNTH_SAMPLES", "256")
    monkeypatch.setenv("DDPX_NO_TQDM", "1")

    amp._state.__init__()
    a = _args(tmp_path, epochs=4, resume=True)
    a.exp_name = "STRAIGHT"
    train(a)
    straight = torch.load(os.path.join(tmp_path, "STRAIGHT.resume.pt"),
                          weights_only=False)["model"]

    amp._state.__init__()
    b = _args(tmp_path, epochs=2, resume=True)
    b.exp_name = "SPLIT"
    train(b)
    amp._state.__init__()
    c = _args(tmp_path, epochs=4, resume=True)
    c.exp_name = "SPLIT"
    train(c)
    split = torch.load(os.path.join(tmp_path, "SPLIT.resume.pt"),
                       weights_only=False)["model"]

    for k in straight:
        assert torch.equal(straight[k], split[k]), k
