"""End-to-end train(args) on CPU/gloo world 1 (BASELINE.json config 1):
plumbing of warmup + plateau + early stopping + checkpoint + TB layout."""
import os
import types

import pytest
import torch
import torch.distributed as dist

from ddp_tricks_amd import amp


@pytest.fixture(autouse=True)
def _reset_amp():
    amp._state.__init__()
    yield
    amp._state.__init__()


def _args(tmp_path, **over):
    a = types.SimpleNamespace(
        exp_name="TEST_run", learning_rate=0.1, batch_size=64, epochs=3,
        warmup_epochs=2, warmup_type="linear", seed_num=42,
        data_path="/nonexistent_data", model_path=str(tmp_path),
        local_rank=0)
    for k, v in over.items():
        setattr(a, k, v)
    return a


@pytest.mark.timeout(300)
def test_train_e2e_cpu(tmp_path, monkeypatch, capsys):
    monkeypatch.setenv("DDPX_SYNTH_SAMPLES", "256")
    if not dist.is_initialized():
        os.environ.setdefault("MASTER_ADDR", "127.0.0.1")
        os.environ.setdefault("MASTER_PORT", "29655")
        dist.init_process_group("gloo", rank=0, world_size=1)
    from ddp_tricks_amd.utils.train import train
    train(_args(tmp_path))
    out = capsys.readouterr().out
    # epoch 0 trains at LR 0
    assert "learning_rate: 0.0," in out
    # epoch line format conformance (reference utils/train.py:101)
    import re
    assert re.search(
        r"epoch: \d{3}/3, time: \d+\.\d{2}s, learning_rate: [\d.e-]+, "
        r"train_loss: \d+\.\d{4}, train_acc: \d+\.\d{4}, "
        r"valid_loss: \d+\.\d{4}, valid_acc: \d+\.\d{4}", out), out
    # checkpoint written with the reference naming scheme
    assert os.path.exists(os.path.join(tmp_path, "TEST_run.pt"))
    sd = torch.load(os.path.join(tmp_path, "TEST_run.pt"))
    assert len(sd) == 37
    # TB layout
    logdir = os.path.join(tmp_path, "logs", "TEST_run")
    subdirs = {d for d in os.listdir(logdir)
               if os.path.isdir(os.path.join(logdir, d))}
    assert {"Loss_train", "Loss_valid", "Acc_train", "Acc_valid"} <= subdirs
    dist.destroy_process_group()


def test_validate_tool(tmp_path, monkeypatch):
    """tools/validate.py evaluates a saved best checkpoint."""
    import subprocess
    import sys
    import types

    import torch.distributed as dist

    from ddp_tricks_amd import amp
    monkeypatch.setenv("DDPX_SYNTH_SAMPLES", "256")
    monkeypatch.setenv("DDPX_NO_TQDM", "1")
    if not dist.is_initialized():
        os.environ.setdefault("MASTER_ADDR", "127.0.0.1")
        os.environ.setdefault("MASTER_PORT", "29684")
        os.environ.setdefault("RANK", "0")
        os.environ.setdefault("WORLD_SIZE", "1")
        dist.init_process_group("gloo", rank=0, world_size=1)
    amp._state.__init__()
    from ddp_tricks_amd.utils.train import train
    args = types.SimpleNamespace(
        exp_name="VAL", learning_rate=0.05, batch_size=64, epochs=1,
        warmup_epochs=1, warmup_type="linear", seed_num=42,
        data_path="/nonexistent", model_path=str(tmp_path), local_rank=0)
    train(args)
    repo = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
    env = dict(os.environ, DDPX_SYNTH_SAMPLES="256")
    r = subprocess.run(
        [sys.executable, os.path.join(repo, "tools", "validate.py"),
         "-n", "VAL", "-p", str(tmp_path), "-d", "/nonexistent"],
        capture_output=True, text=True, timeout=300, env=env, cwd=repo)
    assert r.returncode == 0, r.stderr
    assert "valid_acc" in r.stdout


@pytest.mark.timeout(300)
def test_two_runs_same_seed_identical(tmp_path, monkeypatch):
    """The reference's de-facto verification strategy:
    same seed, same command => identical trained weights."""
    import types

    import torch.distributed as dist

    from ddp_tricks_amd.utils.train import train
    monkeypatch.setenv("DDPX_SYNTH_SAMPLES", "256")
    monkeypatch.setenv("DDPX_NO_TQDM", "1")
    if not dist.is_initialized():
        os.environ.setdefault("MASTER_ADDR", "127.0.0.1")
        os.environ.setdefault("MASTER_PORT", "29656")
        dist.init_process_group("gloo", rank=0, world_size=1)
    sds = []
    for run in range(2):
        amp._state.__init__()
        a = types.SimpleNamespace(
            exp_name=f"DET{run}", learning_rate=0.1, batch_size=64, epochs=2,
            warmup_epochs=1, warmup_type="linear", seed_num=42,
            data_path="/nonexistent", model_path=str(tmp_path), local_rank=0,
            resume=True)
        train(a)
        sds.append(torch.load(os.path.join(tmp_path, f"DET{run}.resume.pt"),
                              weights_only=False)["model"])
    for k in sds[0]:
        assert torch.equal(sds[0][k], sds[1][k]), k


def test_synthetic_train_never_pairs_with_real_valid(tmp_path, capsys):
    """Coherence guard: when train images are absent (synthetic) but real
    t10k files exist, the trainer must fall back to a synthetic valid set
    (validating synthetic training on real digits reports chance acc and
    distorts EarlyStopping/plateau/checkpoint gates)."""
    import gzip
    import shutil
    import struct

    import numpy as np

    raw = tmp_path / "MNIST" / "raw"
    raw.mkdir(parents=True)
    # forge tiny REAL-format t10k idx files (images + labels), no train files
    n = 64
    img = np.random.default_rng(0).integers(0, 255, (n, 28, 28),
                                            dtype=np.uint8)
    with gzip.open(raw / "t10k-images-idx3-ubyte.gz", "wb") as f:
        f.write(struct.pack(">IIII", 0x00000803, n, 28, 28))
        f.write(img.tobytes())
    lbl = np.random.default_rng(1).integers(0, 10, (n,), dtype=np.uint8)
    with gzip.open(raw / "t10k-labels-idx1-ubyte.gz", "wb") as f:
        f.write(struct.pack(">II", 0x00000801, n))
        f.write(lbl.tobytes())

    import argparse

    from ddp_tricks_amd.utils.train import train
    args = argparse.Namespace(
        exp_name="coherence", learning_rate=0.1, batch_size=32, epochs=1,
        warmup_epochs=1, warmup_type="linear", seed_num=42,
        data_path=str(tmp_path), model_path=str(tmp_path / "out"),
        local_rank=0)
    os.environ["DDPX_SYNTH_SAMPLES"] = "64"
    try:
        train(args)
    finally:
        del os.environ["DDPX_SYNTH_SAMPLES"]
    out = capsys.readouterr().out
    assert "synthetic valid split" in out
