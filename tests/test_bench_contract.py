"""Driver-contract guard for bench.py: one JSON line on stdout with the
required fields (the round driver parses this verbatim).  Runs the real
script as a subprocess on CPU/gloo with a tiny config."""
import json
import os
import subprocess
import sys

import pytest

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))

REQUIRED = ["metric", "value", "unit", "n_gpus", "steps", "warmup",
            "ms_per_step", "higher_is_better", "scaling", "vs_baseline",
            "dtype", "data", "config"]


@pytest.mark.timeout(300)
def test_bench_json_contract():
    env = dict(os.environ)
    env.update({"MASTER_ADDR": "127.0.0.1", "MASTER_PORT": "29677",
                "DDPX_NO_TQDM": "1"})
    out = subprocess.run(
        [sys.executable, os.path.join(REPO, "bench.py"),
         "--steps", "2", "--warmup", "1", "--batch-size", "8"],
        env=env, cwd=REPO, capture_output=True, text=True, timeout=280)
    assert out.returncode == 0, out.stderr[-2000:]
    json_lines = [ln for ln in out.stdout.splitlines()
                  if ln.startswith("{")]
    assert len(json_lines) == 1, out.stdout
    rec = json.loads(json_lines[0])
    for k in REQUIRED:
        assert k in rec, k
    assert rec["metric"] == "images/sec"
    assert rec["n_gpus"] == 1
    assert rec["steps"] == 2 and rec["warmup"] == 1
    assert rec["higher_is_better"] is True and rec["scaling"] == "weak"
    assert rec["value"] > 0 and rec["ms_per_step"] > 0
    # headline model reports vs_baseline against BASELINE.md's number
    assert rec["vs_baseline"] == pytest.approx(rec["value"] / 12000.0)
    cfg = rec["config"]
    assert cfg["model"] == "Toy_Net" and cfg["parallelism"] == "dp1"
    assert cfg["global_batch"] == 8
    assert rec["data"] == "synthetic"
