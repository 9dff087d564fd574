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


@pytest.mark.timeout(300)
def test_bench_world2_rank0_prints_once():
    """The driver's N>1 launch shape: one process per rank with
    RANK/LOCAL_RANK/WORLD_SIZE/MASTER_* env — only rank 0 prints the JSON
    line and it reports the whole-job aggregate (n_gpus=2)."""
    procs = []
    for rank in range(2):
        env = dict(os.environ)
        env.update({"MASTER_ADDR": "127.0.0.1", "MASTER_PORT": "29678",
                    "RANK": str(rank), "LOCAL_RANK": str(rank),
                    "WORLD_SIZE": "2", "DDPX_NO_TQDM": "1"})
        procs.append(subprocess.Popen(
            [sys.executable, os.path.join(REPO, "bench.py"),
             "--steps", "2", "--warmup", "1", "--batch-size", "8"],
            env=env, cwd=REPO, stdout=subprocess.PIPE,
            stderr=subprocess.PIPE, text=True))
    outs = [p.communicate(timeout=280) for p in procs]
    assert all(p.returncode == 0 for p in procs), \
        [o[1][-1000:] for o in outs]
    json_lines = [ln for o, _ in outs for ln in o.splitlines()
                  if ln.startswith("{")]
    assert len(json_lines) == 1, json_lines   # rank 0 only
    rec = json.loads(json_lines[0])
    assert rec["n_gpus"] == 2
    assert rec["config"]["parallelism"] == "dp2"
    assert rec["config"]["global_batch"] == 16      # whole-job aggregate
    assert rec["config"]["per_gpu_batch"] == 8
