"""bench.py is a driver contract: one JSON line on stdout with the
whole-job metric. Pin the schema on CPU (tiny config) so a refactor
can't silently break the round-end measurement."""

import json
import math
import os
import subprocess
import sys

import pytest


@pytest.mark.timeout(300)
def test_bench_json_contract():
    repo = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
    out = subprocess.run(
        [sys.executable, os.path.join(repo, "bench.py"),
         "--model", "tiny-cpu", "--steps", "2", "--warmup", "1",
         "--batch", "4"],
        env=dict(os.environ, PYTHONPATH=repo), cwd=repo,
        capture_output=True, text=True, timeout=240)
    assert out.returncode == 0, out.stdout[-1500:] + out.stderr[-1500:]
    lines = [l for l in out.stdout.splitlines() if l.startswith("{")]
    assert len(lines) == 1, out.stdout  # exactly ONE JSON line
    rec = json.loads(lines[0])
    assert rec["metric"] == "train tokens/sec (whole node)"
    assert rec["unit"] == "tokens/s"
    assert rec["value"] > 0
    assert rec["n_gpus"] == 1
    assert rec["steps"] == 2 and rec["warmup"] == 1
    assert rec["ms_per_step"] > 0
    assert rec["higher_is_better"] is True
    assert rec["scaling"] == "weak"
    assert rec["vs_baseline"] is None
    assert rec["data"] == "synthetic"
    assert math.isfinite(rec["final_loss"])  # trajectory honesty field
    cfg = rec["config"]
    assert set(cfg) >= {"model", "global_batch", "seq_len", "parallelism"}
    assert cfg["parallelism"] == "dp1"
    # whole-job aggregate consistency: value == tokens/step / (ms/1000)
    toks = cfg["global_batch"] * cfg["seq_len"]
    assert abs(rec["value"] - toks / (rec["ms_per_step"] / 1e3)) / rec["value"] < 1e-6
