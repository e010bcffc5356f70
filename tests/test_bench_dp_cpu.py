"""bench.py under world_size=2 (gloo, CPU): the exact path the driver's
multi-GPU scale bench takes — env rendezvous, per-rank model, DP
all-reduce, MAX-over-ranks timing, rank-0-only JSON — minus RCCL."""

import json
import os
import socket
import subprocess
import sys

import pytest


def _free_port() -> int:
    with socket.socket() as s:
        s.bind(("127.0.0.1", 0))
        return s.getsockname()[1]


@pytest.mark.timeout(300)
def test_bench_dp_world2(tmp_path):
    repo = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
    port = _free_port()
    procs = []
    for rank in range(2):
        env = dict(os.environ, WORLD_SIZE="2", RANK=str(rank),
                   LOCAL_RANK=str(rank), MASTER_ADDR="127.0.0.1",
                   MASTER_PORT=str(port), PYTHONPATH=repo)
        procs.append(subprocess.Popen(
            [sys.executable, os.path.join(repo, "bench.py"),
             "--model", "tiny-cpu", "--steps", "2", "--warmup", "1",
             "--batch", "2", "--gpus", "2"],
            cwd=repo, env=env, stdout=subprocess.PIPE,
            stderr=subprocess.STDOUT, text=True))
    outs = [p.communicate(timeout=240)[0] for p in procs]
    assert procs[0].returncode == 0, outs[0][-1500:]
    assert procs[1].returncode == 0, outs[1][-1500:]
    json0 = [l for l in outs[0].splitlines() if l.startswith("{")]
    json1 = [l for l in outs[1].splitlines() if l.startswith("{")]
    assert len(json0) == 1  # rank 0 prints exactly one line
    assert json1 == []      # rank 1 prints none
    rec = json.loads(json0[0])
    assert rec["n_gpus"] == 2
    assert rec["config"]["parallelism"] == "dp2"
    # value is the WHOLE-JOB aggregate: global batch = 2 ranks x 2
    assert rec["config"]["global_batch"] == 4
    toks = rec["config"]["global_batch"] * rec["config"]["seq_len"]
    assert abs(rec["value"] - toks / (rec["ms_per_step"] / 1e3)) \
        / rec["value"] < 1e-6
    assert rec["final_loss"] == rec["final_loss"]  # finite (all ranks reduced)
