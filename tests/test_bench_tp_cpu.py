"""tools/bench_tp.py end-to-end under gloo world_size=2 on a tiny config:
the TP bench path (shard, train, reduce, report) runs and prints the
contract JSON line."""

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
def test_bench_tp_world2(tmp_path):
    repo = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
    (tmp_path / "configs" / "model").mkdir(parents=True)
    # bench_tp resolves configs relative to the repo, so drop a tiny
    # config there is not possible — use the checked-in small model? Its
    # dim/heads must divide by 2: small.toml has heads=8 — fine but big
    # for CI; craft a tiny one inside the repo configs would pollute.
    # Instead: run with the checked-in "tiny" config written under the
    # repo for the duration of the test.
    tiny = os.path.join(repo, "configs", "model", "_tp_test_tiny.toml")
    with open(tiny, "w") as f:
        f.write("""
num_tokens = 64
dim = 16
depth = 2
dim_head = 4
heads = 4
window_size = 8
seq_len = 32
ff_glu = true
global_mlp_depth = 1
""")
    try:
        port = _free_port()
        procs = []
        for rank in range(2):
            env = dict(os.environ, WORLD_SIZE="2", RANK=str(rank),
                       LOCAL_RANK=str(rank), MASTER_ADDR="127.0.0.1",
                       MASTER_PORT=str(port), PYTHONPATH=repo)
            procs.append(subprocess.Popen(
                [sys.executable, os.path.join(repo, "tools", "bench_tp.py"),
                 "--steps", "2", "--warmup", "1", "--batch", "2",
                 "--model", "_tp_test_tiny"],
                env=env, stdout=subprocess.PIPE,
                stderr=subprocess.STDOUT, text=True))
        outs = [p.communicate(timeout=240)[0] for p in procs]
        assert procs[0].returncode == 0, outs[0][-1500:]
        assert procs[1].returncode == 0, outs[1][-1500:]
        line = [l for l in outs[0].splitlines() if l.startswith("{")][-1]
        rec = json.loads(line)
        assert rec["config"]["parallelism"] == "tp2"
        assert rec["value"] > 0
    finally:
        os.remove(tiny)
