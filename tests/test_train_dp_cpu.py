"""End-to-end train.py under world_size=2 (gloo, CPU): the full CLI loop
— per-rank batch sharding, loss all-reduce, rank-0 checkpointing —
exercised the way torchrun would launch it."""

import os
import random
import socket
import subprocess
import sys

import pytest

AA = "ACDEFGHIKLMNPQRSTVWY"


def _free_port() -> int:
    with socket.socket() as s:
        s.bind(("127.0.0.1", 0))
        return s.getsockname()[1]


def _make_data(tmp_path):
    rng = random.Random(0)
    fasta = tmp_path / "t.fasta"
    with open(fasta, "w") as f:
        for i in range(40):
            L = rng.randint(10, 40)
            seq = "".join(rng.choice(AA) for _ in range(L))
            f.write(f">U{i} x Tax=Escherichia coli TaxID=1 RepID=U{i}\n{seq}\n")
    (tmp_path / "configs" / "data").mkdir(parents=True)
    (tmp_path / "configs" / "model").mkdir(parents=True)
    (tmp_path / "configs" / "data" / "tiny.toml").write_text(f"""
read_from = "{fasta}"
write_to = "./train_data"
num_samples = 40
max_seq_len = 64
prob_invert_seq_annotation = 0.5
fraction_valid_data = 0.2
num_sequences_per_file = 100
sort_annotations = true
""")
    (tmp_path / "configs" / "model" / "tiny.toml").write_text("""
num_tokens = 256
dim = 16
depth = 2
dim_head = 8
heads = 2
window_size = 16
seq_len = 64
global_mlp_depth = 1
""")


@pytest.mark.timeout(300)
def test_train_cli_world2_gloo(tmp_path):
    repo = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
    _make_data(tmp_path)
    subprocess.run(
        [sys.executable, os.path.join(repo, "generate_data.py"),
         "--data_dir", "./configs/data", "--name", "tiny"],
        cwd=tmp_path, check=True, capture_output=True, timeout=120)

    port = _free_port()
    procs = []
    for rank in range(2):
        env = dict(os.environ,
                   WORLD_SIZE="2", RANK=str(rank), LOCAL_RANK=str(rank),
                   MASTER_ADDR="127.0.0.1", MASTER_PORT=str(port),
                   PYTHONPATH=repo)
        procs.append(subprocess.Popen(
            [sys.executable, os.path.join(repo, "train.py"),
             "--config_path", "./configs/model", "--model_name", "tiny",
             "--data_path", "./train_data", "--batch_size", "2",
             "--grad_accum_every", "2", "--wandb_off", "--yes",
             "--checkpoint_every", "1", "--validate_every", "1",
             "--sample_every", "100000", "--max_steps", "2"],
            cwd=tmp_path, env=env,
            stdout=subprocess.PIPE, stderr=subprocess.STDOUT, text=True))

    outs = []
    for p in procs:
        out, _ = p.communicate(timeout=240)
        outs.append(out)
    assert procs[0].returncode == 0, outs[0][-2000:]
    assert procs[1].returncode == 0, outs[1][-2000:]
    # rank 0 logs the all-reduced loss and checkpoints; rank 1 is silent
    assert "loss:" in outs[0]
    assert "valid_loss:" in outs[0]
    assert "checkpoint to start at" in outs[0]
    assert "loss:" not in outs[1]
    ckpts = list((tmp_path / "ckpts").glob("ckpt_*.pkl"))
    assert len(ckpts) >= 1


@pytest.mark.timeout(300)
def test_graph_flag_says_why_disabled(tmp_path):
    """--graph must NEVER silently run eager: the r02 evidence scripts
    lost the graphed path for days because --grad_accum_every defaults
    to 4 and graph_ok quietly became False."""
    repo = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
    _make_data(tmp_path)
    subprocess.run(
        [sys.executable, os.path.join(repo, "generate_data.py"),
         "--data_dir", "./configs/data", "--name", "tiny"],
        cwd=tmp_path, check=True, capture_output=True, timeout=120)
    out = subprocess.run(
        [sys.executable, os.path.join(repo, "train.py"),
         "--config_path", "./configs/model", "--model_name", "tiny",
         "--data_path", "./train_data", "--batch_size", "2",
         "--graph", "--wandb_off", "--yes",
         "--checkpoint_every", "100000", "--validate_every", "100000",
         "--sample_every", "100000", "--max_steps", "1"],
        cwd=tmp_path, env=dict(os.environ, PYTHONPATH=repo),
        capture_output=True, text=True, timeout=240)
    assert out.returncode == 0, out.stdout[-2000:] + out.stderr[-2000:]
    assert "--graph disabled (" in out.stdout
    assert "no GPU" in out.stdout
    assert "--grad_accum_every 4 != 1" in out.stdout
