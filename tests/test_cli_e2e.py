"""End-to-end slice: synthetic fasta -> generate_data -> train (CPU,
tiny) -> checkpoint -> resume -> sample  (SURVEY.md §7.3)."""

import random

import pytest
from click.testing import CliRunner

AA = "ACDEFGHIKLMNPQRSTVWY"


def _write_fasta(path, n=30, maxlen=40):
    rng = random.Random(0)
    with open(path, "w") as f:
        for i in range(n):
            L = rng.randint(10, maxlen)
            seq = "".join(rng.choice(AA) for _ in range(L))
            f.write(f">UniRef50_A{i} Some protein n=1 Tax=Escherichia coli "
                    f"TaxID=562 RepID=A{i}_ECOLI\n")
            for j in range(0, L, 12):
                f.write(seq[j:j + 12] + "\n")
    return path


@pytest.fixture()
def workdir(tmp_path, monkeypatch):
    monkeypatch.chdir(tmp_path)
    fasta = _write_fasta(tmp_path / "tiny.fasta")
    (tmp_path / "configs" / "data").mkdir(parents=True)
    (tmp_path / "configs" / "model").mkdir(parents=True)
    (tmp_path / "configs" / "data" / "tiny.toml").write_text(f"""
read_from = "{fasta}"
write_to = "./train_data"
num_samples = 30
max_seq_len = 64
prob_invert_seq_annotation = 0.5
fraction_valid_data = 0.2
num_sequences_per_file = 20
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
    return tmp_path


@pytest.mark.timeout(300)
def test_generate_train_resume_sample(workdir):
    import generate_data
    import sample as sample_cli
    import train as train_cli

    r = CliRunner().invoke(generate_data.main,
                           ["--data_dir", "./configs/data", "--name", "tiny"])
    assert r.exit_code == 0, r.output
    shards = list((workdir / "train_data").glob("*.tfrecord.gz"))
    assert any(".train." in s.name for s in shards)
    assert any(".valid." in s.name for s in shards)
    # filename count contract
    for s in shards:
        assert s.name.split(".")[-4].isdigit()

    common = ["--config_path", "./configs/model", "--model_name", "tiny",
              "--data_path", "./train_data", "--batch_size", "2",
              "--grad_accum_every", "2", "--wandb_off",
              "--checkpoint_every", "1", "--validate_every", "1",
              "--sample_every", "1000", "--max_steps", "2",
              "--prime_length", "4"]
    r = CliRunner().invoke(train_cli.main, common)
    assert r.exit_code == 0, r.output
    assert "loss:" in r.output and "valid_loss:" in r.output
    ckpts = list((workdir / "ckpts").glob("ckpt_*.pkl"))
    assert len(ckpts) >= 1

    # resume: model config comes from checkpoint, data skips ahead
    r = CliRunner().invoke(train_cli.main, common)
    assert r.exit_code == 0, r.output
    assert "starting from sequence" in r.output
    start = [ln for ln in r.output.splitlines()
             if ln.startswith("starting from sequence")][0]
    assert int(start.split()[-1]) > 0

    # sample from the checkpoint — all three decoder paths
    r = CliRunner().invoke(sample_cli.main, ["--prime", "# M"])
    assert r.exit_code == 0, r.output
    assert "params:" in r.output
    r = CliRunner().invoke(sample_cli.main, ["--prime", "# M", "--fast"])
    assert r.exit_code == 0, r.output
    r = CliRunner().invoke(sample_cli.main, ["--prime", "# M", "--cached"])
    assert r.exit_code == 0, r.output
