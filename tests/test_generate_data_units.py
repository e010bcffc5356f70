"""Unit tests for generate_data.py's parsing/emission functions
(reference parity: generate_data.py:36-74) — the CLI e2e test covers
the pipeline; these pin the per-record behavior."""

import random as pyrandom

import generate_data as G


def test_read_fasta_multiline_and_case(tmp_path):
    p = tmp_path / "t.fasta"
    p.write_text(">A desc one\nacd\nefg\n>B other\nHIK\n")
    rows = list(G.read_fasta(str(p)))
    assert rows == [("A desc one", "ACDEFG"), ("B other", "HIK")]


def test_tax_regex_parity():
    # the reference regex captures the Tax= genus/species words
    # (reference: generate_data.py:36-43)
    cfg = {}
    d = ("UniRef50_P0A7G6 Recombination protein n=1 "
         "Tax=Escherichia coli TaxID=562 RepID=RECA_ECOLI")
    ann = G.get_annotations_from_description(cfg, d)
    assert ann == {"tax": "Escherichia coli"}
    # no Tax= field -> empty
    assert G.get_annotations_from_description(cfg, "plain desc") == {}


def test_row_to_sequence_strings_dual_emission():
    cfg = {"sort_annotations": True, "prob_invert_seq_annotation": 0.0}
    d = "X Tax=Escherichia coli TaxID=1 RepID=X"
    out = G.row_to_sequence_strings(cfg, d, "MKV")
    # annotated variant + plain variant (reference: generate_data.py:45-74)
    assert out == [b"[tax=Escherichia coli] # MKV", b"# MKV"]

    # no annotations -> plain only
    out2 = G.row_to_sequence_strings(cfg, "no tax here", "MKV")
    assert out2 == [b"# MKV"]


def test_row_to_sequence_strings_inversion():
    cfg = {"sort_annotations": True, "prob_invert_seq_annotation": 1.0}
    d = "X Tax=Escherichia coli TaxID=1 RepID=X"
    pyrandom.seed(0)
    out = G.row_to_sequence_strings(cfg, d, "MKV")
    # always-invert puts the sequence before the annotation
    assert out[0] == b"MKV # [tax=Escherichia coli]"
    assert out[1] == b"# MKV"


def test_parallel_shard_writing_matches_serial(tmp_path, monkeypatch):
    """--workers N writes the same shard set with the same records per
    shard as the serial run (shards are independent units of work)."""
    import numpy as np

    import generate_data as G
    from progen_amd.data import iter_tfrecord_file

    monkeypatch.chdir(tmp_path)
    fasta = tmp_path / "t.fasta"
    with open(fasta, "w") as f:
        for i in range(30):
            f.write(f">U{i} x Tax=Escherichia coli TaxID=1 RepID=U{i}\n"
                    f"{'ACDEFGHIK'[:(i % 8) + 2] * 3}\n")
    cfg = dict(read_from=str(fasta), write_to="./out", num_samples=30,
               max_seq_len=128, prob_invert_seq_annotation=0.5,
               fraction_valid_data=0.2, num_sequences_per_file=7,
               sort_annotations=True)
    G.fasta_to_tmp_files(cfg)

    def snapshot(workers):
        np.random.seed(123)  # same permutation/split both runs
        G.files_to_tfrecords(dict(cfg), workers=workers)
        out = {}
        for p in sorted((tmp_path / "out").iterdir()):
            out[p.name] = list(iter_tfrecord_file(str(p)))
        return out

    serial = snapshot(1)
    parallel = snapshot(3)
    assert len(serial) > 2  # multiple shards, else the test is vacuous
    assert serial == parallel
