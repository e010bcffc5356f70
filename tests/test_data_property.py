"""Property-based tests (hypothesis) for the TFRecord wire format and
tokenizer — SURVEY.md §4's recommended round-trip coverage."""

import io

import numpy as np
from hypothesis import given, settings, strategies as st

from progen_amd import data as D


@given(st.binary(min_size=0, max_size=4096))
@settings(max_examples=50, deadline=None)
def test_example_roundtrip_any_bytes(payload):
    assert D.decode_example(D.encode_example(payload)) == payload


@given(st.lists(st.binary(min_size=1, max_size=300), min_size=1, max_size=20))
@settings(max_examples=30, deadline=None)
def test_record_stream_roundtrip(seqs):
    buf = io.BytesIO()
    for s in seqs:
        D.write_record(buf, D.encode_example(s))
    buf.seek(0)
    got = [D.decode_example(p) for p in D.read_records(buf)]
    assert got == seqs


@given(st.integers(min_value=0, max_value=2**31 - 1))
@settings(max_examples=50, deadline=None)
def test_varint_roundtrip(n):
    enc = D._varint(n)
    val, pos = D._read_varint(enc, 0)
    assert val == n and pos == len(enc)


@given(st.text(alphabet=st.characters(min_codepoint=1, max_codepoint=254),
               min_size=0, max_size=200))
@settings(max_examples=50, deadline=None)
def test_tokenizer_roundtrip_any_text(s):
    toks = D.encode_tokens(s)
    assert all(1 <= t <= 255 for t in toks)
    assert D.decode_tokens(np.array(toks)) == s


@given(st.lists(st.binary(min_size=1, max_size=100), min_size=1, max_size=8),
       st.integers(min_value=4, max_value=64))
@settings(max_examples=30, deadline=None)
def test_collate_invariants(batch, pad_length):
    out = D.collate_fn(batch, pad_length=pad_length, offset=1)
    assert out.shape == (len(batch), pad_length)
    assert out.dtype == np.uint16
    for i, b in enumerate(batch):
        L = min(len(b), pad_length)
        np.testing.assert_array_equal(
            out[i, :L], np.frombuffer(b, dtype=np.uint8)[:L].astype(np.uint16) + 1)
        assert (out[i, L:] == 0).all()
