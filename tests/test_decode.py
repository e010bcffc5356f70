"""Cached incremental decode (progen_amd/decode.py) vs the full-forward
reference path: per-step logits parity and token-identical sampling."""

import torch

from progen_amd import decode, utils
from progen_amd.config import ProGenConfig
from progen_amd.models.progen import ProGenBase


def _tiny(global_mlp_depth=1, shift_tokens=True, ff_glu=True):
    torch.manual_seed(7)
    cfg = ProGenConfig(num_tokens=256, dim=16, depth=3, dim_head=8, heads=2,
                       window_size=8, seq_len=48, ff_glu=ff_glu,
                       global_mlp_depth=global_mlp_depth,
                       shift_tokens=shift_tokens)
    return ProGenBase(cfg).eval()


def _stepwise_logits(model, seq):
    cache = decode.DecodeCache(model, batch=1)
    rows = [decode.forward_step(model, seq[p:p + 1], cache)
            for p in range(seq.shape[0])]
    return torch.cat(rows, dim=0)


def test_forward_step_matches_full_forward():
    model = _tiny()
    seq = torch.randint(1, 256, (48,))
    with torch.no_grad():
        full = model(seq.unsqueeze(0))[0]
    inc = _stepwise_logits(model, seq)
    torch.testing.assert_close(inc, full, rtol=1e-4, atol=1e-4)


def test_forward_step_prefix_parity_mid_window():
    # row p must only depend on tokens <= p (incl. the window-0 zero-key
    # quirk and the lookback window boundary at p = window_size)
    model = _tiny(global_mlp_depth=2, ff_glu=False)
    seq = torch.randint(1, 256, (21,))  # not a window multiple
    padded = torch.nn.functional.pad(seq, (0, 24 - 21))  # causal zero tail
    with torch.no_grad():
        full = model(padded.unsqueeze(0))[0, :21]
    inc = _stepwise_logits(model, seq)
    torch.testing.assert_close(inc, full, rtol=1e-4, atol=1e-4)


def test_forward_step_no_shift():
    model = _tiny(shift_tokens=False)
    seq = torch.randint(1, 256, (16,))
    with torch.no_grad():
        full = model(seq.unsqueeze(0))[0]
    inc = _stepwise_logits(model, seq)
    torch.testing.assert_close(inc, full, rtol=1e-4, atol=1e-4)


def test_sample_cached_matches_sample():
    model = _tiny()
    prime = torch.randint(1, 256, (5,))

    def fn(seq):
        with torch.no_grad():
            return model(seq.unsqueeze(0))[0]

    g1 = torch.Generator().manual_seed(11)
    ref = utils.sample(fn, prime.clone(), 48, top_k=20, generator=g1)
    g2 = torch.Generator().manual_seed(11)
    got = decode.sample_cached(model, prime.clone(), 48, top_k=20, generator=g2)
    assert torch.equal(ref, got)


def test_sample_cached_eos_and_add_bos():
    model = _tiny()
    prime = torch.randint(1, 256, (3,))
    g1 = torch.Generator().manual_seed(3)
    ref = utils.sample(model_fn(model), prime.clone(), 32, top_k=25,
                       add_bos=True, generator=g1)
    g2 = torch.Generator().manual_seed(3)
    got = decode.sample_cached(model, prime.clone(), 32, top_k=25,
                               add_bos=True, generator=g2)
    assert torch.equal(ref, got)
    # nothing after the 2nd pad
    z = (got == 0).long().cumsum(-1) > 1
    assert (got[z] == 0).all()


def model_fn(model):
    def fn(seq):
        with torch.no_grad():
            return model(seq.unsqueeze(0))[0]
    return fn


def test_sample_cached_batch_greedy_parity():
    from progen_amd.decode import sample_cached_batch, DecodeCache, forward_step
    model = _tiny()
    primes = [torch.randint(1, 256, (4,)), torch.randint(1, 256, (7,))]
    out = sample_cached_batch(model, primes, 24)

    # each row must equal its own single-row greedy decode
    for i, p in enumerate(primes):
        cache = DecodeCache(model, batch=1)
        seq = torch.zeros(24, dtype=torch.long)
        seq[:p.shape[0]] = p
        for pos in range(23):
            logits = forward_step(model, seq[pos:pos + 1], cache)
            nxt = logits[0].float().argmax()
            if pos + 1 >= p.shape[0]:
                seq[pos + 1] = nxt
        z = (seq == 0).long().cumsum(-1) > 1
        seq = seq * (~z).long()
        assert torch.equal(out[i], seq)


def test_forward_step_static_matches_dynamic():
    """The position-static step (fixed shapes, tensor indexing — the
    hipGraph-capturable form) must be numerically identical to
    forward_step, including the window-0 zero-halo quirk and window
    boundaries."""
    from progen_amd.decode import forward_step_static
    model = _tiny(global_mlp_depth=2)
    seq = torch.randint(1, 256, (48,))

    cache_d = decode.DecodeCache(model, batch=1)
    cache_s = decode.DecodeCache(model, batch=1)
    for p in range(48):
        ref = decode.forward_step(model, seq[p:p + 1], cache_d)
        pos = torch.tensor(p, dtype=torch.long)
        got = forward_step_static(model, seq[p:p + 1], cache_s, pos)
        # fixed-band vs variable-band einsum: same math, different fp32
        # reduction order -> ulp-level noise only
        torch.testing.assert_close(got, ref, rtol=1e-5, atol=1e-5)
