"""GPU: hipGraph-captured decode step (decode.GraphedDecodeStep) must
reproduce the uncaptured static step's tokens exactly — same kernels,
pure replay, host-side sampling."""

import pytest
import torch

pytestmark = [pytest.mark.gpu,
              pytest.mark.skipif(not torch.cuda.is_available(),
                                 reason="needs MI355X")]


def _model():
    from progen_amd import ProGenBase, ProGenConfig
    cfg = ProGenConfig(num_tokens=256, dim=256, depth=3, heads=4,
                       dim_head=64, window_size=64, seq_len=256,
                       ff_glu=False, global_mlp_depth=1)
    torch.manual_seed(3)
    m = ProGenBase(cfg).to(device="cuda", dtype=torch.bfloat16).eval()
    m.rotary_sin = m.rotary_sin.float()
    m.rotary_cos = m.rotary_cos.float()
    return m


def _greedy_static_eager(m, prime, n):
    from progen_amd.decode import DecodeCache, forward_step, forward_step_static
    cache = DecodeCache(m, batch=1)
    logits = None
    for p in range(len(prime)):
        logits = forward_step(m, prime[p:p + 1].cuda(), cache)
    pos_dev = torch.tensor(len(prime), dtype=torch.long, device="cuda")
    toks = []
    tok = logits.cpu().float().argmax(dim=-1)
    for _ in range(n):
        toks.append(int(tok))
        logits = forward_step_static(m, tok.cuda(), cache, pos_dev)
        pos_dev += 1
        tok = logits.cpu().float().argmax(dim=-1)
    return toks


def _greedy_graphed(m, prime, n):
    from progen_amd.decode import DecodeCache, GraphedDecodeStep, forward_step
    cache = DecodeCache(m, batch=1)
    logits = None
    for p in range(len(prime)):
        logits = forward_step(m, prime[p:p + 1].cuda(), cache)
    first = logits.cpu().float().argmax(dim=-1)
    g = GraphedDecodeStep(m, cache, start_pos=len(prime))
    toks = []
    tok = first
    for _ in range(n):
        toks.append(int(tok))
        out = g.step(tok)
        tok = out.cpu().float().argmax(dim=-1)
    return toks


def test_graphed_decode_matches_eager_static():
    m = _model()
    torch.manual_seed(11)
    prime = torch.randint(1, 256, (17,))
    want = _greedy_static_eager(m, prime, 40)
    got = _greedy_graphed(m, prime, 40)
    assert got == want, (got, want)


def test_graphed_sample_cached_batch_runs():
    from progen_amd.decode import sample_cached_batch
    m = _model()
    torch.manual_seed(5)
    primes = [torch.randint(1, 256, (9,)), torch.randint(1, 256, (5,))]
    out = sample_cached_batch(m, primes, length=64, graph=True)
    assert out.shape == (2, 64)
    assert (out[0, :9].cpu() == primes[0]).all()
