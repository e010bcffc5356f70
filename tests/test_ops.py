"""Unit tests for the op layer vs independent numpy/f64 oracles.

Each oracle here is written directly from the reference math
(progen_transformer/progen.py, utils.py), NOT by calling the library code
under test, so a bug in ops/reference.py cannot self-certify.
"""

import math

import numpy as np
import pytest
import torch

from progen_amd.ops import reference as R


# ---------------------------------------------------------------------------
# numpy oracles (float64)
# ---------------------------------------------------------------------------

def np_fixed_pos_embedding(seq, dim):
    inv_freq = 1.0 / (10000 ** (np.arange(0, dim, 2) / dim))
    sinusoid = np.einsum("i,j->ij", np.arange(seq), inv_freq)
    sinusoid = np.repeat(sinusoid, 2, axis=-1)  # 'b n -> b (n r)', r=2
    return np.sin(sinusoid), np.cos(sinusoid)


def np_rotate_every_two(x):
    x1 = x[..., 0::2]
    x2 = x[..., 1::2]
    out = np.stack((-x2, x1), axis=-1)
    return out.reshape(*x.shape[:-1], -1)


def np_apply_rotary(x, sin, cos):
    return x * cos + np_rotate_every_two(x) * sin


def np_shift_tokens(x):
    # x: (n, d); np.array_split puts the extra channel in the first half
    d = x.shape[-1]
    split = -(-d // 2)
    x_shift, x_pass = x[:, :split], x[:, split:]
    x_shift = np.pad(x_shift, ((1, 0), (0, 0)))[:-1]
    return np.concatenate((x_shift, x_pass), axis=-1)


def np_layernorm_nobias(x, g, eps=1e-5):
    mu = x.mean(-1, keepdims=True)
    var = x.var(-1, keepdims=True)
    return (x - mu) / np.sqrt(var + eps) * g


def np_local_attention(qkv, heads, wsz):
    """Literal transcription of reference progen.py:83-103 in f64, per
    sequence (unbatched), with explicit windowing."""
    n = qkv.shape[0]
    inner = qkv.shape[1] // 3
    dh = inner // heads
    w = n // wsz
    scale = dh ** -0.5
    sin, cos = np_fixed_pos_embedding(n, dh)

    q, k, v = np.split(qkv, 3, axis=-1)
    # 'n (h d) -> h n d'
    def to_heads(t):
        return t.reshape(n, heads, dh).transpose(1, 0, 2)
    q, k, v = map(to_heads, (q, k, v))
    q, k, v = (np_apply_rotary(t, sin[None], cos[None]) for t in (q, k, v))
    # 'h (w n) d -> h w n d'
    q = q.reshape(heads, w, wsz, dh)
    k = k.reshape(heads, w, wsz, dh)
    v = v.reshape(heads, w, wsz, dh)
    # lookback pad + concat
    k = np.pad(k, ((0, 0), (1, 0), (0, 0), (0, 0)))
    v = np.pad(v, ((0, 0), (1, 0), (0, 0), (0, 0)))
    k = np.concatenate((k[:, :-1], k[:, 1:]), axis=2)
    v = np.concatenate((v[:, :-1], v[:, 1:]), axis=2)

    sim = np.einsum("hwid,hwjd->hwij", q, k) * scale
    mask = np.tril(np.ones((wsz, 2 * wsz)), wsz).astype(bool)
    sim = np.where(mask, sim, -1e10)
    sim = sim - sim.max(-1, keepdims=True)
    attn = np.exp(sim)
    attn = attn / attn.sum(-1, keepdims=True)
    out = np.einsum("hwij,hwjd->hwid", attn, v)
    # 'h w n d -> (w n) (h d)'
    out = out.transpose(1, 2, 0, 3).reshape(n, heads * dh)
    return out


def np_cross_entropy(logits, targets):
    """reference utils.py:45-59 (per-sequence), f64."""
    x = logits - logits.max(-1, keepdims=True)
    logprobs = x - np.log(np.exp(x).sum(-1, keepdims=True))
    nll = np.take_along_axis(logprobs, targets[:, None], axis=-1).squeeze(-1)
    mask = targets != 0
    eos_mask = (~mask).cumsum(-1) == 1
    mask = mask | eos_mask
    return -(nll * mask).sum() / mask.sum()


# ---------------------------------------------------------------------------
# tests
# ---------------------------------------------------------------------------

def test_rotary_table_matches_reference_construction():
    sin, cos = R.fixed_pos_embedding(64, 16)
    nsin, ncos = np_fixed_pos_embedding(64, 16)
    np.testing.assert_allclose(sin.numpy(), nsin, atol=1e-6)
    np.testing.assert_allclose(cos.numpy(), ncos, atol=1e-6)


def test_rotate_every_two():
    x = torch.randn(3, 8)
    got = R.rotate_every_two(x).numpy()
    np.testing.assert_allclose(got, np_rotate_every_two(x.numpy()), atol=1e-6)


@pytest.mark.parametrize("d", [8, 7])
def test_shift_tokens(d):
    x = torch.randn(1, 5, d)
    got = R.shift_tokens(x)[0].numpy()
    np.testing.assert_allclose(got, np_shift_tokens(x[0].numpy()), atol=1e-6)
    # first ceil(d/2) channels shifted: row 0 of shifted half is zero
    assert np.allclose(got[0, : -(-d // 2)], 0.0)


def test_layernorm_scale_only():
    x = torch.randn(2, 5, 16, dtype=torch.float64)
    g = torch.randn(16, dtype=torch.float64)
    got = R.layernorm_nobias(x, g).numpy()
    want = np_layernorm_nobias(x.numpy(), g.numpy())
    np.testing.assert_allclose(got, want, atol=1e-5)


def test_local_attention_matches_oracle():
    torch.manual_seed(1)
    B, N, h, dh, wsz = 2, 64, 2, 8, 16
    qkv = torch.randn(B, N, 3 * h * dh, dtype=torch.float64)
    sin, cos = R.fixed_pos_embedding(N, dh, dtype=torch.float64)
    got = R.local_attention(qkv, sin, cos, h, wsz).numpy()
    for b in range(B):
        want = np_local_attention(qkv[b].numpy(), h, wsz)
        np.testing.assert_allclose(got[b], want, atol=1e-8)


def test_window0_zero_lookback_keys_enter_softmax():
    """Window 0's lookback keys are zero vectors with logit 0 — UNMASKED —
    so they dilute the softmax (reference quirk, progen.py:90-96)."""
    torch.manual_seed(2)
    h, dh, wsz = 1, 4, 4
    N = wsz  # single window
    qkv = torch.randn(1, N, 3 * h * dh, dtype=torch.float64)
    sin, cos = R.fixed_pos_embedding(N, dh, dtype=torch.float64)
    out = R.local_attention(qkv, sin, cos, h, wsz)[0]

    # manual: row 0 attends to [wsz zero keys ‖ key 0]; zero keys have
    # logit 0 and v=0, real key has its own logit
    q, k, v = qkv[0].chunk(3, dim=-1)
    q = R.apply_rotary_pos_emb(q.view(N, dh), sin, cos)
    k = R.apply_rotary_pos_emb(k.view(N, dh), sin, cos)
    v = R.apply_rotary_pos_emb(v.view(N, dh), sin, cos)
    scale = dh ** -0.5
    logit00 = (q[0] @ k[0]) * scale
    denom = wsz * 1.0 + math.exp(float(logit00))  # wsz zero-logit terms: e^0 each
    want_row0 = (math.exp(float(logit00)) / denom) * v[0]
    np.testing.assert_allclose(out[0].numpy(), want_row0.numpy(), atol=1e-8)


def test_local_attention_locality():
    """Output at window w depends only on tokens in windows w-1 and w."""
    torch.manual_seed(3)
    h, dh, wsz = 2, 8, 8
    N = 4 * wsz
    qkv = torch.randn(1, N, 3 * h * dh, dtype=torch.float64)
    sin, cos = R.fixed_pos_embedding(N, dh, dtype=torch.float64)
    base = R.local_attention(qkv, sin, cos, h, wsz)

    qkv2 = qkv.clone()
    qkv2[0, 0] += 100.0  # perturb window 0
    out2 = R.local_attention(qkv2, sin, cos, h, wsz)
    # windows 2,3 (positions >= 2*wsz) unaffected
    np.testing.assert_allclose(base[0, 2 * wsz:].numpy(),
                               out2[0, 2 * wsz:].numpy(), atol=1e-10)
    # window 0 and 1 affected
    assert not np.allclose(base[0, :2 * wsz].numpy(), out2[0, :2 * wsz].numpy())


def test_local_attention_causal():
    torch.manual_seed(4)
    h, dh, wsz = 1, 8, 8
    N = 2 * wsz
    qkv = torch.randn(1, N, 3 * h * dh, dtype=torch.float64)
    sin, cos = R.fixed_pos_embedding(N, dh, dtype=torch.float64)
    base = R.local_attention(qkv, sin, cos, h, wsz)
    p = 10
    qkv2 = qkv.clone()
    qkv2[0, p] += 100.0
    out2 = R.local_attention(qkv2, sin, cos, h, wsz)
    np.testing.assert_allclose(base[0, :p].numpy(), out2[0, :p].numpy(), atol=1e-10)
    assert not np.allclose(base[0, p:].numpy(), out2[0, p:].numpy())


def test_cross_entropy_eos_mask():
    torch.manual_seed(5)
    B, N, V = 1, 10, 12
    logits = torch.randn(B, N, V, dtype=torch.float64)
    targets = torch.tensor([[3, 4, 5, 0, 0, 0, 0, 0, 0, 0]])
    got = R.cross_entropy(logits, targets).item()
    want = np_cross_entropy(logits[0].numpy(), targets[0].numpy())
    assert got == pytest.approx(want, abs=1e-10)
    # changing logits at a position after the first pad must not change loss
    logits2 = logits.clone()
    logits2[0, 5:, 3] += 10.0
    got2 = R.cross_entropy(logits2, targets).item()
    assert got2 == pytest.approx(got, abs=1e-10)
    # but the first pad position IS learned (EOS)
    logits3 = logits.clone()
    logits3[0, 3, 0] += 1.0
    assert R.cross_entropy(logits3, targets).item() != pytest.approx(got, abs=1e-6)


def test_cross_entropy_batch_reduction_order():
    """Per-sequence masked mean then plain batch mean (utils.py:67,75-76),
    not a global masked mean."""
    torch.manual_seed(6)
    V = 8
    logits = torch.randn(2, 6, V, dtype=torch.float64)
    targets = torch.tensor([[1, 2, 0, 0, 0, 0],      # 3 masked positions
                            [1, 2, 3, 4, 5, 0]])     # 6 masked positions
    got = R.cross_entropy(logits, targets).item()
    want = 0.5 * (np_cross_entropy(logits[0].numpy(), targets[0].numpy())
                  + np_cross_entropy(logits[1].numpy(), targets[1].numpy()))
    assert got == pytest.approx(want, abs=1e-10)


def test_select_top_k_strict_gt_quirk():
    """Ties with the k-th value are EXCLUDED (strict >, utils.py:99) and
    excluded logits become 0, not -inf (utils.py:100)."""
    t = torch.tensor([1.0, 2.0, 2.0, 3.0])
    mask, out = R.select_top_k(t, 2)
    assert mask.tolist() == [False, False, False, True]  # 2.0 ties excluded
    assert out.tolist() == [0.0, 0.0, 0.0, 3.0]


def test_glu_gelu():
    x = torch.randn(2, 3, 8, dtype=torch.float64)
    a, g = x.chunk(2, dim=-1)
    want = a * 0.5 * g * (1.0 + torch.tanh(math.sqrt(2.0 / math.pi) * (g + 0.044715 * g ** 3)))
    np.testing.assert_allclose(R.glu_gelu(x).numpy(), want.numpy(), atol=1e-7)


def test_sgu_causality_and_bias_init():
    torch.manual_seed(7)
    B, N, H = 1, 6, 8
    x = torch.randn(B, N, H, dtype=torch.float64)
    g = torch.ones(H // 2, dtype=torch.float64)
    W = torch.randn(N, N, dtype=torch.float64)
    b = torch.ones(N, 1, dtype=torch.float64)
    out = R.sgu_gate(x, g, W, b)
    assert out.shape == (B, N, H // 2)
    # causality: perturbing row p affects only rows >= p of the output
    x2 = x.clone()
    p = 3
    x2[0, p] += 10.0
    out2 = R.sgu_gate(x2, g, W, b)
    np.testing.assert_allclose(out[0, :p].numpy(), out2[0, :p].numpy(), atol=1e-10)
    assert not np.allclose(out[0, p:].numpy(), out2[0, p:].numpy())
    # manual row: out[m] = x_a[m] * (sum_{n<=m} W[m,n]*LN(gate)[n] + 1)
    xa, gate = x[0].chunk(2, dim=-1)
    ln = torch.from_numpy(np_layernorm_nobias(gate.numpy(), g.numpy()))
    m = 2
    want = xa[m] * ((W[m, : m + 1].unsqueeze(0) @ ln[: m + 1]).squeeze(0) + 1.0)
    np.testing.assert_allclose(out[0, m].numpy(), want.numpy(), atol=1e-8)


def test_eager_ops_env_gates_dispatch(monkeypatch):
    """PROGEN_EAGER_OPS routes only the named ops to the torch reference
    (the per-kernel bisect lever for the graphed-replay investigation)."""
    from progen_amd.ops import dispatch

    class FakeCuda:
        is_cuda = True

    t = FakeCuda()
    monkeypatch.delenv("PROGEN_FORCE_EAGER", raising=False)
    monkeypatch.setenv("PROGEN_EAGER_OPS", "attn, sgu")
    assert dispatch.use_hip(t, "attn") is False
    assert dispatch.use_hip(t, "sgu") is False
    assert dispatch.use_hip(t, "ln") is True
    assert dispatch.use_hip(t) is True
    monkeypatch.delenv("PROGEN_EAGER_OPS")
    assert dispatch.use_hip(t, "attn") is True
    monkeypatch.setenv("PROGEN_FORCE_EAGER", "1")
    assert dispatch.use_hip(t, "ln") is False
