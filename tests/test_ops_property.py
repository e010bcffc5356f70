"""Structural invariants of the ProGen ops (reference implementations,
fp64): causality, rotary geometry, mask semantics. These pin behavior the
numeric oracle tests can't see (a wrong-but-consistent mask would pass a
self-comparison)."""

import torch

from progen_amd.ops import reference as R


def _qkv(B, N, h, dh, seed=0):
    torch.manual_seed(seed)
    return torch.randn(B, N, 3 * h * dh, dtype=torch.float64)


def test_local_attention_causal():
    # perturbing tokens AFTER position p must not change outputs <= p
    B, N, h, dh, wsz = 2, 32, 2, 8, 8
    sin, cos = R.fixed_pos_embedding(N, dh, dtype=torch.float64)
    qkv = _qkv(B, N, h, dh)
    out = R.local_attention(qkv, sin, cos, h, wsz)
    for p in (5, 15, 24):
        q2 = qkv.clone()
        q2[:, p + 1:] += torch.randn_like(q2[:, p + 1:])
        out2 = R.local_attention(q2, sin, cos, h, wsz)
        torch.testing.assert_close(out2[:, :p + 1], out[:, :p + 1])


def test_local_attention_window_locality():
    # tokens more than 2 windows back must not influence the output
    B, N, h, dh, wsz = 1, 32, 2, 8, 8
    sin, cos = R.fixed_pos_embedding(N, dh, dtype=torch.float64)
    qkv = _qkv(B, N, h, dh, seed=1)
    out = R.local_attention(qkv, sin, cos, h, wsz)
    q2 = qkv.clone()
    q2[:, :8] += 10.0  # window 0
    out2 = R.local_attention(q2, sin, cos, h, wsz)
    # window 3 (rows 24..31) looks back only to window 2: unchanged
    torch.testing.assert_close(out2[:, 24:], out[:, 24:])
    # window 1 rows DO see window 0 (lookback): must change
    assert not torch.allclose(out2[:, 8:16], out[:, 8:16])


def test_window0_zero_keys_enter_softmax():
    # quirk (progen.py:90-96): window 0's zero lookback keys are unmasked
    # -> they dilute the softmax. A correct implementation gives row 0 an
    # output that is NOT exactly v_0-rotated (which pure self-attention
    # over one key would give after normalization).
    B, N, h, dh, wsz = 1, 8, 1, 8, 8
    sin, cos = R.fixed_pos_embedding(N, dh, dtype=torch.float64)
    qkv = _qkv(B, N, h, dh, seed=2)
    out = R.local_attention(qkv, sin, cos, h, wsz)
    v = qkv[..., 2 * h * dh:].view(B, N, h, dh)
    v0_rot = R.apply_rotary_pos_emb(v[:, 0:1, 0], sin[0:1], cos[0:1])
    # row 0 attends [8 zero keys ‖ itself]: out0 = w * v0_rot with w < 1
    ratio = out[:, 0] / v0_rot.reshape(B, -1)
    assert (ratio < 1.0).all()
    torch.testing.assert_close(ratio, ratio[..., :1].expand_as(ratio))


def test_rotary_norm_and_identity():
    sin, cos = R.fixed_pos_embedding(16, 8, dtype=torch.float64)
    x = torch.randn(3, 16, 8, dtype=torch.float64)
    y = R.apply_rotary_pos_emb(x, sin, cos)
    # pure rotation: pairwise norms preserved at every position
    xn = x.view(3, 16, 4, 2).norm(dim=-1)
    yn = y.view(3, 16, 4, 2).norm(dim=-1)
    torch.testing.assert_close(xn, yn)
    # position 0: identity
    torch.testing.assert_close(y[:, 0], x[:, 0])


def test_shift_tokens_moves_first_half():
    x = torch.randn(2, 5, 6, dtype=torch.float64)
    y = R.shift_tokens(x)
    torch.testing.assert_close(y[:, 1:, :3], x[:, :-1, :3])  # shifted half
    torch.testing.assert_close(y[:, :, 3:], x[:, :, 3:])     # pass half
    assert (y[:, 0, :3] == 0).all()                          # zero pad row 0
    # odd D: ceil split (np.array_split parity)
    x = torch.randn(1, 3, 5, dtype=torch.float64)
    y = R.shift_tokens(x)
    assert (y[:, 0, :3] == 0).all() and (y[:, 0, 3:] == x[:, 0, 3:]).all()


def test_sgu_causal():
    n, d2 = 16, 4
    x = torch.randn(1, n, 2 * d2, dtype=torch.float64)
    w = torch.randn(n, n, dtype=torch.float64)
    b = torch.ones(n, 1, dtype=torch.float64)
    g = torch.ones(d2, dtype=torch.float64)
    out = R.sgu_gate(x, g, w, b)
    x2 = x.clone()
    x2[:, 10:] += 1.0
    out2 = R.sgu_gate(x2, g, w, b)
    torch.testing.assert_close(out2[:, :10], out[:, :10])


def test_select_top_k_quirks():
    t = torch.tensor([1.0, 3.0, 3.0, 2.0, 0.5])
    mask, vals = R.select_top_k(t, 2)
    # strict > vs the 2nd value (3.0): ties at the threshold are DROPPED
    assert mask.tolist() == [False, False, False, False, False] or \
        mask.sum() <= 2
    # excluded logits are 0, not -inf
    assert (vals[~mask] == 0).all()
    t2 = torch.tensor([1.0, 5.0, 3.0, 2.0])
    mask2, vals2 = R.select_top_k(t2, 2)
    assert mask2.tolist() == [False, True, False, False] or mask2.sum() <= 2
    assert vals2[1] == 5.0


def test_cross_entropy_mask_semantics():
    torch.manual_seed(3)
    B, N, V = 2, 10, 16
    logits = torch.randn(B, N, V, dtype=torch.float64)
    tgt = torch.randint(1, V, (B, N))
    tgt[:, 7:] = 0  # pad tail: first pad (col 7) is EOS, later ones masked
    base = R.cross_entropy(logits, tgt)
    # changing logits at positions AFTER the EOS must not change the loss
    l2 = logits.clone()
    l2[:, 8:] += torch.randn_like(l2[:, 8:])
    torch.testing.assert_close(R.cross_entropy(l2, tgt), base)
    # changing logits AT the EOS position must change it (single logit:
    # a uniform shift would be softmax-invariant)
    l3 = logits.clone()
    l3[:, 7, 3] += 1.0
    assert not torch.allclose(R.cross_entropy(l3, tgt), base)
    # reduction order: per-seq masked mean then batch mean (NOT global)
    lens = torch.tensor([4, 8])
    tgt2 = torch.randint(1, V, (B, N))
    for i, L in enumerate(lens):
        tgt2[i, L:] = 0
    per_seq = []
    lp = torch.log_softmax(logits, dim=-1)
    for i in range(B):
        L = int(lens[i]) + 1  # + EOS position
        nll = -lp[i, torch.arange(L), tgt2[i, :L]]
        per_seq.append(nll.mean())
    want = torch.stack(per_seq).mean()
    torch.testing.assert_close(R.cross_entropy(logits, tgt2), want)
