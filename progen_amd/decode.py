"""Cached incremental decoding (serving path).

The reference's sampler (reference: progen_transformer/utils.py:106-135)
re-runs a full-length forward per emitted token — O(L * N^2). This module
maintains per-layer recurrent state so each new token costs one O(window)
attention row, one O(n) SGU row and the per-token GEMV projections:

  - token shift needs the previous position's LN'd row (progen.py:43-46);
  - windowed attention needs the rotated k/v of the current + previous
    window only (progen.py:88-96) — window 0's zero lookback keys enter
    the softmax denominator unmasked, which the step reproduces exactly;
  - the SGU gate row m is sum_{n<=m} W[m,n]*gate_ln[n] + b[m]
    (progen.py:179-182), so the LN'd gate history is the cache.

Emitted tokens are identical to ``utils.sample`` up to fp reduction-order
noise (tests/test_decode.py pins exact token equality on seeded fp32
models). This is the O(1)-per-token analog of a KV cache for this
architecture; ``sample_cached`` mirrors the sampler quirks (strict-> top-k
with zeros-not-inf, gumbel-max, zero-after-second-pad).
"""

from __future__ import annotations

from typing import Optional

import torch
import torch.nn.functional as F

from .models.progen import ProGenBase
from .ops import reference as R


def _ln_row(x: torch.Tensor, weight: torch.Tensor, eps: float = 1e-5) -> torch.Tensor:
    """Scale-only LN of a (B, D) row (progen.py:22); stats in fp32."""
    return R.layernorm_nobias(x.unsqueeze(1), weight, eps).squeeze(1)


def _shift_row(y: torch.Tensor, prev: Optional[torch.Tensor]) -> torch.Tensor:
    """Token shift at one position: first ceil(D/2) channels come from the
    PREVIOUS position's LN'd row (zeros at position 0)  (progen.py:43-46)."""
    d = y.shape[-1]
    split = -(-d // 2)
    head = torch.zeros_like(y[..., :split]) if prev is None else prev[..., :split]
    return torch.cat((head, y[..., split:]), dim=-1)


class DecodeCache:
    """Recurrent state for one ``ProGenBase`` forward, batch-first (B, ...).

    Buffers are allocated once at ``seq_len`` capacity (288 GB HBM3E makes
    full-length caches the right trade on MI355X — no ring-buffer
    bookkeeping on the hot path)."""

    def __init__(self, model: ProGenBase, batch: int = 1,
                 device=None, dtype=None):
        cfg = model.cfg
        p = next(model.parameters())
        device = p.device if device is None else device
        dtype = p.dtype if dtype is None else dtype
        self.cfg = cfg
        self.pos = 0
        N, H, DH = cfg.seq_len, cfg.heads, cfg.dim_head
        self.k = []    # per attn layer: (B, H, N, DH) rotated keys
        self.v = []
        # previous position's LN row per branch (token-shift halo);
        # preallocated + written in place so a captured static step
        # updates fixed buffers (zeros == the position-0 "no previous
        # row" semantics)
        self.attn_prev = [torch.zeros(batch, cfg.dim, device=device,
                                      dtype=dtype) for _ in range(cfg.depth)]
        self.ff_prev = [torch.zeros(batch, cfg.dim, device=device,
                                    dtype=dtype) for _ in range(cfg.depth)]
        self.gate_hist = []  # per SGU layer: (B, N, d2) LN'd gate rows
        for _attn, ff in model.layers:
            self.k.append(torch.zeros(batch, H, N, DH, device=device, dtype=dtype))
            self.v.append(torch.zeros(batch, H, N, DH, device=device, dtype=dtype))
            if ff.sgu is not None:
                d2 = ff.sgu.norm_weight.shape[0]
                self.gate_hist.append(
                    torch.zeros(batch, N, d2, device=device, dtype=dtype))
            else:
                self.gate_hist.append(None)


def _attn_step(attn, x, sin_p, cos_p, cache: DecodeCache, li: int) -> torch.Tensor:
    """One LocalAttention row at position cache.pos. x: (B, dim)."""
    p = cache.pos
    H, DH, wsz = attn.heads, cache.cfg.dim_head, attn.window_size
    y = _ln_row(x, attn.norm_weight)
    y_in = _shift_row(y, cache.attn_prev[li]) if attn.shift_tokens else y
    cache.attn_prev[li].copy_(y)
    qkv = F.linear(y_in, attn.to_qkv.weight)            # (B, 3*H*DH)
    B = qkv.shape[0]
    q, k, v = qkv.view(B, 3, H, DH).unbind(1)
    # interleaved rotary on q, k AND v at position p (progen.py:87)
    sc, ss = cos_p.to(q.dtype), sin_p.to(q.dtype)
    q, k, v = (t * sc + R.rotate_every_two(t) * ss for t in (q, k, v))
    cache.k[li][:, :, p] = k
    cache.v[li][:, :, p] = v

    win = p // wsz
    start = (win - 1) * wsz if win > 0 else 0
    keys = cache.k[li][:, :, start:p + 1]               # (B, H, n, DH)
    vals = cache.v[li][:, :, start:p + 1]
    s = torch.einsum("bhd,bhnd->bhn", q, keys) * (DH ** -0.5)
    if win == 0:
        # window 0's zero lookback keys are UNMASKED (progen.py:90-96):
        # wsz extra zero logits in the softmax, zero values
        s = F.pad(s, (wsz, 0))
    s = s - s.amax(dim=-1, keepdim=True)
    a = s.softmax(dim=-1)
    if win == 0:
        a = a[..., wsz:]
    out = torch.einsum("bhn,bhnd->bhd", a, vals).reshape(B, H * DH)
    return F.linear(out, attn.to_out.weight, attn.to_out.bias)


def _ff_step(ff, x, cache: DecodeCache, li: int) -> torch.Tensor:
    """One FeedForward row at position cache.pos. x: (B, dim)."""
    y = _ln_row(x, ff.norm_weight)
    y_in = _shift_row(y, cache.ff_prev[li]) if ff.shift_tokens else y
    cache.ff_prev[li].copy_(y)
    h = F.linear(y_in, ff.proj_in.weight, ff.proj_in.bias)
    if ff.glu:
        a, g = h.chunk(2, dim=-1)
        h = a * F.gelu(g, approximate="tanh")
    else:
        h = F.gelu(h, approximate="tanh")
    if ff.sgu is not None:
        p = cache.pos
        xa, gate = h.chunk(2, dim=-1)
        gate_ln = _ln_row(gate, ff.sgu.norm_weight)
        hist = cache.gate_hist[li]
        hist[:, p] = gate_ln
        w_row = ff.sgu.spatial_weights[p, :p + 1]       # causal row (progen.py:179)
        gate_out = torch.einsum("n,bnd->bd", w_row.to(hist.dtype),
                                hist[:, :p + 1]) + ff.sgu.spatial_biases[p]
        h = xa * gate_out
        h = F.linear(h, ff.sgu.proj_out.weight, ff.sgu.proj_out.bias)
    return F.linear(h, ff.proj_out.weight, ff.proj_out.bias)


@torch.no_grad()
def forward_step(model: ProGenBase, token: torch.Tensor,
                 cache: DecodeCache) -> torch.Tensor:
    """Advance the cache by one token; returns (B, V) logits at this
    position — identical (up to fp noise) to row ``cache.pos`` of a full
    ``model(seq)`` forward over the same prefix."""
    p = cache.pos
    assert p < model.cfg.seq_len, "decode past seq_len"
    if model.rotary_sin.dtype != torch.float32:
        sin, cos = R.fixed_pos_embedding(model.cfg.seq_len, model.cfg.dim_head,
                                         device=model.rotary_sin.device)
        model.rotary_sin, model.rotary_cos = sin, cos
    sin_p, cos_p = model.rotary_sin[p], model.rotary_cos[p]
    h = model.embed(token.long().reshape(-1))           # (B, dim)
    for li, (attn, ff) in enumerate(model.layers):
        h = h + _attn_step(attn, h, sin_p, cos_p, cache, li)
        h = h + _ff_step(ff, h, cache, li)
    h = _ln_row(h, model.final_norm_weight)
    logits = F.linear(h, model.to_logits.weight, model.to_logits.bias)
    cache.pos = p + 1
    return logits


@torch.no_grad()
def forward_step_static(model: ProGenBase, token: torch.Tensor,
                        cache: DecodeCache,
                        pos_dev: torch.Tensor) -> torch.Tensor:
    """Position-STATIC decode step: numerically identical to
    ``forward_step`` but every operation has position-independent shapes
    and reads the position from a device tensor — the form a hipGraph
    can capture once and replay per token (TODO.md "Serving"; the
    pure-replay rule of profiles/r01_graph_interleave_bug.md means the
    position increment must itself be in-graph, which this function's
    tensor-only indexing permits).

    Mechanics per layer:
      - attention gathers a FIXED 2*window_size key slice at indices
        win_start - wsz + [0, 2wsz); indices < 0 fetch zeroed k/v and
        stay UNMASKED (exactly the reference's window-0 zero-pad quirk,
        progen.py:90-96); indices > pos are masked -inf;
      - the SGU row uses the full spatial_weights row masked by
        arange(N) <= pos (static O(N*d2) dot);
      - rotary rows, shift halos and cache writes are index_put/gather
        with tensor indices.

    ``pos_dev``: 0-dim int64 tensor holding the current position; the
    caller increments it AFTER the step (in-graph when captured).
    Updates the same ``DecodeCache`` as ``forward_step``."""
    cfg = model.cfg
    wsz = cfg.window_size
    dev = token.device
    p = pos_dev  # 0-dim int64 on device
    sin_p = model.rotary_sin.index_select(0, p.reshape(1))[0]
    cos_p = model.rotary_cos.index_select(0, p.reshape(1))[0]
    h = model.embed(token.long().reshape(-1))

    win_start = (p // wsz) * wsz
    gather_idx = win_start - wsz + torch.arange(2 * wsz, device=dev)
    valid = (gather_idx >= 0)                       # zero-halo quirk keys
    causal = (gather_idx <= p)                      # future keys masked
    safe_idx = gather_idx.clamp(min=0)

    for li, (attn, ff) in enumerate(model.layers):
        # ---- attention branch ----
        x = h
        y = _ln_row(x, attn.norm_weight)
        y_in = _shift_row(y, cache.attn_prev[li]) if attn.shift_tokens else y
        cache.attn_prev[li].copy_(y)
        qkv = F.linear(y_in, attn.to_qkv.weight)
        B = qkv.shape[0]
        H, DH = attn.heads, cfg.dim_head
        q, k, v = qkv.view(B, 3, H, DH).unbind(1)
        sc, ss = cos_p.to(q.dtype), sin_p.to(q.dtype)
        q, k, v = (t * sc + R.rotate_every_two(t) * ss for t in (q, k, v))
        cache.k[li].index_copy_(2, p.reshape(1), k.unsqueeze(2))
        cache.v[li].index_copy_(2, p.reshape(1), v.unsqueeze(2))
        keys = cache.k[li].index_select(2, safe_idx)    # (B, H, 2wsz, DH)
        vals = cache.v[li].index_select(2, safe_idx)
        kmask = valid.view(1, 1, -1, 1).to(keys.dtype)
        keys = keys * kmask                             # zero halo keys
        vals = vals * kmask
        s_row = torch.einsum("bhd,bhnd->bhn", q, keys) * (DH ** -0.5)
        # masked_fill with a python scalar: no host->device transfer, so
        # the op is hipGraph-capturable (torch.tensor(scalar) is not)
        s_row = s_row.masked_fill(~causal.view(1, 1, -1), -1e30)
        s_row = s_row - s_row.amax(dim=-1, keepdim=True)
        a = s_row.softmax(dim=-1)
        out = torch.einsum("bhn,bhnd->bhd", a, vals).reshape(B, H * DH)
        h = h + F.linear(out, attn.to_out.weight, attn.to_out.bias)

        # ---- ff branch ----
        y = _ln_row(h, ff.norm_weight)
        y_in = _shift_row(y, cache.ff_prev[li]) if ff.shift_tokens else y
        cache.ff_prev[li].copy_(y)
        t = F.linear(y_in, ff.proj_in.weight, ff.proj_in.bias)
        if ff.glu:
            a2, g2 = t.chunk(2, dim=-1)
            t = a2 * F.gelu(g2, approximate="tanh")
        else:
            t = F.gelu(t, approximate="tanh")
        if ff.sgu is not None:
            sgu = ff.sgu
            xa, gate = t.chunk(2, dim=-1)
            gate_ln = _ln_row(gate, sgu.norm_weight)
            hist = cache.gate_hist[li]
            hist.index_copy_(1, p.reshape(1), gate_ln.unsqueeze(1))
            n = hist.shape[1]
            w_row = sgu.spatial_weights.index_select(0, p.reshape(1))[0, :n]
            past = (torch.arange(n, device=dev) <= p).to(hist.dtype)
            gate_out = torch.einsum("n,bnd->bd", w_row.to(hist.dtype) * past,
                                    hist) + \
                sgu.spatial_biases.index_select(0, p.reshape(1))[0]
            t = xa * gate_out
            t = F.linear(t, sgu.proj_out.weight, sgu.proj_out.bias)
        h = h + F.linear(t, ff.proj_out.weight, ff.proj_out.bias)

    y = _ln_row(h, model.final_norm_weight)
    return F.linear(y, model.to_logits.weight, model.to_logits.bias)
    # NOTE: cache.pos is NOT advanced — the caller owns pos_dev and
    # increments it (in-graph when captured)


class GraphedDecodeStep:
    """hipGraph-captured decode step (TODO.md "Serving"; VERDICT r1 item
    6): one replay = forward_step_static + in-graph pos_dev increment.
    The eager per-token step is launch-bound (~500 kernel launches); the
    graph replays them as one unit.

    Pure-replay discipline (profiles/r01_graph_interleave_bug.md): no
    eager KERNEL may run between replays. The per-token loop only does
    host<->device MEMCPYs (token in, logits out) and CPU-side sampling,
    which keeps replay state intact — validated by
    tests/test_gpu_decode_graph.py against the eager cached decoder.

    Usage:
        cache = DecodeCache(model, batch=B)
        ... eager prefill via forward_step ...
        g = GraphedDecodeStep(model, cache, start_pos=cache.pos)
        for _ in range(n):
            logits = g.step(tokens)        # (B, V) device tensor
            tokens = <sample on host>      # no device kernels!
    """

    def __init__(self, model: ProGenBase, cache: DecodeCache,
                 start_pos: int, warmup: int = 2):
        dev = next(model.parameters()).device
        batch = cache.attn_prev[0].shape[0]
        self.cache = cache
        self.token = torch.zeros(batch, dtype=torch.long, device=dev)
        self.pos_dev = torch.tensor(start_pos, dtype=torch.long, device=dev)

        # warmup + capture perturb the cache and pos; snapshot & restore
        snap = {
            "k": [t.clone() for t in cache.k],
            "v": [t.clone() for t in cache.v],
            "ap": [t.clone() for t in cache.attn_prev],
            "fp": [t.clone() for t in cache.ff_prev],
            "gh": [t.clone() for t in cache.gate_hist if t is not None],
            "pos": self.pos_dev.clone(),
        }

        side = torch.cuda.Stream()
        side.wait_stream(torch.cuda.current_stream())
        with torch.cuda.stream(side):
            for _ in range(warmup):
                forward_step_static(model, self.token, cache, self.pos_dev)
                self.pos_dev += 1
        torch.cuda.current_stream().wait_stream(side)
        torch.cuda.synchronize()

        self.graph = torch.cuda.CUDAGraph()
        with torch.cuda.graph(self.graph):
            self.static_logits = forward_step_static(
                model, self.token, cache, self.pos_dev)
            self.pos_dev += 1  # in-graph position advance

        with torch.no_grad():
            for dst, src in zip(cache.k, snap["k"]):
                dst.copy_(src)
            for dst, src in zip(cache.v, snap["v"]):
                dst.copy_(src)
            for dst, src in zip(cache.attn_prev, snap["ap"]):
                dst.copy_(src)
            for dst, src in zip(cache.ff_prev, snap["fp"]):
                dst.copy_(src)
            gh = [t for t in cache.gate_hist if t is not None]
            for dst, src in zip(gh, snap["gh"]):
                dst.copy_(src)
            self.pos_dev.copy_(snap["pos"])
        torch.cuda.synchronize()

    def step(self, tokens: torch.Tensor) -> torch.Tensor:
        """Replay one decode step. ``tokens``: (B,) int64 (host or
        device); returns the (B, V) static logits tensor — copy it out
        before the next replay."""
        self.token.copy_(tokens.reshape(-1), non_blocking=False)
        self.graph.replay()
        return self.static_logits


@torch.no_grad()
def sample_cached(
    model: ProGenBase,
    prime: torch.Tensor,
    length: int,
    top_k: Optional[int] = None,
    add_bos: bool = False,
    generator: Optional[torch.Generator] = None,
    device=None,
    graph: bool = False,
) -> torch.Tensor:
    """Drop-in ``utils.sample`` with O(window + n_sgu) per-token cost.

    Same decoding semantics (reference: utils.py:106-135): gumbel-max with
    strict-> top-k masking (excluded logits -> 0, not -inf), pad-to-length
    output, everything after the second pad/EOS zeroed. Stops forwarding
    at EOS instead of emitting to full length."""
    dev = next(model.parameters()).device if device is None else device
    # the sequence and sampling math live on CPU (matching utils.sample's
    # generator semantics — its fn closure returns .cpu() logits); only
    # the forward_step runs on the model device
    prime = torch.as_tensor(prime).long().flatten().cpu()
    start_pos = prime.shape[-1]
    pad = (0, length - start_pos) if not add_bos else (1, length - start_pos - 1)
    seq = F.pad(prime, pad)
    if add_bos:
        start_pos += 1

    cache = DecodeCache(model, batch=1, device=dev)
    logits = None
    for p in range(start_pos):                           # prefill
        logits = forward_step(model, seq[p:p + 1].to(dev), cache)

    graphed = None
    if graph and dev.type == "cuda":
        logits = logits.clone()  # the static buffer will be reused
        graphed = GraphedDecodeStep(model, cache, start_pos=start_pos)

    pads_seen = int((seq[:start_pos] == 0).sum())
    for curr_pos in range(start_pos, length):
        logits_row = logits[0].cpu().float()  # D2H then host cast: no device kernel between replays
        noise = R.gumbel_noise(logits_row.shape, generator=generator,
                               device=logits_row.device)
        if top_k is not None:
            mask, logits_row = R.select_top_k(logits_row, top_k)
            noise = noise * mask
        sampled = (logits_row + noise).argmax(dim=-1)
        seq[curr_pos] = sampled
        if sampled.item() == 0:
            pads_seen += 1
            if pads_seen >= 2:
                break
        if curr_pos + 1 < length:
            if graphed is not None:
                logits = graphed.step(seq[curr_pos:curr_pos + 1])
            else:
                logits = forward_step(model,
                                      seq[curr_pos:curr_pos + 1].to(dev),
                                      cache)

    remove_after_eos = (seq == 0).long().cumsum(dim=-1) > 1
    return seq * (~remove_after_eos).long()


@torch.no_grad()
def sample_cached_batch(
    model: ProGenBase,
    primes,
    length: int,
    top_k: Optional[int] = None,
    generator: Optional[torch.Generator] = None,
    device=None,
    graph: bool = False,
) -> torch.Tensor:
    """Batched incremental decode: one forward_step advances ALL rows
    (the per-layer caches are batch-first), so serving throughput scales
    with batch at the same per-step latency.

    ``primes``: list of 1-D int tensors (ragged). Shorter primes are
    left-padded into the batch by decoding their sampled continuations
    only after their own prime ends — implemented by overwriting
    positions < len(prime_i) with the prime tokens. Rows that hit their
    second pad/EOS keep emitting into a dead tail that is zeroed at the
    end (same zero-after-second-pad rule as ``utils.sample``). With
    ``top_k=None`` and ``generator=None`` decoding is greedy argmax and
    each row's output equals its single-row ``sample_cached`` run
    (tests/test_decode.py::test_sample_cached_batch_greedy_parity).
    """
    dev = next(model.parameters()).device if device is None else device
    primes = [torch.as_tensor(p).long().flatten().cpu() for p in primes]
    B = len(primes)
    lens = [int(p.shape[-1]) for p in primes]
    seq = torch.zeros(B, length, dtype=torch.long)
    for i, p in enumerate(primes):
        seq[i, :lens[i]] = p

    cache = DecodeCache(model, batch=B, device=dev)
    graphed = GraphedDecodeStep(model, cache, start_pos=0) \
        if (graph and dev.type == "cuda") else None
    greedy = top_k is None and generator is None
    # per-row pad counts over WRITTEN tokens only (the unfilled zero tail
    # must not count as EOS); position 0 is written at entry
    pads = [int(seq[i, 0] == 0) for i in range(B)]
    for pos in range(length - 1):
        if graphed is not None:
            logits = graphed.step(seq[:, pos])
        else:
            logits = forward_step(model, seq[:, pos].to(dev), cache)  # (B, V)
        rows = logits.cpu().float()  # no device kernel between replays
        if greedy:
            nxt = rows.argmax(dim=-1)
        else:
            noise = R.gumbel_noise(rows.shape, generator=generator)
            if top_k is not None:
                mask, rows = R.select_top_k(rows, top_k)
                noise = noise * mask
            nxt = (rows + noise).argmax(dim=-1)
        w = pos + 1
        # keep prime tokens where the prime extends past this position
        for i in range(B):
            if w >= lens[i]:
                seq[i, w] = nxt[i]
            pads[i] += int(seq[i, w] == 0)
        if all(p >= 2 for p in pads):
            break  # every row has emitted its EOS

    remove_after_eos = (seq == 0).long().cumsum(dim=-1) > 1
    return seq * (~remove_after_eos).long()
