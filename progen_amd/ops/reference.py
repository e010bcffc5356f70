"""Pure-PyTorch reference implementations of every ProGen op.

These serve two purposes:
  1. the CPU execution path (this container has no GPU), and
  2. the fp32 numerics oracle that every hand-written HIP kernel is
     tested against (tests/test_ops.py, tests/test_gpu_kernels.py).

Semantics mirror the JAX reference exactly, including its quirks:
  - GPT-J interleaved rotary applied to q, k AND v
    (reference: progen_transformer/progen.py:24-41,87)
  - token shift of the first ceil(d/2) channels by +1 position
    (reference: progen.py:43-46)
  - scale-only LayerNorm, no offset (reference: progen.py:22)
  - local window attention with one-window lookback where window 0's
    lookback keys are all-zero and UNMASKED (reference: progen.py:88-96)
  - masked cross-entropy where the first pad token is learned as EOS
    (reference: progen_transformer/utils.py:45-59)

All functions are batch-first: x is (B, N, ...) unlike the reference's
unbatched (N, ...) traced-through-vmap layout.
"""

from __future__ import annotations

from typing import Optional, Tuple

import torch
import torch.nn.functional as F

ATTN_MASK_VALUE = -1e10  # reference: progen.py:18


# ---------------------------------------------------------------------------
# rotary embedding (GPT-J interleaved)
# ---------------------------------------------------------------------------

def fixed_pos_embedding(
    seq: int, dim: int, dtype: torch.dtype = torch.float32,
    device: Optional[torch.device] = None,
) -> Tuple[torch.Tensor, torch.Tensor]:
    """sin/cos tables of shape (seq, dim).

    Each frequency is repeated twice along the last axis, i.e. elements
    2i and 2i+1 share inv_freq[i]  (reference: progen.py:24-28 — the
    ``repeat 'b n -> b (n r)', r=2``).
    """
    inv_freq = 1.0 / (
        10000 ** (torch.arange(0, dim, 2, dtype=torch.float64, device=device) / dim)
    )
    t = torch.arange(seq, dtype=torch.float64, device=device)
    sinusoid = torch.einsum("i,j->ij", t, inv_freq)           # (seq, dim/2)
    sinusoid = sinusoid.repeat_interleave(2, dim=-1)          # (seq, dim)
    return sinusoid.sin().to(dtype), sinusoid.cos().to(dtype)


def rotate_every_two(x: torch.Tensor) -> torch.Tensor:
    """(x0, x1, x2, x3, ...) -> (-x1, x0, -x3, x2, ...)  (reference: progen.py:30-34)."""
    x1 = x[..., 0::2]
    x2 = x[..., 1::2]
    return torch.stack((-x2, x1), dim=-1).flatten(-2)


def apply_rotary_pos_emb(
    x: torch.Tensor, sin: torch.Tensor, cos: torch.Tensor
) -> torch.Tensor:
    """Apply interleaved rotary over the FULL last dim (rot_dim == dim_head in
    the reference, progen.py:36-41). x: (..., n, d); sin/cos: (n, d)."""
    return x * cos + rotate_every_two(x) * sin


# ---------------------------------------------------------------------------
# token shift
# ---------------------------------------------------------------------------

def shift_tokens(x: torch.Tensor) -> torch.Tensor:
    """Shift the first ceil(d/2) channels by +1 position (pad front, drop
    last). x: (B, N, D).  (reference: progen.py:43-46; np.array_split puts
    the extra channel in the FIRST half for odd D.)"""
    d = x.shape[-1]
    split = -(-d // 2)  # ceil — matches np.array_split(x, 2, axis=-1)
    x_shift, x_pass = x[..., :split], x[..., split:]
    x_shift = F.pad(x_shift, (0, 0, 1, 0))[:, :-1]
    return torch.cat((x_shift, x_pass), dim=-1)


# ---------------------------------------------------------------------------
# scale-only LayerNorm (+ optional fused token shift)
# ---------------------------------------------------------------------------

def layernorm_nobias(
    x: torch.Tensor, weight: torch.Tensor, eps: float = 1e-5
) -> torch.Tensor:
    """LayerNorm with learned scale, no offset (reference: progen.py:22 —
    hk.LayerNorm(create_scale=True, create_offset=False, axis=-1)).

    Statistics are computed in fp32 regardless of input dtype."""
    orig_dtype = x.dtype
    if x.dtype in (torch.bfloat16, torch.float16):
        x = x.float()
    mu = x.mean(dim=-1, keepdim=True)
    var = x.var(dim=-1, unbiased=False, keepdim=True)
    y = (x - mu) * torch.rsqrt(var + eps)
    return (y * weight.to(y.dtype)).to(orig_dtype)


def ln_shift(
    x: torch.Tensor, weight: torch.Tensor, shift: bool = True, eps: float = 1e-5
) -> torch.Tensor:
    """Fused LN -> token-shift prologue used by both branches
    (reference: progen.py:74-77 and progen.py:132-135)."""
    y = layernorm_nobias(x, weight, eps)
    if shift:
        y = shift_tokens(y)
    return y


# ---------------------------------------------------------------------------
# local window attention core
# ---------------------------------------------------------------------------

def local_attention(
    qkv: torch.Tensor,
    sin: torch.Tensor,
    cos: torch.Tensor,
    heads: int,
    window_size: int,
) -> torch.Tensor:
    """Windowed causal attention with one-window lookback.

    qkv: (B, N, 3*h*dh) — output of the bias-free QKV projection
         (reference: progen.py:70,83).
    sin/cos: (N, dh) rotary tables.
    Returns (B, N, h*dh) in the merged '(w n) (h d)' layout
    (reference: progen.py:85-102).

    Quirks preserved:
      - rotary is applied to q, k AND v (reference: progen.py:87)
      - window 0's lookback keys are the zero-pad window and are NOT
        masked: their logit is exactly 0 pre-scale and enters the softmax
        denominator (reference: progen.py:90-96)
      - mask = tril(ones(wsz, 2*wsz), k=wsz): full previous window plus
        causal own window (reference: progen.py:95)
      - pre-softmax max-subtraction with stop_gradient (progen.py:98)
    """
    B, N, three_inner = qkv.shape
    dh = three_inner // (3 * heads)
    wsz = window_size
    assert N % wsz == 0, "sequence length must be divisible by the window size"
    w = N // wsz
    scale = dh ** -0.5

    q, k, v = qkv.chunk(3, dim=-1)
    # (B, N, h*dh) -> (B, h, N, dh)
    def to_heads(t: torch.Tensor) -> torch.Tensor:
        return t.view(B, N, heads, dh).transpose(1, 2)

    q, k, v = map(to_heads, (q, k, v))

    sin = sin.to(q.dtype)
    cos = cos.to(q.dtype)
    q, k, v = (apply_rotary_pos_emb(t, sin, cos) for t in (q, k, v))

    # window: (B, h, w, wsz, dh)
    q = q.view(B, heads, w, wsz, dh)
    k = k.view(B, heads, w, wsz, dh)
    v = v.view(B, heads, w, wsz, dh)

    # one-window lookback: pad a zero window in front, build [prev ‖ own]
    # (reference: progen.py:90-91)
    k = F.pad(k, (0, 0, 0, 0, 1, 0))
    v = F.pad(v, (0, 0, 0, 0, 1, 0))
    k = torch.cat((k[:, :, :-1], k[:, :, 1:]), dim=3)   # (B, h, w, 2*wsz, dh)
    v = torch.cat((v[:, :, :-1], v[:, :, 1:]), dim=3)

    sim = torch.einsum("bhwid,bhwjd->bhwij", q, k) * scale

    mask = torch.ones(wsz, 2 * wsz, dtype=torch.bool, device=qkv.device).tril(wsz)
    # masked_fill with a python scalar (not torch.tensor(...): that is an
    # H2D op, illegal under hipGraph capture — needed by the
    # PROGEN_FORCE_EAGER-in-graph bisect path of the replay investigation)
    sim = sim.masked_fill(~mask, ATTN_MASK_VALUE)

    sim = sim - sim.amax(dim=-1, keepdim=True).detach()
    attn = sim.softmax(dim=-1)

    out = torch.einsum("bhwij,bhwjd->bhwid", attn, v)
    # 'h w n d -> (w n) (h d)'
    out = out.permute(0, 2, 3, 1, 4).reshape(B, N, heads * dh)
    return out


# ---------------------------------------------------------------------------
# GLU feedforward epilogue
# ---------------------------------------------------------------------------

def glu_gelu(x: torch.Tensor) -> torch.Tensor:
    """x, gate = split(h, 2); x * gelu(gate)  (reference: progen.py:139-141).

    The reference's jax.nn.gelu is the tanh approximation (JAX default
    approximate=True); we use the same."""
    x, gate = x.chunk(2, dim=-1)
    return x * F.gelu(gate, approximate="tanh")


def gelu(x: torch.Tensor) -> torch.Tensor:
    """Plain GELU branch for non-GLU FF (reference: progen.py:143)."""
    return F.gelu(x, approximate="tanh")


# ---------------------------------------------------------------------------
# SGU — gMLP spatial gating unit
# ---------------------------------------------------------------------------

def sgu_gate(
    x: torch.Tensor,
    norm_weight: torch.Tensor,
    spatial_weights: torch.Tensor,
    spatial_biases: torch.Tensor,
    eps: float = 1e-5,
) -> torch.Tensor:
    """Spatial gating: split hidden in half, LN the gate half, apply the
    causal learned (n, n) spatial matrix, multiply
    (reference: progen.py:166-183).

    x: (B, N, H) with H even; returns (B, N, H/2) — the gated half
    BEFORE the output projection (proj_out is a plain GEMM, done by the
    caller)."""
    xa, gate = x.chunk(2, dim=-1)
    gate = layernorm_nobias(gate, norm_weight, eps)

    n = x.shape[1]
    w = spatial_weights[:n, :n].tril()  # mask = tril(ones(n, n)) (progen.py:179-180)
    # gate_out[m, d] = sum_n W[m, n] * gate[n, d] + b[m]
    gate = torch.einsum("bnd,mn->bmd", gate, w.to(gate.dtype)) + spatial_biases[:n].to(gate.dtype)
    return xa * gate


# ---------------------------------------------------------------------------
# masked cross-entropy with first-pad-as-EOS
# ---------------------------------------------------------------------------

def masked_mean(t: torch.Tensor, mask: torch.Tensor, dim=None) -> torch.Tensor:
    """(reference: utils.py:42-43)"""
    mask = mask.to(t.dtype)
    return (t * mask).sum(dim=dim) / mask.sum(dim=dim)


def cross_entropy(
    logits: torch.Tensor, targets: torch.Tensor, ignore_index: int = 0
) -> torch.Tensor:
    """Per-sequence masked CE, then mean over batch.

    mask = (targets != 0) extended by the first pad position, so the model
    learns the first pad as EOS (reference: utils.py:45-59). The reference
    computes a per-sequence masked mean inside vmap and then a plain mean
    over the batch (utils.py:67,75-76) — we preserve that exact reduction
    order (NOT a global masked mean).

    logits: (B, N, V); targets: (B, N) int64. Softmax in fp32.
    """
    if logits.dtype in (torch.bfloat16, torch.float16):
        logits = logits.float()  # softmax statistics in >= fp32
    logprobs = F.log_softmax(logits, dim=-1)
    nll = logprobs.gather(-1, targets.unsqueeze(-1).long()).squeeze(-1)

    mask = targets != ignore_index
    eos_mask = (~mask).long().cumsum(dim=-1) == 1
    mask = mask | eos_mask

    ce_per_seq = -masked_mean(nll, mask, dim=-1)  # (B,)
    return ce_per_seq.mean()


# ---------------------------------------------------------------------------
# sampling helpers
# ---------------------------------------------------------------------------

def select_top_k(t: torch.Tensor, k: int) -> Tuple[torch.Tensor, torch.Tensor]:
    """Top-k mask with the reference's quirks: strict `>` against the k-th
    value (may select fewer than k on ties) and excluded logits set to 0,
    not -inf (reference: utils.py:97-100)."""
    values, _ = t.topk(k, dim=-1)
    mask = t > values.amin(dim=-1, keepdim=True)
    return mask, torch.where(mask, t, torch.zeros_like(t))


def gumbel_noise(shape, generator=None, device=None, dtype=torch.float32) -> torch.Tensor:
    u = torch.rand(shape, generator=generator, device=device, dtype=dtype)
    eps = 1e-20  # reference: utils.py:20-21 log(t + eps)
    return -torch.log(-torch.log(u + eps) + eps)
