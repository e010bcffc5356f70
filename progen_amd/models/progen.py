"""The ProGen model — MI355X-native PyTorch module tree.

Architecture mirrors the JAX/Haiku reference exactly
(reference: progen_transformer/progen.py):

  embed -> depth x [ x += LocalAttention(x); x += FeedForward(x) ]
        -> LayerNorm -> Linear(num_tokens)   (no weight tying)

with the last ``global_mlp_depth`` layers swapping the GLU feedforward for
an SGU (gMLP spatial gating) feedforward (reference: progen.py:207-233).

Differences from the reference, by design (MI355X-first):
  - batch-first: forward(x) takes (B, N) int tokens, returns (B, N, V)
    logits. The reference's unbatched init/apply API is preserved by the
    ProGen wrapper below (reference API: progen.py:235-243, README.md:29-51).
  - the rotary sin/cos table is precomputed once as a buffer (the
    reference rebuilds it every call, progen.py:227).
  - hot ops run hand-written CDNA4 HIP kernels on GPU (ops/functional.py);
    projections are hipBLASLt GEMMs.
"""

from __future__ import annotations

import math
from typing import Any, Dict, Optional

import torch
import torch.nn as nn

from ..config import ProGenConfig
from ..ops import functional as OF
from ..ops import reference as R


def _haiku_linear_init_(weight: torch.Tensor, bias: Optional[torch.Tensor],
                        generator: Optional[torch.Generator] = None) -> None:
    """Haiku hk.Linear default init: truncated normal, stddev 1/sqrt(fan_in),
    zero bias. weight layout here is (out, in) (torch convention)."""
    fan_in = weight.shape[1]
    std = 1.0 / math.sqrt(fan_in)
    with torch.no_grad():
        weight.normal_(0.0, std, generator=generator).clamp_(-2 * std, 2 * std)
        if bias is not None:
            bias.zero_()


class Linear(nn.Linear):
    """nn.Linear with haiku-style default init; on GPU the weight grad is
    computed on a side stream concurrent with the backward chain
    (ops/overlap.py)."""

    def reset_parameters(self) -> None:
        _haiku_linear_init_(self.weight, self.bias)

    def forward(self, x: torch.Tensor) -> torch.Tensor:
        from ..ops import fp8
        if fp8.fp8_eligible(x, self.weight):
            # PROGEN_FP8=1: big projections run e4m3 hipBLASLt GEMMs
            # (fwd + dgrad at the 2x fp8 MFMA rate, wgrad bf16)
            return fp8.fp8_linear(x, self.weight, self.bias)
        from ..ops.overlap import overlap_linear
        return overlap_linear(x, self.weight, self.bias)


class LocalAttention(nn.Module):
    """Windowed causal attention with one-window lookback
    (reference: progen.py:50-103)."""

    def __init__(self, cfg: ProGenConfig):
        super().__init__()
        self.heads = cfg.heads
        self.window_size = cfg.window_size
        self.shift_tokens = cfg.shift_tokens
        inner = cfg.inner_dim
        self.norm_weight = nn.Parameter(torch.ones(cfg.dim))
        self.to_qkv = Linear(cfg.dim, inner * 3, bias=False)  # progen.py:70
        self.to_out = Linear(inner, cfg.dim, bias=True)       # progen.py:71

    def forward(self, x: torch.Tensor, sin: torch.Tensor, cos: torch.Tensor) -> torch.Tensor:
        x = OF.ln_shift(x, self.norm_weight, shift=self.shift_tokens)
        return self.inner(x, sin, cos)

    def inner(self, y: torch.Tensor, sin: torch.Tensor, cos: torch.Tensor) -> torch.Tensor:
        """Branch body AFTER the LN+shift prologue (the prologue is fused
        with the residual add in ProGenBase.forward)."""
        qkv = self.to_qkv(y)
        out = OF.local_attention(qkv, sin, cos, self.heads, self.window_size)
        return self.to_out(out)


class SGU(nn.Module):
    """gMLP spatial gating unit (reference: progen.py:151-185)."""

    def __init__(self, dim: int, dim_out: int, seq_len: int, eps: float = 1e-3):
        super().__init__()
        self.seq_len = seq_len
        self.norm_weight = nn.Parameter(torch.ones(dim // 2))
        init_scale = eps / seq_len  # progen.py:171-172
        self.spatial_weights = nn.Parameter(
            torch.empty(seq_len, seq_len).uniform_(-init_scale, init_scale))
        self.spatial_biases = nn.Parameter(torch.ones(seq_len, 1))  # progen.py:175
        self.proj_out = Linear(dim // 2, dim_out, bias=True)

    def forward(self, x: torch.Tensor) -> torch.Tensor:
        gated = OF.sgu_gate(x, self.norm_weight, self.spatial_weights,
                            self.spatial_biases)
        return self.proj_out(gated)


class FeedForward(nn.Module):
    """LN -> shift -> proj_in -> GELU/GLU -> [SGU] -> proj_out
    (reference: progen.py:105-149)."""

    def __init__(self, cfg: ProGenConfig, glu: bool, spatial_gate: bool):
        super().__init__()
        assert not (glu and spatial_gate), \
            "glu and sgu cannot be turned on at the same time"  # progen.py:118
        hidden = cfg.dim * cfg.ff_mult * (2 if glu else 1)
        self.glu = glu
        self.shift_tokens = cfg.shift_tokens
        self.norm_weight = nn.Parameter(torch.ones(cfg.dim))
        self.proj_in = Linear(cfg.dim, hidden, bias=True)
        self.sgu = SGU(hidden, hidden // 2, cfg.seq_len) if spatial_gate else None
        # reference proj_out is Linear(dim) applied after SGU's own proj_out
        # (progen.py:123,148): with SGU, x entering proj_out has hidden//2 dims.
        out_in = hidden // 2 if (glu or spatial_gate) else hidden
        self.proj_out = Linear(out_in, cfg.dim, bias=True)

    def forward(self, x: torch.Tensor) -> torch.Tensor:
        x = OF.ln_shift(x, self.norm_weight, shift=self.shift_tokens)
        return self.inner(x)

    def inner(self, x: torch.Tensor) -> torch.Tensor:
        """Branch body AFTER the LN+shift prologue (see LocalAttention.inner)."""
        x = self.proj_in(x)
        if self.glu:
            x = OF.glu_gelu(x)
        else:
            x = OF.gelu(x)
        if self.sgu is not None:
            x = self.sgu(x)
        return self.proj_out(x)


class ProGenBase(nn.Module):
    """Full model (reference: progen.py:187-233). Batch-first."""

    def __init__(self, cfg: ProGenConfig):
        super().__init__()
        self.cfg = cfg
        self.embed = nn.Embedding(cfg.num_tokens, cfg.dim)
        with torch.no_grad():  # haiku hk.Embed default: trunc-normal 1/sqrt(vocab)
            std = 1.0 / math.sqrt(cfg.num_tokens)
            self.embed.weight.normal_(0.0, std).clamp_(-2 * std, 2 * std)

        layers = []
        for i in range(cfg.depth):
            use_gmlp = (cfg.depth - i) <= cfg.global_mlp_depth  # progen.py:211
            use_ff_glu = (not use_gmlp) and cfg.ff_glu          # progen.py:212
            layers.append(nn.ModuleList([
                LocalAttention(cfg),
                FeedForward(cfg, glu=use_ff_glu, spatial_gate=use_gmlp),
            ]))
        self.layers = nn.ModuleList(layers)

        self.final_norm_weight = nn.Parameter(torch.ones(cfg.dim))
        self.to_logits = Linear(cfg.dim, cfg.num_tokens, bias=True)  # progen.py:219-222

        sin, cos = R.fixed_pos_embedding(cfg.seq_len, cfg.dim_head)
        self.register_buffer("rotary_sin", sin, persistent=False)
        self.register_buffer("rotary_cos", cos, persistent=False)

    def forward(self, x: torch.Tensor) -> torch.Tensor:
        if x.dim() == 1:
            x = x.unsqueeze(0)
        n = x.shape[1]
        if self.rotary_sin.dtype != torch.float32:
            # module.to(bf16) casts buffers; rotary tables must stay fp32
            sin, cos = R.fixed_pos_embedding(self.cfg.seq_len,
                                             self.cfg.dim_head,
                                             device=self.rotary_sin.device)
            self.rotary_sin, self.rotary_cos = sin, cos
        sin = self.rotary_sin[:n]
        cos = self.rotary_cos[:n]
        # residual adds are fused into each branch's LN+shift prologue
        # (ops/hip/ln_shift.hip RES variant): ln_shift_res(h, r) returns
        # the LN'd branch input AND the updated residual stream h + r,
        # so `h = h + branch(h)` never runs as a separate pass
        h = self.embed(x.long())
        r = None
        for attn, ff in self.layers:
            y, h = OF.ln_shift_res(h, r, attn.norm_weight,
                                   shift=attn.shift_tokens)
            a = attn.inner(y, sin, cos)
            y, h = OF.ln_shift_res(h, a, ff.norm_weight,
                                   shift=ff.shift_tokens)
            r = ff.inner(y)
        y, _ = OF.ln_shift_res(h, r, self.final_norm_weight, shift=False)
        return self.to_logits(y)

    def num_params(self) -> int:
        return sum(p.numel() for p in self.parameters())


# ---------------------------------------------------------------------------
# reference-parity functional API:  model = ProGen(...); model.init / .apply
# ---------------------------------------------------------------------------

class TransformedProGen:
    """Wrapper preserving the reference's hk.transform-style API
    (reference: progen.py:235-243; usage README.md:29-51):

        model = ProGen(num_tokens=256, dim=512, seq_len=1024, ...)
        params = model.init(rng, seq)            # seq: 1-D int tensor/array
        logits = model.apply(params, rng, seq)   # -> (seq_len, num_tokens)

    ``rng`` may be an int seed, a torch.Generator, or None. ``params`` is a
    flat dict name -> tensor (the module state_dict). The underlying
    batch-first nn.Module is available as ``.module`` for idiomatic use.
    """

    def __init__(self, cfg: ProGenConfig, module: Optional[ProGenBase] = None):
        self.cfg = cfg
        self.module = module if module is not None else ProGenBase(cfg)

    # -- helpers -----------------------------------------------------------
    @staticmethod
    def _seed_everything(rng) -> None:
        if rng is None:
            return
        if isinstance(rng, torch.Generator):
            torch.manual_seed(int(rng.initial_seed()))
        else:
            torch.manual_seed(int(rng))

    @staticmethod
    def _as_tensor(seq) -> torch.Tensor:
        t = torch.as_tensor(seq)
        return t.long()

    # -- reference API ------------------------------------------------------
    def init(self, rng=None, seq=None) -> Dict[str, torch.Tensor]:
        """(Re)initialize parameters; returns the params dict."""
        self._seed_everything(rng)
        self.module = ProGenBase(self.cfg)
        return {k: v.detach().clone() for k, v in self.module.state_dict().items()}

    def apply(self, params: Dict[str, torch.Tensor], rng=None, seq=None) -> torch.Tensor:
        """Forward a 1-D int sequence -> (n, num_tokens) logits (no grad).

        The forward is deterministic (no dropout), so ``rng`` is accepted
        for parity and unused — matching the reference where the rng
        threading exists only for Haiku's API."""
        seq_t = self._as_tensor(seq)
        unbatched = seq_t.dim() == 1
        if params is not None:
            self.module.load_state_dict(
                {k: torch.as_tensor(v) for k, v in params.items()}, strict=True)
        dev = next(self.module.parameters()).device
        with torch.no_grad():
            logits = self.module(seq_t.to(dev))
        return logits[0] if unbatched else logits


def ProGen(mixed_precision: bool = False,
           mixed_precision_policy: Optional[Dict[str, str]] = None,
           **kwargs: Any) -> TransformedProGen:
    """Factory matching the reference constructor surface
    (reference: progen.py:235-243).

    ``mixed_precision`` maps to bf16 compute on MI355X (the reference's
    jmp fp16 policy has no advantage on CDNA4; bf16 is the MFMA-native
    compute dtype). The policy dict is accepted for parity; only its
    intent (mixed precision on/off) is honored."""
    cfg = ProGenConfig.from_dict(kwargs)
    if mixed_precision:
        cfg.compute_dtype = "bf16"
    return TransformedProGen(cfg)
