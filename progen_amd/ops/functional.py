"""Differentiable ProGen ops with CPU-reference / HIP-kernel dispatch.

GPU path: hand-written CDNA4 HIP kernels (progen_amd/ops/hip/) via the
in-tree extension ``progen_amd._C`` — mandatory on GPU (no silent eager
fallback; see ops/dispatch.py). Plain GEMMs (QKV/FF/out projections, the
SGU spatial matmul) go through torch.matmul = hipBLASLt, which is the
library-GEMM path, not a compatibility layer.

CPU path: the pure-PyTorch fp32 reference (ops/reference.py), which is
also the numerics oracle for the kernels.
"""

from __future__ import annotations

import torch

from . import dispatch, reference


# ---------------------------------------------------------------------------
# fused LayerNorm(scale-only) + token shift
# ---------------------------------------------------------------------------

class _LnShiftFn(torch.autograd.Function):
    @staticmethod
    def forward(ctx, x, weight, shift: bool, eps: float):
        C = dispatch.ext()
        y, mean, rstd = C.ln_shift_fwd(x, weight, bool(shift), float(eps))
        ctx.save_for_backward(x, weight, mean, rstd)
        ctx.shift = bool(shift)
        return y

    @staticmethod
    def backward(ctx, dy):
        x, weight, mean, rstd = ctx.saved_tensors
        C = dispatch.ext()
        dx, dweight = C.ln_shift_bwd(dy.contiguous(), x, weight, mean, rstd, ctx.shift)
        return dx, dweight, None, None


def ln_shift(x: torch.Tensor, weight: torch.Tensor, shift: bool = True,
             eps: float = 1e-5) -> torch.Tensor:
    """LN (scale-only, reference: progen.py:22) + optional token shift
    (reference: progen.py:43-46), fused on GPU."""
    if dispatch.use_hip(x, "ln"):
        return _LnShiftFn.apply(x, weight, shift, eps)
    return reference.ln_shift(x, weight, shift, eps)


class _LnShiftResFn(torch.autograd.Function):
    """Residual-add-fused LN+shift: forms s = x + res in-kernel (bf16
    rounding identical to an eager add), normalizes s, and returns
    (y, s) — s is the new residual stream. Backward returns the SAME
    gradient tensor for x and res (d(x+res) fans out identically); both
    are intermediate activations here, never leaves, so sharing is safe
    and saves the autograd accumulation pass."""

    @staticmethod
    def forward(ctx, x, res, weight, shift: bool, eps: float):
        C = dispatch.ext()
        y, s, mean, rstd = C.ln_shift_res_fwd(x, res, weight, bool(shift),
                                              float(eps))
        ctx.save_for_backward(s, weight, mean, rstd)
        ctx.shift = bool(shift)
        ctx.set_materialize_grads(False)
        return y, s

    @staticmethod
    def backward(ctx, dy, ds):
        s, weight, mean, rstd = ctx.saved_tensors
        C = dispatch.ext()
        ds = ds.contiguous() if ds is not None else None
        dx, dweight = C.ln_shift_res_bwd(dy.contiguous(), ds, s, weight,
                                         mean, rstd, ctx.shift)
        return dx, dx, dweight, None, None


def ln_shift_res(x: torch.Tensor, res, weight: torch.Tensor,
                 shift: bool = True, eps: float = 1e-5):
    """Residual add + LN + shift in one pass: returns (y, s) with
    s = x + res (the updated residual stream) and y = ln_shift(s).
    ``res=None`` degenerates to plain ln_shift with s = x."""
    if dispatch.use_hip(x, "ln"):
        if res is None:
            return _LnShiftFn.apply(x, weight, shift, eps), x
        return _LnShiftResFn.apply(x, res.contiguous(), weight, shift, eps)
    s = x if res is None else x + res
    return reference.ln_shift(s, weight, shift, eps), s


# ---------------------------------------------------------------------------
# fused local window attention (rotary + window + softmax + AV)
# ---------------------------------------------------------------------------

class _LocalAttnFn(torch.autograd.Function):
    @staticmethod
    def forward(ctx, qkv, sin, cos, heads: int, window_size: int, halo):
        C = dispatch.ext()
        # pre-rotation pass (rotary on q, k AND v — progen.py:87); the
        # attention kernels then stage pure bf16 copies
        qkv_rot = C.rope_qkv(qkv.contiguous(), sin, cos)
        out, lse = C.attn_fwd(qkv_rot, int(heads), int(window_size),
                              halo=halo)
        ctx.save_for_backward(qkv_rot, sin, cos, out, lse,
                              *([halo] if halo is not None else []))
        ctx.heads = int(heads)
        ctx.window_size = int(window_size)
        ctx.has_halo = halo is not None
        return out

    @staticmethod
    def backward(ctx, dout):
        qkv_rot, sin, cos, out, lse = ctx.saved_tensors[:5]
        halo = ctx.saved_tensors[5] if ctx.has_halo else None
        C = dispatch.ext()
        # bwd finalize applies the inverse rotation (rotary is linear)
        res = C.attn_bwd(dout.contiguous(), qkv_rot, sin, cos, out, lse,
                         ctx.heads, ctx.window_size, halo=halo)
        dqkv = res[0]
        # the halo holds PRE-ROTATED [k|v]: its grad stays in rotated
        # space (the peer applies the inverse rotation after receiving
        # it, with its own absolute positions — parallel/cp.py)
        dhalo = res[1].to(dout.dtype) if ctx.has_halo else None
        return dqkv, None, None, None, None, dhalo


def local_attention(qkv: torch.Tensor, sin: torch.Tensor, cos: torch.Tensor,
                    heads: int, window_size: int,
                    halo: torch.Tensor = None) -> torch.Tensor:
    """Fused windowed-causal attention core (reference: progen.py:83-103).

    Applies interleaved rotary to q, k AND v (quirk, progen.py:87), windows
    the sequence with one-window lookback (window 0's zero lookback keys
    UNMASKED, progen.py:90-96), runs online-softmax attention on MFMA and
    returns the merged (B, N, h*dh) context.

    ``halo``: optional (B, wsz, 2*H*dh) ROTATED [k|v] band that replaces
    window 0's zero lookback (context parallelism, parallel/cp.py); its
    gradient (same shape/space) is returned to the autograd graph."""
    if dispatch.use_hip(qkv, "attn"):
        return _LocalAttnFn.apply(qkv, sin, cos, heads, window_size, halo)
    assert halo is None, "halo is a kernel-path (GPU) feature"
    return reference.local_attention(qkv, sin, cos, heads, window_size)


# ---------------------------------------------------------------------------
# GLU-GELU epilogue
# ---------------------------------------------------------------------------

class _GluGeluFn(torch.autograd.Function):
    @staticmethod
    def forward(ctx, h):
        C = dispatch.ext()
        y = C.glu_fwd(h)
        ctx.save_for_backward(h)
        return y

    @staticmethod
    def backward(ctx, dy):
        (h,) = ctx.saved_tensors
        C = dispatch.ext()
        return C.glu_bwd(dy.contiguous(), h)


def glu_gelu(h: torch.Tensor) -> torch.Tensor:
    """x, gate = split(h, 2); x * gelu(gate)  (reference: progen.py:139-141)."""
    if dispatch.use_hip(h, "glu"):
        return _GluGeluFn.apply(h)
    return reference.glu_gelu(h)


class _GeluFn(torch.autograd.Function):
    @staticmethod
    def forward(ctx, h):
        C = dispatch.ext()
        y = C.gelu_fwd(h)
        ctx.save_for_backward(h)
        return y

    @staticmethod
    def backward(ctx, dy):
        (h,) = ctx.saved_tensors
        C = dispatch.ext()
        return C.gelu_bwd(dy.contiguous(), h)


def gelu(h: torch.Tensor) -> torch.Tensor:
    if dispatch.use_hip(h, "glu"):
        return _GeluFn.apply(h)
    return reference.gelu(h)


# ---------------------------------------------------------------------------
# masked cross-entropy (fused log_softmax + NLL gather; mask math on host)
# ---------------------------------------------------------------------------

class _CERowFn(torch.autograd.Function):
    """Per-row fused -log_softmax(logits)[target]; returns nll (B, N) fp32.

    The EOS/pad mask reduction (reference: utils.py:54-59) is cheap
    (B, N)-shaped tensor math and stays in torch; the (B, N, V) work is
    the kernel."""

    @staticmethod
    def forward(ctx, logits, targets):
        C = dispatch.ext()
        nll, lse = C.ce_fwd(logits, targets)
        ctx.save_for_backward(logits, targets, lse)
        return nll

    @staticmethod
    def backward(ctx, dnll):
        logits, targets, lse = ctx.saved_tensors
        C = dispatch.ext()
        dlogits = C.ce_bwd(dnll.contiguous(), logits, targets, lse)
        return dlogits, None


def cross_entropy(logits: torch.Tensor, targets: torch.Tensor,
                  ignore_index: int = 0) -> torch.Tensor:
    """Masked CE with first-pad-as-EOS (reference: utils.py:45-59); the
    per-sequence masked mean then batch mean reduction order is preserved
    (reference: utils.py:67,75-76)."""
    if dispatch.use_hip(logits, "ce"):
        targets = targets.long().contiguous()
        nll = _CERowFn.apply(logits.contiguous(), targets)
        mask = targets != ignore_index
        eos_mask = (~mask).long().cumsum(dim=-1) == 1
        mask = (mask | eos_mask).to(nll.dtype)
        ce_per_seq = (nll * mask).sum(dim=-1) / mask.sum(dim=-1)
        return ce_per_seq.mean()
    return reference.cross_entropy(logits, targets, ignore_index)


# ---------------------------------------------------------------------------
# SGU spatial gating
# ---------------------------------------------------------------------------

_TRI_CACHE = {}


def _tri_tiles(n: int, device) -> tuple:
    """Lower-triangle 64x64 tile index lists for sgu_dw (cached per n)."""
    key = (n, str(device))
    if key not in _TRI_CACHE:
        ms, ks = [], []
        for mt in range(n // 64):
            for kt in range(mt + 1):
                ms.append(mt)
                ks.append(kt)
        _TRI_CACHE[key] = (
            torch.tensor(ms, dtype=torch.int32, device=device),
            torch.tensor(ks, dtype=torch.int32, device=device),
        )
    return _TRI_CACHE[key]


class _SGUFn(torch.autograd.Function):
    """Causal spatial matmul + bias + gate multiply on the hand-written
    CDNA4 kernels (ops/hip/sgu.hip)."""

    @staticmethod
    def forward(ctx, xa, g_ln, w, bias):
        C = dispatch.ext()
        out, gate_out = C.sgu_fwd(xa, g_ln, w, bias)
        ctx.save_for_backward(xa, g_ln, w, gate_out)
        return out

    @staticmethod
    def backward(ctx, dy):
        xa, g_ln, w, gate_out = ctx.saved_tensors
        C = dispatch.ext()
        dy = dy.contiguous()
        dxa = dy * gate_out
        t = (dy * xa).contiguous()
        dg_ln = C.sgu_dgate(t, w)
        n = xa.shape[1]
        tri_m, tri_k = _tri_tiles(n, xa.device)
        dw = C.sgu_dw(t, g_ln, tri_m, tri_k).to(w.dtype)
        db = t.float().sum(dim=(0, 2)).unsqueeze(-1).to(w.dtype)
        return dxa, dg_ln, dw, db


def sgu_gate(x: torch.Tensor, norm_weight: torch.Tensor,
             spatial_weights: torch.Tensor, spatial_biases: torch.Tensor,
             eps: float = 1e-5) -> torch.Tensor:
    """gMLP spatial gating unit core (reference: progen.py:166-183).

    GPU path: fused LN kernel on the gate half + the hand-written causal
    spatial-matmul kernels (ops/hip/sgu.hip — the tril mask is baked into
    the tile iteration). Sequences shorter than 256 (toy configs) use the
    hipBLASLt composite path."""
    if dispatch.use_hip(x, "sgu"):
        xa, gate = x.chunk(2, dim=-1)
        gate_ln = ln_shift(gate.contiguous(), norm_weight, shift=False, eps=eps)
        n = x.shape[1]
        d = xa.shape[-1]
        if (x.dtype == torch.bfloat16 and n % 256 == 0 and d % 64 == 0):
            w = spatial_weights[:n, :n].contiguous()
            return _SGUFn.apply(xa.contiguous(), gate_ln, w,
                                spatial_biases[:n])
        w = spatial_weights[:n, :n].tril().to(gate_ln.dtype)
        gate_ln = torch.einsum("bnd,mn->bmd", gate_ln, w) + \
            spatial_biases[:n].to(gate_ln.dtype)
        return xa * gate_ln
    return reference.sgu_gate(x, norm_weight, spatial_weights, spatial_biases, eps)
