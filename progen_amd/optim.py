"""Optimizer: global-norm clip + AdamW (+ grad accumulation).

Mirrors the reference's optax chain
  clip_by_global_norm(max_grad_norm) -> adamw(lr, wd, mask=ndim>1) ->
  apply_every(grad_accum_every)
(reference: train.py:115-121) with the MI355X-native execution model:

  - ALL trainable parameters live in ONE flat contiguous buffer per state
    kind (param dtype / fp32 master / fp32 exp_avg / fp32 exp_avg_sq /
    grad dtype), ordered so that gradients produced late in backward sit
    early in the buffer (reverse registration order) — the same layout
    the DDP bucketer all-reduces, so optimizer, grad accumulation and
    communication all address the same memory;
  - on GPU the whole update is ONE hand-written HIP kernel pass
    (ops/hip/adamw.hip) over the flat buffers, chunked per-tensor so the
    ndim>1 weight-decay mask (reference: train.py:115) is a per-chunk flag;
  - the global grad norm is a single fused reduction.

Math matches optax:
  clip:  g *= max_norm / max(||g||, max_norm)                (optax clip_by_global_norm)
  adamw: m = b1*m + (1-b1)*g; v = b2*v + (1-b2)*g^2
         mhat = m/(1-b1^t); vhat = v/(1-b2^t)
         p -= lr * (mhat/(sqrt(vhat)+eps) + wd*p*decay_mask)

``accum_mode``:
  - "sum" (default): gradients of k micro-batches accumulate in the flat
    grad buffer; step() applies one update from the mean gradient. This
    is the standard scheme implied by the DP-overlap design.
  - "apply_every": the reference's quirk semantics (optax apply_every
    AFTER adamw, train.py:120): Adam moments advance EVERY micro-batch
    and the resulting updates are summed for k micro-batches, then
    applied at once. Supported for parity (eager path).
"""

from __future__ import annotations

from typing import Dict, List, Optional, Tuple

import torch

from .ops import dispatch

CHUNK = 1 << 16  # elements per kernel chunk (64Ki)


class FlatParamSpace:
    """Re-homes a module's parameters into one flat contiguous buffer and
    pre-assigns .grad as views of a flat grad buffer (autograd accumulates
    in place into pre-set grads)."""

    def __init__(self, module: torch.nn.Module,
                 grad_dtype: Optional[torch.dtype] = None):
        params = [p for p in module.parameters() if p.requires_grad]
        # reverse registration order ≈ backward readiness order: the last
        # layers' grads arrive first, so DDP buckets at low offsets fire early
        params = params[::-1]
        self.params: List[torch.nn.Parameter] = params
        self.numel = sum(p.numel() for p in params)
        dev = params[0].device
        dt = params[0].dtype
        self.flat = torch.empty(self.numel, dtype=dt, device=dev)
        self.flat_grad = torch.zeros(
            self.numel, dtype=grad_dtype or dt, device=dev)
        self.offsets: List[Tuple[int, int]] = []
        off = 0
        for p in params:
            n = p.numel()
            self.flat[off:off + n].copy_(p.data.reshape(-1))
            p.data = self.flat[off:off + n].view(p.shape)
            p.grad = self.flat_grad[off:off + n].view(p.shape)
            self.offsets.append((off, n))
            off += n

    def zero_grad(self) -> None:
        self.flat_grad.zero_()


class ProGenAdamW:
    """clip + AdamW + accumulation over a FlatParamSpace."""

    def __init__(
        self,
        module: torch.nn.Module,
        lr: float = 2e-4,
        betas: Tuple[float, float] = (0.9, 0.999),
        eps: float = 1e-8,
        weight_decay: float = 1e-3,
        max_grad_norm: Optional[float] = 0.5,
        accum_mode: str = "sum",
        grad_accum_every: int = 1,
    ):
        assert accum_mode in ("sum", "apply_every")
        self.lr = lr
        self.betas = betas
        self.eps = eps
        self.weight_decay = weight_decay
        self.max_grad_norm = max_grad_norm
        self.accum_mode = accum_mode
        self.grad_accum_every = grad_accum_every

        self.space = FlatParamSpace(module)
        flat = self.space.flat
        self.is_low_precision = flat.dtype in (torch.bfloat16, torch.float16)
        self.master = flat.float() if self.is_low_precision else flat
        self.exp_avg = torch.zeros_like(self.master)
        self.exp_avg_sq = torch.zeros_like(self.master)
        self.step_count = 0
        self._micro = 0
        self._update_acc: Optional[torch.Tensor] = None  # apply_every mode
        # device-side step counter: the fused kernel increments it and
        # computes bias correction from it, so hipGraph replays keep
        # correct Adam bias correction without host involvement
        self.step_dev = torch.zeros(1, dtype=torch.int32, device=flat.device) \
            if flat.is_cuda else None
        # optional hook returning the GLOBAL grad sum-of-squares (0-dim
        # tensor) for clip-by-global-norm under model parallelism: under
        # TP the local flat buffer holds only this rank's shards, so the
        # true norm needs a cross-rank reduction
        # (progen_amd/parallel/tp_model.py::tp_grad_sumsq_fn)
        self.norm_sumsq_fn = None

        # per-chunk decay flags: weight decay only on ndim>1 params
        # (reference: train.py:115 exclude_norm_and_bias_params)
        starts, ends, decay = [], [], []
        for p, (off, n) in zip(self.space.params, self.space.offsets):
            d = 1 if p.dim() > 1 else 0
            for c in range(off, off + n, CHUNK):
                starts.append(c)
                ends.append(min(c + CHUNK, off + n))
                decay.append(d)
        dev = flat.device
        self.chunk_starts = torch.tensor(starts, dtype=torch.int64, device=dev)
        self.chunk_ends = torch.tensor(ends, dtype=torch.int64, device=dev)
        self.chunk_decay = torch.tensor(decay, dtype=torch.int32, device=dev)
        # flat decay mask for the eager path
        self._decay_mask = torch.zeros_like(self.master)
        for p, (off, n) in zip(self.space.params, self.space.offsets):
            if p.dim() > 1:
                self._decay_mask[off:off + n] = 1.0

    # -- public API ---------------------------------------------------------

    def zero_grad(self) -> None:
        self.space.zero_grad()

    def resync_master(self) -> None:
        """Re-derive the fp32 master from the (possibly just broadcast)
        flat param buffer."""
        self.master.copy_(self.space.flat.float())

    def grad_norm(self) -> torch.Tensor:
        return torch.linalg.vector_norm(self.space.flat_grad.float())

    def micro_step(self) -> bool:
        """Call once per micro-batch AFTER backward. Returns True when the
        parameters were actually updated this call."""
        self._micro += 1
        if self.accum_mode == "apply_every":
            if self.space.flat_grad.is_cuda:
                from .ops.overlap import WgradQueue
                WgradQueue.sync()
            self._apply_every_micro()
            return self._micro % self.grad_accum_every == 0
        if self._micro % self.grad_accum_every == 0:
            self.step(grad_scale=1.0 / self.grad_accum_every)
            self.zero_grad()
            return True
        return False

    def step(self, grad_scale: float = 1.0) -> None:
        """One optimizer update from the (accumulated) flat grad buffer."""
        if self.space.flat_grad.is_cuda:
            from .ops.overlap import WgradQueue
            WgradQueue.sync()  # join side-stream wgrads before the update
        self.step_count += 1
        g = self.space.flat_grad
        if dispatch.use_hip(g, "adamw"):
            self._step_hip(grad_scale)
        else:
            self._step_eager(grad_scale)

    # -- implementations ----------------------------------------------------

    def _clip_coef(self, g32: torch.Tensor,
                   grad_scale: float = 1.0) -> torch.Tensor:
        if self.max_grad_norm is None:
            return torch.ones((), device=g32.device)
        if self.norm_sumsq_fn is not None:
            norm = self.norm_sumsq_fn().sqrt() * grad_scale
        else:
            norm = torch.linalg.vector_norm(g32)
        # optax clip_by_global_norm: g * max_norm / max(norm, max_norm)
        return self.max_grad_norm / torch.clamp_min(norm, self.max_grad_norm)

    def _adamw_update(self, g32: torch.Tensor, t=None) -> torch.Tensor:
        b1, b2 = self.betas
        if t is None:
            t = self.step_count
        self.exp_avg.mul_(b1).add_(g32, alpha=1 - b1)
        self.exp_avg_sq.mul_(b2).addcmul_(g32, g32, value=1 - b2)
        mhat = self.exp_avg / (1 - b1 ** t)
        vhat = self.exp_avg_sq / (1 - b2 ** t)
        upd = mhat / (vhat.sqrt() + self.eps)
        if self.weight_decay:
            upd = upd + self.weight_decay * self._decay_mask * self.master
        return self.lr * upd

    def _step_eager(self, grad_scale: float) -> None:
        g32 = self.space.flat_grad.float() * grad_scale
        coef = self._clip_coef(g32, grad_scale)
        # on GPU the bias-correction step count lives on device (as in
        # the fused kernel) so a captured eager step stays correct under
        # hipGraph replay (the FORCE_EAGER/PROGEN_EAGER_OPS bisect path —
        # a python step_count would be frozen at its capture-time value)
        t_dev = None
        if self.step_dev is not None and g32.is_cuda:
            self.step_dev += 1
            t_dev = self.step_dev.float()
        if self.max_grad_norm is not None:
            if g32.is_cuda and torch.cuda.is_current_stream_capturing():
                # capture-clean variant (no host branch) for the
                # FORCE_EAGER-in-graph bisect path: zero non-finite
                # updates instead of skipping the step
                g32 = (g32 * coef).nan_to_num(0.0, 0.0, 0.0)
            elif not (bool(torch.isfinite(coef)) and float(coef) > 0.0):
                # inf/NaN gradients: skip the step outright (GradScaler
                # semantics; parity with the fused kernel's step_ok guard)
                self.step_count -= 1
                if t_dev is not None:
                    self.step_dev -= 1
                return
            else:
                g32 *= coef
        self.master.sub_(self._adamw_update(g32, t_dev))
        if self.is_low_precision:
            self.space.flat.copy_(self.master.to(self.space.flat.dtype))

    def _apply_every_micro(self) -> None:
        """Reference apply_every semantics: moments advance per micro-batch,
        updates accumulate, applied every k micro-batches
        (reference: train.py:117-121,185-191). Deliberately NO
        non-finite skip here: optax apply_every has none either, and
        this mode exists for reference parity."""
        self.step_count += 1
        g32 = self.space.flat_grad.float()
        g32 *= self._clip_coef(g32)
        upd = self._adamw_update(g32)
        if self._update_acc is None:
            self._update_acc = torch.zeros_like(self.master)
        self._update_acc += upd
        self.zero_grad()
        if self._micro % self.grad_accum_every == 0:
            self.master.sub_(self._update_acc)
            self._update_acc.zero_()
            if self.is_low_precision:
                self.space.flat.copy_(self.master.to(self.space.flat.dtype))

    def _step_hip(self, grad_scale: float) -> None:
        C = dispatch.ext()
        if self.max_grad_norm is None:
            clip_coef = torch.ones(1, device=self.master.device)
        elif self.norm_sumsq_fn is not None:
            norm = self.norm_sumsq_fn().sqrt() * grad_scale
            clip_coef = self.max_grad_norm / torch.clamp_min(norm, self.max_grad_norm)
        else:
            # norm(scale*g) = scale*norm(g): fused sumsq kernel, no fp32
            # grad copy, clip coefficient stays on device (no host sync)
            norm = C.grad_sumsq(self.space.flat_grad).sqrt_() * grad_scale
            clip_coef = self.max_grad_norm / torch.clamp_min(norm, self.max_grad_norm)
        C.fused_adamw(
            self.master, self.space.flat, self.space.flat_grad,
            self.exp_avg, self.exp_avg_sq,
            self.chunk_starts, self.chunk_ends, self.chunk_decay,
            float(self.lr), float(self.betas[0]), float(self.betas[1]),
            float(self.eps), float(self.weight_decay), self.step_dev,
            float(grad_scale), clip_coef,
        )

    # -- checkpoint state ----------------------------------------------------

    def state_dict(self) -> Dict:
        if self.step_dev is not None:
            # graph replays advance only the device counter; sync back
            self.step_count = max(self.step_count, int(self.step_dev.item()))
        return {
            "step_count": self.step_count,
            "micro": self._micro,
            "master": self.master,
            "exp_avg": self.exp_avg,
            "exp_avg_sq": self.exp_avg_sq,
        }

    def load_state_dict(self, sd: Dict) -> None:
        self.step_count = int(sd["step_count"])
        if self.step_dev is not None:
            self.step_dev.fill_(self.step_count)
        self._micro = int(sd.get("micro", 0))
        with torch.no_grad():
            self.master.copy_(torch.as_tensor(sd["master"]).to(self.master.device))
            self.exp_avg.copy_(torch.as_tensor(sd["exp_avg"]).to(self.master.device))
            self.exp_avg_sq.copy_(torch.as_tensor(sd["exp_avg_sq"]).to(self.master.device))
            if self.is_low_precision:
                self.space.flat.copy_(self.master.to(self.space.flat.dtype))
