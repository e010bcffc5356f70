"""ZeRO-1 optimizer-state sharding over the flat parameter space.

The reference has no optimizer sharding (its optax state is fully
replicated under pmap, reference: train.py:117-121). On MI355X the fp32
master + two Adam moments for ProGen-1.2B are ~14 GB per GPU; sharding
them across the DP group divides that by N at the cost of one
all-gather of the updated bf16 params per step.

Design (v1, opt-in via PROGEN_ZERO1=1 in train.py/bench.py):
  - gradients are still bucket-all-reduced by DistributedTrainer (every
    rank holds the full mean gradient — so the global-norm clip needs no
    extra communication);
  - each rank runs AdamW on its contiguous 1/N slice of the flat buffer
    (fp32 master/exp_avg/exp_avg_sq exist only for that slice);
  - the updated param slices are all-gathered back into every rank's
    flat buffer (bf16: 2 bytes/param on the wire).

A reduce-scatter of the gradients (instead of all-reduce) would also
halve the gradient communication — that needs bucket/shard alignment in
the DDP overlap path and RCCL reduce_scatter (gloo, used by the CPU
tests, lacks it), so it stays on the future list (TODO.md).
"""

from __future__ import annotations

import torch
import torch.distributed as dist

from typing import Optional

from ..optim import ProGenAdamW


class Zero1AdamW(ProGenAdamW):
    """ProGenAdamW with optimizer state sharded across the DP group.

    Must be constructed AFTER init_distributed; gradients must be
    all-reduced (mean) before ``step`` — exactly what DistributedTrainer
    already does."""

    def __init__(self, module: torch.nn.Module,
                 group: Optional["dist.ProcessGroup"] = None, **kwargs):
        """``group``: the data-parallel group to shard over (default:
        WORLD). Under a TP x DP mesh pass parallel/tp.py::dp_group() —
        the optimizer state shards across REPLICAS; TP shards are
        already rank-local."""
        super().__init__(module, **kwargs)
        assert dist.is_initialized() and dist.get_world_size(group) > 1, \
            "Zero1AdamW requires an initialized process group (world > 1)"
        assert self.accum_mode == "sum", \
            "Zero1AdamW supports accum_mode='sum' only (apply_every " \
            "advances moments from rank-local grads; with sharded state " \
            "the shapes don't even line up — use the replicated optimizer)"
        self.group = group
        self.world = dist.get_world_size(group)
        self.rank = dist.get_rank(group)
        n = self.space.numel
        self.shard = -(-n // self.world)          # padded shard size
        self.lo = min(self.rank * self.shard, n)
        self.hi = min(self.lo + self.shard, n)

        # replace the full-size fp32 state with the local slice
        sl = slice(self.lo, self.hi)
        self.master = self.master[sl].clone()
        self.exp_avg = torch.zeros_like(self.master)
        self.exp_avg_sq = torch.zeros_like(self.master)
        self._decay_mask = self._decay_mask[sl].clone()

        # shard-clipped chunk tables for the fused kernel: each full-
        # space chunk intersected with [lo, hi); the kernel indexes the
        # fp32 state at (i - shard_off) (ops/hip/adamw.hip)
        starts = self.chunk_starts.clamp(min=self.lo, max=self.hi)
        ends = self.chunk_ends.clamp(min=self.lo, max=self.hi)
        keep = ends > starts
        self.shard_chunk_starts = starts[keep].clone()
        self.shard_chunk_ends = ends[keep].clone()
        self.shard_chunk_decay = self.chunk_decay[keep].clone()

    def _gather_params(self) -> None:
        """All-gather the updated bf16 param slice into every rank's
        flat buffer (the one collective of the ZeRO-1 step)."""
        flat = self.space.flat
        buf = torch.zeros(self.shard, dtype=flat.dtype, device=flat.device)
        buf[: self.hi - self.lo] = flat[self.lo:self.hi] \
            if self.is_low_precision else self.master.to(flat.dtype)
        out = torch.empty(self.shard * self.world, dtype=flat.dtype,
                          device=flat.device)
        dist.all_gather_into_tensor(out, buf, group=self.group)
        flat.copy_(out[: flat.numel()])

    def _step_eager(self, grad_scale: float) -> None:
        g_full = self.space.flat_grad.float() * grad_scale
        coef = self._clip_coef(g_full, grad_scale)  # full-grad norm: no comm
        if self.max_grad_norm is not None and not (
                bool(torch.isfinite(coef)) and float(coef) > 0.0):
            # inf/NaN gradients: skip the whole step (GradScaler
            # semantics; the fused kernel's step_ok guard does the same
            # on the HIP path). The coef comes from the FULL grad norm,
            # identical on every rank, so all ranks skip together.
            self.step_count -= 1
            return
        g32 = g_full[self.lo:self.hi] * coef
        self.master.sub_(self._adamw_update(g32))
        if self.is_low_precision:
            self.space.flat[self.lo:self.hi].copy_(
                self.master.to(self.space.flat.dtype))
        else:
            self.space.flat[self.lo:self.hi].copy_(self.master)
        self._gather_params()

    def _step_hip(self, grad_scale: float) -> None:
        """Sharded fused update: the kernel touches only this rank's
        chunk slices (master/moments indexed at i - lo), writes the
        updated bf16 params in place at [lo, hi), then the slice is
        all-gathered (no more eager fallback)."""
        from ..ops import dispatch
        C = dispatch.ext()
        if self.max_grad_norm is None:
            clip_coef = torch.ones(1, device=self.master.device)
        else:
            norm = C.grad_sumsq(self.space.flat_grad).sqrt_() * grad_scale
            clip_coef = self.max_grad_norm / torch.clamp_min(
                norm, self.max_grad_norm)
        C.fused_adamw(
            self.master, self.space.flat, self.space.flat_grad,
            self.exp_avg, self.exp_avg_sq,
            self.shard_chunk_starts, self.shard_chunk_ends,
            self.shard_chunk_decay,
            float(self.lr), float(self.betas[0]), float(self.betas[1]),
            float(self.eps), float(self.weight_decay), self.step_dev,
            float(grad_scale), clip_coef, shard_off=self.lo,
        )
        self._gather_params()

    def resync_master(self) -> None:
        self.master.copy_(self.space.flat[self.lo:self.hi].float())

    # Checkpoints use the REPLICATED format: state_dict() all-gathers
    # the shards into full master/exp_avg/exp_avg_sq tensors, so a
    # ZeRO-1 checkpoint resumes at any world size, with or without
    # ZeRO-1 (the earlier per-rank format only saved rank
    # 0's shard and could not resume at all). The gather is a
    # COLLECTIVE: train.py must call state_dict() on every rank (see
    # state_dict_is_collective).
    state_dict_is_collective = True

    def _gather_full(self, t: torch.Tensor) -> torch.Tensor:
        buf = torch.zeros(self.shard, dtype=t.dtype, device=t.device)
        buf[: self.hi - self.lo] = t
        out = torch.empty(self.shard * self.world, dtype=t.dtype,
                          device=t.device)
        dist.all_gather_into_tensor(out, buf, group=self.group)
        return out[: self.space.numel].clone()

    def state_dict(self):
        if self.step_dev is not None:
            self.step_count = max(self.step_count, int(self.step_dev.item()))
        return {
            "step_count": self.step_count,
            "micro": self._micro,
            "master": self._gather_full(self.master),
            "exp_avg": self._gather_full(self.exp_avg),
            "exp_avg_sq": self._gather_full(self.exp_avg_sq),
        }

    def load_state_dict(self, sd) -> None:
        if "zero1" in sd:
            raise ValueError(
                "this checkpoint uses the removed per-rank ZeRO-1 shard "
                "format (round 1); it only ever held rank 0's shard and "
                "cannot be resumed — restart from a params-only load")
        self.step_count = int(sd["step_count"])
        if self.step_dev is not None:
            self.step_dev.fill_(self.step_count)
        self._micro = int(sd.get("micro", 0))
        with torch.no_grad():
            dev = self.master.device
            sl = slice(self.lo, self.hi)
            self.master.copy_(torch.as_tensor(sd["master"])[sl].to(dev))
            self.exp_avg.copy_(torch.as_tensor(sd["exp_avg"])[sl].to(dev))
            self.exp_avg_sq.copy_(
                torch.as_tensor(sd["exp_avg_sq"])[sl].to(dev))
            if self.is_low_precision:
                # the fp32 master is authoritative for THIS rank's slice;
                # other slices were already loaded from the checkpoint's
                # params (identical values after the bf16 rounding)
                self.space.flat[sl].copy_(
                    self.master.to(self.space.flat.dtype))
