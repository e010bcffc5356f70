"""hipGraph-captured training step.

The profiled eager step on MI355X spends ~45% of wall time in host-side
dispatch gaps (~500 kernel launches per ProGen-1.2B step). Shapes are
fully static, so the training step is captured once into a hipGraph and
replayed per step — the graph is the compiled program, eager PyTorch is
only the tracer ("HIP streams and graphs instead of a tracing compiler").

The captured step is PURE-REPLAY ONLY and single-GPU only: the whole
iteration (zero-grad + forward + backward + grad-clip + fused AdamW,
with the Adam step counter on device) is one graph, and NOTHING else may
launch kernels between replays. Empirically on this ROCm 7.0 stack,
interleaving ANY eager kernel with replays — even a plain index_select —
poisons subsequent replays (NaN gradients or HSA aperture faults; see
profiles/r01_graph_interleave_bug.md for the isolation). Training loops
that validate/sample/checkpoint between steps must therefore run eager
(train.py defaults to --no-graph); bench.py replays exclusively and is
safe. DP (world>1) raises by default — its all-reduce would be an
eager kernel between replays. With PROGEN_GRAPH_DP=1 the WHOLE DP step
(including the RCCL all-reduce and the optimizer) is captured instead,
keeping the pure-replay rule: tools/rccl_graph_probe.py verified that
RCCL collectives capture and replay correctly on this stack (1-rank
group; multi-rank is unvalidated on this pool's 1-GPU boxes, hence the
opt-in).

Warmup runs on a side stream; optimizer/param state perturbed by warmup
and capture is snapshotted and restored.

KNOWN OPEN ISSUE (late r2): even pure back-to-back replays corrupt
gradients probabilistically (~25%/replay at ProGen-small scale) on this
stack — most likely the same runtime bug class as the interleave
corruption above. The fused AdamW's non-finite step skip keeps the
trajectory finite, and eager is the default everywhere; full
investigation and the bisect plan live in
profiles/r02_graphed_nan_investigation.md.
"""

from __future__ import annotations

import os
from typing import Callable, Optional

import torch
import torch.distributed as dist

from .optim import ProGenAdamW
from .parallel.ddp import DistributedTrainer
from .utils import compute_loss


class GraphedTrainStep:
    """Capture a full training step into a replayable hipGraph.

    data shape: (B, seq_len + 1) int64 on the training device.
    """

    def __init__(self, module: torch.nn.Module, optim: ProGenAdamW,
                 ddp: Optional[DistributedTrainer], batch: int, seq_len: int,
                 device: torch.device, warmup: int = 3,
                 loss_fn: Callable = compute_loss):
        self.module = module
        self.optim = optim
        self.ddp = ddp
        self.loss_fn = loss_fn
        self.world = ddp.world if ddp is not None else 1
        if self.world > 1 and os.environ.get("PROGEN_GRAPH_DP") != "1":
            raise RuntimeError(
                "GraphedTrainStep under DP requires PROGEN_GRAPH_DP=1 "
                "(captures the RCCL all-reduce inside the graph; see the "
                "module docstring) — default DP runs eager")
        self.static_data = torch.zeros(batch, seq_len + 1, dtype=torch.long,
                                       device=device)

        snap = {
            "master": optim.master.clone(),
            "exp_avg": optim.exp_avg.clone(),
            "exp_avg_sq": optim.exp_avg_sq.clone(),
            "flat": optim.space.flat.clone(),
            "step": optim.step_dev.clone() if optim.step_dev is not None else None,
        }

        side = torch.cuda.Stream()
        side.wait_stream(torch.cuda.current_stream())
        with torch.cuda.stream(side):
            for _ in range(warmup):
                self._inner()
        torch.cuda.current_stream().wait_stream(side)
        torch.cuda.synchronize()

        self.graph = torch.cuda.CUDAGraph()
        with torch.cuda.graph(self.graph):
            self.static_loss = self._inner()

        with torch.no_grad():
            optim.master.copy_(snap["master"])
            optim.exp_avg.copy_(snap["exp_avg"])
            optim.exp_avg_sq.copy_(snap["exp_avg_sq"])
            optim.space.flat.copy_(snap["flat"])
            if snap["step"] is not None:
                optim.step_dev.copy_(snap["step"])
            optim.space.flat_grad.zero_()
        torch.cuda.synchronize()

    def _fwd_bwd(self) -> torch.Tensor:
        self.optim.space.flat_grad.zero_()
        loss = self.loss_fn(self.module, self.static_data)
        loss.backward()
        from .ops.overlap import WgradQueue
        WgradQueue.sync()  # join side-stream wgrads inside the graph
        return loss.detach()

    def _inner(self) -> torch.Tensor:
        if self.world > 1:
            # whole DP step in-graph: comm is part of the replay
            assert self.ddp is not None
            with self.ddp.no_sync():
                loss = self._fwd_bwd()
            dist.all_reduce(self.optim.space.flat_grad, op=dist.ReduceOp.SUM)
            self.optim.space.flat_grad.div_(self.world)
        else:
            loss = self._fwd_bwd()
        self.optim.step()
        return loss

    def run(self, data: torch.Tensor) -> torch.Tensor:
        """Replay one training step; returns the (device) loss tensor —
        do not .item() it inside a timed region, and do not launch other
        GPU kernels between replays (see module docstring)."""
        self.static_data.copy_(data, non_blocking=True)
        self.graph.replay()
        return self.static_loss
