#!/bin/bash
# Round-3 triage for the graphed-replay corruption
# (profiles/r02_graphed_nan_investigation.md). Three stages, each a
# separate gpurun-able chunk — comment out what the budget doesn't fit:
#
#  A. failure-odds baseline: the bisect sweep 3x (probabilistic ~25%/replay)
#  B. serializer discriminator: same sweep under AMD_SERIALIZE_KERNEL=3 —
#     if serialization cures pure-replay corruption, this is the same
#     runtime bug class as the r1 eager-interleave corruption
#     (profiles/r01_graph_interleave_bug.md) and an ROCm report is due
#  C. clock evidence for the NaN boost: rocm-smi sampled during a NaN
#     (pre-guard unavailable now — use PROGEN_EAGER_OPS= to re-expose?
#     simplest: log clocks during bench eager vs a known-NaN replay loop)
set -x
cd /root/repo
mkdir -p gpurun_out
{
echo "=== A: bisect sweep x3 (guarded; look for DIVERGED/skips) ==="
for i in 1 2 3; do
  PYTHONPATH=/root/repo timeout 600 python tools/r02_nan_probe3.py
done
echo "=== B: sweep under AMD_SERIALIZE_KERNEL=3 ==="
AMD_SERIALIZE_KERNEL=3 PYTHONPATH=/root/repo timeout 900 \
  python tools/r02_nan_probe3.py
echo "=== C: clocks during a replay loop (background sampler) ==="
( for i in $(seq 60); do
    rocm-smi --showgpuclocks --showpower --csv 2>/dev/null | tail -1
    sleep 1
  done ) > gpurun_out/r03_clocks_replay.csv &
SMI=$!
PYTHONPATH=/root/repo timeout 300 python - <<'PY'
import torch
from progen_amd import ProGenBase, ProGenConfig
from progen_amd.optim import ProGenAdamW
from progen_amd.runtime import GraphedTrainStep
torch.manual_seed(21)
cfg = ProGenConfig(num_tokens=256, dim=1536, depth=36, heads=24,
                   dim_head=64, window_size=256, seq_len=1024,
                   global_mlp_depth=2)
m = ProGenBase(cfg).to(device="cuda:0", dtype=torch.bfloat16)
m.rotary_sin = m.rotary_sin.float(); m.rotary_cos = m.rotary_cos.float()
o = ProGenAdamW(m, lr=2e-4, weight_decay=1e-3, max_grad_norm=0.5)
g = GraphedTrainStep(m, o, None, 64, 1024, torch.device("cuda:0"))
d = torch.randint(1, 256, (64, 1025), device="cuda:0"); d[:, 0] = 0
for i in range(50):
    loss = g.run(d).item()
    if i % 10 == 0:
        print(f"replay {i} loss {loss:.4f}", flush=True)
PY
kill $SMI 2>/dev/null
} > gpurun_out/r03_replay_triage.log 2>&1
tail -120 gpurun_out/r03_replay_triage.log
