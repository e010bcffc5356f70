#!/bin/bash
# Round-2 GPU call 1: wgrad GEMM measurement + attn-bwd re-ablation at
# production occupancy (VERDICT items 1 and 2 groundwork).
set -x
export TMPDIR=/tmp
cd /root/repo
mkdir -p gpurun_out
LOG=gpurun_out/r02_call1.log
{
echo "=== 1. hipBLASLt wgrad baseline (4 real 1.2B shapes, K=65536) ==="
timeout 420 python tools/bench_wgrad.py

echo "=== 2. hand-written wgrad probes V0/V1, S sweep ==="
for MN in "4608 1536" "1536 1536" "12288 1536" "1536 6144"; do
  for v in 0 1; do
    timeout 240 tools/abb_bin/wgrad_v$v $MN 65536 1,2,4,8,16 30
  done
done

echo "=== 3. attn bwd ablation, PRODUCTION grid (B=64 H=24 N=1024 wsz=256 -> 6144 blocks) ==="
for round in 1 2; do
  for b in abb_v0_n4 abb_v1_n4 abb_v2_n4 abb_v3_n4; do
    echo "== r$round $b prod"
    timeout 120 tools/abb_bin/$b 64 24 1024 256 100
  done
done
echo "=== 3b. reference point B=32 (ladder comparison) ==="
timeout 120 tools/abb_bin/abb_v0_n4 32 24 1024 256 100

echo "=== 4. PMC counters on attn_bwd production grid ==="
cd /tmp
timeout 300 rocprofv3 --pmc SQ_WAVE_CYCLES SQ_WAIT_ANY SQ_WAIT_INST_ANY SQ_ACTIVE_INST_ANY SQ_VALU_MFMA_BUSY_CYCLES SQ_LDS_BANK_CONFLICT SQ_LDS_IDX_ACTIVE \
  -d /root/repo/gpurun_out/pmc_abb -o abb0 -- /root/repo/tools/abb_bin/abb_v0_n4 64 24 1024 256 20
ls /root/repo/gpurun_out/pmc_abb/ 2>/dev/null
for f in /root/repo/gpurun_out/pmc_abb/*/abb0_counter_collection.csv /root/repo/gpurun_out/pmc_abb/abb0_counter_collection.csv; do
  [ -f "$f" ] && python3 - "$f" <<'EOF'
import csv, sys, collections
agg = collections.defaultdict(float)
with open(sys.argv[1]) as fh:
    for row in csv.DictReader(fh):
        name = row.get('Kernel_Name', '')[:40]
        cname = row.get('Counter_Name', '')
        agg[(name, cname)] += float(row.get('Counter_Value', 0) or 0)
for (kn, cn), v in sorted(agg.items()):
    print(f"{kn:42s} {cn:28s} {v:.3e}")
EOF
done
} > $LOG 2>&1
tail -150 $LOG
