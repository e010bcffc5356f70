#!/bin/bash
# Round-2 GPU call 3: attn-bwd V4 (2-barrier restructure) A/B + PMC on
# wgrad2 and attn-bwd.
set -x
export TMPDIR=/tmp
cd /root/repo
mkdir -p gpurun_out
LOG=gpurun_out/r02_call3.log
{
echo "=== 1. attn bwd V0 vs V4, interleaved A/B, production grid ==="
for round in 1 2 3; do
  for b in abb_v0_n4 abb_v4_n4; do
    echo "== r$round $b"
    timeout 120 tools/abb_bin/$b 64 24 1024 256 100
  done
done
echo "=== wsz=512 sanity (rounds=2 path, default config wsz) ==="
timeout 120 tools/abb_bin/abb_v0_n4 32 16 1024 512 50
timeout 120 tools/abb_bin/abb_v4_n4 32 16 1024 512 50

echo "=== 2. PMC: wgrad2 V0 12288x1536 S=8 ==="
cd /tmp
timeout 300 rocprofv3 --pmc SQ_WAVE_CYCLES SQ_WAIT_ANY SQ_WAIT_INST_ANY SQ_ACTIVE_INST_ANY SQ_VALU_MFMA_BUSY_CYCLES SQ_LDS_BANK_CONFLICT SQ_LDS_IDX_ACTIVE \
  -d /root/repo/gpurun_out/pmc_wg2 -o wg2 --output-format csv -- /root/repo/tools/abb_bin/wgrad2_v0 12288 1536 65536 8 10
echo "=== 3. PMC: attn bwd V4 production grid ==="
timeout 300 rocprofv3 --pmc SQ_WAVE_CYCLES SQ_WAIT_ANY SQ_WAIT_INST_ANY SQ_ACTIVE_INST_ANY SQ_VALU_MFMA_BUSY_CYCLES SQ_LDS_BANK_CONFLICT SQ_LDS_IDX_ACTIVE \
  -d /root/repo/gpurun_out/pmc_abb4 -o abb4 --output-format csv -- /root/repo/tools/abb_bin/abb_v4_n4 64 24 1024 256 20
for f in /root/repo/gpurun_out/pmc_wg2/*_counter_collection.csv /root/repo/gpurun_out/pmc_abb4/*_counter_collection.csv; do
  echo "== $f"
  [ -f "$f" ] && python3 - "$f" <<'PYEOF'
import csv, sys, collections
agg = collections.defaultdict(float)
with open(sys.argv[1]) as fh:
    for row in csv.DictReader(fh):
        agg[(row.get('Kernel_Name','')[:34], row.get('Counter_Name',''))] += float(row.get('Counter_Value',0) or 0)
for (kn, cn), v in sorted(agg.items()):
    print(f"{kn:36s} {cn:26s} {v:.3e}")
PYEOF
done
} > $LOG 2>&1
tail -120 $LOG
