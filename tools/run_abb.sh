#!/bin/bash
# interleaved A/B rounds (guide rule 24): 3 rounds x all variants
cd "$(dirname "$0")/.."
for round in 1 2 3; do
  for b in tools/abb_bin/abb_v0_n4 tools/abb_bin/abb_v1_n4 tools/abb_bin/abb_v2_n4 \
           tools/abb_bin/abb_v3_n4 tools/abb_bin/abb_v0_n2 tools/abb_bin/abb_v0_n1; do
    echo "== round $round $(basename $b)"
    timeout 120 "$b" 32 16 1024 512 100
  done
done
