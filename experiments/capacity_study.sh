#!/bin/bash
# Capacity/quality study: tokens/s and training CE vs
# capacity_factor on the flagship config, fixed seeds/data stream.
# Run via gpurun; writes one JSON line per factor to gpurun_out/capacity_study.jsonl
set -u
OUT=gpurun_out/capacity_study.jsonl
: > "$OUT"
for CF in 1.0 1.25 1.5 2.0; do
  echo "=== capacity_factor $CF ===" >&2
  python bench.py --steps 200 --warmup 3 --capacity-factor "$CF" --loss-every 50 \
    2> "gpurun_out/capacity_cf${CF}.err" | tail -1 >> "$OUT"
done
cat "$OUT"
