#!/bin/bash
# Profile the flagship fwd/bwd kernels for one or more SKA_FLAGSHIP_V
# values.  Run on the GPU box:  bash benchmarks/diag/prof_flagship.sh 12 13
# Writes per-version kernel-stats summaries under gpurun_out/.
set -u
cd /tmp && export TMPDIR=/tmp && cd "$GRAFT_REPO_ROOT"
mkdir -p gpurun_out
for V in "$@"; do
  D=/tmp/prof_v$V
  rm -rf "$D"
  SKA_FLAGSHIP_V=$V rocprofv3 --kernel-trace --stats --output-format csv \
    -d "$D" -- python bench.py --steps 3 --warmup 1 \
    > gpurun_out/prof_v${V}.log 2>&1
  grep ms_per_step gpurun_out/prof_v${V}.log | head -1
  S=$(find "$D" -name "*kernel_stats.csv" | head -1)
  if [ -n "$S" ]; then
    cp "$S" gpurun_out/v${V}_kernel_stats.csv
    python3 - "$S" <<'EOF'
import csv, sys
rows = list(csv.DictReader(open(sys.argv[1])))
rows.sort(key=lambda r: -float(r.get("TotalDurationNs", r.get("DurationNs", 0))))
for r in rows[:8]:
    name = r["Name"].split("(")[0][:44]
    tot = float(r.get("TotalDurationNs", 0)) / 1e6
    n = r.get("Calls", r.get("Count", "?"))
    avg = float(r.get("AverageNs", 0)) / 1e3
    print(f"  {name:46s} n={n:>5s} tot={tot:9.2f}ms avg={avg:8.1f}us")
EOF
  fi
done
