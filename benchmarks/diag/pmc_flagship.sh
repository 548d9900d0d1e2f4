#!/bin/bash
# PMC counters for the flagship kernels at a given SKA_FLAGSHIP_V.
# Usage (GPU box): bash benchmarks/diag/pmc_flagship.sh 12
set -u
V=${1:-12}
cd /tmp && export TMPDIR=/tmp && cd "$GRAFT_REPO_ROOT"
mkdir -p gpurun_out
run_pass() {  # $1 = tag, rest = counters
  local TAG=$1; shift
  local D=/tmp/pmc_${V}_${TAG}
  rm -rf "$D"
  SKA_FLAGSHIP_V=$V rocprofv3 --pmc "$@" --output-format csv -d "$D" \
    -- python bench.py --steps 1 --warmup 1 \
    > gpurun_out/pmc_v${V}_${TAG}.log 2>&1 || tail -5 gpurun_out/pmc_v${V}_${TAG}.log
  local C
  C=$(find "$D" -name "*counter_collection.csv" | head -1)
  if [ -n "$C" ]; then
    cp "$C" gpurun_out/pmc_v${V}_${TAG}.csv
    python3 - "$C" <<'EOF'
import csv, sys, collections
agg = collections.defaultdict(lambda: collections.defaultdict(float))
n = collections.Counter()
for row in csv.DictReader(open(sys.argv[1])):
    k = row["Kernel_Name"].split("(")[0][:28]
    agg[k][row["Counter_Name"]] += float(row["Counter_Value"])
    n[(k, row["Counter_Name"])] += 1
for k, v in sorted(agg.items()):
    if any(s in k for s in ("fwd_d", "bwd_g", "scores")):
        parts = []
        for c, val in v.items():
            cnt = max(1, n[(k, c)])
            parts.append(f"{c}={val/cnt:.4g}")
        print(f"  {k:30s} " + " ".join(parts))
EOF
  else
    echo "  no counter csv for $TAG"
  fi
}
echo "== pass A (util/occupancy/fetch) v$V =="
run_pass a MfmaUtil VALUBusy OccupancyPercent FETCH_SIZE
echo "== pass B (write/wait/busy) v$V =="
run_pass b WRITE_SIZE SQ_WAIT_ANY SQ_BUSY_CYCLES SQ_INSTS_VMEM
