#!/usr/bin/env bash
# rocprof capture automation for the flagship bench (run ON a GPU box).
#
#   bash tools/profile_bench.sh [outdir] [-- bench args...]
#
# Produces in <outdir> (default gpurun_out/):
#   <stamp>_kernel_stats.csv   per-kernel totals (rocprofv3 --stats)
#   <stamp>_summary.md         top-kernel table + the bench JSON line
#
# Counter collection (separate run — NEVER combined with trace domains,
# that combination is suspected of crashing boxes on this pool):
#   bash tools/profile_bench.sh outdir --pmc
set -euo pipefail

ROOT="$(cd "$(dirname "$0")/.." && pwd)"
OUT="${1:-$ROOT/gpurun_out}"
shift || true
MODE="trace"
if [ "${1:-}" = "--pmc" ]; then MODE="pmc"; shift; fi
if [ "${1:-}" = "--" ]; then shift; fi
BENCH_ARGS=("${@:---gpus 1 --steps 4 --warmup 4}")

STAMP="$(date +%H%M%S)"
WORK="$(mktemp -d /tmp/prof.XXXX)"
mkdir -p "$OUT"
cd /tmp && export TMPDIR=/tmp

if [ "$MODE" = "trace" ]; then
  rocprofv3 --kernel-trace --stats --output-format csv -d "$WORK" -o p \
    -- python "$ROOT/bench.py" ${BENCH_ARGS[@]} > "$WORK/bench.log" 2>&1 || true
  CSV="$(find "$WORK" -name "*kernel_stats.csv" | head -1)"
  [ -n "$CSV" ] && cp "$CSV" "$OUT/${STAMP}_kernel_stats.csv"
  {
    echo "# bench profile $STAMP"
    echo
    grep -a '"metric"' "$WORK/bench.log" | tail -1 || true
    echo
    echo "| total ms | calls | kernel |"
    echo "|---|---|---|"
    python3 - "$CSV" <<'EOF'
import csv, sys
rows = sorted(csv.DictReader(open(sys.argv[1])),
              key=lambda r: -float(r["TotalDurationNs"]))
for r in rows[:20]:
    print(f"| {float(r['TotalDurationNs'])/1e6:.2f} | {r['Calls']} | "
          f"`{r['Name'][:80]}` |")
EOF
  } > "$OUT/${STAMP}_summary.md"
  echo "wrote $OUT/${STAMP}_summary.md"
else
  # MFMA utilisation / LDS conflicts / HBM bytes; counters ONLY
  rocprofv3 --pmc SQ_VALU_MFMA_BUSY_CYCLES SQ_WAVE_CYCLES \
      SQ_LDS_BANK_CONFLICT -d "$WORK" -o pmc --output-format csv \
    -- python "$ROOT/bench.py" ${BENCH_ARGS[@]} > "$WORK/bench.log" 2>&1 || true
  find "$WORK" -name "*.csv" -size -5M -exec cp {} "$OUT/" \;
  echo "counter csvs copied to $OUT"
fi
