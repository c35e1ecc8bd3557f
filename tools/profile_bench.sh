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
  # MFMA utilisation / LDS conflicts (pass 1) and HBM bytes (pass 2) —
  # counters ONLY, never combined with trace domains. The raw
  # per-dispatch CSV is large; aggregate to a per-kernel table here.
  # PMC collection serializes dispatch-by-dispatch: run the MINIMAL
  # per-kernel target set (tools/pmc_targets.py), never the full bench.
  PMC_TARGET="${PMC_TARGET:-$ROOT/tools/pmc_targets.py}"
  timeout 250 rocprofv3 --pmc SQ_VALU_MFMA_BUSY_CYCLES SQ_WAVE_CYCLES \
      SQ_LDS_BANK_CONFLICT -d "$WORK" -o pmc --output-format csv \
    -- python "$PMC_TARGET" > "$WORK/bench.log" 2>&1 || true
  timeout 250 rocprofv3 --pmc FETCH_SIZE WRITE_SIZE -d "$WORK" -o pmc2 \
      --output-format csv \
    -- python "$PMC_TARGET" > "$WORK/bench2.log" 2>&1 || true
  python3 - "$WORK" "$OUT/${STAMP}_pmc_summary.md" <<'EOF'
import csv, glob, sys
from collections import defaultdict
work, outp = sys.argv[1], sys.argv[2]
tables = {}
for f in glob.glob(work + "/*counter_collection.csv"):
    agg = defaultdict(lambda: defaultdict(float))
    calls = defaultdict(int)
    counted = defaultdict(set)
    for r in csv.DictReader(open(f)):
        name = r.get("Kernel_Name", "?")[:70]
        agg[name][r["Counter_Name"]] += float(r["Counter_Value"])
        key = (r.get("Dispatch_Id"),)
        if key not in counted[name]:
            counted[name].add(key)
    for n in agg:
        calls[n] = len(counted[n])
    tables[f] = (agg, calls)
with open(outp, "w") as o:
    o.write("# PMC per-kernel summary (whole bench run)\n\n")
    o.write("MFMA%% = SQ_VALU_MFMA_BUSY_CYCLES / SQ_WAVE_CYCLES;\n")
    o.write("HBM GB = (FETCH_SIZE*2 + WRITE_SIZE) KiB (FETCH halved by\n")
    o.write("rocprof on gfx950 wide reads -- see guide); conflicts are\n")
    o.write("SQ_LDS_BANK_CONFLICT totals.\n\n")
    for f, (agg, calls) in sorted(tables.items()):
        o.write(f"## {f.split('/')[-1]}\n\n")
        o.write("| kernel | calls | MFMA% | LDS-conf | wave-cyc | HBM GB |\n")
        o.write("|---|---|---|---|---|---|\n")
        rows = sorted(agg.items(),
                      key=lambda kv: -kv[1].get("SQ_WAVE_CYCLES",
                                                kv[1].get("FETCH_SIZE", 0)))
        for name, c in rows[:25]:
            wc = c.get("SQ_WAVE_CYCLES", 0)
            mf = c.get("SQ_VALU_MFMA_BUSY_CYCLES", 0)
            lc = c.get("SQ_LDS_BANK_CONFLICT", 0)
            hbm = (c.get("FETCH_SIZE", 0) * 2 + c.get("WRITE_SIZE", 0)) / 1e6
            o.write(f"| `{name}` | {calls[name]} | "
                    f"{100*mf/wc if wc else 0:.1f} | {lc:.3g} | "
                    f"{wc:.3g} | {hbm:.2f} |\n")
        o.write("\n")
print("wrote", outp)
EOF
  echo "pmc summary written to $OUT/${STAMP}_pmc_summary.md"
fi
