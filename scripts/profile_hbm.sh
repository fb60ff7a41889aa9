#!/bin/bash
# Collect per-kernel time + HBM FETCH/WRITE PMC for the bench pipeline on the
# GPU box (run via gpurun). Per MI355X_MICROARCH.md §HBM: counters in their
# own --pmc passes (never combined with trace domains), FETCH_SIZE doubled
# (gfx950 reports half of wide coalesced reads). Writes summaries into
# gpurun_out/ for committing under profiles/.
set -e
cd /tmp && export TMPDIR=/tmp
REPO=${GRAFT_REPO_ROOT:-/root/repo}
BENCH="python $REPO/bench.py --steps 12 --warmup 3 --skip-cpu-baseline"
OUT=$REPO/gpurun_out
rm -rf /tmp/p1 /tmp/p2 /tmp/p3
rocprofv3 --kernel-trace --stats --output-format csv -d /tmp/p1 -o ks -- $BENCH || true
rocprofv3 --pmc FETCH_SIZE --output-format csv -d /tmp/p2 -o f -- $BENCH || true
rocprofv3 --pmc WRITE_SIZE --output-format csv -d /tmp/p3 -o w -- $BENCH || true
cp /tmp/p1/*kernel_stats.csv $OUT/r02_kernel_stats.csv 2>/dev/null || true
python3 - <<'EOF'
import csv, glob, json, os
repo = os.environ.get("GRAFT_REPO_ROOT", "/root/repo")
out = {}
for tag, d, corr in (("fetch", "/tmp/p2", 2.0), ("write", "/tmp/p3", 1.0)):
    agg = {}
    for f in glob.glob(d + "/**/*counter_collection.csv", recursive=True) + \
             glob.glob(d + "/*counter_collection.csv"):
        with open(f) as fh:
            for row in csv.DictReader(fh):
                k = row.get("Kernel_Name") or row.get("Kernel-Name") or ""
                v = float(row.get("Counter_Value") or 0)
                n, s = agg.get(k, (0, 0.0))
                agg[k] = (n + 1, s + v)
    out[tag] = {k: {"launches": n,
                    "kb_per_launch" + ("_x2corrected" if corr > 1 else ""):
                        round(s / n * corr, 1)}
                for k, (n, s) in agg.items() if n}
with open(os.path.join(repo, "gpurun_out", "r02_pmc.json"), "w") as fh:
    json.dump(out, fh, indent=1)
print("pmc summary written")
EOF
