#!/bin/bash
# Runs on the gpurun box: fresh kernel stats + HBM PMC passes for the 1B
# bench at HEAD; writes cropped summaries into gpurun_out/.
set -x
cd /tmp && export TMPDIR=/tmp
R=$GRAFT_REPO_ROOT
mkdir -p "$R/gpurun_out"
# 1) per-kernel time stats
timeout 300 rocprofv3 --kernel-trace --stats -d /tmp/p1 -o stats -- \
  python "$R/bench.py" --steps 3 --warmup 1 > /tmp/b1.json 2>/tmp/b1.err
DB=$(find /tmp/p1 -name "*results.db" | head -1)
python "$R/tools/prof_summary.py" "$DB" "$R/gpurun_out/r01_final_kernel_stats.md"
# 2) HBM bytes: separate passes (TCC slots)
timeout 300 rocprofv3 --pmc FETCH_SIZE --output-format csv -d /tmp/p2 -o fetch -- \
  python "$R/bench.py" --steps 1 --warmup 1 > /dev/null 2>/tmp/b2.err
timeout 300 rocprofv3 --pmc WRITE_SIZE --output-format csv -d /tmp/p3 -o write -- \
  python "$R/bench.py" --steps 1 --warmup 1 > /dev/null 2>/tmp/b3.err
python - <<'PY'
import csv, glob, collections, json, os
R = os.environ["GRAFT_REPO_ROOT"]
def agg(pat, counter):
    tot = collections.Counter(); cnt = collections.Counter()
    for f in glob.glob(pat, recursive=True):
        with open(f) as fh:
            for row in csv.DictReader(fh):
                if row.get("Counter_Name") == counter:
                    k = row["Kernel_Name"].split("(")[0]
                    tot[k] += float(row["Counter_Value"])
                    cnt[k] += 1
    return tot, cnt
import subprocess
subprocess.run("find /tmp/p2 /tmp/p3 -type f | head -20", shell=True)
ft, fc = agg("/tmp/p2/**/*.csv", "FETCH_SIZE")
wt, wc = agg("/tmp/p3/**/*.csv", "WRITE_SIZE")
out = {}
for k in set(ft) | set(wt):
    out[k] = {"fetch_mb_total": round(ft.get(k,0)/1e6,1), "dispatches": fc.get(k,0) or wc.get(k,0),
              "write_mb_total": round(wt.get(k,0)/1e6,1)}
with open(R + "/gpurun_out/r01_final_pmc.json", "w") as f:
    json.dump(out, f, indent=1, sort_keys=True)
print(json.dumps({k:v for k,v in sorted(out.items(), key=lambda kv: -kv[1]["fetch_mb_total"])[:8]}, indent=1))
PY
tail -2 /tmp/b1.json
