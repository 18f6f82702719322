#!/bin/bash
# Round-2 final profiling pass on the GPU box: per-kernel time stats + HBM
# FETCH/WRITE PMC for the 1B bench at HEAD (guide-compliant: --pmc never
# combined with trace domains; FETCH and WRITE in separate passes), plus the
# config-3 1B scan with the thread-cap + mmap decode.
set -x
cd /tmp && export TMPDIR=/tmp
R=$GRAFT_REPO_ROOT
mkdir -p "$R/gpurun_out"
timeout 250 rocprofv3 --kernel-trace --stats -d /tmp/pk -o stats -- \
  python "$R/bench.py" --steps 2 --warmup 1 --skip-cpu-baseline \
  > /dev/null 2>/tmp/pk.err
DB=$(find /tmp/pk -name "*results.db" | head -1)
python "$R/tools/prof_summary.py" "$DB" "$R/gpurun_out/r02_final_kernel_stats.md"
timeout 250 rocprofv3 --pmc FETCH_SIZE --output-format csv -d /tmp/pf -o f -- \
  python "$R/bench.py" --steps 1 --warmup 1 --skip-cpu-baseline \
  > /dev/null 2>/tmp/pf.err
timeout 250 rocprofv3 --pmc WRITE_SIZE --output-format csv -d /tmp/pw -o w -- \
  python "$R/bench.py" --steps 1 --warmup 1 --skip-cpu-baseline \
  > /dev/null 2>/tmp/pw.err
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
ft, fc = agg("/tmp/pf/**/*.csv", "FETCH_SIZE")
wt, wc = agg("/tmp/pw/**/*.csv", "WRITE_SIZE")
out = {}
for k in set(ft) | set(wt):
    if "auron" not in k:
        continue
    out[k] = {"fetch_kb_total": round(ft.get(k, 0), 1),
              "write_kb_total": round(wt.get(k, 0), 1),
              "dispatches": max(fc.get(k, 0), wc.get(k, 0))}
with open(R + "/gpurun_out/r02_final_pmc.json", "w") as f:
    json.dump(out, f, indent=1, sort_keys=True)
print("pmc json written")
PY
timeout 600 python "$R/tools/bench_scan.py" --rows 1000000000 --steps 2 \
  --warmup 1 > "$R/gpurun_out/scan1b_final.json" 2>/tmp/sc.err
tail -2 /tmp/sc.err
