#!/bin/bash
# PMC comparison of the v2 vs v3 scatter kernels on the 1B workload:
# HBM fetch/write traffic + SQ stall breakdown (guide-compliant: --pmc never
# combined with trace domains; FETCH and WRITE in separate passes).
set -x
cd /tmp && export TMPDIR=/tmp
R=$GRAFT_REPO_ROOT
mkdir -p "$R/gpurun_out"
run() { # name env counter
  timeout 280 env $2 rocprofv3 --pmc $3 --output-format csv -d /tmp/$1 -o $1 -- \
    python "$R/bench.py" --steps 1 --warmup 1 --skip-cpu-baseline \
    > /dev/null 2>/tmp/$1.err
}
run v3f AURON_AGG2_V3=1 FETCH_SIZE
run v3w AURON_AGG2_V3=1 WRITE_SIZE
run v2w AURON_AGG2_V3=0 WRITE_SIZE
run v3s AURON_AGG2_V3=1 "SQ_WAIT_ANY,SQ_WAIT_INST_ANY,SQ_ACTIVE_INST_ANY,SQ_BUSY_CYCLES"
run v2s AURON_AGG2_V3=0 "SQ_WAIT_ANY,SQ_WAIT_INST_ANY,SQ_ACTIVE_INST_ANY,SQ_BUSY_CYCLES"
python - <<'PY'
import csv, glob, collections, json, os
R = os.environ["GRAFT_REPO_ROOT"]
out = {}
for tag in ["v3f", "v3w", "v2w", "v3s", "v2s"]:
    agg = collections.defaultdict(collections.Counter)
    cnt = collections.Counter()
    for f in glob.glob(f"/tmp/{tag}/**/*.csv", recursive=True):
        with open(f) as fh:
            for row in csv.DictReader(fh):
                k = row.get("Kernel_Name", "").split("(")[0]
                if "agg" not in k and "gather" not in k:
                    continue
                agg[k][row["Counter_Name"]] += float(row["Counter_Value"])
                cnt[k] += 1
    out[tag] = {k: {"counters": dict(v), "rows": cnt[k]}
                for k, v in agg.items()}
with open(R + "/gpurun_out/r2_pmc_scatter.json", "w") as f:
    json.dump(out, f, indent=1, sort_keys=True)
print("wrote r2_pmc_scatter.json")
PY
tail -2 /tmp/v3f.err /tmp/v2w.err
