#!/usr/bin/env python3
"""Config-3 measurement: Parquet scan + filter(key < c, ~10%
selectivity) + project + hash-agg on 1 GPU. Reports rows/s and effective
bytes/s over the compressed file, plus the engine's agg kernel metric.
Run under rocprofv3 for the per-kernel HBM evidence (profiles/).

Usage: python tools/bench_scan.py [--rows N] [--codec snappy] [--steps K]
"""
import argparse
import json
import os
import sys
import time

import numpy as np
# torch must be imported (and its bundled HIP runtime loaded) BEFORE
# libauron_hip.so, or torch's own HIP init fails with "No HIP GPUs"
import torch  # noqa: F401

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
sys.path.insert(0, REPO)


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--rows", type=int, default=50_000_000)
    ap.add_argument("--codec", default="snappy")
    ap.add_argument("--steps", type=int, default=2)
    ap.add_argument("--warmup", type=int, default=1)
    ap.add_argument("--cutoff", type=int, default=100_000)  # ~10% of 1M keys
    args = ap.parse_args()

    import pyarrow as pa
    import pyarrow.parquet as pq

    import blaze_amd as ba
    from blaze_amd import plan

    path = "/tmp/scan_bench.parquet"
    rng = np.random.default_rng(42)
    t0 = time.perf_counter()
    # chunked generation + ParquetWriter: the 1B-row file (~12 GB raw) never
    # needs the whole table in memory
    schema = pa.schema([("key", pa.int64()), ("val", pa.float64())])
    writer = pq.ParquetWriter(path, schema, compression=args.codec)
    sel_count = 0
    CH = 50_000_000
    for beg in range(0, args.rows, CH):
        m = min(CH, args.rows - beg)
        keys = rng.integers(0, 1_000_000, m).astype(np.int64)
        vals = rng.integers(0, 1_000_000, m).astype(np.float64)
        vv = rng.random(m) >= 0.001
        sel_count += int((keys < args.cutoff).sum())
        writer.write_table(
            pa.table({"key": pa.array(keys, pa.int64()),
                      "val": pa.array(vals, pa.float64(), mask=~vv)},
                     schema=schema),
            row_group_size=4_000_000)
    writer.close()
    size = os.path.getsize(path)
    print(f"[scan-bench] wrote {args.rows} rows, {size/1e6:.0f} MB "
          f"({args.codec}) in {time.perf_counter()-t0:.1f}s", file=sys.stderr)

    td = plan.plan_parquet_filter_agg([(path, size)], cutoff=args.cutoff)

    def step():
        t = ba.Task(td, conf={"BATCH_SIZE": 1 << 20})
        outs = t.run()
        groups = sum(len(ob[0]["values"]) for ob in outs)
        upd = t.metric("agg_update_ns")
        t.finalize()
        return groups, upd

    for _ in range(args.warmup):
        step()
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    groups = upd_ns = 0
    for _ in range(args.steps):
        groups, upd_ns = step()
    torch.cuda.synchronize()
    dt = (time.perf_counter() - t0) / args.steps

    sel = sel_count / args.rows
    out = {
        "metric": "rows/s parquet scan+filter+project+agg (config 3)",
        "value": round(args.rows / dt, 1),
        "unit": "rows/s",
        "n_gpus": 1,
        "steps": args.steps,
        "warmup": args.warmup,
        "ms_per_step": round(dt * 1000, 2),
        "higher_is_better": True,
        "scaling": "strong",
        "vs_baseline": None,
        "dtype": "f64",
        "data": "synthetic",
        "config": {
            "workload": "parquet_scan_filter_agg",
            "rows": args.rows,
            "codec": args.codec,
            "file_mb": round(size / 1e6, 1),
            "selectivity": round(sel, 4),
            "compressed_bytes_per_s": round(size / dt, 0),
            "groups_out": groups,
        },
        "notes": "scan decode: host walks footers/page headers + decodes the "
                 "dict-encoded prefix; snappy PLAIN pages decompress+decode ON "
                 "DEVICE (kernels_pq.hip wave-per-page), filter + agg on GPU",
    }
    print(json.dumps(out))


if __name__ == "__main__":
    main()
