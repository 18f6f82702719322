"""Diagnostic: repeat Task(partial-only) on the SAME device input N times in
one process — isolates the bench's second-step hang."""
import os
import sys

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import numpy as np  # noqa: E402

import blaze_amd as ba  # noqa: E402
from blaze_amd import plan  # noqa: E402


def main():
    import torch

    n = int(sys.argv[1]) if len(sys.argv) > 1 else 10_000_000
    reps = int(sys.argv[2]) if len(sys.argv) > 2 else 4
    g = torch.Generator(device="cuda:0")
    g.manual_seed(42)
    keys = torch.randint(0, 1_000_000, (n,), dtype=torch.int64,
                         device="cuda:0", generator=g)
    vals = torch.randint(0, 1_000_000, (n,), dtype=torch.int64,
                         device="cuda:0", generator=g).to(torch.float64)
    torch.cuda.synchronize()
    db = ba.DeviceBatch([
        {"ptr": keys.data_ptr(), "len": n},
        {"ptr": vals.data_ptr(), "len": n},
    ])
    for r in range(reps):
        import time

        t0 = time.perf_counter()
        t = ba.Task(plan.plan_partial_only(), device_batches=[db.as_input()],
                    conf={"BATCH_SIZE": 1 << 20})
        outs = t.run()
        ng = sum(len(ob[0]["values"]) for ob in outs)
        upd = t.metric("agg_update_ns")
        t.finalize()
        print(f"rep {r}: groups={ng} dt={time.perf_counter()-t0:.3f}s "
              f"update_ms={upd/1e6:.2f}", flush=True)


if __name__ == "__main__":
    main()
