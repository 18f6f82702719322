"""Diagnostic (not a pytest): binary-search the stage1 hang dimensions.
Run variants under individual timeouts; print PASS/FAIL per variant."""
import os
import signal
import subprocess
import sys

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))

VARIANT_SRC = r"""
import sys, numpy as np
sys.path.insert(0, {repo!r})
import blaze_amd as ba
from blaze_amd import plan

mode = sys.argv[1]          # host|dev
n = int(sys.argv[2])
nkeys = int(sys.argv[3])
bs = int(sys.argv[4])
use_valid = sys.argv[5] == "1"

conf = {{"BATCH_SIZE": bs}}
if mode == "host":
    rng = np.random.default_rng(42)
    keys = rng.integers(0, nkeys, n).astype(np.int64)
    vals = rng.integers(0, 1000000, n).astype(np.float64)
    vv = (rng.random(n) >= 0.001) if use_valid else None
    t = ba.Task(plan.plan_partial_only(), batches=[[(keys, None), (vals, vv)]],
                conf=conf)
else:
    import torch
    g = torch.Generator(device="cuda:0"); g.manual_seed(42)
    keys = torch.randint(0, nkeys, (n,), dtype=torch.int64, device="cuda:0", generator=g)
    vals = torch.randint(0, 1000000, (n,), dtype=torch.int64, device="cuda:0", generator=g).to(torch.float64)
    cols = [{{"ptr": keys.data_ptr(), "len": n}}]
    nc = 0
    bitmap = None
    if use_valid:
        nulls = torch.rand(n, device="cuda:0", generator=g) < 0.001
        valid = ~nulls
        pad = (-n) % 8
        if pad: valid = torch.cat([valid, torch.ones(pad, dtype=torch.bool, device="cuda:0")])
        w = torch.tensor([1,2,4,8,16,32,64,128], dtype=torch.uint8, device="cuda:0")
        bitmap = (valid.view(-1,8).to(torch.uint8)*w).sum(dim=1).to(torch.uint8).contiguous()
        nc = int(nulls.sum())
        cols.append({{"ptr": vals.data_ptr(), "len": n, "validity_ptr": bitmap.data_ptr(), "null_count": nc}})
    else:
        cols.append({{"ptr": vals.data_ptr(), "len": n}})
    torch.cuda.synchronize()
    db = ba.DeviceBatch(cols)
    t = ba.Task(plan.plan_partial_only(), device_batches=[db.as_input()], conf=conf)
outs = t.run()
ng = sum(len(ob[0]["values"]) for ob in outs)
print("OK groups=", ng, flush=True)
t.finalize()
"""


def run_variant(name, mode, n, nkeys, bs, use_valid, timeout=60):
    src = VARIANT_SRC.format(repo=REPO)
    cmd = [sys.executable, "-c", src, mode, str(n), str(nkeys), str(bs),
           "1" if use_valid else "0"]
    try:
        r = subprocess.run(cmd, timeout=timeout, capture_output=True, text=True)
        ok = r.returncode == 0 and "OK groups=" in r.stdout
        print(f"{name}: {'PASS' if ok else 'FAIL rc=' + str(r.returncode)} "
              f"{r.stdout.strip()[:80]} {r.stderr.strip()[-200:] if not ok else ''}",
              flush=True)
    except subprocess.TimeoutExpired:
        print(f"{name}: TIMEOUT", flush=True)


if __name__ == "__main__":
    run_variant("host_10m_keys1m_bs1m", "host", 10_000_000, 1_000_000, 1 << 20, True)
    run_variant("dev_1m_keys1m_bs10k", "dev", 1_000_000, 1_000_000, 10_000, True)
    run_variant("dev_10m_keys1k_bs10k", "dev", 10_000_000, 1000, 10_000, True)
    run_variant("dev_10m_keys1m_bs10k_novalid", "dev", 10_000_000, 1_000_000, 10_000, False)
    run_variant("dev_10m_keys1m_bs1m", "dev", 10_000_000, 1_000_000, 1 << 20, True)
    run_variant("host_10m_keys1m_bs10k", "host", 10_000_000, 1_000_000, 10_000, True)
