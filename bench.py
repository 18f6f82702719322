#!/usr/bin/env python3
"""bench.py — the north-star benchmark: rows/sec hash-agg + shuffle over
1B-row int64-key batches (BASELINE.json), on 1..8 MI355X GPUs.

A step = one pass of the hot path over one synthetic batch set already
resident in HBM: partial hash-agg (GPU) -> murmur3 200-way partition ->
RCCL all-to-all exchange of partial records (owner = partition % N) ->
final merge agg (GPU). At N=1 the exchange is a local pass-through.

Workload (config 4 of BASELINE.json, largest single-GPU config): 1B rows
total, key:int64 uniform [0,1e6), val:float64 integer-valued uniform [0,1e6)
with 0.1% nulls, seed 42, synthetic (no network). Total work is FIXED as N
grows (scaling: strong).

Run:  python bench.py [--gpus N] [--steps K] [--warmup W] [--rows R]
N>1 is launched by the driver via torch.distributed.run (one rank per GPU,
RCCL backend); rank/world read from the env.
"""
import argparse
import json
import os
import sys
import time

import numpy as np

REPO = os.path.dirname(os.path.abspath(__file__))
sys.path.insert(0, REPO)

TOTAL_ROWS_DEFAULT = 1_000_000_000
NUM_PARTITIONS = 200
NUM_KEYS = 1_000_000
NULL_FRAC = 0.001
SEED = 42

METRIC = "rows/sec hash-agg+shuffle, 1B-row int64-key Arrow batches, 1/2/4/8 GPU"


def make_device_input(torch, rows, rank, device):
    g = torch.Generator(device=device)
    g.manual_seed(SEED + 1000 * rank)
    keys = torch.randint(0, NUM_KEYS, (rows,), dtype=torch.int64, device=device,
                         generator=g)
    vals = torch.randint(0, 1_000_000, (rows,), dtype=torch.int64, device=device,
                         generator=g).to(torch.float64)
    nulls = torch.rand(rows, device=device, generator=g) < NULL_FRAC
    valid = ~nulls
    pad = (-rows) % 8
    if pad:
        valid = torch.cat([valid, torch.ones(pad, dtype=torch.bool,
                                             device=device)])
    weights = torch.tensor([1, 2, 4, 8, 16, 32, 64, 128], dtype=torch.uint8,
                           device=device)
    bitmap = (valid.view(-1, 8).to(torch.uint8) * weights).sum(
        dim=1).to(torch.uint8).contiguous()
    return keys, vals, bitmap, int(nulls.sum().item())


VERBOSE = os.environ.get("BENCH_VERBOSE", "0") == "1"


def _log(msg):
    if VERBOSE:
        print(f"[bench {time.strftime('%H:%M:%S')}] {msg}", file=sys.stderr,
              flush=True)


class _CudaBuf:
    """Minimal __cuda_array_interface__ wrapper so torch can view raw HBM
    pointers (the engine's device outputs) zero-copy."""

    def __init__(self, ptr, n, typestr):
        self.__cuda_array_interface__ = {
            "shape": (int(n),), "typestr": typestr,
            "data": (int(ptr), False), "version": 2}


def run_step(ba, plan, dev_input, world, rank, torch, dist, device, stats):
    """One step, fully device-resident: partial hash-agg -> device
    repartition (murmur3 pids + stable sort + a11 gather into dest-rank-major
    partition order, buffered_data.rs:284-351 analog) -> RCCL all-to-all of
    the partial records (N>1) -> final merge agg. The partial output batch
    never round-trips through host memory.

    BATCH_SIZE is the reference's own conf knob (conf.rs:32), raised so the
    ~1M-group partial output is ONE device batch. Table sized for the
    1M-group workload (2^22 slots; the engine still grows 4x on overflow)."""
    conf = {"BATCH_SIZE": 4 << 20, "AURON_HIP_AGG_TABLE_SLOTS": 1 << 22,
            "AURON_HIP_DEVICE_OUTPUT": 1}

    def mark(name, t0):
        dt = time.perf_counter() - t0
        stats.setdefault("stage_s", {}).setdefault(name, 0.0)
        stats["stage_s"][name] += dt
        _log(f"{name}: {dt*1000:.1f} ms")
        return time.perf_counter()

    tm = time.perf_counter()
    t = ba.Task(plan.plan_partial_only(), device_batches=[dev_input],
                conf=conf)
    tm = mark("s1_create", tm)
    t.run()
    tm = mark("s1_run", tm)
    stats["agg_update_ns"] += t.metric("agg_update_ns")
    stats["agg_update_rows"] += t.metric("agg_update_rows")
    assert len(t.device_outputs) == 1, "partial output must be one batch"
    ob = t.device_outputs[0]
    kc, bc = ob["cols"]
    n = ob["num_rows"]

    # stage 2: device repartition — 200-way murmur3 partition ids + stable
    # sort + gather into dest-rank-major partition order (the a11 gather)
    rp = ba.DeviceRepartition(n, kc["ptr"], bc["offsets_ptr"], bc["ptr"],
                              NUM_PARTITIONS, world,
                              key_validity_ptr=kc["validity_ptr"])
    t.finalize()  # repartition copied out of the partial task's buffers
    tm = mark("s2_repartition", tm)

    if world > 1:
        # RCCL all-to-all of (keys, validity bytes, lens, data bytes) by
        # destination rank; splits from the repartition counters
        send_rows = torch.tensor(rp.rank_rows, device=device)
        send_bytes = torch.tensor(rp.rank_bytes, device=device)
        recv_rows_t = torch.empty_like(send_rows)
        recv_bytes_t = torch.empty_like(send_bytes)
        dist.all_to_all_single(recv_rows_t, send_rows)
        dist.all_to_all_single(recv_bytes_t, send_bytes)
        recv_rows = recv_rows_t.cpu().tolist()
        recv_bytes = recv_bytes_t.cpu().tolist()

        keys_s = torch.as_tensor(_CudaBuf(rp.keys_ptr, n, "<i8"),
                                 device=device)
        offs_s = torch.as_tensor(_CudaBuf(rp.offsets_ptr, n + 1, "<i4"),
                                 device=device)
        data_s = torch.as_tensor(
            _CudaBuf(rp.data_ptr, max(int(offs_s[-1].item()), 1), "|u1"),
            device=device)
        lens_s = (offs_s[1:] - offs_s[:-1]).contiguous()
        # validity bitmap -> byte-per-row for the ragged exchange
        vbytes_s = None
        if rp.key_validity_ptr:
            bm = torch.as_tensor(
                _CudaBuf(rp.key_validity_ptr, (n + 7) // 8, "|u1"),
                device=device)
            bits = torch.tensor([1, 2, 4, 8, 16, 32, 64, 128],
                                dtype=torch.uint8, device=device)
            vbytes_s = ((bm.unsqueeze(1) & bits) != 0).reshape(-1)[:n] \
                .to(torch.uint8).contiguous()

        def a2a(src_t, out_n, splits_out, splits_in):
            dst = torch.empty(max(int(out_n), 1), dtype=src_t.dtype,
                              device=device)
            dist.all_to_all_single(dst[:int(out_n)], src_t,
                                   [int(x) for x in splits_in],
                                   [int(x) for x in splits_out])
            return dst

        nr = sum(recv_rows)
        keys_r = a2a(keys_s, nr, rp.rank_rows, recv_rows)
        lens_r = a2a(lens_s, nr, rp.rank_rows, recv_rows)
        data_r = a2a(data_s[:int(lens_s.sum().item())], sum(recv_bytes),
                     rp.rank_bytes, recv_bytes)
        vb_r = a2a(vbytes_s, nr, rp.rank_rows, recv_rows) \
            if vbytes_s is not None else None
        offs_r = torch.zeros(nr + 1, dtype=torch.int32, device=device)
        offs_r[1:] = torch.cumsum(lens_r.to(torch.int64), 0).to(torch.int32)
        null_count = 0
        valid_bm_r = None
        if vb_r is not None:
            null_count = int((vb_r == 0).sum().item())
            if null_count:
                pad = (-nr) % 8
                vb_pad = torch.cat([vb_r, torch.ones(pad, dtype=torch.uint8,
                                                     device=device)]) \
                    if pad else vb_r
                bits = torch.tensor([1, 2, 4, 8, 16, 32, 64, 128],
                                    dtype=torch.uint8, device=device)
                valid_bm_r = ((vb_pad.view(-1, 8) != 0).to(torch.uint8) *
                              bits).sum(dim=1).to(torch.uint8).contiguous()
        torch.cuda.synchronize()
        rp.free()
        keep = [keys_r, offs_r, data_r, valid_bm_r]
        fin_cols = [
            {"ptr": keys_r.data_ptr(), "len": nr,
             "validity_ptr": valid_bm_r.data_ptr() if valid_bm_r is not None
             else None, "null_count": null_count},
            {"ptr": data_r.data_ptr(), "offsets_ptr": offs_r.data_ptr(),
             "len": nr},
        ]
        tm = mark("s2_exchange", tm)
    else:
        keep = [rp]
        fin_cols = [
            {"ptr": rp.keys_ptr, "len": n,
             "validity_ptr": rp.key_validity_ptr, "null_count": -1},
            {"ptr": rp.data_ptr, "offsets_ptr": rp.offsets_ptr, "len": n},
        ]

    # stage 3: final merge agg of the partition-ordered records (device in)
    fin_in = ba.DeviceBatch(fin_cols, device_id=int(device.split(":")[1])) \
        .as_input()
    t2 = ba.Task(plan.plan_final_only(),
                 conf={"BATCH_SIZE": 4 << 20,
                       "AURON_HIP_AGG_TABLE_SLOTS": 1 << 22},
                 device_batches=[fin_in])
    t2.run()
    tm = mark("s3_final", tm)
    nfinal = t2.metric("num_groups")
    stats["num_groups"] = nfinal
    t2.finalize()
    if world == 1:
        keep[0].free()
    del keep
    return nfinal


def cpu_baseline_leg(total_rows):
    """The oracle (CPU restatement of the reference's operators) on ALL host
    cores in the reference's own map/reduce topology: per-core map tasks run
    partial agg + freeze + partition ids over a row slice and sort their
    frozen records into partition order (the shuffle write side); per-
    PARTITION reduce tasks then merge every map's slice of that partition
    (the shuffle read + final agg side). Threads, not processes: the
    oracle's ctypes calls and numpy release the GIL, and the frozen records
    move by slicing shared arrays — like the reference's in-memory shuffle
    on one box. 3 warm + 5 timed runs, median (BASELINE.md protocol);
    single-core figure reported alongside."""
    from concurrent.futures import ThreadPoolExecutor

    from oracle import pywrap as oracle

    ncores = os.cpu_count() or 1
    rng = np.random.default_rng(SEED)
    cal_n = 2_000_000
    k = rng.integers(0, NUM_KEYS, cal_n).astype(np.int64)
    v = rng.integers(0, 1_000_000, cal_n).astype(np.float64)
    m = rng.random(cal_n) >= NULL_FRAC
    t0 = time.perf_counter()
    a = oracle.Agg()
    a.update(k, v, val_valid=m)
    data, offsets = a.freeze()
    g = a.output()
    h = oracle.hash_cols([(g["keys"], None)])
    oracle.partition_ids(h, NUM_PARTITIONS)
    b = oracle.Agg()
    b.merge_frozen(g["keys"], data, offsets)
    single_rate = cal_n / (time.perf_counter() - t0)

    # sample: bounded to keep the default bench within minutes
    sample = int(min(2e8, max(cal_n * ncores, single_rate * ncores * 1.0)))
    per_w = max(1, sample // ncores)
    sample = per_w * ncores
    # pregenerate ONCE (generation is not part of the measured pass)
    keys = rng.integers(0, NUM_KEYS, sample).astype(np.int64)
    vals = rng.integers(0, 1_000_000, sample).astype(np.float64)
    vv = rng.random(sample) >= NULL_FRAC

    P = NUM_PARTITIONS

    def map_task(w):
        lo, hi = w * per_w, (w + 1) * per_w
        a = oracle.Agg()
        a.update(keys[lo:hi], vals[lo:hi], val_valid=vv[lo:hi])
        data, offs = a.freeze()
        g = a.output()
        h = oracle.hash_cols([(g["keys"], None)])
        pids = oracle.partition_ids(h, P)
        order = np.argsort(pids, kind="stable")
        keys_s = g["keys"][order]
        lens = (offs[1:] - offs[:-1]).astype(np.int64)
        lens_s = lens[order]
        offs_s = np.concatenate([[0], np.cumsum(lens_s)]).astype(np.int64)
        total = int(offs_s[-1])
        pos = np.repeat(offs[order][: len(lens_s)] - offs_s[:-1], lens_s) + \
            np.arange(total, dtype=np.int64)
        data_np = np.asarray(data, np.uint8)
        data_s = data_np[pos] if total else np.empty(0, np.uint8)
        bounds = np.searchsorted(pids[order], np.arange(P + 1))
        return keys_s, offs_s, data_s, bounds

    def reduce_task(p, maps):
        fin = oracle.Agg()
        for keys_s, offs_s, data_s, bounds in maps:
            b0, b1 = int(bounds[p]), int(bounds[p + 1])
            if b0 == b1:
                continue
            fin.merge_frozen(keys_s[b0:b1], data_s, offs_s[b0:b1 + 1])
        return fin.num_groups

    times = []
    with ThreadPoolExecutor(ncores) as ex:
        for it in range(8):  # 3 warm + 5 timed
            t0 = time.perf_counter()
            maps = list(ex.map(map_task, range(ncores)))
            list(ex.map(lambda p: reduce_task(p, maps), range(P)))
            dt = time.perf_counter() - t0
            if it >= 3:
                times.append(dt)
    times.sort()
    med = times[len(times) // 2]
    return {
        "value": sample / med,
        "unit": "rows/s",
        "cores": ncores,
        "kind": "port",
        "single_core_value": round(single_rate, 1),
        "sample": f"{sample} rows of the {total_rows}-row workload "
                  f"(map: per-core partial agg + freeze + partition sort; "
                  f"reduce: per-partition frozen merges across maps), "
                  f"{ncores} worker threads, median of 5 timed runs after "
                  f"3 warmups",
    }


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--gpus", type=int, default=1)
    ap.add_argument("--steps", type=int, default=3)
    ap.add_argument("--warmup", type=int, default=1)
    ap.add_argument("--rows", type=int, default=TOTAL_ROWS_DEFAULT,
                    help="TOTAL rows across all ranks (strong scaling)")
    ap.add_argument("--skip-cpu-baseline", action="store_true")
    args = ap.parse_args()

    import torch

    import blaze_amd as ba
    from blaze_amd import plan

    world = int(os.environ.get("WORLD_SIZE", "1"))
    rank = int(os.environ.get("RANK", "0"))
    local_rank = int(os.environ.get("LOCAL_RANK", "0"))
    if world == 1 and args.gpus > 1:
        print("N>1 must be launched via torch.distributed.run", file=sys.stderr)
        sys.exit(2)

    dist = None
    if world > 1:
        import torch.distributed as dist_mod

        dist = dist_mod
        dist.init_process_group("nccl")
    torch.cuda.set_device(local_rank)
    device = f"cuda:{local_rank}"

    my_rows = args.rows // world
    _log(f"generating {my_rows} rows on {device}")
    keys, vals, bitmap, null_count = make_device_input(torch, my_rows, rank,
                                                       device)
    torch.cuda.synchronize()
    _log("input ready")
    dev_input = ba.DeviceBatch([
        {"ptr": keys.data_ptr(), "len": my_rows},
        {"ptr": vals.data_ptr(), "len": my_rows,
         "validity_ptr": bitmap.data_ptr(), "null_count": null_count},
    ], device_id=local_rank).as_input()

    stats = {"agg_update_ns": 0, "agg_update_rows": 0, "num_groups": 0}
    for _ in range(args.warmup):
        run_step(ba, plan, dev_input, world, rank, torch, dist, device, stats)

    stats = {"agg_update_ns": 0, "agg_update_rows": 0, "num_groups": 0}
    if dist:
        dist.barrier()
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(args.steps):
        run_step(ba, plan, dev_input, world, rank, torch, dist, device, stats)
    torch.cuda.synchronize()
    if dist:
        dist.barrier()
    elapsed = time.perf_counter() - t0
    # max over ranks
    if dist:
        t = torch.tensor([elapsed], device=device)
        dist.all_reduce(t, op=dist.ReduceOp.MAX)
        elapsed = float(t.item())

    if rank == 0:
        print(f"[bench] stage seconds over {args.steps} steps: "
              f"{stats.get('stage_s', {})}", file=sys.stderr, flush=True)
        value = args.rows * args.steps / elapsed
        upd_ns = max(stats["agg_update_ns"], 1)
        upd_rows = stats["agg_update_rows"]
        achieved_gbs = upd_rows * 16.0 / upd_ns  # bytes/ns == GB/s
        roofline = {
            "bound": "hbm",
            "achieved": round(achieved_gbs, 1),
            "peak": 8000.0,
            "unit": "GB/s",
            "frac": round(achieved_gbs / 8000.0, 4),
            "traffic": None,  # PMC traffic in profiles/ (rocprofv3), see DESIGN.md
        }
        cpu = None if args.skip_cpu_baseline or world > 1 else \
            cpu_baseline_leg(args.rows)
        out = {
            "metric": METRIC,
            "value": round(value, 1),
            "unit": "rows/s",
            "n_gpus": world,
            "steps": args.steps,
            "warmup": args.warmup,
            "ms_per_step": round(elapsed / args.steps * 1000, 2),
            "higher_is_better": True,
            "scaling": "strong",
            "vs_baseline": None,
            "dtype": "f64",
            "data": "synthetic",
            "config": {
                "workload": "hashagg_shuffle_1b",
                "rows_total": args.rows,
                "distinct_keys": NUM_KEYS,
                "null_frac": NULL_FRAC,
                "num_partitions": NUM_PARTITIONS,
                "num_groups": stats["num_groups"],
                "exchange": "rccl_all_to_all" if world > 1 else "local",
            },
            "roofline": roofline,
            "cpu_baseline": cpu,
        }
        print(json.dumps(out))
    if dist:
        dist.destroy_process_group()


if __name__ == "__main__":
    main()
