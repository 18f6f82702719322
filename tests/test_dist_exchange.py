"""Multi-process (world_size=2, gloo on CPU) coverage of the bench's RCCL
exchange control flow: partial records sorted into dest-rank-major partition
order (the numpy restatement of auron_repartition_device, which is itself
GPU-parity-tested in test_gpu_device_output.py), exchanged with
all_to_all_single using the per-rank splits, merged — the result must equal
the single-process oracle aggregate. Pins the N>1 flow without a GPU."""
import os

import numpy as np
import pytest


def _worker(rank, world, q):
    os.environ["MASTER_ADDR"] = "127.0.0.1"
    os.environ["MASTER_PORT"] = "29511"
    import torch
    import torch.distributed as dist

    from oracle import pywrap as oracle

    dist.init_process_group("gloo", rank=rank, world_size=world)
    try:
        # per-rank slice of a deterministic workload
        rng = np.random.default_rng(123 + rank)
        n = 20_000
        keys = rng.integers(0, 500, n).astype(np.int64)
        vals = rng.integers(0, 1000, n).astype(np.float64)
        part = oracle.Agg()
        part.update(keys, vals)
        g = part.output()
        data, offs64 = part.freeze()
        lens = (offs64[1:] - offs64[:-1]).astype(np.int32)

        # dest-rank-major partition-ordered sort + gather (the
        # auron_repartition_device contract, restated on host)
        hashes = oracle.hash_cols([(g["keys"], None)])
        pids = oracle.partition_ids(hashes, 200)
        dest = (pids % world).astype(np.int64)
        order = np.argsort(dest * 200 + pids, kind="stable")
        keys_s = g["keys"][order]
        lens_s = lens[order]
        offs = np.concatenate([[0], np.cumsum(lens)]).astype(np.int64)
        out_base = np.concatenate([[0], np.cumsum(lens_s)])[:-1]
        data_np = np.asarray(data, np.uint8)
        pos = np.repeat(offs[order] - out_base, lens_s) + \
            np.arange(int(lens_s.sum()), dtype=np.int64)
        data_s = data_np[pos]
        row_splits = np.bincount(dest, minlength=world).astype(np.int64)
        byte_splits = np.zeros(world, dtype=np.int64)
        np.add.at(byte_splits, dest, lens.astype(np.int64))

        def a2a(src_np, splits_out, splits_in):
            s = torch.from_numpy(np.ascontiguousarray(src_np))
            dst = torch.empty(int(splits_in.sum()), dtype=s.dtype)
            dist.all_to_all_single(dst, s, splits_in.tolist(),
                                   splits_out.tolist())
            return dst.numpy()

        t_rows = torch.tensor(row_splits)
        r_rows_t = torch.empty_like(t_rows)
        dist.all_to_all_single(r_rows_t, t_rows)
        in_rows = r_rows_t.numpy()
        t_bytes = torch.tensor(byte_splits)
        r_bytes_t = torch.empty_like(t_bytes)
        dist.all_to_all_single(r_bytes_t, t_bytes)
        in_bytes = r_bytes_t.numpy()
        rk = a2a(keys_s, row_splits, in_rows)
        rl = a2a(lens_s, row_splits, in_rows)
        rd = a2a(data_s, byte_splits, in_bytes)
        # every received key must belong to this rank
        rh = oracle.hash_cols([(rk, None)])
        rp = oracle.partition_ids(rh, 200)
        assert ((rp % world) == rank).all()

        # final merge of received records
        roffs = np.concatenate([[0], np.cumsum(rl)]).astype(np.int64)
        fin = oracle.Agg()
        fin.merge_frozen(rk, rd, roffs)
        out = fin.output()
        q.put((rank, out["keys"], out["sums"], out["counts"], keys, vals))
    finally:
        dist.destroy_process_group()


def test_exchange_two_ranks_gloo():
    import torch.multiprocessing as mp

    ctx = mp.get_context("spawn")
    q = ctx.Queue()
    procs = [ctx.Process(target=_worker, args=(r, 2, q)) for r in range(2)]
    for p in procs:
        p.start()
    results = [q.get(timeout=180) for _ in range(2)]
    for p in procs:
        p.join(timeout=60)
        assert p.exitcode == 0

    from oracle import pywrap as oracle

    # union of rank outputs == whole-data aggregate
    whole = oracle.Agg()
    for _, _, _, _, keys, vals in sorted(results):
        whole.update(keys, vals)
    ref = whole.output()
    got_keys = np.concatenate([r[1] for r in results])
    got_sums = np.concatenate([r[2] for r in results])
    got_cnts = np.concatenate([r[3] for r in results])
    assert len(got_keys) == whole.num_groups
    oi = np.argsort(ref["keys"], kind="stable")
    gi = np.argsort(got_keys, kind="stable")
    np.testing.assert_array_equal(got_keys[gi], ref["keys"][oi])
    np.testing.assert_array_equal(got_sums[gi], ref["sums"][oi])
    np.testing.assert_array_equal(got_cnts[gi], ref["counts"][oi])


if __name__ == "__main__":
    test_exchange_two_ranks_gloo()
    print("ok")
