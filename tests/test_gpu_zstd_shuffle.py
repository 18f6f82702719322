"""Shuffle codec = zstd (ipc_compression.rs:189-196: the reference selects
lz4 or zstd via spark.io.compression.codec, zstd level via
SPARK_IO_COMPRESSION_ZSTD_LEVEL, default 1). Byte parity with the
oracle-constructed files at codec=zstd, and IpcReader read-back of the
zstd blocks."""
import numpy as np
import pytest

import blaze_amd
from blaze_amd import plan
from oracle import pywrap as oracle

pytestmark = pytest.mark.gpu

from test_gpu_parity import batches_of, gen_northstar, run_oracle  # noqa


def test_shuffle_write_files_zstd(tmp_path):
    P = 16
    keys, vals, vv = gen_northstar(80_000, nkeys=1500)
    data_file = str(tmp_path / "z.data")
    index_file = str(tmp_path / "z.index")
    t = blaze_amd.Task(
        plan.plan_agg_shuffle(data_file, index_file, num_partitions=P),
        batches=batches_of(keys, vals, vv),
        conf={"SPARK_IO_COMPRESSION_CODEC": "zstd"})
    assert t.run() == []
    t.finalize()

    orc = run_oracle(keys, vals, vv)
    g = orc.output()
    fz_data, fz_offs = orc.freeze()
    hashes = oracle.hash_cols([(g["keys"], None)])
    pids = oracle.partition_ids(hashes, P)
    order = np.argsort(pids, kind="stable")
    exp_index = []
    exp_bytes = bytearray()
    pos = 0
    for p in range(P):
        exp_index.append(pos)
        rows = order[pids[order] == p]
        if len(rows) == 0:
            continue
        w = oracle.IpcWriter(codec=1)
        for beg in range(0, len(rows), 10_000):
            chunk = rows[beg:beg + 10_000]
            kcol = g["keys"][chunk]
            lens = (fz_offs[chunk + 1] - fz_offs[chunk])
            offs = np.concatenate([[0], np.cumsum(lens)]).astype(np.int64)
            bb = np.concatenate([fz_data[fz_offs[r]:fz_offs[r + 1]]
                                 for r in chunk]) if len(chunk) else \
                np.array([], np.uint8)
            payload = oracle.serde_batch(len(chunk), [
                ("prim", kcol), ("bytes", bb, offs)])
            w.write_payload(payload)
        w.finish_block()
        blob = w.bytes()
        exp_bytes += blob
        pos += len(blob)
    exp_index.append(pos)

    got_data = open(data_file, "rb").read()
    got_index = np.frombuffer(open(index_file, "rb").read(), dtype="<u8")
    np.testing.assert_array_equal(got_index, np.array(exp_index, np.uint64))
    assert got_data == bytes(exp_bytes)

    # read-back: IpcReader plan over the zstd block stream of partition 0
    seg = got_data[got_index[0]:got_index[1]]
    if len(seg):
        td = plan.plan_ipc_final_named(("sum", "count"))
        t2 = blaze_amd.Task(td, ipc_segments=[bytes(seg)],
                            conf={"SPARK_IO_COMPRESSION_CODEC": "zstd"})
        outs = t2.run()
        n0 = int((pids == 0).sum())
        assert sum(len(o[0]["values"]) for o in outs) == n0
        t2.finalize()
