"""Fully-native multi-stage roundtrip (§8f.3): stage 1 writes shuffle files
through the engine (partial agg + 200-way hash partition + lz4 block data
file + u64 index), the harness plays Spark's block fetcher (slicing the data
file by the index), and stage 2 reads the raw blocks back through
IpcReaderExec into a Final agg — results must equal the oracle."""
import numpy as np
import pytest

import blaze_amd
from blaze_amd import plan
from oracle import pywrap as oracle

pytestmark = pytest.mark.gpu


def test_shuffle_write_then_ipc_read_roundtrip(tmp_path):
    P = 200
    rng = np.random.default_rng(21)
    n = 300_000
    keys = rng.integers(0, 20_000, n).astype(np.int64)
    vals = rng.integers(0, 1_000_000, n).astype(np.float64)
    vv = rng.random(n) >= 0.001
    data_file = str(tmp_path / "s.data")
    index_file = str(tmp_path / "s.index")

    # stage 1: native shuffle write
    t = blaze_amd.Task(
        plan.plan_agg_shuffle(data_file, index_file, num_partitions=P),
        batches=[[(keys[i:i + 50_000], None),
                  (vals[i:i + 50_000], vv[i:i + 50_000])]
                 for i in range(0, n, 50_000)])
    assert t.run() == []
    t.finalize()

    # harness = Spark's shuffle reader: slice the data file per the index
    index = np.frombuffer(open(index_file, "rb").read(), dtype="<u8")
    blob = open(data_file, "rb").read()
    assert index[-1] == len(blob)
    segments = [blob[index[p]:index[p + 1]] for p in range(P)
                if index[p + 1] > index[p]]
    assert len(segments) > 1

    # stage 2: fully-native IpcReader -> Final agg
    t2 = blaze_amd.Task(plan.plan_ipc_final(), ipc_segments=segments)
    outs = t2.run()
    got_keys = np.concatenate([ob[0]["values"] for ob in outs])
    got_sums = np.concatenate([ob[1]["values"] for ob in outs])
    got_cnts = np.concatenate([ob[2]["values"] for ob in outs])
    t2.finalize()

    orc = oracle.Agg()
    for i in range(0, n, 50_000):
        orc.update(keys[i:i + 50_000], vals[i:i + 50_000],
                   val_valid=vv[i:i + 50_000])
    ref = orc.output()
    assert len(got_keys) == orc.num_groups
    gi = np.argsort(got_keys, kind="stable")
    oi = np.argsort(ref["keys"], kind="stable")
    np.testing.assert_array_equal(got_keys[gi], ref["keys"][oi])
    np.testing.assert_array_equal(got_sums[gi], ref["sums"][oi])
    np.testing.assert_array_equal(got_cnts[gi], ref["counts"][oi])
