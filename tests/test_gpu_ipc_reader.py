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


def test_new_agg_families_through_shuffle_and_ipc(tmp_path):
    """MIN/MAX/FIRST + i64-typed freeze parts survive the full multi-stage
    wire: partial agg -> a8 freeze -> batch_serde -> lz4 shuffle blocks ->
    data/index files -> IpcReader unfreeze -> final merge."""
    P = 50
    rng = np.random.default_rng(29)
    n = 150_000
    keys = rng.integers(0, 8_000, n).astype(np.int64)
    vals = rng.integers(-(1 << 45), 1 << 45, n).astype(np.int64)
    vv = rng.random(n) >= 0.2
    names = ["min", "max", "sum", "first", "first_ignores_null", "count"]
    data_file = str(tmp_path / "m.data")
    index_file = str(tmp_path / "m.index")

    t = blaze_amd.Task(
        plan.plan_agg_shuffle_named(data_file, index_file, names,
                                    num_partitions=P,
                                    val_dt=plan.DT_INT64),
        batches=[[(keys[i:i + 30_000], None),
                  (vals[i:i + 30_000], vv[i:i + 30_000])]
                 for i in range(0, n, 30_000)])
    assert t.run() == []
    t.finalize()

    index = np.frombuffer(open(index_file, "rb").read(), dtype="<u8")
    blob = open(data_file, "rb").read()
    segments = [blob[index[p]:index[p + 1]] for p in range(P)
                if index[p + 1] > index[p]]
    assert len(segments) > 1

    t2 = blaze_amd.Task(
        plan.plan_ipc_final_named(names, val_dt=plan.DT_INT64),
        ipc_segments=segments)
    outs = t2.run()
    got = {}
    got["key"] = np.concatenate([ob[0]["values"] for ob in outs])
    for j, nm in enumerate(names):
        v = np.concatenate([ob[1 + j]["values"] for ob in outs])
        val = np.concatenate(
            [ob[1 + j]["valid"] if ob[1 + j]["valid"] is not None
             else np.ones(len(ob[1 + j]["values"]), bool) for ob in outs])
        got[nm] = (v, val)
    t2.finalize()

    ok, mins, maxs = oracle.minmax_groups(keys, vals, vv)
    _, firsts, firsts_nn = oracle.first_groups(keys, vals, vv)
    _, sums, cnts = oracle.int_sum_groups(keys, vals, vv)
    k = got["key"]
    gi = np.argsort(k, kind="stable")
    oi = np.argsort(np.array(ok), kind="stable")
    np.testing.assert_array_equal(k[gi], np.array(ok)[oi])
    assert got["sum"][0].dtype == np.int64
    np.testing.assert_array_equal(got["sum"][0][gi][got["sum"][1][gi]],
                                  np.array(sums, np.int64)[oi][got["sum"][1][gi]])
    np.testing.assert_array_equal(got["count"][0][gi], np.array(cnts)[oi])
    exp_min = np.array([np.iinfo(np.int64).min if m is None else m
                        for m in mins], np.int64)
    exp_valid = np.array([m is not None for m in mins])
    np.testing.assert_array_equal(got["min"][1][gi], exp_valid[oi])
    np.testing.assert_array_equal(got["min"][0][gi][exp_valid[oi]],
                                  exp_min[oi][exp_valid[oi]])
    exp_max = np.array([0 if m is None else m for m in maxs], np.int64)
    np.testing.assert_array_equal(got["max"][0][gi][exp_valid[oi]],
                                  exp_max[oi][exp_valid[oi]])
    exp_f = np.array([np.iinfo(np.int64).min if f[1] is None else f[1]
                      for f in firsts], np.int64)
    exp_fv = np.array([f[1] is not None for f in firsts])
    np.testing.assert_array_equal(got["first"][1][gi], exp_fv[oi])
    np.testing.assert_array_equal(got["first"][0][gi][exp_fv[oi]],
                                  exp_f[oi][exp_fv[oi]])
    exp_n = np.array([np.iinfo(np.int64).min if f is None else f
                      for f in firsts_nn], np.int64)
    exp_nv = np.array([f is not None for f in firsts_nn])
    np.testing.assert_array_equal(got["first_ignores_null"][1][gi],
                                  exp_nv[oi])
    np.testing.assert_array_equal(
        got["first_ignores_null"][0][gi][exp_nv[oi]], exp_n[oi][exp_nv[oi]])
