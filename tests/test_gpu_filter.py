"""GPU parity for FilterExec + ProjectionExec (a15 in-engine half):
FFIReader -> Filter(key < c) -> Project -> Agg chain vs the oracle applied to
the numpy-filtered input (filter semantics per filter_exec.rs:174-198:
predicate -> selection -> take; null compares false)."""
import numpy as np
import pytest

import blaze_amd
from blaze_amd import plan
from oracle import pywrap as oracle

pytestmark = pytest.mark.gpu


def test_filter_project_agg_parity():
    rng = np.random.default_rng(5)
    n = 500_000
    keys = rng.integers(0, 1_000_000, n).astype(np.int64)
    vals = rng.integers(0, 1_000_000, n).astype(np.float64)
    vv = rng.random(n) >= 0.01
    cutoff = 500_000
    t = blaze_amd.Task(
        plan.plan_filter_project_agg(cutoff=cutoff, cutoff_dtype="int64"),
        batches=[[(keys[i:i + 50_000], None),
                  (vals[i:i + 50_000], vv[i:i + 50_000])]
                 for i in range(0, n, 50_000)])
    outs = t.run()
    got_keys = np.concatenate([ob[0]["values"] for ob in outs])
    got_sums = np.concatenate([ob[1]["values"] for ob in outs])
    got_cnts = np.concatenate([ob[2]["values"] for ob in outs])
    t.finalize()

    sel = keys < cutoff
    orc = oracle.Agg()
    for i in range(0, n, 50_000):
        s = sel[i:i + 50_000]
        orc.update(keys[i:i + 50_000][s], vals[i:i + 50_000][s],
                   val_valid=vv[i:i + 50_000][s])
    ref = orc.output()
    np.testing.assert_array_equal(got_keys, ref["keys"])
    np.testing.assert_array_equal(got_sums, ref["sums"])
    np.testing.assert_array_equal(got_cnts, ref["counts"])
    assert (got_keys < cutoff).all()


def test_filter_selects_nothing_and_everything():
    keys = np.arange(1000, dtype=np.int64)
    vals = np.ones(1000)
    # nothing passes
    t = blaze_amd.Task(plan.plan_filter_project_agg(cutoff=0),
                       batches=[[(keys, None), (vals, None)]])
    assert t.run() == []
    t.finalize()
    # everything passes
    t = blaze_amd.Task(plan.plan_filter_project_agg(cutoff=10**9),
                       batches=[[(keys, None), (vals, None)]])
    outs = t.run()
    got = np.concatenate([ob[0]["values"] for ob in outs])
    np.testing.assert_array_equal(np.sort(got), keys)
    t.finalize()


def test_filter_float_literal():
    import pyarrow as pa  # noqa: F401

    rng = np.random.default_rng(9)
    n = 100_000
    keys = rng.integers(0, 1000, n).astype(np.int64)
    vals = rng.random(n) * 100
    # Filter(val > 50.0) over the f64 column
    reader = plan.ffi_reader(plan.northstar_input_fields(), "input0")
    filt = plan.filter_node(reader, [
        plan.binary_expr(plan.column("val", 1),
                         plan.literal(50.0, "float64"), "Gt")])
    partial = plan.agg(filt, [plan.column("key", 0)], plan.sum_count_aggs(1),
                       [plan.MODE_PARTIAL] * 2, ["key"], ["sum", "cnt"])
    final = plan.agg(partial, [plan.column("key", 0)], plan.sum_count_aggs(1),
                     [plan.MODE_FINAL] * 2, ["key"], ["sum", "cnt"])
    t = blaze_amd.Task(plan.task_definition(final),
                       batches=[[(keys, None), (vals, None)]])
    outs = t.run()
    got_keys = np.concatenate([ob[0]["values"] for ob in outs])
    got_cnts = np.concatenate([ob[2]["values"] for ob in outs])
    t.finalize()
    sel = vals > 50.0
    orc = oracle.Agg()
    orc.update(keys[sel], vals[sel])
    ref = orc.output()
    np.testing.assert_array_equal(got_keys, ref["keys"])
    np.testing.assert_array_equal(got_cnts, ref["counts"])
