"""COLLECT_LIST parity tests (collect.rs restated; oracle/pywrap.py
collect_groups). GPU-only."""
import numpy as np
import pytest

import blaze_amd
from blaze_amd import plan
from oracle import pywrap as oracle

pytestmark = pytest.mark.gpu


def _run(names, keys, vals, vv, val_dt=plan.DT_FLOAT64, batch=10_000,
         conf=None):
    batches = []
    for i in range(0, len(keys), batch):
        batches.append([(keys[i:i + batch], None),
                        (vals[i:i + batch],
                         vv[i:i + batch] if vv is not None else None)])
    t = blaze_amd.Task(plan.plan_partial_final_named(names, val_dt=val_dt),
                       batches=batches, conf=conf or {})
    outs = t.run()
    got = {"key": np.concatenate([ob[0]["values"] for ob in outs])}
    for j, nm in enumerate(names):
        got[nm] = [ob[1 + j] for ob in outs]
    t.finalize()
    return got


def _concat_lists(parts):
    """Concatenate list-column output chunks -> python list of np arrays."""
    out = []
    for p in parts:
        off = p["offsets"]
        for i in range(len(off) - 1):
            out.append(p["values"][off[i]:off[i + 1]])
    return out


def test_collect_list_basic():
    rng = np.random.default_rng(41)
    n = 50_000
    keys = rng.integers(0, 700, n).astype(np.int64)
    vals = rng.random(n) * 100
    vv = rng.random(n) >= 0.2
    got = _run(["collect_list", "count"], keys, vals, vv)
    ok, lists = oracle.collect_groups(keys, vals, vv)
    np.testing.assert_array_equal(got["key"], np.array(ok))
    gl = _concat_lists(got["collect_list"])
    assert len(gl) == len(lists)
    for g, e in zip(gl, lists):
        np.testing.assert_array_equal(g, np.array(e))


def test_collect_list_special_keys_and_int64():
    """Null keys, i64::MIN keys, i64 values, multi-batch arrival order."""
    rng = np.random.default_rng(43)
    n = 20_000
    keys = rng.integers(-5, 5, n).astype(np.int64)
    keys[keys == -5] = np.iinfo(np.int64).min
    kv = rng.random(n) >= 0.1  # null keys
    vals = rng.integers(-(1 << 50), 1 << 50, n).astype(np.int64)
    vv = rng.random(n) >= 0.3
    names = ["collect_list", "sum"]
    batches = []
    for i in range(0, n, 999):
        batches.append([(keys[i:i + 999], kv[i:i + 999]),
                        (vals[i:i + 999], vv[i:i + 999])])
    reader = plan.ffi_reader([plan.field("key", plan.DT_INT64, True),
                              plan.field("val", plan.DT_INT64, True)],
                             "input0")
    partial = plan.agg(reader, [plan.column("key", 0)],
                       plan.named_aggs(names, val_dt=plan.DT_INT64),
                       [plan.MODE_PARTIAL] * 2, ["key"], names)
    final = plan.agg(partial, [plan.column("key", 0)],
                     plan.named_aggs(names, val_dt=plan.DT_INT64),
                     [plan.MODE_FINAL] * 2, ["key"], names)
    t = blaze_amd.Task(plan.task_definition(final), batches=batches)
    outs = t.run()
    got_keys = np.concatenate([ob[0]["values"] for ob in outs])
    got_kv = np.concatenate(
        [ob[0]["valid"] if ob[0]["valid"] is not None
         else np.ones(len(ob[0]["values"]), bool) for ob in outs])
    gl = _concat_lists([ob[1] for ob in outs])
    t.finalize()
    ok, lists = oracle.collect_groups(keys, vals, vv, key_valid=kv)
    exp_keys = np.array([0 if k is None else k for k in ok], np.int64)
    exp_kv = np.array([k is not None for k in ok])
    np.testing.assert_array_equal(got_kv, exp_kv)
    np.testing.assert_array_equal(got_keys[exp_kv], exp_keys[exp_kv])
    assert len(gl) == len(lists)
    for g, e in zip(gl, lists):
        np.testing.assert_array_equal(g, np.array(e, np.int64))


def test_collect_list_grow_and_skipping():
    """Table grows (pool stores keys, not slots) and partial-skipping
    single-row pass-through records."""
    rng = np.random.default_rng(47)
    n = 60_000
    keys = np.arange(n, dtype=np.int64) % (n // 3)
    vals = rng.random(n)
    vv = np.ones(n, bool)
    got = _run(["collect_list"], keys, vals, vv,
               conf={"AURON_HIP_AGG_TABLE_SLOTS": 1 << 10})
    ok, lists = oracle.collect_groups(keys, vals, vv)
    np.testing.assert_array_equal(got["key"], np.array(ok))
    gl = _concat_lists(got["collect_list"])
    for g, e in zip(gl, lists):
        np.testing.assert_array_equal(g, np.array(e))

    # skipping: distinct keys flip pass-through after 20k rows
    keys2 = np.arange(n, dtype=np.int64)
    reader = plan.ffi_reader(plan.northstar_input_fields(), "input0")
    names = ["collect_list"]
    partial = plan.agg(reader, [plan.column("key", 0)],
                       plan.named_aggs(names),
                       [plan.MODE_PARTIAL], ["key"], names,
                       supports_partial_skipping=True)
    final = plan.agg(partial, [plan.column("key", 0)],
                     plan.named_aggs(names),
                     [plan.MODE_FINAL], ["key"], names)
    t = blaze_amd.Task(plan.task_definition(final),
                       batches=[[(keys2[i:i + 10_000], None),
                                 (vals[i:i + 10_000], vv[i:i + 10_000])]
                                for i in range(0, n, 10_000)])
    outs = t.run()
    k2 = np.concatenate([ob[0]["values"] for ob in outs])
    gl2 = _concat_lists([ob[1] for ob in outs])
    t.finalize()
    gi = np.argsort(k2, kind="stable")
    np.testing.assert_array_equal(k2[gi], keys2)
    flat = np.concatenate([gl2[int(i)] for i in gi])
    np.testing.assert_array_equal(flat, vals)


def test_collect_set_dedup():
    """COLLECT_SET: distinct values per group, FIRST-occurrence order
    (AccSet.append appends only novel values)."""
    rng = np.random.default_rng(53)
    n = 40_000
    keys = rng.integers(0, 300, n).astype(np.int64)
    vals = rng.integers(0, 25, n).astype(np.float64)  # heavy duplication
    vv = rng.random(n) >= 0.1
    got = _run(["collect_set", "count"], keys, vals, vv)
    ok, sets = oracle.collect_groups(keys, vals, vv, distinct=True)
    np.testing.assert_array_equal(got["key"], np.array(ok))
    gl = _concat_lists(got["collect_set"])
    assert len(gl) == len(sets)
    for g, e in zip(gl, sets):
        np.testing.assert_array_equal(g, np.array(e))


def test_collect_set_i64_null_keys_multistage():
    """SET over i64 values with null keys, chained partial->final (dedup
    happens at both stages; the frozen set wire re-merges losslessly)."""
    rng = np.random.default_rng(59)
    n = 15_000
    keys = rng.integers(0, 40, n).astype(np.int64)
    kv = rng.random(n) >= 0.15
    vals = rng.integers(-8, 8, n).astype(np.int64)
    vv = rng.random(n) >= 0.25
    names = ["collect_set"]
    batches = [[(keys[i:i + 777], kv[i:i + 777]),
                (vals[i:i + 777], vv[i:i + 777])]
               for i in range(0, n, 777)]
    reader = plan.ffi_reader([plan.field("key", plan.DT_INT64, True),
                              plan.field("val", plan.DT_INT64, True)],
                             "input0")
    partial = plan.agg(reader, [plan.column("key", 0)],
                       plan.named_aggs(names, val_dt=plan.DT_INT64),
                       [plan.MODE_PARTIAL], ["key"], names)
    final = plan.agg(partial, [plan.column("key", 0)],
                     plan.named_aggs(names, val_dt=plan.DT_INT64),
                     [plan.MODE_FINAL], ["key"], names)
    t = blaze_amd.Task(plan.task_definition(final), batches=batches)
    outs = t.run()
    got_keys = np.concatenate([ob[0]["values"] for ob in outs])
    got_kv = np.concatenate(
        [ob[0]["valid"] if ob[0]["valid"] is not None
         else np.ones(len(ob[0]["values"]), bool) for ob in outs])
    gl = _concat_lists([ob[1] for ob in outs])
    t.finalize()
    ok, sets = oracle.collect_groups(keys, vals, vv, distinct=True,
                                     key_valid=kv)
    exp_kv = np.array([k is not None for k in ok])
    np.testing.assert_array_equal(got_kv, exp_kv)
    np.testing.assert_array_equal(
        got_keys[exp_kv], np.array([k for k in ok if k is not None]))
    assert len(gl) == len(sets)
    for g, e in zip(gl, sets):
        np.testing.assert_array_equal(g, np.array(e, np.int64))


def test_collect_spill_drain():
    """Forced table spill with COLLECT_LIST: the pool drains into the frozen
    spill records (chained a8 wire) and the bucket-merged output reproduces
    every group's items in arrival order; null-key and i64::MIN-key groups
    stay resident with their items intact."""
    rng = np.random.default_rng(47)
    n = 300_000
    keys = rng.integers(0, 90_000, n).astype(np.int64)
    keys[::5000] = -2**63
    kv = np.ones(n, bool)
    kv[::4999] = False
    vals = (rng.random(n) * 100).round(3)
    vv = rng.random(n) >= 0.1
    conf = {"AURON_HIP_MEM_BUDGET": 1 << 20,
            "AURON_HIP_AGG_TABLE_SLOTS": 1 << 14}
    batches = [[(keys[i:i + 50_000], kv[i:i + 50_000]),
                (vals[i:i + 50_000], vv[i:i + 50_000])]
               for i in range(0, n, 50_000)]
    t = blaze_amd.Task(plan.plan_partial_final_named(["collect_list",
                                                      "count"]),
                       batches=batches, conf=conf)
    outs = t.run()
    assert t.metric("spill_count") > 0
    t.finalize()
    got_keys = np.concatenate([o[0]["values"] for o in outs])
    got_kv = np.concatenate(
        [o[0].get("valid") if o[0].get("valid") is not None
         else np.ones(len(o[0]["values"]), bool) for o in outs])
    gl = _concat_lists([o[1] for o in outs])
    got_cnt = np.concatenate([o[2]["values"] for o in outs])

    ok, lists = oracle.collect_groups(keys, vals, vv, key_valid=kv)
    assert len(got_keys) == len(ok)
    # spill output order is per-bucket: compare as key-indexed dict, but
    # each group's ITEM order must still be exact arrival order
    ref = {k: lst for k, lst in zip(ok, lists)}
    seen = set()
    for i in range(len(got_keys)):
        k = None if not got_kv[i] else int(got_keys[i])
        assert k in ref and k not in seen
        seen.add(k)
        np.testing.assert_allclose(gl[i], np.array(ref[k]), rtol=1e-12)
        assert got_cnt[i] == int(np.sum([1 for _ in ref[k]]))
    assert seen == set(ref.keys())


def test_collect_set_spill_drain():
    """Same with COLLECT_SET: dedup happens per spill freeze AND at the
    final merge; first-occurrence order survives the spill chain."""
    rng = np.random.default_rng(48)
    n = 250_000
    keys = rng.integers(0, 80_000, n).astype(np.int64)
    vals = rng.integers(0, 5, n).astype(np.float64)  # heavy duplication
    conf = {"AURON_HIP_MEM_BUDGET": 1 << 20,
            "AURON_HIP_AGG_TABLE_SLOTS": 1 << 14}
    batches = [[(keys[i:i + 50_000], None), (vals[i:i + 50_000], None)]
               for i in range(0, n, 50_000)]
    t = blaze_amd.Task(plan.plan_partial_final_named(["collect_set"]),
                       batches=batches, conf=conf)
    outs = t.run()
    assert t.metric("spill_count") > 0
    t.finalize()
    got_keys = np.concatenate([o[0]["values"] for o in outs])
    gl = _concat_lists([o[1] for o in outs])
    ok, sets_ = oracle.collect_groups(keys, vals, distinct=True)
    ref = {k: lst for k, lst in zip(ok, sets_)}
    assert len(got_keys) == len(ref)
    for i, k in enumerate(got_keys):
        np.testing.assert_allclose(gl[i], np.array(ref[int(k)]), rtol=1e-12)
