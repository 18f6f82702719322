"""Oracle hash-agg semantics vs independent references (pyarrow group_by,
numpy) and the reference's fuzztest recipe (agg_exec.rs:714-843)."""
import numpy as np
import pytest

from oracle import pywrap as oracle


def _np_reference(keys, vals, val_valid=None):
    """Independent numpy restatement: SUM skips nulls (null if no valid
    values), COUNT counts non-null args, record order = first occurrence."""
    order = {}
    sums, cnts, have = {}, {}, {}
    for i, k in enumerate(keys.tolist()):
        if k not in order:
            order[k] = len(order)
            sums[k] = 0.0
            cnts[k] = 0
            have[k] = False
        if val_valid is None or val_valid[i]:
            sums[k] += float(vals[i])
            cnts[k] += 1
            have[k] = True
    ks = sorted(order, key=lambda k: order[k])
    return (np.array(ks, dtype=np.int64),
            np.array([sums[k] for k in ks]),
            np.array([have[k] for k in ks]),
            np.array([cnts[k] for k in ks], dtype=np.int64))


def test_agg_basic_insertion_order():
    keys = np.array([7, 8, 7, 9, 8, 7], dtype=np.int64)
    vals = np.array([1.5, 2.0, 2.5, -1.0, 0.0, 10.0])
    a = oracle.Agg()
    a.update(keys, vals)
    out = a.output()
    np.testing.assert_array_equal(out["keys"], [7, 8, 9])
    np.testing.assert_allclose(out["sums"], [14.0, 2.0, -1.0])
    np.testing.assert_array_equal(out["counts"], [3, 2, 1])
    assert out["sum_valid"].all()
    assert out["key_valid"].all()


def test_agg_null_vals():
    # fuzztest semantics: SUM null iff no non-null contribution; COUNT 0 then
    keys = np.array([1, 1, 2, 3], dtype=np.int64)
    vals = np.array([5.0, 7.0, 9.0, 11.0])
    val_valid = np.array([True, False, False, True])
    a = oracle.Agg()
    a.update(keys, vals, val_valid=val_valid)
    out = a.output()
    np.testing.assert_array_equal(out["keys"], [1, 2, 3])
    np.testing.assert_array_equal(out["counts"], [1, 0, 1])
    np.testing.assert_array_equal(out["sum_valid"], [True, False, True])
    np.testing.assert_allclose(out["sums"][[0, 2]], [5.0, 11.0])


def test_agg_null_key_is_a_group():
    keys = np.array([1, 0, 1], dtype=np.int64)
    key_valid = np.array([True, False, True])
    vals = np.array([1.0, 2.0, 3.0])
    a = oracle.Agg()
    a.update(keys, vals, key_valid=key_valid)
    out = a.output()
    assert a.num_groups == 2
    np.testing.assert_array_equal(out["key_valid"], [True, False])
    np.testing.assert_allclose(out["sums"], [4.0, 2.0])


def test_agg_fuzz_vs_numpy_and_pyarrow():
    rng = np.random.default_rng(42)
    n = 200_000
    keys = rng.integers(0, 5000, n).astype(np.int64)
    vals = (rng.integers(0, 1_000_000, n)).astype(np.float64)
    val_valid = rng.random(n) > 0.001  # ~0.1% nulls per BASELINE.md config
    a = oracle.Agg()
    # feed in batches of 10k like the reference (batch size conf)
    for i in range(0, n, 10_000):
        a.update(keys[i:i + 10_000], vals[i:i + 10_000],
                 val_valid=val_valid[i:i + 10_000])
    out = a.output()
    rk, rs, rhave, rc = _np_reference(keys, vals, val_valid)
    np.testing.assert_array_equal(out["keys"], rk)
    np.testing.assert_array_equal(out["counts"], rc)
    np.testing.assert_array_equal(out["sum_valid"], rhave)
    np.testing.assert_allclose(out["sums"], rs, rtol=1e-12)

    pa = pytest.importorskip("pyarrow")
    tbl = pa.table({"k": keys, "v": np.where(val_valid, vals, np.nan)})
    tbl = tbl.set_column(1, "v", pa.array(vals, mask=~val_valid))
    g = tbl.group_by("k").aggregate([("v", "sum"), ("v", "count")])
    g = g.sort_by("k")
    idx = np.argsort(out["keys"], kind="stable")
    np.testing.assert_array_equal(out["keys"][idx], g.column("k").to_numpy())
    np.testing.assert_allclose(out["sums"][idx],
                               g.column("v_sum").to_numpy(zero_copy_only=False),
                               rtol=1e-12)
    np.testing.assert_array_equal(out["counts"][idx],
                                  g.column("v_count").to_numpy())


def test_agg_freeze_unfreeze_roundtrip():
    # a8 Binary agg-buf: [u8 valid][8B LE sum]? ++ varint(count)
    keys = np.array([10, 20, 10, 30], dtype=np.int64)
    vals = np.array([1.25, 2.5, 3.25, 4.0])
    val_valid = np.array([True, True, True, False])
    a = oracle.Agg()
    a.update(keys, vals, val_valid=val_valid)
    data, offsets = a.freeze()
    # record 0: key 10, sum 4.5 valid, count 2
    rec0 = bytes(data[offsets[0]:offsets[1]])
    assert rec0[0] == 1
    assert np.frombuffer(rec0[1:9], dtype=np.float64)[0] == 4.5
    assert rec0[9:] == b"\x02"
    # record 2: key 30, sum invalid, count 0
    rec2 = bytes(data[offsets[2]:offsets[3]])
    assert rec2 == b"\x00\x00"

    # merge into a fresh (final-stage) agg
    out_p = a.output()
    b = oracle.Agg()
    b.merge_frozen(out_p["keys"], data, offsets)
    out = b.output()
    np.testing.assert_array_equal(out["keys"], [10, 20, 30])
    np.testing.assert_allclose(out["sums"][:2], [4.5, 2.5])
    np.testing.assert_array_equal(out["counts"], [2, 1, 0])
    np.testing.assert_array_equal(out["sum_valid"], [True, True, False])


def test_agg_partial_shards_merge_equals_single():
    """Partial agg on 4 shards + final merge == single-pass agg (the
    multi-GPU config-4 topology)."""
    rng = np.random.default_rng(7)
    n = 40_000
    keys = rng.integers(0, 997, n).astype(np.int64)
    vals = rng.random(n)
    whole = oracle.Agg()
    whole.update(keys, vals)
    ref = whole.output()

    final = oracle.Agg()
    for s in range(4):
        part = oracle.Agg()
        part.update(keys[s::4], vals[s::4])
        data, offsets = part.freeze()
        pk = part.output()["keys"]
        final.merge_frozen(pk, data, offsets)
    out = final.output()
    idx_r = np.argsort(ref["keys"], kind="stable")
    idx_o = np.argsort(out["keys"], kind="stable")
    np.testing.assert_array_equal(out["keys"][idx_o], ref["keys"][idx_r])
    np.testing.assert_array_equal(out["counts"][idx_o], ref["counts"][idx_r])
    np.testing.assert_allclose(out["sums"][idx_o], ref["sums"][idx_r], rtol=1e-12)


def test_agg_empty():
    a = oracle.Agg()
    assert a.num_groups == 0
    out = a.output()
    assert len(out["keys"]) == 0
    data, offsets = a.freeze()
    assert len(data) == 0 and offsets.tolist() == [0]


def test_agg_extreme_keys():
    keys = np.array([2**63 - 1, -2**63, 0, -1, 2**63 - 1], dtype=np.int64)
    vals = np.ones(5)
    a = oracle.Agg()
    a.update(keys, vals)
    out = a.output()
    np.testing.assert_array_equal(out["keys"], [2**63 - 1, -2**63, 0, -1])
    np.testing.assert_array_equal(out["counts"], [2, 1, 1, 1])
