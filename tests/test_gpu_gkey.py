"""Generalized grouping keys (a4 closure): Utf8 and multi-column key tuples
through the full Partial -> a8 agg-buf -> Final chain, parity vs the oracle
restatement. The engine's key ENCODING is its own (SURVEY.md §8c(i) permits
substitution); what must round-trip — and is checked bit-exact here — are the
grouping column VALUES, the group SET, the insertion ORDER and the agg
values."""
import numpy as np
import pytest

import blaze_amd
from blaze_amd import plan
from oracle import pywrap as oracle

pytestmark = pytest.mark.gpu


def _utf8_col(strings):
    data = b"".join(s.encode() for s in strings)
    offs = np.zeros(len(strings) + 1, dtype=np.int32)
    for i, s in enumerate(strings):
        offs[i + 1] = offs[i] + len(s.encode())
    return np.frombuffer(data, dtype=np.uint8).copy(), offs


def _run(td, cols, conf=None):
    t = blaze_amd.Task(td, batches=[cols], conf=conf or {})
    outs = t.run()
    t.finalize()
    return outs


def test_utf8_single_key():
    rng = np.random.default_rng(31)
    vocab = [f"key_{i:05d}" for i in range(800)] + ["", "x" * 40, "天地人"]
    n = 60_000
    idx = rng.integers(0, len(vocab), n)
    strs = [vocab[i] for i in idx]
    vals = rng.integers(0, 1000, n).astype(np.float64)
    vv = rng.random(n) >= 0.02
    data, offs = _utf8_col(strs)

    td = plan.plan_partial_final_gkey([("k", plan.DT_UTF8, False)],
                                      ("sum", "count", "min", "max"))
    outs = _run(td, [("binary", data, offs, None), (vals, vv)])
    keys = np.concatenate([o[0]["data"] for o in outs]).tobytes()
    koffs = outs[0][0]["offsets"]
    got_keys = [keys[koffs[i]:koffs[i + 1]].decode()
                for i in range(len(koffs) - 1)]
    got_sum = np.concatenate([o[1]["values"] for o in outs])
    got_cnt = np.concatenate([o[2]["values"] for o in outs])
    got_min = np.concatenate([o[3]["values"] for o in outs])
    got_max = np.concatenate([o[4]["values"] for o in outs])

    ok, sums, cnts, mins, maxs = oracle.gkey_agg_groups(
        [strs], vals, val_valid=vv)
    assert got_keys == [k[0] for k in ok]
    np.testing.assert_allclose(got_sum, sums, rtol=1e-12)
    np.testing.assert_array_equal(got_cnt, cnts)
    np.testing.assert_array_equal(got_min, mins)
    np.testing.assert_array_equal(got_max, maxs)


def test_two_col_i64_i32_keys_with_nulls():
    rng = np.random.default_rng(32)
    n = 80_000
    k1 = rng.integers(-50, 50, n).astype(np.int64)
    k2 = rng.integers(0, 7, n).astype(np.int32)
    k1v = rng.random(n) >= 0.05   # nulls in a KEY column group together
    vals = rng.integers(0, 100, n).astype(np.float64)

    td = plan.plan_partial_final_gkey(
        [("a", plan.DT_INT64, True), ("b", plan.DT_INT32, False)],
        ("sum", "count"))
    outs = _run(td, [(k1, k1v), (k2, None), (vals, None)])
    got_a = np.concatenate([o[0]["values"] for o in outs])
    av = np.concatenate([o[0].get("valid")
                         if o[0].get("valid") is not None
                         else np.ones(len(o[0]["values"]), bool)
                         for o in outs])
    got_b = np.concatenate([o[1]["values"] for o in outs])
    got_sum = np.concatenate([o[2]["values"] for o in outs])
    got_cnt = np.concatenate([o[3]["values"] for o in outs])

    ok, sums, cnts, _, _ = oracle.gkey_agg_groups(
        [k1.tolist(), k2.tolist()], vals, key_valids=[k1v, None])
    assert len(got_a) == len(ok)
    for i, kt in enumerate(ok):
        if kt[0] is None:
            assert not av[i]
        else:
            assert av[i] and got_a[i] == kt[0]
        assert got_b[i] == kt[1]
        assert got_cnt[i] == cnts[i]
        np.testing.assert_allclose(got_sum[i], sums[i], rtol=1e-12)


def test_mixed_utf8_i32_keys_with_growth():
    rng = np.random.default_rng(33)
    n = 50_000
    vocab = [f"s{i}" for i in range(3000)]
    idx = rng.integers(0, len(vocab), n)
    strs = [vocab[i] for i in idx]
    sv = rng.random(n) >= 0.03
    k2 = rng.integers(0, 5, n).astype(np.int32)
    vals = rng.integers(0, 10, n).astype(np.float64)
    data, offs = _utf8_col(strs)

    td = plan.plan_partial_final_gkey(
        [("s", plan.DT_UTF8, True), ("b", plan.DT_INT32, False)],
        ("sum", "count"))
    # tiny initial table: forces several 4x growths + gkey rebuilds
    outs = _run(td, [("binary", data, offs, sv), (k2, None), (vals, None)],
                conf={"AURON_HIP_AGG_TABLE_SLOTS": 1 << 10})
    koffs = np.concatenate(
        [o[0]["offsets"][:-1] + sum(int(p[0]["offsets"][-1])
                                    for p in outs[:j])
         for j, o in enumerate(outs)] +
        [[sum(int(p[0]["offsets"][-1]) for p in outs)]]).astype(np.int64)
    kdata = np.concatenate([o[0]["data"] for o in outs]).tobytes()
    sv_out = np.concatenate(
        [o[0].get("valid") if o[0].get("valid") is not None
         else np.ones(len(o[0]["offsets"]) - 1, bool) for o in outs])
    got_b = np.concatenate([o[1]["values"] for o in outs])
    got_sum = np.concatenate([o[2]["values"] for o in outs])
    got_cnt = np.concatenate([o[3]["values"] for o in outs])

    ok, sums, cnts, _, _ = oracle.gkey_agg_groups(
        [strs, k2.tolist()], vals, key_valids=[sv, None])
    assert len(got_b) == len(ok)
    for i, kt in enumerate(ok):
        s = kdata[koffs[i]:koffs[i + 1]].decode()
        if kt[0] is None:
            assert not sv_out[i]
        else:
            assert sv_out[i] and s == kt[0]
        assert got_b[i] == kt[1]
        assert got_cnt[i] == cnts[i]
        np.testing.assert_allclose(got_sum[i], sums[i], rtol=1e-12)


def test_gkey_partial_freeze_roundtrip_two_tasks():
    """Stage split like Spark: Partial task emits (key cols + a8 Binary),
    a SEPARATE Final task merges the frozen records (host hop)."""
    rng = np.random.default_rng(34)
    n = 30_000
    vocab = [f"g{i}" for i in range(500)]
    strs = [vocab[i] for i in rng.integers(0, len(vocab), n)]
    vals = rng.integers(0, 50, n).astype(np.float64)
    data, offs = _utf8_col(strs)

    td1 = plan.plan_partial_only_gkey([("k", plan.DT_UTF8, False)],
                                      ("sum", "count"))
    outs = _run(td1, [("binary", data, offs, None), (vals, None)])
    # feed the partial records into a Final-mode task
    fields = [plan.field("k", plan.DT_UTF8, True),
              plan.field("#9223372036854775807", plan.DT_BINARY, False)]
    reader = plan.ffi_reader(fields, "input0")
    aggs = [plan.agg_expr(plan.AGG_SUM, [plan.column("val", 1)],
                          plan.DT_FLOAT64),
            plan.agg_expr(plan.AGG_COUNT, [plan.column("val", 1)],
                          plan.DT_INT64)]
    fin = plan.agg(reader, [plan.column("k", 0)], aggs,
                   [plan.MODE_FINAL] * 2, ["k"], ["sum", "count"])
    td2 = plan.task_definition(fin)
    batches = []
    for o in outs:
        batches.append([("binary", o[0]["data"], o[0]["offsets"], None),
                        ("binary", o[1]["data"], o[1]["offsets"], None)])
    t2 = blaze_amd.Task(td2, batches=batches)
    out2 = t2.run()
    t2.finalize()
    koffs = out2[0][0]["offsets"]
    kdata = np.concatenate([o[0]["data"] for o in out2]).tobytes()
    got_keys = [kdata[koffs[i]:koffs[i + 1]].decode()
                for i in range(len(koffs) - 1)]
    got_sum = np.concatenate([o[1]["values"] for o in out2])
    got_cnt = np.concatenate([o[2]["values"] for o in out2])
    ok, sums, cnts, _, _ = oracle.gkey_agg_groups([strs], vals)
    assert got_keys == [k[0] for k in ok]
    np.testing.assert_allclose(got_sum, sums, rtol=1e-12)
    np.testing.assert_array_equal(got_cnt, cnts)


def test_utf8_shuffle_partition_ids(tmp_path):
    """Utf8 hash partitioning end-to-end: ShuffleWriter(hash(k), P) writes
    files whose per-partition contents match the oracle's murmur3(seed 42)
    byte-path routing (mur.rs:19-30, golden-pinned in hash_vectors.json)."""
    rng = np.random.default_rng(35)
    n = 5_000
    vocab = ["hello", "bar", "", "天地", "abcdefgh", "x"]
    strs = [vocab[i] for i in rng.integers(0, len(vocab), n)]
    data, offs = _utf8_col(strs)
    vals = rng.integers(0, 100, n).astype(np.float64)
    P = 8
    dfile = str(tmp_path / "g.data")
    ifile = str(tmp_path / "g.index")
    fields = [plan.field("k", plan.DT_UTF8, False),
              plan.field("v", plan.DT_FLOAT64, True)]
    td = plan.plan_shuffle_only(dfile, ifile, num_partitions=P,
                                fields=fields, hash_col=("k", 0))
    t = blaze_amd.Task(td, batches=[[("binary", data, offs, None),
                                     (vals, None)]])
    t.run()
    t.finalize()
    index = np.fromfile(ifile, dtype=np.uint64)
    assert len(index) == P + 1
    # oracle routing: murmur3 bytes seed 42 -> pmod
    h = np.array([oracle.murmur3(s.encode(), 42) for s in strs],
                 dtype=np.int64).astype(np.int32)
    pids = oracle.partition_ids(h, P)
    counts = np.bincount(pids, minlength=P)
    # partitions with zero rows have equal adjacent offsets; nonzero
    # partitions must be nonempty in the data file
    sizes = np.diff(index.astype(np.int64))
    for p in range(P):
        assert (sizes[p] > 0) == (counts[p] > 0), (p, sizes[p], counts[p])


def test_utf8_keys_at_scale():
    """1M rows / 120K distinct utf8 keys: pool growth + probe behavior at
    load, full partial->final chain vs the oracle reference."""
    rng = np.random.default_rng(36)
    n = 1_000_000
    nk = 120_000
    idx = rng.integers(0, nk, n)
    # varying lengths incl. empty and long keys
    vocab = [("k%06x" % i) * (1 + i % 3) for i in range(nk)]
    vocab[0] = ""
    strs = [vocab[i] for i in idx]
    vals = rng.integers(0, 1000, n).astype(np.float64)
    data, offs = _utf8_col(strs)

    td = plan.plan_partial_final_gkey([("k", plan.DT_UTF8, False)],
                                      ("sum", "count"))
    t = blaze_amd.Task(td, batches=[[("binary", data, offs, None),
                                     (vals, None)]],
                       conf={"BATCH_SIZE": 1 << 20,
                             "AURON_HIP_AGG_TABLE_SLOTS": 1 << 14})
    outs = t.run()
    t.finalize()
    koffs = outs[0][0]["offsets"]
    kdata = np.concatenate([o[0]["data"] for o in outs]).tobytes()
    got_sum = np.concatenate([o[1]["values"] for o in outs])
    got_cnt = np.concatenate([o[2]["values"] for o in outs])

    import collections
    rs = collections.defaultdict(float)
    rc = collections.Counter()
    order = []
    seen = set()
    for i in range(n):
        s = strs[i]
        if s not in seen:
            seen.add(s)
            order.append(s)
        rs[s] += vals[i]
        rc[s] += 1
    assert len(got_cnt) == len(order)
    got_keys = [kdata[koffs[i]:koffs[i + 1]].decode()
                for i in range(len(koffs) - 1)]
    assert got_keys == order[:len(got_keys)]
    for i, s in enumerate(got_keys):
        assert got_cnt[i] == rc[s]
        np.testing.assert_allclose(got_sum[i], rs[s], rtol=1e-9)
