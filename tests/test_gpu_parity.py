"""GPU parity tests: the HIP engine (through the C ABI) vs the CPU oracle.

Parity bar (BASELINE.json north_star): bit-exact for integer keys/counts and
for f64 sums of integer-valued inputs (exact in f64 regardless of addition
order while < 2^53); 1e-6 relative for fractional f64 sums. Record ORDER is
also checked: the engine reproduces the reference's first-occurrence
insertion order (agg_hash_map.rs:77-168).
"""
import os

import numpy as np
import pytest

import blaze_amd
from blaze_amd import plan
from oracle import pywrap as oracle

pytestmark = pytest.mark.gpu


def gen_northstar(n, seed=42, nkeys=1_000_000, null_frac=0.001,
                  fractional=False):
    rng = np.random.default_rng(seed)
    keys = rng.integers(0, nkeys, n).astype(np.int64)
    vals = rng.integers(0, 1_000_000, n).astype(np.float64)
    if fractional:
        vals += rng.random(n)
    val_valid = rng.random(n) >= null_frac
    return keys, vals, val_valid


def batches_of(keys, vals, val_valid, batch=10_000):
    out = []
    for i in range(0, len(keys), batch):
        out.append([(keys[i:i + batch], None),
                    (vals[i:i + batch], val_valid[i:i + batch])])
    return out


def run_oracle(keys, vals, val_valid):
    a = oracle.Agg()
    for i in range(0, len(keys), 10_000):
        a.update(keys[i:i + 10_000], vals[i:i + 10_000],
                 val_valid=val_valid[i:i + 10_000] if val_valid is not None else None)
    return a


def engine_final(task_outputs):
    """Concatenate final-agg output batches into flat arrays."""
    keys, sums, sum_valid, cnts = [], [], [], []
    for ob in task_outputs:
        keys.append(ob[0]["values"])
        sums.append(ob[1]["values"])
        v = ob[1]["valid"]
        sum_valid.append(v if v is not None else np.ones(len(ob[1]["values"]), bool))
        cnts.append(ob[2]["values"])
    return (np.concatenate(keys), np.concatenate(sums),
            np.concatenate(sum_valid), np.concatenate(cnts))


def assert_agg_parity(outputs, orc, fractional):
    k, s, sv, c = engine_final(outputs)
    ref = orc.output()
    assert len(k) == orc.num_groups
    # ORDER parity: first-occurrence insertion order
    np.testing.assert_array_equal(k, ref["keys"])
    np.testing.assert_array_equal(c, ref["counts"])
    np.testing.assert_array_equal(sv, ref["sum_valid"])
    if fractional:
        np.testing.assert_allclose(s[sv], ref["sums"][ref["sum_valid"]],
                                   rtol=1e-6)
    else:
        np.testing.assert_array_equal(s[sv], ref["sums"][ref["sum_valid"]])


@pytest.mark.parametrize("n,nkeys", [(100_000, 5000), (2_000_000, 1_000_000)])
def test_partial_final_parity(n, nkeys):
    keys, vals, vv = gen_northstar(n, nkeys=nkeys)
    t = blaze_amd.Task(plan.plan_partial_final(),
                       batches=batches_of(keys, vals, vv))
    outputs = t.run()
    assert_agg_parity(outputs, run_oracle(keys, vals, vv), fractional=False)
    t.finalize()


def test_partial_final_parity_10m_config2():
    # BASELINE.json config 2: 10M rows, key in [0,1e6), val uniform [0,1e6)
    # 0.1% nulls, seed 42
    keys, vals, vv = gen_northstar(10_000_000)
    t = blaze_amd.Task(plan.plan_partial_final(),
                       batches=batches_of(keys, vals, vv, batch=100_000))
    outputs = t.run()
    assert_agg_parity(outputs, run_oracle(keys, vals, vv), fractional=False)
    t.finalize()


def test_fractional_sums_tolerance():
    keys, vals, vv = gen_northstar(500_000, nkeys=1000, fractional=True)
    t = blaze_amd.Task(plan.plan_partial_final(),
                       batches=batches_of(keys, vals, vv))
    assert_agg_parity(t.run(), run_oracle(keys, vals, vv), fractional=True)
    t.finalize()


def test_partial_binary_aggbuf_format():
    """a8 wire format: the engine's Binary agg-buf column must be
    byte-identical to the oracle's freeze output (integer-valued sums)."""
    keys, vals, vv = gen_northstar(50_000, nkeys=300)
    t = blaze_amd.Task(plan.plan_partial_only(),
                       batches=batches_of(keys, vals, vv))
    outputs = t.run()
    orc = run_oracle(keys, vals, vv)
    exp_data, exp_offsets = orc.freeze()
    exp_keys = orc.output()["keys"]
    got_keys = np.concatenate([ob[0]["values"] for ob in outputs])
    got_data = np.concatenate([ob[1]["data"] for ob in outputs])
    # rebase chunked offsets
    parts = []
    base = 0
    for ob in outputs:
        parts.append(ob[1]["offsets"][:-1] + base)
        base += ob[1]["offsets"][-1]
    got_offsets = np.concatenate(parts + [np.array([base])])
    np.testing.assert_array_equal(got_keys, exp_keys)
    np.testing.assert_array_equal(got_offsets, exp_offsets)
    np.testing.assert_array_equal(got_data, exp_data)
    t.finalize()


def test_final_only_merge():
    """Final-stage agg consuming frozen partial rows (stage-2 topology)."""
    keys, vals, vv = gen_northstar(100_000, nkeys=777)
    # two oracle partials = two 'map tasks'
    final_in = []
    merged = oracle.Agg()
    for half in (slice(0, 50_000), slice(50_000, None)):
        p = oracle.Agg()
        p.update(keys[half], vals[half], val_valid=vv[half])
        data, offs = p.freeze()
        pk = p.output()["keys"]
        final_in.append([(pk, None), ("binary", data, offs.astype(np.int32), None)])
        merged.merge_frozen(pk, data, offs)
    t = blaze_amd.Task(plan.plan_final_only(), batches=final_in)
    assert_agg_parity(t.run(), merged, fractional=False)
    t.finalize()


def test_edge_cases():
    # empty input
    t = blaze_amd.Task(plan.plan_partial_final(), batches=[])
    assert t.run() == []
    t.finalize()
    # single group, extreme keys, null-key group
    keys = np.array([2**63 - 1, -2**63, 0, -1, 2**63 - 1, -2**63], np.int64)
    vals = np.arange(6, dtype=np.float64)
    kv = np.array([True, True, True, True, False, True])
    t = blaze_amd.Task(
        plan.plan_partial_final(),
        batches=[[(keys, kv), (vals, None)]])
    outputs = t.run()
    k, s, sv, c = engine_final(outputs)
    orc = oracle.Agg()
    orc.update(keys, vals, key_valid=kv)
    ref = orc.output()
    np.testing.assert_array_equal(k[ref["key_valid"]], ref["keys"][ref["key_valid"]])
    np.testing.assert_array_equal(c, ref["counts"])
    np.testing.assert_array_equal(s, ref["sums"])
    t.finalize()


def test_partial_skipping_end_to_end():
    """supports_partial_skipping: high-cardinality input flips the partial
    stage to pass-through after 20k rows (agg_table.rs:109-120); the final
    stage must still produce the exact aggregate."""
    n = 60_000
    keys = np.arange(n, dtype=np.int64)  # all distinct -> ratio 1.0
    vals = np.ones(n)
    vv = np.ones(n, bool)
    t = blaze_amd.Task(plan.plan_partial_final(resource_id="input0"),
                       batches=batches_of(keys, vals, vv))
    outputs = t.run()
    k, s, sv, c = engine_final(outputs)
    assert len(k) == n
    np.testing.assert_array_equal(np.sort(k), keys)
    np.testing.assert_array_equal(c, np.ones(n, np.int64))
    np.testing.assert_array_equal(s, np.ones(n))
    t.finalize()


def test_table_grow():
    """Force the hash table to grow (4x rebuild at 3/4 load) several times via
    a tiny AURON_HIP_AGG_TABLE_SLOTS conf; results must be unaffected."""
    keys, vals, vv = gen_northstar(300_000, nkeys=100_000)
    t = blaze_amd.Task(plan.plan_partial_final(),
                       batches=batches_of(keys, vals, vv),
                       conf={"AURON_HIP_AGG_TABLE_SLOTS": 1024})
    outputs = t.run()
    assert_agg_parity(outputs, run_oracle(keys, vals, vv), fractional=False)
    t.finalize()


def test_shuffle_write_files(tmp_path):
    """Config-4 stage 1 on one GPU: partial agg + 200-way hash shuffle write.
    The data/index files must byte-match the oracle-constructed expectation
    (same murmur3 ids, same stable row order, same serde + lz4 framing)."""
    P = 200
    keys, vals, vv = gen_northstar(200_000, nkeys=3000)
    data_file = str(tmp_path / "shuffle.data")
    index_file = str(tmp_path / "shuffle.index")
    t = blaze_amd.Task(
        plan.plan_agg_shuffle(data_file, index_file, num_partitions=P),
        batches=batches_of(keys, vals, vv))
    out = t.run()
    assert out == []  # shuffle writer emits no batches
    t.finalize()

    # oracle expectation
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
        w = oracle.IpcWriter()
        for beg in range(0, len(rows), 10_000):
            chunk = rows[beg:beg + 10_000]
            kcol = g["keys"][chunk]
            lens = (fz_offs[chunk + 1] - fz_offs[chunk])
            offs = np.concatenate([[0], np.cumsum(lens)]).astype(np.int64)
            bb = np.concatenate([fz_data[fz_offs[r]:fz_offs[r + 1]]
                                 for r in chunk]) if len(chunk) else np.array([], np.uint8)
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


def test_avg_agg_layouts():
    """Generalized agg lists over one shared column: [AVG], [SUM,COUNT,AVG]
    (AVG freeze = sum ++ count, avg.rs:208-217; final = sum/count)."""
    keys, vals, vv = gen_northstar(100_000, nkeys=500)
    orc = run_oracle(keys, vals, vv)
    ref = orc.output()
    exp_avg = np.where(ref["counts"] > 0, ref["sums"] / np.maximum(ref["counts"], 1), 0)

    def avg_aggs():
        return [plan.agg_expr(plan.AGG_AVG, [plan.column("val", 1)],
                              plan.DT_FLOAT64)]

    def sca_aggs():
        return [plan.agg_expr(plan.AGG_SUM, [plan.column("val", 1)],
                              plan.DT_FLOAT64),
                plan.agg_expr(plan.AGG_COUNT, [plan.column("val", 1)],
                              plan.DT_INT64),
                plan.agg_expr(plan.AGG_AVG, [plan.column("val", 1)],
                              plan.DT_FLOAT64)]

    for aggs_fn, names in ((avg_aggs, ["avg"]),
                           (sca_aggs, ["sum", "cnt", "avg"])):
        reader = plan.ffi_reader(plan.northstar_input_fields(), "input0")
        partial = plan.agg(reader, [plan.column("key", 0)], aggs_fn(),
                           [plan.MODE_PARTIAL] * len(names), ["key"], names)
        final = plan.agg(partial, [plan.column("key", 0)], aggs_fn(),
                         [plan.MODE_FINAL] * len(names), ["key"], names)
        t = blaze_amd.Task(plan.task_definition(final),
                           batches=batches_of(keys, vals, vv))
        outs = t.run()
        got_keys = np.concatenate([ob[0]["values"] for ob in outs])
        np.testing.assert_array_equal(got_keys, ref["keys"])
        got_avg = np.concatenate([ob[len(names)]["values"] for ob in outs])
        np.testing.assert_allclose(got_avg[ref["counts"] > 0],
                                   exp_avg[ref["counts"] > 0], rtol=1e-13)
        if len(names) == 3:
            got_sums = np.concatenate([ob[1]["values"] for ob in outs])
            got_cnts = np.concatenate([ob[2]["values"] for ob in outs])
            np.testing.assert_array_equal(got_sums, ref["sums"])
            np.testing.assert_array_equal(got_cnts, ref["counts"])
        t.finalize()


def test_spill_to_host():
    """a9 analog: a tiny AURON_HIP_MEM_BUDGET forces the table to spill
    frozen records to host buckets mid-stream (agg_table.rs:540-588
    semantics); the bucket-merged output must still be the exact aggregate
    (order is per-bucket once spilling happens, like the reference)."""
    keys, vals, vv = gen_northstar(400_000, nkeys=120_000)
    conf = {"AURON_HIP_MEM_BUDGET": 1 << 20,       # ~16k-slot table cap
            "AURON_HIP_AGG_TABLE_SLOTS": 1 << 14}
    t = blaze_amd.Task(plan.plan_partial_final(),
                       batches=batches_of(keys, vals, vv), conf=conf)
    outputs = t.run()
    assert t.metric("spill_count") > 0
    k, s, sv, c = engine_final(outputs)
    ref = run_oracle(keys, vals, vv).output()
    assert len(k) == len(ref["keys"])
    gi = np.argsort(k, kind="stable")
    oi = np.argsort(ref["keys"], kind="stable")
    np.testing.assert_array_equal(k[gi], ref["keys"][oi])
    np.testing.assert_array_equal(c[gi], ref["counts"][oi])
    np.testing.assert_array_equal(s[gi], ref["sums"][oi])
    t.finalize()


def test_spill_preserves_special_groups():
    """Null-key and i64::MIN-key groups stay resident across spills and are
    emitted exactly once."""
    n = 120_000
    rng = np.random.default_rng(3)
    keys = rng.integers(0, 50_000, n).astype(np.int64)
    keys[::1000] = -2**63
    kv = np.ones(n, bool)
    kv[::997] = False
    vals = np.ones(n)
    conf = {"AURON_HIP_MEM_BUDGET": 1 << 20,
            "AURON_HIP_AGG_TABLE_SLOTS": 1 << 14}
    t = blaze_amd.Task(plan.plan_partial_final(),
                       batches=[[(keys, kv), (vals, None)]], conf=conf)
    outputs = t.run()
    assert t.metric("spill_count") > 0
    k, s, sv, c = engine_final(outputs)
    orc = oracle.Agg()
    orc.update(keys, vals, key_valid=kv)
    ref = orc.output()
    assert len(k) == orc.num_groups
    np.testing.assert_array_equal(np.sort(c), np.sort(ref["counts"]))
    assert c.sum() == n
    t.finalize()


def test_two_phase_with_special_keys():
    """Chunks >= 4M rows take the two-phase (radix+LDS) path; null keys and
    i64::MIN keys must route through the special slots there too (regression:
    the special counter once aliased the leftover counter)."""
    n = 5_000_000
    rng = np.random.default_rng(77)
    keys = rng.integers(0, 200_000, n).astype(np.int64)
    keys[::5000] = -2**63
    kv = np.ones(n, bool)
    kv[::4999] = False
    vals = rng.integers(0, 1000, n).astype(np.float64)
    t = blaze_amd.Task(plan.plan_partial_final(),
                       batches=[[(keys, kv), (vals, None)]])
    outputs = t.run()
    k, s, sv, c = engine_final(outputs)
    orc = oracle.Agg()
    orc.update(keys, vals, key_valid=kv)
    ref = orc.output()
    assert len(k) == orc.num_groups
    np.testing.assert_array_equal(c, ref["counts"])
    np.testing.assert_array_equal(s, ref["sums"])
    np.testing.assert_array_equal(k[ref["key_valid"]],
                                  ref["keys"][ref["key_valid"]])
    t.finalize()


def test_i32_grouping_keys():
    """Int32 grouping column (the agg_exec.rs:493-681 golden groups by an
    int32 'c'): widened to the i64 table internally, emitted as Int32."""
    rng = np.random.default_rng(88)
    n = 300_000
    keys = rng.integers(-50_000, 50_000, n).astype(np.int32)
    vals = rng.integers(0, 1000, n).astype(np.float64)
    t = blaze_amd.Task(plan.plan_partial_final(key_dt=plan.DT_INT32),
                       batches=[[(keys, None), (vals, None)]])
    outputs = t.run()
    got_keys = np.concatenate([ob[0]["values"] for ob in outputs])
    got_sums = np.concatenate([ob[1]["values"] for ob in outputs])
    got_cnts = np.concatenate([ob[2]["values"] for ob in outputs])
    assert got_keys.dtype == np.int32
    orc = oracle.Agg()
    orc.update(keys.astype(np.int64), vals)
    ref = orc.output()
    np.testing.assert_array_equal(got_keys.astype(np.int64), ref["keys"])
    np.testing.assert_array_equal(got_sums, ref["sums"])
    np.testing.assert_array_equal(got_cnts, ref["counts"])
    t.finalize()


def test_round_robin_shuffle_write(tmp_path):
    """RoundRobin partitioning (shuffle/mod.rs:190-202): part id chain starts
    at partition_id*1000193 %% P (buffered_data.rs:291-293)."""
    P, pid = 4, 3
    n = 10_000
    keys = np.arange(n, dtype=np.int64)
    vals = np.ones(n)
    data_file = str(tmp_path / "rr.data")
    index_file = str(tmp_path / "rr.index")
    t = blaze_amd.Task(plan.plan_shuffle_robin(data_file, index_file, P, pid),
                       batches=[[(keys, None), (vals, None)]])
    assert t.run() == []
    t.finalize()
    index = np.frombuffer(open(index_file, "rb").read(), dtype="<u8")
    blob = open(data_file, "rb").read()
    start = (pid * 1000193) % P
    ids = (start + np.arange(n)) % P
    for p in range(P):
        seg = blob[index[p]:index[p + 1]]
        payload = oracle.ipc_decode(seg)
        # first batch in the segment: varint rows, then key col
        rows, used = oracle.read_len(payload)
        exp_rows = np.nonzero(ids == p)[0]
        # engine sub-batches at BATCH_SIZE=10000; single batch here
        assert rows == len(exp_rows)
        body = payload[used:]
        has_null, k2 = oracle.read_len(body)
        assert has_null == 0
        planes = np.frombuffer(body[k2:k2 + 8 * rows],
                               dtype=np.uint8).reshape(8, rows)
        got_keys = np.ascontiguousarray(planes.T).reshape(-1).view(np.int64)
        np.testing.assert_array_equal(got_keys, keys[exp_rows])


def _minmax_cols(outs, names):
    """Concat output cols: returns {name: (values, valid)} keyed by agg name
    (col 0 is the key)."""
    res = {"key": (np.concatenate([ob[0]["values"] for ob in outs]), None)}
    for j, nm in enumerate(names):
        vals = np.concatenate([ob[1 + j]["values"] for ob in outs])
        vv = np.concatenate(
            [ob[1 + j]["valid"] if ob[1 + j]["valid"] is not None
             else np.ones(len(ob[1 + j]["values"]), bool) for ob in outs])
        res[nm] = (vals, vv)
    return res


def test_minmax_agg_family():
    """MIN/MAX aggregates (maxmin.rs:104-119 partial_update, :196-216
    partial_merge) in a mixed list with SUM/COUNT/AVG, partial->final.
    MIN/MAX select actual input values, so parity is bit-exact even for
    fractional f64."""
    keys, vals, vv = gen_northstar(300_000, nkeys=2000, fractional=True)
    names = ["min", "max", "sum", "count", "avg"]
    t = blaze_amd.Task(plan.plan_partial_final_named(names),
                       batches=batches_of(keys, vals, vv))
    outs = t.run()
    got = _minmax_cols(outs, names)
    ref = run_oracle(keys, vals, vv).output()
    ok, mins, maxs = oracle.minmax_groups(keys, vals, vv)
    np.testing.assert_array_equal(got["key"][0], ref["keys"])
    np.testing.assert_array_equal(np.array(ok), ref["keys"])
    exp_valid = np.array([m is not None for m in mins])
    np.testing.assert_array_equal(got["min"][1], exp_valid)
    np.testing.assert_array_equal(got["max"][1], exp_valid)
    np.testing.assert_array_equal(
        got["min"][0][exp_valid],
        np.array([m for m in mins if m is not None]))
    np.testing.assert_array_equal(
        got["max"][0][exp_valid],
        np.array([m for m in maxs if m is not None]))
    np.testing.assert_array_equal(got["count"][0], ref["counts"])
    t.finalize()


def test_minmax_only_with_all_null_groups():
    """[MIN] alone: the frozen record carries only the min part (no count on
    the wire) and all-null groups emit null (maxmin.rs nullable:85-87)."""
    rng = np.random.default_rng(7)
    n = 2000
    keys = rng.integers(0, 500, n).astype(np.int64)
    vals = rng.random(n) * 1000 - 500  # negative values exercise the omap
    vv = rng.random(n) >= 0.5
    names = ["min"]
    t = blaze_amd.Task(plan.plan_partial_final_named(names),
                       batches=batches_of(keys, vals, vv, batch=100))
    outs = t.run()
    got = _minmax_cols(outs, names)
    ok, mins, maxs = oracle.minmax_groups(keys, vals, vv)
    np.testing.assert_array_equal(got["key"][0], np.array(ok))
    exp_valid = np.array([m is not None for m in mins])
    assert (~exp_valid).sum() > 0, "test data must produce all-null groups"
    np.testing.assert_array_equal(got["min"][1], exp_valid)
    np.testing.assert_array_equal(
        got["min"][0][exp_valid],
        np.array([m for m in mins if m is not None]))
    t.finalize()


def test_minmax_spill_and_grow():
    """MIN/MAX accumulators survive table grow (mm pairs carried by the
    rebuild) and host spill (mm parts ride the frozen a8 records)."""
    keys, vals, vv = gen_northstar(400_000, nkeys=120_000, fractional=True)
    names = ["min", "max", "count"]
    conf = {"AURON_HIP_MEM_BUDGET": 1 << 20,
            "AURON_HIP_AGG_TABLE_SLOTS": 1 << 10}
    t = blaze_amd.Task(plan.plan_partial_final_named(names),
                       batches=batches_of(keys, vals, vv), conf=conf)
    outs = t.run()
    assert t.metric("spill_count") > 0
    got = _minmax_cols(outs, names)
    ok, mins, maxs = oracle.minmax_groups(keys, vals, vv)
    k = got["key"][0]
    assert len(k) == len(ok)
    gi = np.argsort(k, kind="stable")
    oi = np.argsort(np.array(ok), kind="stable")
    np.testing.assert_array_equal(k[gi], np.array(ok)[oi])
    exp_min = np.array([np.nan if m is None else m for m in mins])
    exp_max = np.array([np.nan if m is None else m for m in maxs])
    exp_valid = ~np.isnan(exp_min)
    np.testing.assert_array_equal(got["min"][1][gi], exp_valid[oi])
    np.testing.assert_array_equal(got["min"][0][gi][exp_valid[oi]],
                                  exp_min[oi][exp_valid[oi]])
    np.testing.assert_array_equal(got["max"][0][gi][exp_valid[oi]],
                                  exp_max[oi][exp_valid[oi]])
    t.finalize()


def test_minmax_partial_skipping():
    """Partial-skipping pass-through freezes each row as its own record with
    min=max=value (agg_ctx.rs:428-462 analog); the final stage must still
    produce the exact min/max."""
    n = 60_000
    rng = np.random.default_rng(11)
    keys = np.arange(n, dtype=np.int64) % (n // 2)  # 2 rows/group
    vals = rng.random(n) * 100
    vv = np.ones(n, bool)

    reader = plan.ffi_reader(plan.northstar_input_fields(), "input0")
    names = ["min", "max"]
    partial = plan.agg(reader, [plan.column("key", 0)],
                       plan.named_aggs(names),
                       [plan.MODE_PARTIAL] * 2, ["key"], names,
                       supports_partial_skipping=True)
    final = plan.agg(partial, [plan.column("key", 0)],
                     plan.named_aggs(names),
                     [plan.MODE_FINAL] * 2, ["key"], names)
    t = blaze_amd.Task(plan.task_definition(final),
                       batches=batches_of(keys, vals, vv))
    outs = t.run()
    got = _minmax_cols(outs, names)
    ok, mins, maxs = oracle.minmax_groups(keys, vals, vv)
    k = got["key"][0]
    assert len(k) == len(ok)
    gi = np.argsort(k, kind="stable")
    oi = np.argsort(np.array(ok), kind="stable")
    np.testing.assert_array_equal(k[gi], np.array(ok)[oi])
    np.testing.assert_array_equal(got["min"][0][gi], np.array(mins)[oi])
    np.testing.assert_array_equal(got["max"][0][gi], np.array(maxs)[oi])
    t.finalize()


def test_first_agg_family():
    """FIRST / FIRST_IGNORES_NULL (first.rs, first_ignores_null.rs) in a
    mixed list, partial->final. The engine's two-pass capture must reproduce
    the reference's sequential latch semantics exactly."""
    keys, vals, vv = gen_northstar(200_000, nkeys=1500, null_frac=0.3,
                                   fractional=True)
    names = ["first", "first_ignores_null", "count"]
    t = blaze_amd.Task(plan.plan_partial_final_named(names),
                       batches=batches_of(keys, vals, vv))
    outs = t.run()
    got = _minmax_cols(outs, names)
    ok, firsts, firsts_nn = oracle.first_groups(keys, vals, vv)
    np.testing.assert_array_equal(got["key"][0], np.array(ok))
    exp_f_valid = np.array([f[1] is not None for f in firsts])
    exp_fn_valid = np.array([f is not None for f in firsts_nn])
    assert (~exp_f_valid).sum() > 0      # some groups start with a null value
    np.testing.assert_array_equal(got["first"][1], exp_f_valid)
    np.testing.assert_array_equal(
        got["first"][0][exp_f_valid],
        np.array([f[1] for f in firsts if f[1] is not None]))
    np.testing.assert_array_equal(got["first_ignores_null"][1], exp_fn_valid)
    np.testing.assert_array_equal(
        got["first_ignores_null"][0][exp_fn_valid],
        np.array([f for f in firsts_nn if f is not None]))
    t.finalize()


def test_first_spill_grow_and_skipping():
    """FIRST survives table grow + host spill (frozen parts + preserved
    first_row priorities) and partial-skipping pass-through."""
    keys, vals, vv = gen_northstar(300_000, nkeys=90_000, null_frac=0.2,
                                   fractional=True)
    names = ["first", "first_ignores_null"]
    conf = {"AURON_HIP_MEM_BUDGET": 1 << 20,
            "AURON_HIP_AGG_TABLE_SLOTS": 1 << 10}
    t = blaze_amd.Task(plan.plan_partial_final_named(names),
                       batches=batches_of(keys, vals, vv), conf=conf)
    outs = t.run()
    assert t.metric("spill_count") > 0
    got = _minmax_cols(outs, names)
    ok, firsts, firsts_nn = oracle.first_groups(keys, vals, vv)
    k = got["key"][0]
    assert len(k) == len(ok)
    gi = np.argsort(k, kind="stable")
    oi = np.argsort(np.array(ok), kind="stable")
    np.testing.assert_array_equal(k[gi], np.array(ok)[oi])
    exp_f = np.array([np.nan if f[1] is None else f[1] for f in firsts])
    exp_fv = ~np.isnan(exp_f)
    np.testing.assert_array_equal(got["first"][1][gi], exp_fv[oi])
    np.testing.assert_array_equal(got["first"][0][gi][exp_fv[oi]],
                                  exp_f[oi][exp_fv[oi]])
    exp_n = np.array([np.nan if f is None else f for f in firsts_nn])
    exp_nv = ~np.isnan(exp_n)
    np.testing.assert_array_equal(got["first_ignores_null"][1][gi], exp_nv[oi])
    np.testing.assert_array_equal(
        got["first_ignores_null"][0][gi][exp_nv[oi]], exp_n[oi][exp_nv[oi]])

    # skipping: distinct keys flip the partial stage to pass-through
    n = 50_000
    keys2 = np.arange(n, dtype=np.int64)
    rng = np.random.default_rng(5)
    vals2 = rng.random(n)
    vv2 = rng.random(n) >= 0.3
    reader = plan.ffi_reader(plan.northstar_input_fields(), "input0")
    partial = plan.agg(reader, [plan.column("key", 0)],
                       plan.named_aggs(names),
                       [plan.MODE_PARTIAL] * 2, ["key"], names,
                       supports_partial_skipping=True)
    final = plan.agg(partial, [plan.column("key", 0)],
                     plan.named_aggs(names),
                     [plan.MODE_FINAL] * 2, ["key"], names)
    t2 = blaze_amd.Task(plan.task_definition(final),
                        batches=batches_of(keys2, vals2, vv2))
    outs2 = t2.run()
    got2 = _minmax_cols(outs2, names)
    k2 = got2["key"][0]
    gi2 = np.argsort(k2, kind="stable")
    np.testing.assert_array_equal(k2[gi2], keys2)
    np.testing.assert_array_equal(got2["first"][1][gi2], vv2)
    np.testing.assert_array_equal(got2["first"][0][gi2][vv2], vals2[vv2])
    np.testing.assert_array_equal(got2["first_ignores_null"][1][gi2], vv2)
    t2.finalize()


def test_int64_typed_aggregation():
    """i64 accumulator mode (sum.rs:78-88: acc type = declared data type):
    SUM over bigint is bit-exact WRAPPING i64 arithmetic, MIN/MAX use the
    i64 order, outputs are Int64 columns. Includes values large enough that
    f64 accumulation would lose precision (> 2^53)."""
    rng = np.random.default_rng(17)
    n = 200_000
    keys = rng.integers(0, 3000, n).astype(np.int64)
    # values near 2^60: three adds exceed 2^62, wrapping differences from
    # f64 rounding would be visible immediately
    vals = (rng.integers(0, 1 << 60, n) - (1 << 59)).astype(np.int64)
    vv = rng.random(n) >= 0.05
    names = ["sum", "min", "max", "count"]
    t = blaze_amd.Task(
        plan.plan_partial_final_named(names, val_dt=plan.DT_INT64),
        batches=batches_of(keys, vals, vv))
    outs = t.run()
    got = _minmax_cols(outs, names)
    ok, sums, cnts = oracle.int_sum_groups(keys, vals, vv)
    np.testing.assert_array_equal(got["key"][0], np.array(ok))
    assert got["sum"][0].dtype == np.int64
    np.testing.assert_array_equal(got["sum"][0][got["sum"][1]],
                                  np.array(sums, np.int64)[got["sum"][1]])
    np.testing.assert_array_equal(got["count"][0], np.array(cnts))
    omin, omax = oracle.minmax_groups(keys, vals, vv)[1:3]
    exp_valid = np.array([m is not None for m in omin])
    np.testing.assert_array_equal(got["min"][1], exp_valid)
    assert got["min"][0].dtype == np.int64
    np.testing.assert_array_equal(
        got["min"][0][exp_valid],
        np.array([m for m in omin if m is not None], np.int64))
    np.testing.assert_array_equal(
        got["max"][0][exp_valid],
        np.array([m for m in omax if m is not None], np.int64))
    t.finalize()


def test_int64_sum_two_phase_1m_rows():
    """The two-phase (LDS bucket) path in i64 mode: large single batch so
    chunks exceed AGG2_MIN_CHUNK; wrapping sums must stay bit-exact."""
    rng = np.random.default_rng(23)
    n = 6_000_000
    keys = rng.integers(0, 50_000, n).astype(np.int64)
    vals = rng.integers(-(1 << 50), 1 << 50, n).astype(np.int64)
    names = ["sum", "count"]
    t = blaze_amd.Task(
        plan.plan_partial_final_named(names, val_dt=plan.DT_INT64),
        batches=[[(keys, None), (vals, None)]])
    outs = t.run()
    got = _minmax_cols(outs, names)
    # ground truth via numpy bincount-style aggregation (wrapping)
    order = {}
    for k in keys:
        if int(k) not in order:
            order[int(k)] = len(order)
    exp_keys = np.array(sorted(order, key=order.get), np.int64)
    with np.errstate(over="ignore"):
        exp_sums = np.zeros(len(order), np.int64)
        np.add.at(exp_sums, np.array([order[int(k)] for k in keys]), vals)
    np.testing.assert_array_equal(got["key"][0], exp_keys)
    np.testing.assert_array_equal(got["sum"][0], exp_sums)
    assert got["count"][0].sum() == n
    t.finalize()


def test_agg_argument_casts():
    """sum.rs:78-88 prepare_partial_args: the argument column is cast to the
    accumulator type. Int32 input with f64 aggs, and Int32 input with i64
    aggs, both via device-side widening."""
    rng = np.random.default_rng(31)
    n = 100_000
    keys = rng.integers(0, 1000, n).astype(np.int64)
    vals32 = rng.integers(-(1 << 30), 1 << 30, n).astype(np.int32)
    vv = rng.random(n) >= 0.1

    # i32 arg -> f64 accs
    names = ["sum", "count"]
    reader = plan.ffi_reader([plan.field("key", plan.DT_INT64, False),
                              plan.field("val", plan.DT_INT32, True)],
                             "input0")
    partial = plan.agg(reader, [plan.column("key", 0)], plan.named_aggs(names),
                       [plan.MODE_PARTIAL] * 2, ["key"], names)
    final = plan.agg(partial, [plan.column("key", 0)], plan.named_aggs(names),
                     [plan.MODE_FINAL] * 2, ["key"], names)
    t = blaze_amd.Task(plan.task_definition(final),
                       batches=batches_of(keys, vals32, vv))
    outs = t.run()
    got = _minmax_cols(outs, names)
    ref = run_oracle(keys, vals32.astype(np.float64), vv).output()
    np.testing.assert_array_equal(got["key"][0], ref["keys"])
    np.testing.assert_array_equal(got["sum"][0][got["sum"][1]],
                                  ref["sums"][ref["sum_valid"]])
    t.finalize()

    # i32 arg -> i64 accs (SUM(int) returns bigint in Spark)
    partial2 = plan.agg(reader, [plan.column("key", 0)],
                        plan.named_aggs(names, val_dt=plan.DT_INT64),
                        [plan.MODE_PARTIAL] * 2, ["key"], names)
    final2 = plan.agg(partial2, [plan.column("key", 0)],
                      plan.named_aggs(names, val_dt=plan.DT_INT64),
                      [plan.MODE_FINAL] * 2, ["key"], names)
    t2 = blaze_amd.Task(plan.task_definition(final2),
                        batches=batches_of(keys, vals32, vv))
    outs2 = t2.run()
    got2 = _minmax_cols(outs2, names)
    ok, sums, cnts = oracle.int_sum_groups(keys, vals32.astype(np.int64), vv)
    np.testing.assert_array_equal(got2["key"][0], np.array(ok))
    assert got2["sum"][0].dtype == np.int64
    np.testing.assert_array_equal(got2["sum"][0][got2["sum"][1]],
                                  np.array(sums, np.int64)[got2["sum"][1]])
    t2.finalize()


@pytest.mark.parametrize("seed", [101, 202, 303, 404, 505, 616, 727, 838])
def test_agg_fuzz_randomized(seed):
    """Randomized end-to-end parity in the spirit of the reference's own
    agg fuzz test (agg_exec.rs:714-843): random agg lists, accumulator
    modes, cardinalities, null fractions, batch sizes and table confs, each
    checked against the python/numpy oracle restatements."""
    rng = np.random.default_rng(seed)
    n = int(rng.integers(5_000, 120_000))
    nkeys = int(rng.integers(10, max(11, n // 2)))
    null_frac = float(rng.choice([0.0, 0.01, 0.3]))
    batch = int(rng.choice([77, 1000, 8192]))
    int_mode = bool(rng.integers(0, 2))
    pool = ["sum", "count", "min", "max", "first", "first_ignores_null"] + \
        ([] if int_mode else ["avg"])
    names = list(rng.choice(pool, size=int(rng.integers(1, 5)), replace=False))
    keys = rng.integers(0, nkeys, n).astype(np.int64)
    if int_mode:
        vals = rng.integers(-(1 << 40), 1 << 40, n).astype(np.int64)
    else:
        vals = (rng.random(n) * 1e6 - 5e5)
    vv = rng.random(n) >= null_frac
    conf = {"BATCH_SIZE": batch}
    if rng.integers(0, 2):
        conf["AURON_HIP_AGG_TABLE_SLOTS"] = 1 << 10  # force grows
    t = blaze_amd.Task(
        plan.plan_partial_final_named(
            names, val_dt=plan.DT_INT64 if int_mode else plan.DT_FLOAT64),
        batches=batches_of(keys, vals, vv, batch=max(512, batch)),
        conf=conf)
    outs = t.run()
    got = _minmax_cols(outs, names)
    ok, mins, maxs = oracle.minmax_groups(keys, vals, vv)
    np.testing.assert_array_equal(got["key"][0], np.array(ok))
    fok, firsts, firsts_nn = oracle.first_groups(keys, vals, vv)
    for nm in names:
        v, valid = got[nm]
        if nm == "count":
            _, _, cnts = oracle.int_sum_groups(keys, np.zeros(n), vv)
            np.testing.assert_array_equal(v, np.array(cnts))
        elif nm == "sum":
            if int_mode:
                _, sums, _ = oracle.int_sum_groups(keys, vals, vv)
                np.testing.assert_array_equal(v[valid],
                                              np.array(sums, np.int64)[valid])
            else:
                agg = np.zeros(len(ok))
                idx = {k: i for i, k in enumerate(ok)}
                np.add.at(agg, [idx[int(k)] for k in keys[vv]], vals[vv])
                np.testing.assert_allclose(v[valid], agg[valid], rtol=1e-9)
        elif nm == "avg":
            _, _, cnts = oracle.int_sum_groups(keys, np.zeros(n), vv)
            agg = np.zeros(len(ok))
            idx = {k: i for i, k in enumerate(ok)}
            np.add.at(agg, [idx[int(k)] for k in keys[vv]], vals[vv])
            cnts = np.array(cnts, float)
            exp = np.divide(agg, np.maximum(cnts, 1))
            np.testing.assert_allclose(v[valid], exp[valid], rtol=1e-9)
        elif nm in ("min", "max"):
            src = mins if nm == "min" else maxs
            exp_valid = np.array([m is not None for m in src])
            np.testing.assert_array_equal(valid, exp_valid)
            np.testing.assert_array_equal(
                v[exp_valid], np.array([m for m in src if m is not None]))
        elif nm == "first":
            exp_valid = np.array([f[1] is not None for f in firsts])
            np.testing.assert_array_equal(valid, exp_valid)
            np.testing.assert_array_equal(
                v[exp_valid],
                np.array([f[1] for f in firsts if f[1] is not None]))
        elif nm == "first_ignores_null":
            exp_valid = np.array([f is not None for f in firsts_nn])
            np.testing.assert_array_equal(valid, exp_valid)
            np.testing.assert_array_equal(
                v[exp_valid],
                np.array([f for f in firsts_nn if f is not None]))
    t.finalize()
