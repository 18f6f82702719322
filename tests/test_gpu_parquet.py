"""GPU parity for ParquetScanExec (a15/f.1, config-3 subset): pyarrow-written
files (plain + dictionary encodings, nulls, several row groups, multiple
codecs) scanned by the engine must aggregate to exactly what the oracle
computes over the same data read back with pyarrow."""
import numpy as np
import pytest

import blaze_amd
from blaze_amd import plan
from oracle import pywrap as oracle

pa = pytest.importorskip("pyarrow")
pq = pytest.importorskip("pyarrow.parquet")

pytestmark = pytest.mark.gpu


def _write(tmp_path, name, keys, vals, vv, codec, dictionary, row_group_size,
           data_page_version="1.0"):
    path = str(tmp_path / name)
    t = pa.table({
        "key": pa.array(keys, type=pa.int64()),
        "val": pa.array(np.where(vv, vals, 0.0), type=pa.float64(),
                        mask=~vv),
    })
    pq.write_table(t, path, compression=codec, use_dictionary=dictionary,
                   row_group_size=row_group_size,
                   data_page_version=data_page_version)
    import os
    return path, os.path.getsize(path)


def _run_and_check(path_size, keys, vals, vv, cutoff=None):
    t = blaze_amd.Task(plan.plan_parquet_filter_agg([path_size],
                                                    cutoff=cutoff))
    outs = t.run()
    got_keys = np.concatenate([ob[0]["values"] for ob in outs]) if outs else \
        np.empty(0, np.int64)
    got_sums = np.concatenate([ob[1]["values"] for ob in outs]) if outs else \
        np.empty(0)
    got_cnts = np.concatenate([ob[2]["values"] for ob in outs]) if outs else \
        np.empty(0, np.int64)
    t.finalize()
    sel = np.ones(len(keys), bool) if cutoff is None else keys < cutoff
    orc = oracle.Agg()
    orc.update(keys[sel], vals[sel], val_valid=vv[sel])
    ref = orc.output()
    gi = np.argsort(got_keys, kind="stable")
    oi = np.argsort(ref["keys"], kind="stable")
    np.testing.assert_array_equal(got_keys[gi], ref["keys"][oi])
    np.testing.assert_array_equal(got_sums[gi], ref["sums"][oi])
    np.testing.assert_array_equal(got_cnts[gi], ref["counts"][oi])


@pytest.mark.parametrize("codec", ["none", "snappy", "zstd", "lz4"])
@pytest.mark.parametrize("dictionary", [True, False])
def test_parquet_codecs_encodings(tmp_path, codec, dictionary):
    rng = np.random.default_rng(31)
    n = 120_000
    keys = rng.integers(0, 500, n).astype(np.int64)  # low-card: dict-friendly
    vals = rng.integers(0, 1_000_000, n).astype(np.float64)
    vv = rng.random(n) >= 0.01
    ps = _write(tmp_path, f"t_{codec}_{dictionary}.parquet", keys, vals, vv,
                codec, dictionary, row_group_size=50_000)
    _run_and_check(ps, keys, vals, vv)


def test_parquet_high_cardinality_plain(tmp_path):
    rng = np.random.default_rng(32)
    n = 200_000
    keys = rng.integers(0, 10**12, n).astype(np.int64)  # defeats dictionary
    vals = rng.integers(0, 1_000_000, n).astype(np.float64)
    vv = np.ones(n, bool)
    ps = _write(tmp_path, "hc.parquet", keys, vals, vv, "snappy", True,
                row_group_size=64_000)
    _run_and_check(ps, keys, vals, vv)


def test_parquet_with_filter_config3(tmp_path):
    rng = np.random.default_rng(33)
    n = 150_000
    keys = rng.integers(0, 1_000_000, n).astype(np.int64)
    vals = rng.integers(0, 1_000_000, n).astype(np.float64)
    vv = rng.random(n) >= 0.001
    ps = _write(tmp_path, "f.parquet", keys, vals, vv, "zstd", True,
                row_group_size=40_000)
    # ~10% selectivity like config 3
    _run_and_check(ps, keys, vals, vv, cutoff=100_000)


def test_parquet_data_page_v2(tmp_path):
    rng = np.random.default_rng(34)
    n = 60_000
    keys = rng.integers(0, 300, n).astype(np.int64)
    vals = rng.integers(0, 1000, n).astype(np.float64)
    vv = rng.random(n) >= 0.05
    ps = _write(tmp_path, "v2.parquet", keys, vals, vv, "snappy", True,
                row_group_size=20_000, data_page_version="2.0")
    _run_and_check(ps, keys, vals, vv)


def test_parquet_utf8_column_through_shuffle(tmp_path):
    """BYTE_ARRAY/UTF8 columns decode (plain + dictionary) and round-trip
    through a Single-partition shuffle write."""
    import os
    rng = np.random.default_rng(41)
    n = 50_000
    keys = rng.integers(0, 1000, n).astype(np.int64)
    names = np.array([f"name_{k % 97}" for k in keys], dtype=object)
    names[::113] = None
    t = pa.table({"key": pa.array(keys, pa.int64()),
                  "name": pa.array(names, pa.utf8())})
    path = str(tmp_path / "s.parquet")
    pq.write_table(t, path, compression="zstd", row_group_size=20_000)
    size = os.path.getsize(path)

    fields = [plan.field("key", plan.DT_INT64, True),
              plan.field("name", plan.DT_UTF8, True)]
    scan = plan.parquet_scan([(path, size)], fields)
    rep = plan.single_repartition()
    data_file = str(tmp_path / "s.data")
    index_file = str(tmp_path / "s.index")
    sw = plan.shuffle_writer(scan, rep, data_file, index_file)
    tk = blaze_amd.Task(plan.task_definition(sw))
    assert tk.run() == []
    tk.finalize()

    payload = oracle.ipc_decode(open(data_file, "rb").read())
    # decode batches: [rows][key col][name col (bytes)]
    got_keys, got_names = [], []
    pos = 0
    while pos < len(payload):
        rows, k = oracle.read_len(payload[pos:])
        pos += k
        # key col: has_null header + transposed i64
        hn, k2 = oracle.read_len(payload[pos:])
        pos += k2
        if hn:
            pos += (rows + 7) // 8
        planes = np.frombuffer(payload[pos:pos + 8 * rows],
                               dtype=np.uint8).reshape(8, rows)
        got_keys.append(np.ascontiguousarray(planes.T).reshape(-1).view(np.int64))
        pos += 8 * rows
        # name col: has_null + bitmap + transposed lens + data
        hn, k2 = oracle.read_len(payload[pos:])
        pos += k2
        valid = None
        if hn:
            bm = np.frombuffer(payload[pos:pos + (rows + 7) // 8], np.uint8)
            valid = np.unpackbits(bm, bitorder="little")[:rows].astype(bool)
            pos += (rows + 7) // 8
        lp = np.frombuffer(payload[pos:pos + 4 * rows],
                           dtype=np.uint8).reshape(4, rows)
        lens = np.ascontiguousarray(lp.T).reshape(-1).view(np.int32)
        pos += 4 * rows
        total = int(lens.sum())
        data = payload[pos:pos + total]
        pos += total
        offs = np.concatenate([[0], np.cumsum(lens)])
        for i in range(rows):
            if valid is not None and not valid[i]:
                got_names.append(None)
            else:
                got_names.append(data[offs[i]:offs[i + 1]].decode())
    got_keys = np.concatenate(got_keys)
    np.testing.assert_array_equal(got_keys, keys)
    exp = [None if v is None else v for v in names.tolist()]
    assert got_names == exp


def test_parquet_row_group_pruning(tmp_path):
    """pruning_predicates against row-group min/max stats: sorted keys give
    disjoint per-group ranges; key < c must drop whole groups."""
    import os
    n = 100_000
    keys = np.arange(n, dtype=np.int64)  # sorted -> rg k covers [k*25k,(k+1)*25k)
    vals = np.ones(n)
    t = pa.table({"key": pa.array(keys, pa.int64()),
                  "val": pa.array(vals, pa.float64())})
    path = str(tmp_path / "p.parquet")
    pq.write_table(t, path, compression="snappy", row_group_size=25_000)
    size = os.path.getsize(path)
    fields = [plan.field("key", plan.DT_INT64, True),
              plan.field("val", plan.DT_FLOAT64, True)]
    cutoff = 30_000
    pruning = [plan.binary_expr(plan.column("key", 0),
                                plan.literal(cutoff, "int64"), "Lt")]
    scan = plan.parquet_scan([(path, size)], fields, pruning=pruning)
    partial = plan.agg(scan, [plan.column("key", 0)], plan.sum_count_aggs(1),
                       [plan.MODE_PARTIAL] * 2, ["key"], ["sum", "cnt"])
    final = plan.agg(partial, [plan.column("key", 0)], plan.sum_count_aggs(1),
                     [plan.MODE_FINAL] * 2, ["key"], ["sum", "cnt"])
    tk = blaze_amd.Task(plan.task_definition(final))
    outs = tk.run()
    got = np.concatenate([ob[0]["values"] for ob in outs])
    tk.finalize()
    # groups 0 (0..25k) and 1 (25k..50k) survive; 2,3 pruned
    assert len(got) == 50_000
    assert got.max() == 49_999


def test_parquet_gpu_page_decompress_ab(tmp_path):
    """The device page-decompression path (kernels_pq.hip, default-on for
    snappy PLAIN suffixes) must produce byte-identical aggregation results
    to the forced host decode (AURON_PARQUET_GPUCOMP=0) — dict prefix +
    PLAIN fallback, 3 row groups, 10% nulls to stress the on-device
    def-level walk across unaligned page bit boundaries."""
    import os
    rng = np.random.default_rng(33)
    n = 2_000_000
    keys = rng.integers(0, 200_000, n).astype(np.int64)
    vals = rng.integers(0, 1000, n).astype(np.float64)
    vv = rng.random(n) >= 0.1
    ps = _write(tmp_path, "ab.parquet", keys, vals, vv, "snappy", True,
                700_000)
    _run_and_check(ps, keys, vals, vv)          # device path (default)
    os.environ["AURON_PARQUET_GPUCOMP"] = "0"   # forced host decode
    try:
        _run_and_check(ps, keys, vals, vv)
    finally:
        os.environ.pop("AURON_PARQUET_GPUCOMP", None)
