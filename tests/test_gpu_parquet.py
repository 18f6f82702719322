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
