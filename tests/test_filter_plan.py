"""Filter/Projection plan decode + the Arrow-IPC scalar literal reader
(CPU-only): the C++ flatbuffer walker must agree with pyarrow-produced
ScalarValue.ipc_bytes (the reference's literal encoding,
auron-serde/src/lib.rs:447-456)."""
import ctypes

import pytest

import blaze_amd
from blaze_amd import plan

pa = pytest.importorskip("pyarrow")


def _decode_scalar(blob: bytes) -> str:
    lib = blaze_amd.lib()
    lib.auron_debug_decode_scalar.restype = ctypes.c_int32
    lib.auron_debug_decode_scalar.argtypes = [ctypes.c_char_p, ctypes.c_size_t,
                                              ctypes.c_char_p, ctypes.c_size_t]
    out = ctypes.create_string_buffer(512)
    rc = lib.auron_debug_decode_scalar(blob, len(blob), out, 512)
    assert rc >= 0
    return out.value.decode()


def test_ipc_scalar_int32():
    blob = plan.literal_ipc(123456, pa.int32())
    assert _decode_scalar(blob) == "i:123456 dtype=8"


def test_ipc_scalar_int64_negative():
    blob = plan.literal_ipc(-(2**40), pa.int64())
    assert _decode_scalar(blob) == f"i:{-(2**40)} dtype=10"


def test_ipc_scalar_float64():
    blob = plan.literal_ipc(3.5, pa.float64())
    assert _decode_scalar(blob) == "f:3.5"


def test_ipc_scalar_utf8():
    blob = plan.literal_ipc("hello world", pa.utf8())
    assert _decode_scalar(blob) == "str:hello world"


def test_ipc_scalar_null():
    blob = plan.literal_ipc(None, pa.int64())
    assert _decode_scalar(blob).startswith("null")


def test_filter_project_plan_decodes():
    td = plan.plan_filter_project_agg(cutoff=777)
    s = blaze_amd.debug_decode_plan(td)
    # chain shape is intact down to the reader
    assert "Agg(mode=2" in s and "Agg(mode=0" in s
    assert "FFIReader(nfields=2,rid=input0)" in s


def test_parquet_host_decode_summary(tmp_path):
    """Host-side parquet footer/page decode vs pyarrow metadata (no GPU)."""
    import numpy as np
    pq = pytest.importorskip("pyarrow.parquet")
    rng = np.random.default_rng(17)
    n = 30_000
    keys = rng.integers(0, 100, n).astype(np.int64)
    vals = rng.integers(0, 1000, n).astype(np.float64)
    vv = rng.random(n) >= 0.1
    t = pa.table({"key": pa.array(keys, pa.int64()),
                  "val": pa.array(vals, pa.float64(), mask=~vv)})
    path = str(tmp_path / "t.parquet")
    pq.write_table(t, path, compression="snappy", row_group_size=10_000)

    lib = blaze_amd.lib()
    lib.auron_debug_parquet_summary.restype = ctypes.c_int32
    lib.auron_debug_parquet_summary.argtypes = [ctypes.c_char_p,
                                                ctypes.c_char_p,
                                                ctypes.c_size_t]
    out = ctypes.create_string_buffer(1 << 14)
    rc = lib.auron_debug_parquet_summary(path.encode(), out, len(out))
    s = out.value.decode()
    assert rc > 0, s
    assert "ERROR" not in s, s
    assert "cols=key:2?,val:5?," in s
    assert "rgs=3" in s
    total_nulls = int((~vv).sum())
    import re
    nulls = sum(int(m) for m in re.findall(r"c1\{n=\d+,nulls=(\d+)", s))
    assert nulls == total_nulls
    rows = sum(int(m) for m in re.findall(r"rows=(\d+)", s))
    assert rows == n


def test_parquet_snappy_overlap_and_runs(tmp_path):
    """CPU regression for the bulk-copy snappy decoder and the RLE fast
    paths: highly repetitive values force overlapping back-references
    (offset < length pattern copies), long RLE runs, and nulls."""
    pa = pytest.importorskip("pyarrow")
    import pyarrow.parquet as pq
    import numpy as np

    n = 400_000
    rng = np.random.default_rng(3)
    # long constant runs (RLE dict indices + snappy pattern copies) mixed
    # with random spans
    vals = np.repeat(rng.integers(0, 50, n // 100), 100).astype(np.int64)[:n]
    mask = rng.random(n) < 0.05
    path = str(tmp_path / "rep.parquet")
    pq.write_table(
        pa.table({"v": pa.array(vals, pa.int64(), mask=mask)}), path,
        compression="snappy", row_group_size=123_457)
    lib = blaze_amd.lib()
    import ctypes
    lib.auron_debug_parquet_summary.restype = ctypes.c_int32
    lib.auron_debug_parquet_summary.argtypes = [ctypes.c_char_p,
                                               ctypes.c_char_p,
                                               ctypes.c_size_t]
    out = ctypes.create_string_buffer(1 << 14)
    rc = lib.auron_debug_parquet_summary(path.encode(), out, len(out))
    assert rc > 0, "summary failed"
    s = out.value.decode()
    # the summary decodes every chunk; its per-row-group null counts must
    # sum to the numpy ground truth (proves the RLE def levels and the
    # snappy-decompressed pages parsed correctly)
    import re
    rows = [int(m) for m in re.findall(r"n=(\d+)", s)]
    nulls = [int(m) for m in re.findall(r"nulls=(\d+)", s)]
    assert sum(rows) == n, s
    assert sum(nulls) == int(mask.sum()), s
    assert "dict=50" in s, s  # 50 distinct values -> dict encoding held


def test_parquet_decode_fuzz_cpu():
    """Randomized parquet files (codecs x schemas x row-group sizes x null
    fractions x page sizes) must decode on the host with exact row/null
    counts vs pyarrow's ground truth."""
    pa = pytest.importorskip("pyarrow")
    import pyarrow.parquet as pq
    import re
    import numpy as np
    import ctypes

    lib = blaze_amd.lib()
    lib.auron_debug_parquet_summary.restype = ctypes.c_int32
    lib.auron_debug_parquet_summary.argtypes = [ctypes.c_char_p,
                                               ctypes.c_char_p,
                                               ctypes.c_size_t]
    rng = np.random.default_rng(83)
    import tempfile, os
    for trial in range(8):
        n = int(rng.integers(1, 60_000))
        codec = ["snappy", "zstd", "lz4", "none"][trial % 4]
        cols, null_counts = {}, []
        ncols = int(rng.integers(1, 4))
        for ci in range(ncols):
            kind = int(rng.integers(0, 4))
            nf = float(rng.choice([0.0, 0.02, 0.5]))
            mask = rng.random(n) < nf
            if kind == 0:
                arr = pa.array(rng.integers(-2**31, 2**31, n, dtype=np.int64)
                               .astype(np.int32), pa.int32(), mask=mask)
            elif kind == 1:
                arr = pa.array(rng.integers(0, int(rng.choice([50, n + 1])),
                                            n).astype(np.int64),
                               pa.int64(), mask=mask)
            elif kind == 2:
                arr = pa.array(rng.random(n), pa.float64(), mask=mask)
            else:
                arr = pa.array(rng.random(n).astype(np.float32),
                               pa.float32(), mask=mask)
            cols[f"c{ci}"] = arr
            null_counts.append(int(mask.sum()))
        with tempfile.TemporaryDirectory() as td:
            path = os.path.join(td, "f.parquet")
            kw = {}
            mode = int(rng.integers(0, 3))
            if mode == 1:  # delta int columns (v2)
                kw = dict(use_dictionary=False, version="2.6",
                          data_page_version="2.0",
                          column_encoding={k: "DELTA_BINARY_PACKED"
                                           for k, a in cols.items()
                                           if pa.types.is_integer(a.type)})
            elif mode == 2:  # byte-stream-split float columns (v2)
                bss = [k for k, a in cols.items()
                       if pa.types.is_floating(a.type)]
                if bss:
                    kw = dict(use_dictionary=False, version="2.6",
                              use_byte_stream_split=bss)
            else:
                kw = dict(use_dictionary=bool(rng.integers(0, 2)))
            pq.write_table(
                pa.table(cols), path, compression=codec,
                row_group_size=int(rng.integers(100, n + 1)),
                data_page_size=int(rng.choice([1024, 64 * 1024, 1 << 20])),
                **kw)
            out = ctypes.create_string_buffer(1 << 16)
            rc = lib.auron_debug_parquet_summary(path.encode(), out, len(out))
            assert rc > 0, (trial, out.value)
            s = out.value.decode()
            per_col_rows = [0] * ncols
            per_col_nulls = [0] * ncols
            for m in re.finditer(r"c(\d+)\{n=(\d+),nulls=(\d+)", s):
                per_col_rows[int(m.group(1))] += int(m.group(2))
                per_col_nulls[int(m.group(1))] += int(m.group(3))
            assert per_col_rows == [n] * ncols, (trial, s[:200])
            assert per_col_nulls == null_counts, (trial, s[:200])


def test_parquet_delta_binary_packed_cpu():
    """DELTA_BINARY_PACKED (encoding 5, parquet delta spec) decode for
    INT32/INT64 incl. negatives, nulls, multi-block streams — verified
    against pyarrow-written files via the host decode summary."""
    pa = pytest.importorskip("pyarrow")
    import pyarrow.parquet as pq
    import numpy as np
    import re
    import ctypes
    import tempfile, os

    lib = blaze_amd.lib()
    lib.auron_debug_parquet_summary.restype = ctypes.c_int32
    lib.auron_debug_parquet_summary.argtypes = [ctypes.c_char_p,
                                               ctypes.c_char_p,
                                               ctypes.c_size_t]
    rng = np.random.default_rng(101)
    for n in (1, 100, 5000, 70_000):
        a64 = rng.integers(-(1 << 40), 1 << 40, n).astype(np.int64)
        a32 = rng.integers(-(1 << 30), 1 << 30, n).astype(np.int32)
        mask = rng.random(n) < 0.1
        with tempfile.TemporaryDirectory() as td:
            path = os.path.join(td, "d.parquet")
            pq.write_table(
                pa.table({"a": pa.array(a64, pa.int64(), mask=mask),
                          "b": pa.array(a32, pa.int32())}),
                path, compression="snappy", use_dictionary=False,
                column_encoding={"a": "DELTA_BINARY_PACKED",
                                 "b": "DELTA_BINARY_PACKED"},
                version="2.6", data_page_version="2.0")
            encs = pq.ParquetFile(path).metadata.row_group(0).column(0).encodings
            assert "DELTA_BINARY_PACKED" in encs
            out = ctypes.create_string_buffer(1 << 16)
            rc = lib.auron_debug_parquet_summary(path.encode(), out, len(out))
            assert rc > 0, (n, out.value)
            s = out.value.decode()
            rows = sum(int(m) for m in re.findall(r"c0\{n=(\d+)", s))
            nulls = sum(int(m) for m in re.findall(r"c0\{n=\d+,nulls=(\d+)", s))
            assert rows == n and nulls == int(mask.sum()), (n, s[:200])
            # VALUE verification: wrapping i64 checksums of the decoded
            # non-null value streams vs numpy ground truth
            with np.errstate(over="ignore"):
                cs0 = np.int64(a64[~mask].sum()) if (~mask).any() else np.int64(0)
                cs1 = np.int64(a32.astype(np.int64).sum()) if n else np.int64(0)
            got0 = sum(int(m) for m in re.findall(r"c0\{[^}]*csum=(-?\d+)", s))
            got1 = sum(int(m) for m in re.findall(r"c1\{[^}]*csum=(-?\d+)", s))
            assert np.int64(got0) == cs0, (n, got0, int(cs0))
            assert np.int64(got1) == cs1, (n, got1, int(cs1))


def test_parquet_delta_byte_array_cpu():
    """DELTA_BYTE_ARRAY (prefix+suffix) and DELTA_LENGTH_BYTE_ARRAY string
    encodings (v2 writers) — values verified via the position-weighted
    checksum against python ground truth."""
    pa = pytest.importorskip("pyarrow")
    import pyarrow.parquet as pq
    import numpy as np
    import re
    import ctypes
    import tempfile, os

    lib = blaze_amd.lib()
    lib.auron_debug_parquet_summary.restype = ctypes.c_int32
    lib.auron_debug_parquet_summary.argtypes = [ctypes.c_char_p,
                                               ctypes.c_char_p,
                                               ctypes.c_size_t]
    rng = np.random.default_rng(113)
    n = 20_000
    base = ["prefix_shared_", "prefix_other_", "", "x"]
    strs = [base[int(rng.integers(0, 4))] + str(int(rng.integers(0, 500)))
            for _ in range(n)]
    mask = rng.random(n) < 0.1
    for enc in ("DELTA_BYTE_ARRAY", "DELTA_LENGTH_BYTE_ARRAY"):
        with tempfile.TemporaryDirectory() as td:
            path = os.path.join(td, "s.parquet")
            pq.write_table(
                pa.table({"s": pa.array(
                    [None if mask[i] else strs[i] for i in range(n)],
                    pa.string())}),
                path, compression="snappy", use_dictionary=False,
                column_encoding={"s": enc}, version="2.6",
                data_page_version="2.0",
                row_group_size=7_777)
            encs = pq.ParquetFile(path).metadata.row_group(0).column(0).encodings
            assert enc in encs, encs
            out = ctypes.create_string_buffer(1 << 16)
            rc = lib.auron_debug_parquet_summary(path.encode(), out, len(out))
            assert rc > 0, out.value
            s = out.value.decode()
            rows = sum(int(m) for m in re.findall(r"c0\{n=(\d+)", s))
            assert rows == n
            # ground truth: per row group, position-weighted lens + bytes
            exp_total = 0
            rg = 7_777
            for beg in range(0, n, rg):
                end = min(n, beg + rg)
                lens, data = [], bytearray()
                for i in range(beg, end):
                    b = b"" if mask[i] else strs[i].encode()
                    lens.append(len(b))
                    data.extend(b)
                cs = 0
                for i, l in enumerate(lens):
                    cs = (cs + l * (i + 1)) % (1 << 64)
                for j, byte in enumerate(data):
                    cs = (cs + byte * (j + 1)) % (1 << 64)
                exp_total += cs if cs < (1 << 63) else cs - (1 << 64)
            got = sum(int(m) for m in re.findall(r"c0\{[^}]*csum=(-?\d+)", s))
            assert got == exp_total, (got, exp_total)


def test_parquet_byte_stream_split_cpu():
    """BYTE_STREAM_SPLIT (encoding 9; float planes) — bit-exact via the
    value checksum of the raw f64/f32 bit patterns."""
    pa = pytest.importorskip("pyarrow")
    import pyarrow.parquet as pq
    import numpy as np
    import re
    import ctypes
    import tempfile, os

    lib = blaze_amd.lib()
    lib.auron_debug_parquet_summary.restype = ctypes.c_int32
    lib.auron_debug_parquet_summary.argtypes = [ctypes.c_char_p,
                                               ctypes.c_char_p,
                                               ctypes.c_size_t]
    rng = np.random.default_rng(127)
    n = 30_000
    vals = rng.random(n) * 1e6 - 5e5
    mask = rng.random(n) < 0.05
    with tempfile.TemporaryDirectory() as td:
        path = os.path.join(td, "b.parquet")
        pq.write_table(
            pa.table({"f": pa.array(vals, pa.float64(), mask=mask)}), path,
            compression="zstd", use_dictionary=False,
            use_byte_stream_split=["f"], version="2.6",
            row_group_size=9_999)
        encs = pq.ParquetFile(path).metadata.row_group(0).column(0).encodings
        assert "BYTE_STREAM_SPLIT" in encs
        out = ctypes.create_string_buffer(1 << 16)
        rc = lib.auron_debug_parquet_summary(path.encode(), out, len(out))
        assert rc > 0, out.value
        s = out.value.decode()
        bits = vals.view(np.int64)[~mask]
        with np.errstate(over="ignore"):
            exp = int(np.int64(bits.sum()))
        got = sum(int(m) for m in re.findall(r"c0\{[^}]*csum=(-?\d+)", s))
        # per-rg wrapping sums add without wrap at this scale
        assert np.int64(got) == np.int64(exp), (got, exp)
