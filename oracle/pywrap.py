"""ctypes wrapper over liboracle.so — TEST INFRASTRUCTURE ONLY.

Importable only from tests/, bench.py's cpu_baseline leg, and
__graft_entry__.smoke(). The product path (blaze_amd) must never import this.
"""
import ctypes
import os
import subprocess

import numpy as np

_DIR = os.path.dirname(os.path.abspath(__file__))
_SO = os.path.join(_DIR, "liboracle.so")


def _ensure_built():
    if not os.path.exists(_SO) or os.path.getmtime(_SO) < os.path.getmtime(
        os.path.join(_DIR, "oracle.c")
    ):
        subprocess.run(["make", "-C", _DIR], check=True, capture_output=True)


_ensure_built()
_lib = ctypes.CDLL(_SO)

_lib.oracle_murmur3.restype = ctypes.c_int32
_lib.oracle_murmur3.argtypes = [ctypes.c_char_p, ctypes.c_size_t, ctypes.c_int32]
_lib.oracle_murmur3_long.restype = ctypes.c_int32
_lib.oracle_murmur3_long.argtypes = [ctypes.c_int64, ctypes.c_int32]
_lib.oracle_xxhash64.restype = ctypes.c_int64
_lib.oracle_xxhash64.argtypes = [ctypes.c_char_p, ctypes.c_size_t, ctypes.c_int64]

_p = ctypes.POINTER


def _np_ptr(arr, ctype):
    if arr is None:
        return None
    return arr.ctypes.data_as(_p(ctype))


def murmur3(data: bytes, seed: int = 42) -> int:
    return _lib.oracle_murmur3(data, len(data), seed)


def murmur3_long(value: int, seed: int = 42) -> int:
    return _lib.oracle_murmur3_long(value, seed)


def xxhash64(data: bytes, seed: int = 42) -> int:
    return _lib.oracle_xxhash64(data, len(data), seed)


def _as_bitmap(valid, n):
    """bool array -> Arrow LSB validity bitmap (or None)."""
    if valid is None:
        return None
    v = np.asarray(valid, dtype=bool)
    assert v.shape == (n,)
    return np.packbits(v, bitorder="little")


def hash_cols(cols, seed=42, n=None):
    """create_murmur3_hashes (spark_hash.rs:28-57) over a list of
    (numpy values, validity bool array or None)."""
    n = n if n is not None else len(cols[0][0])
    hashes = np.full(n, seed, dtype=np.int32)
    for vals, valid in cols:
        bm = _as_bitmap(valid, n)
        bmp = _np_ptr(bm, ctypes.c_uint8)
        if vals.dtype == np.int64:
            _lib.oracle_hash_col_i64(_np_ptr(vals, ctypes.c_int64), bmp, n,
                                     _np_ptr(hashes, ctypes.c_int32))
        elif vals.dtype == np.int32:
            _lib.oracle_hash_col_i32(_np_ptr(vals, ctypes.c_int32), bmp, n,
                                     _np_ptr(hashes, ctypes.c_int32))
        elif vals.dtype == np.float64:
            _lib.oracle_hash_col_f64(_np_ptr(vals, ctypes.c_double), bmp, n,
                                     _np_ptr(hashes, ctypes.c_int32))
        else:
            raise TypeError(vals.dtype)
    return hashes


def xxhash_cols_i64(vals, valid=None, seed=42):
    n = len(vals)
    hashes = np.full(n, seed, dtype=np.int64)
    bm = _as_bitmap(valid, n)
    _lib.oracle_xxhash_col_i64(_np_ptr(vals, ctypes.c_int64),
                               _np_ptr(bm, ctypes.c_uint8), n,
                               _np_ptr(hashes, ctypes.c_int64))
    return hashes


def partition_ids(hashes, num_partitions):
    n = len(hashes)
    out = np.empty(n, dtype=np.uint32)
    _lib.oracle_partition_ids(_np_ptr(hashes, ctypes.c_int32), n,
                              num_partitions, _np_ptr(out, ctypes.c_uint32))
    return out


def radix_sort_triples(triples, num_keys):
    """Exact restatement of rdx_sort.rs:24-74 over (part,batch,row) u32
    triples; returns (sorted triples, counts)."""
    t = np.ascontiguousarray(triples, dtype=np.uint32)
    assert t.ndim == 2 and t.shape[1] == 3
    counts = np.zeros(num_keys, dtype=np.uintp)
    _lib.oracle_radix_sort_triples(_np_ptr(t, ctypes.c_uint32), t.shape[0],
                                   num_keys, _np_ptr(counts, ctypes.c_size_t))
    return t, counts.astype(np.int64)


class Agg:
    """North-star GROUP BY i64 -> SUM(f64), COUNT(val) oracle."""

    def __init__(self):
        _lib.oracle_agg_new.restype = ctypes.c_void_p
        self._h = ctypes.c_void_p(_lib.oracle_agg_new())

    def __del__(self):
        if getattr(self, "_h", None):
            _lib.oracle_agg_free(self._h)
            self._h = None

    def update(self, keys, vals, key_valid=None, val_valid=None):
        n = len(keys)
        keys = np.ascontiguousarray(keys, dtype=np.int64)
        vals = np.ascontiguousarray(vals, dtype=np.float64)
        kb = _as_bitmap(key_valid, n)
        vb = _as_bitmap(val_valid, n)
        _lib.oracle_agg_update(self._h, _np_ptr(keys, ctypes.c_int64),
                               _np_ptr(kb, ctypes.c_uint8),
                               _np_ptr(vals, ctypes.c_double),
                               _np_ptr(vb, ctypes.c_uint8), n)

    def merge_frozen(self, keys, acc_data, acc_offsets, key_valid=None):
        n = len(keys)
        keys = np.ascontiguousarray(keys, dtype=np.int64)
        acc_data = np.ascontiguousarray(acc_data, dtype=np.uint8)
        acc_offsets = np.ascontiguousarray(acc_offsets, dtype=np.int64)
        kb = _as_bitmap(key_valid, n)
        _lib.oracle_agg_merge_frozen(self._h, _np_ptr(keys, ctypes.c_int64),
                                     _np_ptr(kb, ctypes.c_uint8),
                                     _np_ptr(acc_data, ctypes.c_uint8),
                                     _np_ptr(acc_offsets, ctypes.c_int64), n)

    @property
    def num_groups(self):
        _lib.oracle_agg_num_groups.restype = ctypes.c_size_t
        return _lib.oracle_agg_num_groups(self._h)

    def output(self):
        g = self.num_groups
        keys = np.empty(g, dtype=np.int64)
        key_valid = np.empty(g, dtype=np.uint8)
        sums = np.empty(g, dtype=np.float64)
        sum_valid = np.empty(g, dtype=np.uint8)
        counts = np.empty(g, dtype=np.int64)
        _lib.oracle_agg_output(self._h, _np_ptr(keys, ctypes.c_int64),
                               _np_ptr(key_valid, ctypes.c_uint8),
                               _np_ptr(sums, ctypes.c_double),
                               _np_ptr(sum_valid, ctypes.c_uint8),
                               _np_ptr(counts, ctypes.c_int64))
        return dict(keys=keys, key_valid=key_valid.astype(bool), sums=sums,
                    sum_valid=sum_valid.astype(bool), counts=counts)

    def freeze(self):
        g = self.num_groups
        _lib.oracle_agg_freeze.restype = ctypes.c_size_t
        offsets = np.empty(g + 1, dtype=np.int64)
        total = _lib.oracle_agg_freeze(self._h, None,
                                       _np_ptr(offsets, ctypes.c_int64))
        data = np.empty(total, dtype=np.uint8)
        _lib.oracle_agg_freeze(self._h, _np_ptr(data, ctypes.c_uint8),
                               _np_ptr(offsets, ctypes.c_int64))
        return data, offsets


def write_len(n):
    buf = (ctypes.c_uint8 * 16)()
    _lib.oracle_write_len.restype = ctypes.c_size_t
    k = _lib.oracle_write_len(ctypes.c_uint64(n), buf)
    return bytes(buf[:k])


def read_len(data):
    out = ctypes.c_uint64()
    _lib.oracle_read_len.restype = ctypes.c_size_t
    k = _lib.oracle_read_len(bytes(data), len(data), ctypes.byref(out))
    return out.value, k


def serde_col_prim(values, valid=None):
    """batch_serde.rs primitive column bytes."""
    v = np.ascontiguousarray(values)
    n = len(v)
    w = v.dtype.itemsize
    bm = _as_bitmap(valid, n)
    _lib.oracle_serde_col_prim.restype = ctypes.c_size_t
    raw = v.view(np.uint8).reshape(-1)
    total = _lib.oracle_serde_col_prim(_np_ptr(raw, ctypes.c_uint8), w, n,
                                       _np_ptr(bm, ctypes.c_uint8), None)
    out = np.empty(total, dtype=np.uint8)
    _lib.oracle_serde_col_prim(_np_ptr(raw, ctypes.c_uint8), w, n,
                               _np_ptr(bm, ctypes.c_uint8),
                               _np_ptr(out, ctypes.c_uint8))
    return out.tobytes()


def serde_col_bytes(data, offsets, valid=None):
    data = np.ascontiguousarray(data, dtype=np.uint8)
    offsets = np.ascontiguousarray(offsets, dtype=np.int64)
    n = len(offsets) - 1
    bm = _as_bitmap(valid, n)
    _lib.oracle_serde_col_bytes.restype = ctypes.c_size_t
    total = _lib.oracle_serde_col_bytes(_np_ptr(data, ctypes.c_uint8),
                                        _np_ptr(offsets, ctypes.c_int64), n,
                                        _np_ptr(bm, ctypes.c_uint8), None)
    out = np.empty(total, dtype=np.uint8)
    _lib.oracle_serde_col_bytes(_np_ptr(data, ctypes.c_uint8),
                                _np_ptr(offsets, ctypes.c_int64), n,
                                _np_ptr(bm, ctypes.c_uint8),
                                _np_ptr(out, ctypes.c_uint8))
    return out.tobytes()


def serde_batch(num_rows, cols):
    """write_batch (batch_serde.rs:66-77): varint(num_rows) ++ columns.
    cols: list of ('prim', values, valid) or ('bytes', data, offsets, valid)."""
    out = [write_len(num_rows)]
    for c in cols:
        if c[0] == "prim":
            out.append(serde_col_prim(c[1], c[2] if len(c) > 2 else None))
        elif c[0] == "bytes":
            out.append(serde_col_bytes(c[1], c[2], c[3] if len(c) > 3 else None))
        else:
            raise ValueError(c[0])
    return b"".join(out)


class IpcWriter:
    def __init__(self, target=0, codec=0, zstd_level=1):
        """codec 0 = lz4 frames, 1 = zstd (ipc_compression.rs:189-196)."""
        _lib.oracle_ipc_writer_new2.restype = ctypes.c_void_p
        self._h = ctypes.c_void_p(
            _lib.oracle_ipc_writer_new2(target, codec, zstd_level))

    def __del__(self):
        if getattr(self, "_h", None):
            _lib.oracle_ipc_writer_free(self._h)
            self._h = None

    def write_payload(self, payload: bytes):
        rc = _lib.oracle_ipc_write_payload(self._h, payload, len(payload))
        assert rc == 0

    def finish_block(self):
        assert _lib.oracle_ipc_finish_block(self._h) == 0

    def bytes(self) -> bytes:
        ptr = ctypes.POINTER(ctypes.c_uint8)()
        _lib.oracle_ipc_bytes.restype = ctypes.c_size_t
        ln = _lib.oracle_ipc_bytes(self._h, ctypes.byref(ptr))
        return bytes(bytearray(ptr[i] for i in range(ln))) if ln else b""


def ipc_decode(blob: bytes, codec=0) -> bytes:
    fn = _lib.oracle_ipc_decode_zstd if codec == 1 else _lib.oracle_ipc_decode
    fn.restype = ctypes.c_size_t
    need = fn(blob, len(blob), None, 0)
    assert need != ctypes.c_size_t(-1).value, "ipc decode failed"
    out = np.empty(need, dtype=np.uint8)
    got = fn(blob, len(blob), _np_ptr(out, ctypes.c_uint8), need)
    assert got == need
    return out.tobytes()


# ---- MIN/MAX aggregation restatement (pure python / numpy) -----------------
# CPU oracle for the MIN/MAX agg family (maxmin.rs). The C oracle above
# covers the north-star [SUM, COUNT] set; MIN/MAX parity is pinned by this
# restatement of maxmin.rs:
#  - partial_update (maxmin.rs:104-119): for each non-null arg, keep the
#    current acc iff acc.partial_cmp(new) == Ordering(Less for MIN / Greater
#    for MAX), else take the new value; null args are skipped.
#  - partial_merge (maxmin.rs:196-216): same compare over non-null merged accs.
#  - freeze: the generic prim column save [u8 valid][8B LE f64]? — the same
#    AccPrimColumn format SUM uses (acc.rs:335-347; maxmin.rs:91-93
#    create_acc_column returns the same generic column as sum.rs).
# Record order and key handling follow the C oracle (first-occurrence,
# special null/i64::MIN groups last-inserted-by-arrival like OracleAgg).
# NaN note: the reference's NaN outcome is arrival-order dependent
# (partial_cmp == None always takes the new value); this restatement and the
# GPU engine use the total-order map instead — parity tests use NaN-free data.
def minmax_groups(keys, vals, val_valid=None, key_valid=None):
    """Insertion-ordered dict key->(mn, mx, any_valid); None key = null-key
    group. Returns (ordered_keys, mins, maxs, valid_mask) with mins/maxs None
    entries for all-null groups."""
    import math
    keys = list(keys)
    n = len(keys)
    groups = {}
    for i in range(n):
        k = None if (key_valid is not None and not key_valid[i]) else int(keys[i])
        if k not in groups:
            groups[k] = [None, None]
        vv = val_valid is None or bool(val_valid[i])
        if vv:
            v = vals[i]  # type-preserving: np.int64 stays exact (i64 mode)
            g = groups[k]
            g[0] = v if g[0] is None or v < g[0] else g[0]
            g[1] = v if g[1] is None or v > g[1] else g[1]
    ordered = list(groups.keys())
    mins = [groups[k][0] for k in ordered]
    maxs = [groups[k][1] for k in ordered]
    return ordered, mins, maxs


def minmax_freeze_rec(parts):
    """Freeze one record: parts is a list of ('prim', value_or_None) /
    ('cnt', int) in layout order; returns the a8 wire bytes for the record."""
    import struct
    out = b""
    for kind, v in parts:
        if kind == "prim":
            out += b"\x00" if v is None else b"\x01" + struct.pack("<d", v)
        else:
            out += write_len(int(v))
    return out


# ---- FIRST / FIRST_IGNORES_NULL restatement (pure python) ------------------
# first.rs:91-148 partial_update: the group's FIRST acc latches on the first
# processed row — value = that row's arg (null included), flag = touched.
# first_ignores_null.rs:83-117: the acc latches on the first NON-NULL arg.
# Merge (first.rs:198-207 / first_ignores_null.rs:146-170): the earliest
# touched (resp. valid) partial wins, in sequential merge order.
def first_groups(keys, vals, val_valid=None, key_valid=None):
    """Insertion-ordered: returns (ordered_keys, firsts, firsts_ignore_null)
    where firsts[i] is (touched, value_or_None) and firsts_ignore_null[i] is
    value_or_None."""
    n = len(keys)
    groups = {}
    for i in range(n):
        k = None if (key_valid is not None and not key_valid[i]) else int(keys[i])
        vv = val_valid is None or bool(val_valid[i])
        v = float(vals[i]) if vv else None
        if k not in groups:
            groups[k] = [True, v, v]           # touched, first, first_nonnull
        elif groups[k][2] is None and v is not None:
            groups[k][2] = v
    ordered = list(groups.keys())
    firsts = [(groups[k][0], groups[k][1]) for k in ordered]
    firsts_nn = [groups[k][2] for k in ordered]
    return ordered, firsts, firsts_nn


# ---- typed (Int64) aggregation restatement ---------------------------------
# sum.rs:78-88: SUM's accumulator column is the agg's declared data type and
# inputs are cast to it; for bigint that is i64 with WRAPPING addition
# (release-mode Rust `v + x`). numpy int64 adds wrap the same way.
def int_sum_groups(keys, vals, val_valid=None):
    """Insertion-ordered (keys, sums, counts) with np.int64 wrapping sums."""
    import numpy as np
    groups = {}
    n = len(keys)
    for i in range(n):
        k = int(keys[i])
        if k not in groups:
            groups[k] = [np.int64(0), 0]
        if val_valid is None or val_valid[i]:
            with np.errstate(over="ignore"):
                groups[k][0] = np.int64(groups[k][0] + np.int64(vals[i]))
            groups[k][1] += 1
    ordered = list(groups.keys())
    return (ordered, [groups[k][0] for k in ordered],
            [groups[k][1] for k in ordered])


# ---- COLLECT_LIST / COLLECT_SET restatement (pure python) ------------------
# collect.rs: COLLECT_LIST accumulates every non-null arg in arrival order
# (collect.rs:119-138 partial_update appends non-null scalars); COLLECT_SET
# dedups, keeping FIRST-occurrence order (AccSet.append appends only novel
# values). Freeze (collect.rs:237-241 save_raw) = write_len(raw_len) ++ the
# concatenated non-nullable scalar encodings — for i64/f64 args that is the
# raw LE 8-byte values (scalar_serde.rs:35-47 write_prim non-nullable).
# Merge (collect.rs:139-158) concatenates in merge arrival order (set: with
# dedup). CPU parity anchor for the round-2 device implementation.
def collect_groups(keys, vals, val_valid=None, distinct=False,
                   key_valid=None):
    """Insertion-ordered (keys, lists): lists hold non-null values in
    arrival order; distinct=True dedups keeping first occurrence. None key
    = the null-key group."""
    groups = {}
    n = len(keys)
    for i in range(n):
        k = None if (key_valid is not None and not key_valid[i]) \
            else int(keys[i])
        if k not in groups:
            groups[k] = ([], set())
        if val_valid is None or val_valid[i]:
            v = vals[i]
            lst, seen = groups[k]
            if distinct:
                if v in seen:
                    continue
                seen.add(v)
            lst.append(v)
    ordered = list(groups.keys())
    return ordered, [groups[k][0] for k in ordered]


def collect_freeze_rec(values, fmt="<d"):
    """Freeze one COLLECT record: varint(byte-len) ++ packed values."""
    import struct
    raw = b"".join(struct.pack(fmt, v) for v in values)
    return write_len(len(raw)) + raw


# ---- generalized grouping keys (Utf8 / multi-column tuples) ----------------
# Reference semantics: GROUP BY over the key TUPLE with null values grouping
# together per column (agg_ctx.rs grouping rows; arrow-row encoding makes
# null a distinct point in the ordering). Insertion order = first occurrence,
# like the single-key paths. Pure-python restatement for test sizes.
def gkey_agg_groups(key_cols, vals, val_valid=None, key_valids=None):
    """key_cols: list of per-column value sequences (str/int/float);
    key_valids: optional list of per-column validity arrays. Returns
    (ordered key tuples with None for nulls, sums, counts, mins, maxs)."""
    n = len(vals)
    groups = {}
    for i in range(n):
        kt = []
        for c, col in enumerate(key_cols):
            valid = key_valids is None or key_valids[c] is None or \
                bool(key_valids[c][i])
            kt.append(col[i] if valid else None)
        kt = tuple(kt)
        if kt not in groups:
            groups[kt] = [0.0, 0, None, None]
        if val_valid is None or val_valid[i]:
            g = groups[kt]
            v = vals[i]
            g[0] += v
            g[1] += 1
            g[2] = v if g[2] is None or v < g[2] else g[2]
            g[3] = v if g[3] is None or v > g[3] else g[3]
    ordered = list(groups.keys())
    return (ordered, [groups[k][0] for k in ordered],
            [groups[k][1] for k in ordered],
            [groups[k][2] for k in ordered],
            [groups[k][3] for k in ordered])
