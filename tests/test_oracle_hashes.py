"""Pin the oracle's hash restatement against the reference's own golden
vectors (spark_hash.rs:369-482, mur.rs:94-103), transcribed into
tests/golden/hash_vectors.json."""
import json
import os

import numpy as np

from oracle import pywrap as oracle

GOLDEN = json.load(open(os.path.join(os.path.dirname(__file__), "golden",
                                     "hash_vectors.json")))


def _i32(u):
    return np.int32(np.uint32(u))


def test_murmur3_strings():
    g = GOLDEN["murmur3_str_seed42"]
    got = [oracle.murmur3(s.encode(), 42) for s in g["inputs"]]
    assert got == g["expected"]


def test_murmur3_i8_col():
    g = GOLDEN["murmur3_i8_col_seed42"]
    # i8 values widen to i32 before hashing (spark_hash.rs hash_array_primitive
    # `as i32` for Int8Array)
    vals = np.array(g["inputs"], dtype=np.int32)
    got = oracle.hash_cols([(vals, None)])
    exp = np.array([_i32(u) for u in g["expected_u32"]], dtype=np.int32)
    np.testing.assert_array_equal(got, exp)


def test_murmur3_i32_col():
    g = GOLDEN["murmur3_i32_col_seed42"]
    for v, e in zip(g["inputs"], g["expected"]):
        got = oracle.hash_cols([(np.array([v], dtype=np.int32), None)])
        assert got[0] == e


def test_murmur3_i64_col():
    g = GOLDEN["murmur3_i64_col_seed42"]
    vals = np.array(g["inputs"], dtype=np.int64)
    got = oracle.hash_cols([(vals, None)])
    exp = np.array([_i32(u) for u in g["expected_u32"]], dtype=np.int32)
    np.testing.assert_array_equal(got, exp)
    # hash_long (mur.rs:76-87) must agree with the 8-byte LE path
    for v, e in zip(g["inputs"], exp):
        assert oracle.murmur3_long(int(v), 42) == e


def test_xxhash64_i64_col():
    g = GOLDEN["xxhash64_i64_col_seed42"]
    vals = np.array(g["inputs"], dtype=np.int64)
    got = oracle.xxhash_cols_i64(vals)
    np.testing.assert_array_equal(got, np.array(g["expected"], dtype=np.int64))


def test_murmur3_utf8_col():
    g = GOLDEN["murmur3_utf8_col_seed42"]
    got = [oracle.murmur3(s.encode("utf-8"), 42) for s in g["inputs"]]
    exp = [int(_i32(u)) for u in g["expected_u32"]]
    assert got == exp


def test_xxhash64_utf8_col():
    g = GOLDEN["xxhash64_utf8_col_seed42"]
    got = [oracle.xxhash64(s.encode("utf-8"), 42) for s in g["inputs"]]
    assert got == g["expected"]


def test_null_rows_keep_previous_hash():
    # spark_hash.rs hash_array_primitive: null rows keep the running hash
    vals = np.array([1, 99, 3], dtype=np.int64)
    valid = np.array([True, False, True])
    got = oracle.hash_cols([(vals, valid)])
    assert got[1] == 42  # untouched seed
    assert got[0] == oracle.murmur3_long(1, 42)
    assert got[2] == oracle.murmur3_long(3, 42)


def test_partition_ids_pmod():
    # shuffle/mod.rs:178-188 rem_euclid semantics incl. negative hashes
    hashes = np.array([-559580957, 1765031574, -1, 0, -200], dtype=np.int32)
    ids = oracle.partition_ids(hashes, 200)
    exp = np.array([h % 200 for h in hashes.tolist()], dtype=np.uint32)
    np.testing.assert_array_equal(ids, exp)
    assert (ids < 200).all()
