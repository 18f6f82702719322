"""Pure-python restatement of the reference's aggregate semantics, pinned by
the transcribed golden table of agg_exec.rs:493-681 test_agg (partial→final
over 8 int32 columns, grouped by c). Covers SUM/AVG/MAX/MIN/COUNT/
FIRST_IGNORES_NULL — wider than the engine's current SUM/COUNT/AVG set; this
is the semantic anchor the engine extends against (CollectList/CollectSet are
transcribed in comments only — variable-length accs are out of hot-path
scope)."""
import numpy as np

# inputs transcribed from agg_exec.rs:509-520
A = [2, 9, 3, 1, 0, 4, 6]
B = [1, 0, 0, 3, 5, 6, 3]
C = [7, 8, 7, 8, 9, 2, 5]
D = [-7, 86, 71, 83, 90, -2, 5]
E = [-7, 86, 71, 83, 90, -2, 5]
F = [0, 1, 2, 3, 4, 5, 6]
G = [6, 3, 6, 3, 1, 5, 4]
H = [6, 3, 6, 3, 1, 5, 4]

# expected (agg_exec.rs:667-681, sorted by c; collect columns omitted)
EXPECTED = {
    2: dict(sum_a=4, avg_b=6.0, max_d=-2, min_e=-2, cnt_f=1, firstign_h=5),
    5: dict(sum_a=6, avg_b=3.0, max_d=5, min_e=5, cnt_f=1, firstign_h=4),
    7: dict(sum_a=5, avg_b=0.5, max_d=71, min_e=-7, cnt_f=2, firstign_h=6),
    8: dict(sum_a=10, avg_b=1.5, max_d=86, min_e=83, cnt_f=2, firstign_h=3),
    9: dict(sum_a=0, avg_b=5.0, max_d=90, min_e=90, cnt_f=1, firstign_h=1),
}


def restate_aggregates(rows):
    """Row-arrival-order aggregation with the reference's null rules:
    SUM/AVG skip null args (null result iff no non-null arg, sum.rs:90-115,
    avg.rs), MIN/MAX skip nulls (maxmin semantics), COUNT counts rows where
    ALL args are non-null (count.rs:90-149), FIRST_IGNORES_NULL takes the
    first non-null (first.rs)."""
    groups = {}
    order = []
    for r in rows:
        key = r["c"]
        if key not in groups:
            groups[key] = dict(sum_a=None, sum_b=None, cnt_b=0, max_d=None,
                               min_e=None, cnt_f=0, firstign_h=None)
            order.append(key)
        g = groups[key]
        if r["a"] is not None:
            g["sum_a"] = r["a"] if g["sum_a"] is None else g["sum_a"] + r["a"]
        if r["b"] is not None:
            g["sum_b"] = r["b"] if g["sum_b"] is None else g["sum_b"] + r["b"]
            g["cnt_b"] += 1
        if r["d"] is not None:
            g["max_d"] = r["d"] if g["max_d"] is None else max(g["max_d"], r["d"])
        if r["e"] is not None:
            g["min_e"] = r["e"] if g["min_e"] is None else min(g["min_e"], r["e"])
        if r["f"] is not None:
            g["cnt_f"] += 1
        if r["h"] is not None and g["firstign_h"] is None:
            g["firstign_h"] = r["h"]
    return groups, order


def test_reference_golden_agg_table():
    rows = [dict(a=A[i], b=B[i], c=C[i], d=D[i], e=E[i], f=F[i], h=H[i])
            for i in range(7)]
    groups, order = restate_aggregates(rows)
    assert sorted(order) == sorted(EXPECTED)
    for c, exp in EXPECTED.items():
        g = groups[c]
        assert g["sum_a"] == exp["sum_a"], c
        avg = g["sum_b"] / g["cnt_b"]
        assert avg == exp["avg_b"], c
        assert g["max_d"] == exp["max_d"], c
        assert g["min_e"] == exp["min_e"], c
        assert g["cnt_f"] == exp["cnt_f"], c
        assert g["firstign_h"] == exp["firstign_h"], c


def test_null_rules_match_fuzztest():
    """agg_exec.rs:714-843 recipe: SUM null iff all args null; COUNT 0 then."""
    rows = [dict(a=None, b=None, c=1, d=None, e=None, f=None, h=None),
            dict(a=None, b=None, c=1, d=None, e=None, f=None, h=None)]
    groups, _ = restate_aggregates(rows)
    g = groups[1]
    assert g["sum_a"] is None
    assert g["cnt_f"] == 0
    assert g["firstign_h"] is None


def test_minmax_restatement_hand_case():
    """oracle.pywrap.minmax_groups vs a hand-computed maxmin.rs trace:
    non-null values only participate (maxmin.rs:106-108 is_valid guard);
    first-occurrence group order; all-null group -> None."""
    from oracle import pywrap as oracle
    keys = [5, 3, 5, 3, 9, 5]
    vals = [2.5, -1.0, 7.0, 4.0, 0.0, -3.5]
    vv = [True, True, True, False, False, True]
    ok, mins, maxs = oracle.minmax_groups(keys, vals, vv)
    assert ok == [5, 3, 9]
    assert mins == [-3.5, -1.0, None]
    assert maxs == [7.0, -1.0, None]


def test_minmax_freeze_rec_format():
    """MIN/MAX freeze parts are the SUM-style prim format [u8 valid][8B LE]
    (acc.rs:335-347 generic prim save; maxmin.rs:91-93)."""
    import struct
    from oracle import pywrap as oracle
    rec = oracle.minmax_freeze_rec([("prim", 1.5), ("prim", None),
                                    ("cnt", 300)])
    assert rec[:9] == b"\x01" + struct.pack("<d", 1.5)
    assert rec[9:10] == b"\x00"
    assert rec[10:] == oracle.write_len(300)


def test_first_restatement_hand_case():
    """oracle.pywrap.first_groups vs a hand trace of first.rs:91-148 /
    first_ignores_null.rs:83-117: FIRST latches the first row's value even
    when null; FIRST_IGNORES_NULL latches the first non-null."""
    from oracle import pywrap as oracle
    keys = [1, 1, 2, 2, 3]
    vals = [9.0, 4.0, 5.0, 6.0, 7.0]
    vv = [False, True, True, True, False]
    ok, firsts, firsts_nn = oracle.first_groups(keys, vals, vv)
    assert ok == [1, 2, 3]
    assert firsts == [(True, None), (True, 5.0), (True, None)]
    assert firsts_nn == [4.0, 5.0, None]


def test_collect_restatement_hand_case():
    """collect.rs semantics: LIST keeps arrival order incl. duplicates; SET
    dedups keeping first occurrence; null args skipped."""
    from oracle import pywrap as oracle
    keys = [1, 2, 1, 1, 2, 1]
    vals = [5.0, 7.0, 5.0, 3.0, 7.0, 9.0]
    vv = [True, True, True, True, True, False]
    ok, lists = oracle.collect_groups(keys, vals, vv)
    assert ok == [1, 2]
    assert lists == [[5.0, 5.0, 3.0], [7.0, 7.0]]
    ok2, sets = oracle.collect_groups(keys, vals, vv, distinct=True)
    assert sets == [[5.0, 3.0], [7.0]]


def test_collect_freeze_format():
    """COLLECT freeze = varint(raw_len) ++ raw LE scalar bytes
    (collect.rs:237-241 save_raw; scalar_serde.rs:35-47)."""
    import struct
    from oracle import pywrap as oracle
    rec = oracle.collect_freeze_rec([1.5, -2.0])
    assert rec == oracle.write_len(16) + struct.pack("<d", 1.5) + \
        struct.pack("<d", -2.0)
    assert oracle.collect_freeze_rec([]) == oracle.write_len(0)
    reci = oracle.collect_freeze_rec([7, -9], fmt="<q")
    assert reci == oracle.write_len(16) + struct.pack("<q", 7) + \
        struct.pack("<q", -9)
