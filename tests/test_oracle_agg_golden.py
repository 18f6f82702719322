"""Reference-held end-to-end agg golden, transcribed from the reference's own
test (agg_exec.rs:493-681 `test_agg`): 7-row 8-column Int32 table, GROUP BY c,
ten aggregates with INDEPENDENT argument columns and mixed declared types
(SUM->Int64, AVG->Float64, MAX/MIN/FIRST_IGNORES_NULL->Int32, COUNT->Int64,
COLLECT_LIST/SET->list<Int32>, plus two COLLECTs over a NULL Utf8 literal).

The pipeline under test is the oracle restatement of the reference's
Partial -> [a8 Binary agg-buf wire] -> Final chain:
  partial_update per agg (sum.rs / avg.rs / maxmin.rs / count.rs /
  collect.rs / first_ignores_null.rs) -> freeze_to_rows (acc.rs:335-347 prim,
  count.rs:193-203 varint, collect.rs:236-241 save_raw,
  first_ignores_null prim-only) -> unfreeze + partial_merge -> final output,
and the expected output table is the reference's own fixture (the one
reference-held golden that pins the full pipeline including the Binary
agg-buf semantics, not just restatement-vs-restatement agreement).

The input is split into two partial batches (rows 0..4 / 4..7) so the Final
stage really merges two frozen records per surviving group, exercising
partial_merge for every agg family.
"""
import struct
import sys
from pathlib import Path

import pytest

sys.path.insert(0, str(Path(__file__).resolve().parent.parent))

from oracle import pywrap as oracle  # noqa: E402

# ---- transcribed input table (agg_exec.rs:497-506) -------------------------
COLS = {
    "a": [2, 9, 3, 1, 0, 4, 6],
    "b": [1, 0, 0, 3, 5, 6, 3],
    "c": [7, 8, 7, 8, 9, 2, 5],
    "d": [-7, 86, 71, 83, 90, -2, 5],
    "e": [-7, 86, 71, 83, 90, -2, 5],
    "f": [0, 1, 2, 3, 4, 5, 6],
    "g": [6, 3, 6, 3, 1, 5, 4],
    "h": [6, 3, 6, 3, 1, 5, 4],
}

# ---- transcribed expected output (agg_exec.rs:662-675), keyed by c ---------
# columns: sum(a), avg(b), max(d), min(e), count(f), collect_list(g),
#          collect_set(h), collect_list(nil), collect_set(nil), firstign(h)
EXPECTED = {
    2: (4, 6.0, -2, -2, 1, [5], [5], [], [], 5),
    5: (6, 3.0, 5, 5, 1, [4], [4], [], [], 4),
    7: (5, 0.5, 71, -7, 2, [6, 6], [6], [], [], 6),
    8: (10, 1.5, 86, 83, 2, [3, 3], [3], [], [], 3),
    9: (0, 5.0, 90, 90, 1, [1], [1], [], [], 1),
}


def freeze_prim(value, fmt):
    """AccPrimColumn freeze (acc.rs:335-347): [u8 valid][LE value]."""
    if value is None:
        return b"\x00"
    return b"\x01" + struct.pack(fmt, value)


def parse_prim(buf, pos, fmt):
    valid = buf[pos]
    pos += 1
    if valid == 0:
        return None, pos
    w = struct.calcsize(fmt)
    v = struct.unpack_from(fmt, buf, pos)[0]
    return v, pos + w


def parse_len(buf, pos):
    """read_len varint (io/mod.rs:60-79)."""
    n = 0
    shift = 0
    while True:
        b = buf[pos]
        pos += 1
        n |= (b & 0x7F) << shift
        if not (b & 0x80):
            return n, pos
        shift += 7


def freeze_collect(values, fmt):
    """collect.rs:236-241 save_raw: write_len(raw len) ++ non-nullable items
    (scalar_serde.rs:35-47 write_prim)."""
    raw = b"".join(struct.pack(fmt, v) for v in values)
    return oracle.write_len(len(raw)) + raw


def parse_collect(buf, pos, fmt):
    ln, pos = parse_len(buf, pos)
    w = struct.calcsize(fmt)
    assert ln % w == 0
    vals = [struct.unpack_from(fmt, buf, pos + i * w)[0]
            for i in range(ln // w)]
    return vals, pos + ln


class PartialState:
    """One group's partial accumulators, per the golden's ten aggregates."""

    def __init__(self):
        self.sum_a = None          # SUM(a) -> Int64 (sum.rs:78-88 cast)
        self.avg_sum = None        # AVG(b): prim f64 sum
        self.avg_cnt = 0           # AVG(b): varint count
        self.max_d = None          # MAX(d) -> Int32
        self.min_e = None          # MIN(e) -> Int32
        self.count_f = 0           # COUNT(f) varint
        self.clist_g = []          # COLLECT_LIST(g) Int32 items
        self.cset_h = []           # COLLECT_SET(h): first-occurrence dedup
        self.firstign_h = None     # FIRST_IGNORES_NULL(h) -> Int32

    def update(self, a, b, d, e, f, g, h):
        self.sum_a = a if self.sum_a is None else self.sum_a + a
        self.avg_sum = float(b) if self.avg_sum is None \
            else self.avg_sum + float(b)
        self.avg_cnt += 1
        self.max_d = d if self.max_d is None else max(self.max_d, d)
        self.min_e = e if self.min_e is None else min(self.min_e, e)
        self.count_f += 1
        self.clist_g.append(g)
        if h not in self.cset_h:
            self.cset_h.append(h)
        if self.firstign_h is None:
            self.firstign_h = h

    def freeze(self):
        """a8 record: per-agg freezes concatenated in declaration order.
        The two nil COLLECTs freeze as empty lists (all appends skipped on
        the NULL literal arg, collect.rs:119-138)."""
        return (freeze_prim(self.sum_a, "<q") +
                freeze_prim(self.avg_sum, "<d") + oracle.write_len(self.avg_cnt) +
                freeze_prim(self.max_d, "<i") +
                freeze_prim(self.min_e, "<i") +
                oracle.write_len(self.count_f) +
                freeze_collect(self.clist_g, "<i") +
                freeze_collect(self.cset_h, "<i") +
                oracle.write_len(0) +      # collect_list(Utf8 NULL literal)
                oracle.write_len(0) +      # collect_set(Utf8 NULL literal)
                freeze_prim(self.firstign_h, "<i"))


class FinalState:
    """Final-mode merge of frozen partial records (partial_merge per agg)."""

    def __init__(self):
        self.p = PartialState()

    def merge_record(self, buf):
        pos = 0
        sum_a, pos = parse_prim(buf, pos, "<q")
        avg_sum, pos = parse_prim(buf, pos, "<d")
        avg_cnt, pos = parse_len(buf, pos)
        max_d, pos = parse_prim(buf, pos, "<i")
        min_e, pos = parse_prim(buf, pos, "<i")
        count_f, pos = parse_len(buf, pos)
        clist, pos = parse_collect(buf, pos, "<i")
        cset, pos = parse_collect(buf, pos, "<i")
        nil1, pos = parse_len(buf, pos)
        nil2, pos = parse_len(buf, pos)
        firstign, pos = parse_prim(buf, pos, "<i")
        assert pos == len(buf), "a8 record over/under-parse"
        assert nil1 == 0 and nil2 == 0
        s = self.p
        if sum_a is not None:
            s.sum_a = sum_a if s.sum_a is None else s.sum_a + sum_a
        if avg_sum is not None:
            s.avg_sum = avg_sum if s.avg_sum is None else s.avg_sum + avg_sum
        s.avg_cnt += avg_cnt
        if max_d is not None:
            s.max_d = max_d if s.max_d is None else max(s.max_d, max_d)
        if min_e is not None:
            s.min_e = min_e if s.min_e is None else min(s.min_e, min_e)
        s.count_f += count_f
        s.clist_g.extend(clist)
        for v in cset:
            if v not in s.cset_h:
                s.cset_h.append(v)
        if s.firstign_h is None and firstign is not None:
            s.firstign_h = firstign

    def final_values(self):
        s = self.p
        avg = None if s.avg_cnt == 0 else s.avg_sum / s.avg_cnt
        return (s.sum_a, avg, s.max_d, s.min_e, s.count_f, s.clist_g,
                s.cset_h, [], [], s.firstign_h)


def run_pipeline(batch_splits):
    """Partial-agg each batch, freeze to a8 records, merge all records in
    arrival order into the Final stage; returns {c: final tuple}."""
    rows = list(zip(*(COLS[c] for c in "abcdefgh")))
    final = {}
    final_order = []
    for lo, hi in batch_splits:
        partial = {}
        order = []
        for (a, b, c, d, e, f, g, h) in rows[lo:hi]:
            if c not in partial:
                partial[c] = PartialState()
                order.append(c)
            partial[c].update(a, b, d, e, f, g, h)
        # freeze + hand to final in the partial stage's emit order
        for c in order:
            rec = partial[c].freeze()
            if c not in final:
                final[c] = FinalState()
                final_order.append(c)
            final[c].merge_record(rec)
    return {c: final[c].final_values() for c in final_order}


@pytest.mark.parametrize("splits", [[(0, 7)], [(0, 4), (4, 7)],
                                    [(0, 2), (2, 5), (5, 7)]])
def test_agg_golden_pipeline(splits):
    got = run_pipeline(splits)
    assert set(got.keys()) == set(EXPECTED.keys())
    for c, exp in EXPECTED.items():
        assert got[c] == exp, f"group c={c}: {got[c]} != {exp}"


def test_agg_golden_oracle_helpers_agree():
    """The oracle's own group helpers reproduce the same reference fixture
    for the families they cover (pins minmax_groups/collect_groups/
    first_groups/int_sum_groups against the reference-held table)."""
    import numpy as np

    keys = np.array(COLS["c"], dtype=np.int64)
    ok, mins, maxs = oracle.minmax_groups(keys, np.array(COLS["d"]))
    for k, mx in zip(ok, maxs):
        assert mx == EXPECTED[k][2]
    ok, mins, _ = oracle.minmax_groups(keys, np.array(COLS["e"]))
    for k, mn in zip(ok, mins):
        assert mn == EXPECTED[k][3]
    ok, sums, _ = oracle.int_sum_groups(keys, np.array(COLS["a"]))
    for k, s in zip(ok, sums):
        assert int(s) == EXPECTED[k][0]
    ok, lists = oracle.collect_groups(keys, COLS["g"])
    for k, lst in zip(ok, lists):
        assert lst == EXPECTED[k][5]
    ok, sets_ = oracle.collect_groups(keys, COLS["h"], distinct=True)
    for k, st in zip(ok, sets_):
        assert st == EXPECTED[k][6]
    ok, _, firsts_nn = oracle.first_groups(keys, np.array(COLS["h"]))
    for k, v in zip(ok, firsts_nn):
        assert int(v) == EXPECTED[k][9]
