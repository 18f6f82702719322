"""Two-phase pipeline parity at sizes that actually trigger it (chunks >= 4M
rows, AGG2_MIN_CHUNK): the bench exercises the v3 scatter/bucket kernels at
1B rows, this pins their CORRECTNESS against the oracle at CI scale, both
the v3 (default) and v2 (AURON_AGG2_V3=0) pipelines, with nulls and both
special key groups in-stream."""
import os

import numpy as np
import pytest

import blaze_amd
from blaze_amd import plan
from oracle import pywrap as oracle

pytestmark = pytest.mark.gpu


def _gen(n, nkeys, seed):
    rng = np.random.default_rng(seed)
    keys = rng.integers(0, nkeys, n).astype(np.int64)
    keys[::100_001] = -2**63          # i64::MIN special group
    kv = np.ones(n, bool)
    kv[::99_991] = False              # null-key special group
    vals = rng.integers(0, 1000, n).astype(np.float64)
    vv = rng.random(n) >= 0.001
    return keys, kv, vals, vv


@pytest.mark.parametrize("v3", ["1", "0"])
def test_two_phase_parity_6m(v3):
    os.environ["AURON_AGG2_V3"] = v3
    try:
        n = 6_000_000
        keys, kv, vals, vv = _gen(n, 300_000, 51)
        t = blaze_amd.Task(plan.plan_partial_final(),
                           batches=[[(keys, kv), (vals, vv)]],
                           conf={"BATCH_SIZE": 1 << 20,
                                 "AURON_HIP_AGG_TABLE_SLOTS": 1 << 20})
        outs = t.run()
        t.finalize()
        got_k = np.concatenate([o[0]["values"] for o in outs])
        got_s = np.concatenate([o[1]["values"] for o in outs])
        got_c = np.concatenate([o[2]["values"] for o in outs])
        orc = oracle.Agg()
        orc.update(keys, vals, key_valid=kv, val_valid=vv)
        ref = orc.output()
        assert len(got_k) == orc.num_groups
        # insertion order must match the oracle's exactly (first_row sort)
        np.testing.assert_array_equal(got_k, ref["keys"])
        np.testing.assert_array_equal(got_c, ref["counts"])
        np.testing.assert_array_equal(got_s, ref["sums"])  # integer-valued
    finally:
        os.environ.pop("AURON_AGG2_V3", None)


@pytest.mark.parametrize("skew", [False, True])
def test_two_phase_v4_ring_scatter(skew):
    """The experimental barrier-free ring scatter (AURON_AGG2_V4=1,
    kernels_agg3.hip k_agg4_scatter) must match the oracle, uniform and
    under adversarial skew (pending FIFO + pre-reserve bypass paths)."""
    os.environ["AURON_AGG2_V4"] = "1"
    try:
        n = 6_000_000
        if skew:
            rng = np.random.default_rng(53)
            keys = np.full(n, 4242, dtype=np.int64)
            spread = rng.integers(0, 50, n)
            keys[spread > 10] = rng.integers(0, 100, int((spread > 10).sum()))
            kv = vv = None
            vals = np.ones(n, dtype=np.float64)
        else:
            keys, kv, vals, vv = _gen(n, 300_000, 54)
        t = blaze_amd.Task(plan.plan_partial_final(),
                           batches=[[(keys, kv), (vals, vv)]],
                           conf={"BATCH_SIZE": 1 << 20,
                                 "AURON_HIP_AGG_TABLE_SLOTS": 1 << 20})
        outs = t.run()
        t.finalize()
        got_k = np.concatenate([o[0]["values"] for o in outs])
        got_s = np.concatenate([o[1]["values"] for o in outs])
        got_c = np.concatenate([o[2]["values"] for o in outs])
        orc = oracle.Agg()
        orc.update(keys, vals, key_valid=kv, val_valid=vv)
        ref = orc.output()
        np.testing.assert_array_equal(got_k, ref["keys"])
        np.testing.assert_array_equal(got_c, ref["counts"])
        np.testing.assert_array_equal(got_s, ref["sums"])
    finally:
        os.environ.pop("AURON_AGG2_V4", None)


def test_two_phase_skewed_buckets():
    """Adversarial skew: one dominant key floods one partition bucket (the
    leftover path and packet flushes must stay correct)."""
    os.environ["AURON_AGG2_V3"] = "1"
    try:
        n = 5_000_000
        rng = np.random.default_rng(52)
        keys = np.full(n, 7777, dtype=np.int64)
        spread = rng.integers(0, 50, n)
        keys[spread > 10] = rng.integers(0, 100, int((spread > 10).sum()))
        vals = np.ones(n, dtype=np.float64)
        t = blaze_amd.Task(plan.plan_partial_final(),
                           batches=[[(keys, None), (vals, None)]],
                           conf={"BATCH_SIZE": 1 << 20})
        outs = t.run()
        t.finalize()
        got_k = np.concatenate([o[0]["values"] for o in outs])
        got_c = np.concatenate([o[2]["values"] for o in outs])
        orc = oracle.Agg()
        orc.update(keys, vals)
        ref = orc.output()
        np.testing.assert_array_equal(got_k, ref["keys"])
        np.testing.assert_array_equal(got_c, ref["counts"])
    finally:
        os.environ.pop("AURON_AGG2_V3", None)


def test_two_phase_packed16_adaptive():
    """Adaptive 16B partition records: chunk 1 probes the key range (hist
    min/max) and later chunks pack keys as u32 offsets. Three 4M-row chunks
    (AURON_AGG2_CHUNK_M=4): chunk 1 narrow (decides packed), chunk 2 mixes
    in keys FAR outside the observed range (must take the counted leftover
    bypass), chunk 3 narrow again — parity against the oracle."""
    os.environ["AURON_AGG2_CHUNK_M"] = "4"
    try:
        rng = np.random.default_rng(63)
        n = 4_000_000
        base = 7_000_000_000  # away from zero: exercises non-zero key_base
        k1 = base + rng.integers(0, 200_000, n)
        k2 = base + rng.integers(0, 200_000, n)
        k2[::5000] = rng.integers(0, 50, (k2[::5000]).shape[0]) * 2**40
        k3 = base + rng.integers(0, 200_000, n)
        batches = []
        vals_all, keys_all = [], []
        for k in (k1, k2, k3):
            v = rng.integers(0, 1000, n).astype(np.float64)
            batches.append([(k.astype(np.int64), None), (v, None)])
            keys_all.append(k.astype(np.int64))
            vals_all.append(v)
        t = blaze_amd.Task(plan.plan_partial_final(), batches=batches,
                           conf={"BATCH_SIZE": 1 << 22,
                                 "AURON_HIP_AGG_TABLE_SLOTS": 1 << 20})
        outs = t.run()
        t.finalize()
        got_k = np.concatenate([o[0]["values"] for o in outs])
        got_s = np.concatenate([o[1]["values"] for o in outs])
        got_c = np.concatenate([o[2]["values"] for o in outs])
        orc = oracle.Agg()
        for k, v in zip(keys_all, vals_all):
            orc.update(k, v)
        ref = orc.output()
        np.testing.assert_array_equal(got_k, ref["keys"])
        np.testing.assert_array_equal(got_c, ref["counts"])
        np.testing.assert_array_equal(got_s, ref["sums"])
    finally:
        os.environ.pop("AURON_AGG2_CHUNK_M", None)
