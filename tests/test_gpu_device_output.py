"""Device-resident OUTPUT path (AURON_HIP_DEVICE_OUTPUT): the partial-agg
freeze output stays in HBM and is exported through import_device_batch;
auron_repartition_device then sorts/gathers the records into dest-rank-major
partition order (buffered_data.rs:284-351 analog) and the Final stage
consumes them as a zero-copy ArrowDeviceArray. Parity: the whole chain must
be bit-exact vs the host-output chain and the oracle on the same input."""
import ctypes

import numpy as np
import pytest

import blaze_amd
from blaze_amd import plan
from oracle import pywrap as oracle

pytestmark = pytest.mark.gpu


def _make_input(n, distinct, seed, null_frac=0.01):
    import torch

    g = torch.Generator(device="cuda:0")
    g.manual_seed(seed)
    keys = torch.randint(0, distinct, (n,), dtype=torch.int64,
                         device="cuda:0", generator=g)
    vals = torch.randint(0, 1000, (n,), dtype=torch.int64, device="cuda:0",
                         generator=g).to(torch.float64)
    nulls = torch.rand(n, device="cuda:0", generator=g) < null_frac
    valid = ~nulls
    pad = (-n) % 8
    v = torch.cat([valid, torch.ones(pad, dtype=torch.bool, device="cuda:0")])
    w = torch.tensor([1, 2, 4, 8, 16, 32, 64, 128], dtype=torch.uint8,
                     device="cuda:0")
    bitmap = (v.view(-1, 8).to(torch.uint8) * w).sum(dim=1).to(
        torch.uint8).contiguous()
    torch.cuda.synchronize()
    db = blaze_amd.DeviceBatch([
        {"ptr": keys.data_ptr(), "len": n},
        {"ptr": vals.data_ptr(), "len": n, "validity_ptr": bitmap.data_ptr(),
         "null_count": int(nulls.sum().item())},
    ])
    return keys, vals, bitmap, nulls, db


def _d2h(ptr, nbytes):
    buf = (ctypes.c_uint8 * nbytes)()
    rc = ctypes.CDLL("libamdhip64.so").hipMemcpy(
        buf, ctypes.c_void_p(ptr), nbytes, 2)  # hipMemcpyDeviceToHost
    assert rc == 0
    return np.frombuffer(buf, dtype=np.uint8).copy()


def test_device_output_matches_host_output():
    n = 300_000
    keys, vals, bitmap, nulls, db = _make_input(n, 5000, 21)
    conf_dev = {"AURON_HIP_DEVICE_OUTPUT": 1, "BATCH_SIZE": 1 << 20}
    t = blaze_amd.Task(plan.plan_partial_only(), device_batches=[db.as_input()],
                       conf=conf_dev)
    t.run()
    assert len(t.device_outputs) == 1 and not t.outputs
    ob = t.device_outputs[0]
    kc, bc = ob["cols"]
    ng = ob["num_rows"]

    # host-output reference chain on the same input
    db2 = blaze_amd.DeviceBatch([
        {"ptr": keys.data_ptr(), "len": n},
        {"ptr": vals.data_ptr(), "len": n, "validity_ptr": bitmap.data_ptr(),
         "null_count": int(nulls.sum().item())},
    ])
    th = blaze_amd.Task(plan.plan_partial_only(),
                        device_batches=[db2.as_input()],
                        conf={"BATCH_SIZE": 1 << 20})
    href = th.run()
    assert len(href) == 1  # BATCH_SIZE > #groups: single batch
    h_keys = href[0][0]["values"]
    h_offs = href[0][1]["offsets"].astype(np.int32)
    h_data = href[0][1]["data"]

    d_keys = _d2h(kc["ptr"], ng * 8).view(np.int64)
    d_offs = _d2h(bc["offsets_ptr"], (ng + 1) * 4).view(np.int32)
    d_data = _d2h(bc["ptr"], bc["data_len"])
    np.testing.assert_array_equal(d_keys, h_keys)
    np.testing.assert_array_equal(d_offs, h_offs)
    np.testing.assert_array_equal(d_data, h_data)
    th.finalize()

    # repartition + final merge, fully device-resident
    rp = blaze_amd.DeviceRepartition(ng, kc["ptr"], bc["offsets_ptr"],
                                     bc["ptr"], 16, 1,
                                     key_validity_ptr=kc["validity_ptr"])
    t.finalize()
    assert sum(rp.rank_rows) == ng
    assert sum(rp.rank_bytes) == int(d_offs[-1])
    fin = blaze_amd.DeviceBatch([
        {"ptr": rp.keys_ptr, "len": ng, "validity_ptr": rp.key_validity_ptr,
         "null_count": -1},
        {"ptr": rp.data_ptr, "offsets_ptr": rp.offsets_ptr, "len": ng},
    ])
    t2 = blaze_amd.Task(plan.plan_final_only(),
                        device_batches=[fin.as_input()],
                        conf={"BATCH_SIZE": 1 << 20})
    out2 = t2.run()
    rp.free()

    # oracle on the same rows
    orc = oracle.Agg()
    orc.update(keys.cpu().numpy(), vals.cpu().numpy(),
               val_valid=~nulls.cpu().numpy())
    ref = orc.output()
    got_keys = np.concatenate([o[0]["values"] for o in out2])
    got_sums = np.concatenate([o[1]["values"] for o in out2])
    got_cnts = np.concatenate([o[2]["values"] for o in out2])
    # final output arrives partition-ordered (the repartition sorted the
    # records); compare as key-indexed dicts
    ref_by_key = dict(zip(ref["keys"], zip(ref["sums"], ref["counts"])))
    assert len(got_keys) == len(ref["keys"])
    for k, s, cn in zip(got_keys, got_sums, got_cnts):
        rs, rc2 = ref_by_key[int(k)]
        assert rs == s and rc2 == cn, (k, s, cn, rs, rc2)
    t2.finalize()


def test_repartition_partition_order():
    """Records come back dest-rank-major, partition-ordered within rank, and
    arrival-stable within partition (the a11 stable gather contract)."""
    n = 100_000
    P, world = 32, 4
    keys, vals, bitmap, nulls, db = _make_input(n, 700, 22, null_frac=0.0)
    t = blaze_amd.Task(plan.plan_partial_only(),
                       device_batches=[db.as_input()],
                       conf={"AURON_HIP_DEVICE_OUTPUT": 1,
                             "BATCH_SIZE": 1 << 20})
    t.run()
    ob = t.device_outputs[0]
    kc, bc = ob["cols"]
    ng = ob["num_rows"]
    in_keys = _d2h(kc["ptr"], ng * 8).view(np.int64)
    rp = blaze_amd.DeviceRepartition(ng, kc["ptr"], bc["offsets_ptr"],
                                     bc["ptr"], P, world,
                                     key_validity_ptr=kc["validity_ptr"])
    t.finalize()
    out_keys = _d2h(rp.keys_ptr, ng * 8).view(np.int64)
    # oracle pids for both layouts
    pid_in = oracle.partition_ids(oracle.hash_cols([(in_keys, None)]), P)
    pid_out = oracle.partition_ids(oracle.hash_cols([(out_keys, None)]), P)
    ord_out = (pid_out % world).astype(np.int64) * P + pid_out
    assert (np.diff(ord_out) >= 0).all(), "not dest-rank-major ordered"
    # stable within (dest, pid): arrival order preserved
    order = np.argsort((pid_in % world).astype(np.int64) * P + pid_in,
                       kind="stable")
    np.testing.assert_array_equal(out_keys, in_keys[order])
    # splits match
    rows = np.bincount(pid_in % world, minlength=world)
    np.testing.assert_array_equal(np.array(rp.rank_rows), rows)
    rp.free()


def test_cuda_buf_wrap_and_exchange_ops():
    """bench.py's N>1 exchange helpers on ONE GPU: the _CudaBuf
    __cuda_array_interface__ wrapper must view raw HBM pointers zero-copy
    (it only runs at world>1, so no other GPU test covers it), and the
    validity-bitmap <-> byte-per-row round trip used around the all-to-all
    must be exact."""
    import torch

    import bench

    t = torch.arange(1000, dtype=torch.int64, device="cuda:0") * 3
    w = torch.as_tensor(bench._CudaBuf(t.data_ptr(), 1000, "<i8"),
                        device="cuda:0")
    assert w.data_ptr() == t.data_ptr()  # zero-copy view
    torch.testing.assert_close(w, t)
    t[5] = -7
    assert int(w[5].item()) == -7  # same memory

    b = torch.randint(0, 256, (125,), dtype=torch.uint8, device="cuda:0")
    wb = torch.as_tensor(bench._CudaBuf(b.data_ptr(), 125, "|u1"),
                         device="cuda:0")
    n = 1000
    bits = torch.tensor([1, 2, 4, 8, 16, 32, 64, 128], dtype=torch.uint8,
                        device="cuda:0")
    vbytes = ((wb.unsqueeze(1) & bits) != 0).reshape(-1)[:n] \
        .to(torch.uint8).contiguous()
    # repack (the receive side of the exchange) and compare
    pad = (-n) % 8
    vb_pad = torch.cat([vbytes, torch.ones(pad, dtype=torch.uint8,
                                           device="cuda:0")]) if pad else \
        vbytes
    repacked = ((vb_pad.view(-1, 8) != 0).to(torch.uint8) *
                bits).sum(dim=1).to(torch.uint8)
    torch.testing.assert_close(repacked, b)
