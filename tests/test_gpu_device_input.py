"""Device-resident (ArrowDeviceArray) input path: zero-copy HBM batches from
torch tensors through the engine."""
import numpy as np
import pytest

import blaze_amd
from blaze_amd import plan
from oracle import pywrap as oracle

pytestmark = pytest.mark.gpu


def test_device_input_small():
    import torch

    n = 100_000
    g = torch.Generator(device="cuda:0")
    g.manual_seed(7)
    keys = torch.randint(0, 1000, (n,), dtype=torch.int64, device="cuda:0",
                         generator=g)
    vals = torch.randint(0, 1000, (n,), dtype=torch.int64, device="cuda:0",
                         generator=g).to(torch.float64)
    torch.cuda.synchronize()
    db = blaze_amd.DeviceBatch([
        {"ptr": keys.data_ptr(), "len": n},
        {"ptr": vals.data_ptr(), "len": n},
    ])
    t = blaze_amd.Task(plan.plan_partial_final(), device_batches=[db.as_input()])
    outs = t.run()
    got_keys = np.concatenate([ob[0]["values"] for ob in outs])
    got_sums = np.concatenate([ob[1]["values"] for ob in outs])
    got_cnts = np.concatenate([ob[2]["values"] for ob in outs])
    orc = oracle.Agg()
    orc.update(keys.cpu().numpy(), vals.cpu().numpy())
    ref = orc.output()
    np.testing.assert_array_equal(got_keys, ref["keys"])
    np.testing.assert_array_equal(got_sums, ref["sums"])
    np.testing.assert_array_equal(got_cnts, ref["counts"])
    t.finalize()


def test_device_input_with_validity_bitmap():
    import torch

    n = 80_000
    g = torch.Generator(device="cuda:0")
    g.manual_seed(11)
    keys = torch.randint(0, 500, (n,), dtype=torch.int64, device="cuda:0",
                         generator=g)
    vals = torch.randint(0, 1000, (n,), dtype=torch.int64, device="cuda:0",
                         generator=g).to(torch.float64)
    nulls = torch.rand(n, device="cuda:0", generator=g) < 0.01
    valid = ~nulls
    pad = (-n) % 8
    v = torch.cat([valid, torch.ones(pad, dtype=torch.bool, device="cuda:0")])
    weights = torch.tensor([1, 2, 4, 8, 16, 32, 64, 128], dtype=torch.uint8,
                           device="cuda:0")
    bitmap = (v.view(-1, 8).to(torch.uint8) * weights).sum(dim=1).to(
        torch.uint8).contiguous()
    torch.cuda.synchronize()
    db = blaze_amd.DeviceBatch([
        {"ptr": keys.data_ptr(), "len": n},
        {"ptr": vals.data_ptr(), "len": n, "validity_ptr": bitmap.data_ptr(),
         "null_count": int(nulls.sum())},
    ])
    t = blaze_amd.Task(plan.plan_partial_final(), device_batches=[db.as_input()])
    outs = t.run()
    got_keys = np.concatenate([ob[0]["values"] for ob in outs])
    got_cnts = np.concatenate([ob[2]["values"] for ob in outs])
    orc = oracle.Agg()
    orc.update(keys.cpu().numpy(), vals.cpu().numpy(),
               val_valid=valid.cpu().numpy())
    ref = orc.output()
    np.testing.assert_array_equal(got_keys, ref["keys"])
    np.testing.assert_array_equal(got_cnts, ref["counts"])
    t.finalize()
