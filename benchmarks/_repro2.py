import os, sys
sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import torch
t = torch.empty(2, dtype=torch.int64, device="cuda")
try:
    t[0] = 2**63 - 1
    t[1] = -(2**63)
    torch.cuda.synchronize()
    print("fill ok", t.cpu().tolist())
except Exception as e:
    print("fill FAIL", str(e).splitlines()[0])
# emulate bindings steps with minmax path but small
from fugue_amd.hip.ext import get_ext
ext = get_ext()
g = torch.Generator(device="cuda").manual_seed(1)
keys = torch.randint(0, 1000, (100000,), device="cuda", generator=g)
vals = torch.rand((1, 100000), device="cuda", dtype=torch.float64, generator=g)
ops = torch.tensor([0], dtype=torch.int32, device="cuda")
try:
    k, a, c, ovf = ext.gb_aggregate_partitioned(keys, vals, ops, 512, 1 << 12, 0, 0, 0, -1)
    torch.cuda.synchronize()
    print("auto ok")
except Exception as e:
    print("auto FAIL", str(e).splitlines()[0])
# overflow path: keys above 2^31 must be detected and redone wide
import torch as _t
from fugue_amd.hip import ops as dops
from fugue_amd.hip.frame import DeviceColumn, HipDataFrame
from fugue_amd.schema import Schema
import pyarrow as pa
big = keys + (1 << 40)
bigvals = vals.clone()
cols = {"k": DeviceColumn(big, None, pa.int64()), "v": DeviceColumn(bigvals[0], None, pa.float64())}
df = HipDataFrame.from_columns(cols, Schema("k:long,v:double"), "cuda")
ok_, aggs_, cnt_, meta_ = dops.groupby_aggregate(df, ["k"], [("v", dops.AGG_SUM, "s")], expected_groups=200000)
import pandas as pd
exp = pd.DataFrame({"k": big.cpu().numpy(), "v": bigvals[0].cpu().numpy()}).groupby("k")["v"].sum()
got = pd.Series(aggs_["s"].cpu().numpy(), index=ok_.cpu().numpy()).sort_index()
assert len(got) == len(exp), (len(got), len(exp))
import numpy as np
np.testing.assert_allclose(got.values, exp.sort_index().values, rtol=1e-9)
print("overflow redo path OK")
