import os, sys, time
sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import torch
from fugue_amd.hip.ext import get_ext

ext = get_ext()
dev = torch.device("cuda:0")
n = 60_000_000
g = torch.Generator(device=dev).manual_seed(0)
cols = [
    torch.randint(0, 1 << 40, (n,), device=dev, generator=g),          # orderkey
    torch.rand((n,), device=dev, dtype=torch.float64, generator=g),    # price
    torch.rand((n,), device=dev, dtype=torch.float64, generator=g),    # discount
    torch.randint(8000, 10000, (n,), device=dev, generator=g),         # shipdate
]
mask = cols[3] > 9204

def t(name, fn, iters=5):
    fn(); torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(iters):
        fn()
    torch.cuda.synchronize()
    print(f"{name}: {(time.perf_counter()-t0)/iters*1000:.2f} ms", flush=True)

out_n = int(mask.sum().item())
print("selectivity:", out_n / n, flush=True)
t("compact_columns (fused)", lambda: ext.compact_columns(mask, cols, out_n))
def torch_path():
    idx = mask.nonzero(as_tuple=True)[0]
    return [c.index_select(0, idx) for c in cols]
t("torch nonzero+index_select", torch_path)
t("mask.sum().item (sync)", lambda: int(mask.sum().item()))
# engine-level filter
import pandas as pd
from fugue_amd.hip.execution_engine import HipExecutionEngine
from fugue_amd.hip.frame import DeviceColumn, HipDataFrame
from fugue_amd.schema import Schema
import pyarrow as pa
e = HipExecutionEngine()
fr = HipDataFrame.from_columns(
    {"a": DeviceColumn(cols[0], None, pa.int64()),
     "b": DeviceColumn(cols[1], None, pa.float64()),
     "c": DeviceColumn(cols[2], None, pa.float64()),
     "d": DeviceColumn(cols[3], None, pa.int64())},
    Schema("a:long,b:double,c:double,d:long"), "cuda")
from fugue_amd.column.expressions import col
t("engine.filter", lambda: e.filter(fr, col("d") > 9204))
