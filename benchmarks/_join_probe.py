import os, sys, time
sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import torch
from benchmarks.q3_bench import gen_tables
from fugue_amd.hip.execution_engine import HipExecutionEngine
from fugue_amd.column.expressions import col

e = HipExecutionEngine()
c, o, l, _ = gen_tables(10.0, e.device, 0)
cf = e.filter(c, col("mktsegment") == "BUILDING")
of = e.filter(o, col("orderdate") < 9204)
lf = e.filter(l, col("shipdate") > 9204)
j1 = e.join(cf, of, how="inner")
torch.cuda.synchronize()

def t(name, fn, iters=5):
    fn(); torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(iters):
        fn()
    torch.cuda.synchronize()
    print(f"{name}: {(time.perf_counter()-t0)/iters*1000:.2f} ms", flush=True)

print("j1 rows:", j1.count(), "lf rows:", lf.count(), flush=True)
t("join j1 x lf", lambda: e.join(j1, lf, how="inner"))
t("filter lineitem (re-timed)", lambda: e.filter(l, col("shipdate") > 9204))

# decompose: hash + indices + gathers
from fugue_amd.hip import ops as dops
k1 = [j1.col("orderkey")]
k2 = [lf.col("orderkey")]
t("hash keys", lambda: (dops.hash_rows(k1), dops.hash_rows(k2)))
h1 = dops.hash_rows(k1); h2k = dops.hash_rows(k2)
t("join indices", lambda: dops.hash_join_indices(j1.col("orderkey").data, lf.col("orderkey").data, "inner"))
pi, bi = dops.hash_join_indices(j1.col("orderkey").data, lf.col("orderkey").data, "inner")
print("out rows:", pi.numel(), flush=True)
t("gather probe side", lambda: j1.gather_rows(pi))
t("gather build side", lambda: lf.gather_rows(bi))
