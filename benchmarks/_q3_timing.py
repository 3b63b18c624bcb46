import sys, time, os
sys.path.insert(0, os.environ.get("GRAFT_REPO_ROOT", "/root/repo"))
import torch
from benchmarks.q3_bench import Q3, gen_tables
import fugue_amd.api as fa
from fugue_amd.hip.execution_engine import HipExecutionEngine
from fugue_amd.sql.executor import parse_select
from fugue_amd.sql.planner import execute_plan

e = HipExecutionEngine()
c, o, l, _ = gen_tables(10.0, e.device, 0)

def sync():
    torch.cuda.synchronize()

def timeit(name, fn, n=3):
    fn(); sync()
    t0 = time.perf_counter()
    for _ in range(n):
        r = fn()
    sync()
    print(f"{name}: {(time.perf_counter()-t0)/n*1000:.2f} ms")
    return r

stmt = parse_select(Q3.replace("customer","c_t").replace("orders","o_t").replace("lineitem","l_t"))
tables = dict(c_t=c, o_t=o, l_t=l)
timeit("full plan", lambda: execute_plan(stmt, tables, e))
timeit("fugue_sql e2e", lambda: fa.fugue_sql(Q3, customer=c, orders=o, lineitem=l, engine=e, as_fugue=True))
# phases
from fugue_amd.column.expressions import col
cf = timeit("filter customer (string eq)", lambda: e.filter(c, col("mktsegment") == "BUILDING"))
of = timeit("filter orders date", lambda: e.filter(o, col("orderdate") < 9204))
lf = timeit("filter lineitem date", lambda: e.filter(l, col("shipdate") > 9204))
j1 = timeit("join c x o", lambda: e.join(cf, of, how="inner"))
j2 = timeit("join (cxo) x l", lambda: e.join(j1, lf, how="inner"))
from fugue_amd.column import functions as F
from fugue_amd.collections.partition import PartitionSpec
agg = timeit("aggregate", lambda: e.aggregate(j2, PartitionSpec(by=["orderkey","orderdate","shippriority"]), [F.sum(col("extendedprice")*(1-col("discount"))).alias("revenue")]))
timeit("take10", lambda: e.take(agg, 10, presort="revenue desc"))
