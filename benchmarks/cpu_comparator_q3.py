"""CPU comparator for the Q3-like bench shape (pandas; median of 3)."""
import time

import numpy as np
import pandas as pd

SF = 10.0
rng = np.random.default_rng(13)
n_cust, n_ord, n_li = int(150_000 * SF), int(1_500_000 * SF), int(6_000_000 * SF)
SEGMENTS = ["AUTOMOBILE", "BUILDING", "FURNITURE", "MACHINERY", "HOUSEHOLD"]
customer = pd.DataFrame(dict(
    custkey=np.arange(n_cust),
    mktsegment=np.array(SEGMENTS)[rng.integers(0, 5, n_cust)],
))
orders = pd.DataFrame(dict(
    orderkey=np.arange(n_ord),
    custkey=rng.integers(0, n_cust, n_ord),
    orderdate=rng.integers(8766, 9587, n_ord),
    shippriority=np.zeros(n_ord, dtype=np.int64),
))
lineitem = pd.DataFrame(dict(
    orderkey=rng.integers(0, n_ord, n_li),
    extendedprice=rng.random(n_li) * 100000,
    discount=rng.random(n_li) * 0.1,
    shipdate=rng.integers(8766, 9587, n_li),
))

times = []
for it in range(3):
    t0 = time.perf_counter()
    c = customer[customer.mktsegment == "BUILDING"]
    o = orders[orders.orderdate < 9204]
    l = lineitem[lineitem.shipdate > 9204]
    j = c.merge(o, on="custkey").merge(l, on="orderkey")
    j["revenue"] = j.extendedprice * (1 - j.discount)
    g = j.groupby(["orderkey", "orderdate", "shippriority"], as_index=False)[
        "revenue"
    ].sum()
    top = g.nlargest(10, "revenue")
    t1 = time.perf_counter()
    times.append(t1 - t0)
    print(f"run {it}: {t1-t0:.2f}s", flush=True)

med = sorted(times)[1]
total = n_cust + n_ord + n_li
print(f"median: {med:.2f}s -> {total/med/1e6:.1f}M rows/s", flush=True)
