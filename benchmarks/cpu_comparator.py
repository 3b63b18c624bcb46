"""CPU comparator for the headline bench shape (BASELINE.md protocol).

The reference's engines are not installable in this offline image
(triad/duckdb/dask absent), so the comparator is the computation its
always-available NativeExecutionEngine backend (pandas) performs for the
same step: groupby(sum,count) on 125M rows / 1M groups + inner merge with
a 1M-row dim table + filter.  Median of 5 runs.
"""
import time

import numpy as np
import pandas as pd

N = 125_000_000
G = 1_000_000

rng = np.random.default_rng(42)
keys = rng.integers(0, G, N)
vals = rng.random(N)
df = pd.DataFrame(dict(k=keys, v=vals))
dims = pd.DataFrame(dict(k=np.arange(G), w=rng.random(G)))

times = []
for it in range(5):
    t0 = time.perf_counter()
    agg = df.groupby("k", sort=False, as_index=False).agg(
        s=("v", "sum"), n=("v", "count")
    )
    joined = agg.merge(dims, on="k", how="inner")
    out = joined[joined["s"] > joined["w"]]
    t1 = time.perf_counter()
    times.append(t1 - t0)
    print(f"run {it}: {t1-t0:.2f}s out={len(out)}", flush=True)

med = sorted(times)[2]
print(f"median: {med:.2f}s -> {N/med/1e6:.1f}M rows/s", flush=True)
