"""Long-run stability soak: repeated mixed FugueSQL workloads with
device + host memory tracking (plan cache / stats memos must not grow
memory unboundedly).  Prints a line every N iterations."""
import os
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import numpy as np
import pandas as pd
import psutil
import torch


def main():
    iters = int(sys.argv[1]) if len(sys.argv) > 1 else 500
    import pyarrow as pa

    import fugue_amd.api as fa
    from fugue_amd.hip.execution_engine import HipExecutionEngine
    from fugue_amd.hip.frame import DeviceColumn, HipDataFrame
    from fugue_amd.schema import Schema

    engine = HipExecutionEngine()
    device = torch.device(engine.device)
    gen = torch.Generator(device=device)
    gen.manual_seed(1)
    n = 20_000_000
    fact = HipDataFrame.from_columns(
        {"k": DeviceColumn(
            torch.randint(0, 200_000, (n,), dtype=torch.int64,
                          device=device, generator=gen), None, pa.int64()),
         "v": DeviceColumn(
            torch.rand(n, dtype=torch.float64, device=device,
                       generator=gen), None, pa.float64())},
        Schema("k:long,v:double"), engine.device)
    dims = HipDataFrame.from_columns(
        {"k": DeviceColumn(
            torch.arange(0, 200_000, dtype=torch.int64, device=device),
            None, pa.int64()),
         "w": DeviceColumn(
            torch.rand(200_000, dtype=torch.float64, device=device,
                       generator=gen), None, pa.float64())},
        Schema("k:long,w:double"), engine.device)

    def scale(df: HipDataFrame) -> HipDataFrame:
        v = df.col("v")
        return HipDataFrame.from_columns(
            {"k": df.col("k"),
             "v": DeviceColumn(v.data * 1.0001, v.valid, pa.float64())},
            Schema("k:long,v:double"), df.device)

    proc = psutil.Process()
    SQLS = [
        "t = TRANSFORM fact USING scale SCHEMA k:long,v:double\n"
        "agg = SELECT k, SUM(v) AS s, COUNT(v) AS n FROM t GROUP BY k\n"
        "SELECT agg.k, s, n, w FROM agg INNER JOIN dims ON agg.k = dims.k"
        " WHERE s > w\nYIELD DATAFRAME AS result\n",
        "SELECT k, MIN(v) AS lo, MAX(v) AS hi FROM fact GROUP BY k\n"
        "YIELD DATAFRAME AS result\n",
        "SELECT fact.k, v, w FROM fact INNER JOIN dims ON fact.k = dims.k"
        " WHERE v > 0.99 ORDER BY v DESC LIMIT 10\n"
        "YIELD DATAFRAME AS result\n",
    ]
    rss0 = vram0 = None
    t0 = time.perf_counter()
    for i in range(iters):
        fa.fugue_sql(SQLS[i % len(SQLS)], fact=fact, dims=dims, scale=scale,
                     engine=engine, as_fugue=True)
        if i % 100 == 0 or i == iters - 1:
            torch.cuda.synchronize()
            rss = proc.memory_info().rss / 1e6
            vram = torch.cuda.memory_allocated() / 1e6
            if rss0 is None:
                rss0, vram0 = rss, vram
            print(f"iter {i:5d}  rss {rss:9.1f} MB (+{rss-rss0:7.1f})  "
                  f"vram {vram:9.1f} MB (+{vram-vram0:7.1f})  "
                  f"{(time.perf_counter()-t0):7.1f}s", flush=True)
    print("SOAK OK")


if __name__ == "__main__":
    main()
