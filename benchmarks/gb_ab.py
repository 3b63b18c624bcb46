"""A/B harness for the partitioned group-by variants (FUGUE_GB_PARTS):
checks numerics on 5M rows vs pandas, then times the aggregate-only step
at the headline shape (125M rows / 1M groups).

Usage: FUGUE_GB_PARTS=1024 python benchmarks/gb_ab.py [--rows N] [--steps K]
"""
import argparse
import os
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import numpy as np
import torch


def main() -> None:
    ap = argparse.ArgumentParser()
    ap.add_argument("--rows", type=int, default=125_000_000)
    ap.add_argument("--groups", type=int, default=1_000_000)
    ap.add_argument("--steps", type=int, default=30)
    ap.add_argument("--warmup", type=int, default=5)
    args = ap.parse_args()

    import pyarrow as pa

    from fugue_amd.collections.partition import PartitionSpec
    from fugue_amd.column import functions as f
    from fugue_amd.column.expressions import col
    from fugue_amd.hip.execution_engine import HipExecutionEngine
    from fugue_amd.hip.frame import DeviceColumn, HipDataFrame
    from fugue_amd.schema import Schema

    engine = HipExecutionEngine()
    device = torch.device(engine.device)
    parts = os.environ.get("FUGUE_GB_PARTS", "512")

    # numerics check at 5M rows
    gen = torch.Generator(device=device)
    gen.manual_seed(3)
    nn = 5_000_000
    k_small = torch.randint(0, args.groups, (nn,), dtype=torch.int64,
                            device=device, generator=gen)
    v_small = torch.rand(nn, dtype=torch.float64, device=device,
                         generator=gen)
    small = HipDataFrame.from_columns(
        {"k": DeviceColumn(k_small, None, pa.int64()),
         "v": DeviceColumn(v_small, None, pa.float64())},
        Schema("k:long,v:double"), engine.device,
    )
    spec = PartitionSpec(by=["k"])
    agg_cols = [f.sum(col("v")).alias("s"), f.count(col("v")).alias("n")]
    got = engine.aggregate(small, spec, agg_cols).as_pandas()
    got = got.sort_values("k").reset_index(drop=True)
    import pandas as pd

    exp = (
        pd.DataFrame(dict(k=k_small.cpu().numpy(), v=v_small.cpu().numpy()))
        .groupby("k", as_index=False)
        .agg(s=("v", "sum"), n=("v", "count"))
        .sort_values("k")
        .reset_index(drop=True)
    )
    assert got["k"].tolist() == exp["k"].tolist(), "keys mismatch"
    np.testing.assert_allclose(got["s"], exp["s"], rtol=1e-9)
    assert got["n"].tolist() == exp["n"].tolist(), "count mismatch"
    print(f"numerics ok (parts={parts})", flush=True)

    # timing at headline shape
    gen.manual_seed(42)
    keys = torch.randint(0, args.groups, (args.rows,), dtype=torch.int64,
                         device=device, generator=gen)
    vals = torch.rand(args.rows, dtype=torch.float64, device=device,
                      generator=gen)
    fact = HipDataFrame.from_columns(
        {"k": DeviceColumn(keys, None, pa.int64()),
         "v": DeviceColumn(vals, None, pa.float64())},
        Schema("k:long,v:double"), engine.device,
    )

    def step():
        return engine.aggregate(fact, spec, agg_cols)

    for _ in range(args.warmup):
        step()
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(args.steps):
        step()
    torch.cuda.synchronize()
    el = (time.perf_counter() - t0) / args.steps * 1000
    print(f"parts={parts} rows={args.rows} agg_ms_per_step={el:.3f}",
          flush=True)


if __name__ == "__main__":
    main()
