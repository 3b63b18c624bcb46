"""UDF-boundary benchmark: rows/s through a pandas
identity transformer at 1e8 rows — measures the staged D2H → UDF → H2D
pipeline (pinned double-buffered copies + overlapped result uploads).

Usage: python benchmarks/udf_bench.py [--rows N] [--steps K] [--naive]
``--naive`` disables the staged pipeline (whole-shard as_pandas) for an
A/B of the overlap win.
"""
import argparse
import json
import os
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import numpy as np
import pandas as pd
import torch


def main() -> None:
    ap = argparse.ArgumentParser()
    ap.add_argument("--rows", type=int, default=100_000_000)
    ap.add_argument("--steps", type=int, default=3)
    ap.add_argument("--warmup", type=int, default=1)
    ap.add_argument("--parts", type=int, default=32,
                    help="logical partitions per rank (pipeline depth)")
    ap.add_argument("--naive", action="store_true")
    args = ap.parse_args()

    import pyarrow as pa

    import fugue_amd.api as fa
    from fugue_amd.hip.execution_engine import HipExecutionEngine
    from fugue_amd.hip.frame import DeviceColumn, HipDataFrame
    from fugue_amd.schema import Schema

    if args.naive:
        import fugue_amd.hip.staging as staging

        staging.can_fast_stage = lambda df: False  # type: ignore

    engine = HipExecutionEngine()
    device = torch.device(engine.device)
    gen = torch.Generator(device=device)
    gen.manual_seed(5)
    n = args.rows
    df = HipDataFrame.from_columns(
        {
            "k": DeviceColumn(
                torch.randint(0, 1 << 30, (n,), dtype=torch.int64,
                              device=device, generator=gen),
                None, pa.int64(),
            ),
            "v": DeviceColumn(
                torch.rand(n, dtype=torch.float64, device=device,
                           generator=gen),
                None, pa.float64(),
            ),
        },
        Schema("k:long,v:double"),
        engine.device,
    )

    def identity(pdf: pd.DataFrame) -> pd.DataFrame:
        return pdf

    def step():
        return fa.transform(
            df, identity, schema="*", engine=engine, as_fugue=True,
            partition=dict(num=args.parts),
        )

    for _ in range(args.warmup):
        step()
    if torch.cuda.is_available():
        torch.cuda.synchronize()
    t0 = time.perf_counter()
    out = None
    for _ in range(args.steps):
        out = step()
    if torch.cuda.is_available():
        torch.cuda.synchronize()
    el = time.perf_counter() - t0
    rows_per_sec = n * args.steps / el
    print(
        json.dumps(
            dict(
                metric="rows_per_sec_udf_boundary",
                value=rows_per_sec,
                unit="rows/s",
                n_gpus=1,
                steps=args.steps,
                warmup=args.warmup,
                ms_per_step=el / args.steps * 1000.0,
                higher_is_better=True,
                naive=bool(args.naive),
                data="synthetic",
                config=dict(
                    model="pandas-identity-transform",
                    parts=args.parts,
                    rows=n,
                    out_rows=out.count() if out is not None else 0,
                ),
            )
        ),
        flush=True,
    )


if __name__ == "__main__":
    main()
