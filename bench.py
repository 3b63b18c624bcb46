"""Flagship benchmark: rows/sec for the groupby-aggregate + join pipeline
(BASELINE.json metric: "rows/sec for transform()+FugueSQL groupby-join at
1e9 rows, 1/2/4/8 MI355X").

Weak scaling: each GPU owns a fixed shard (default 1e9/8 rows), so at 8
GPUs the global table is the named 1e9-row config.  Synthetic data
(random int64 keys + fp64 values) is generated on-device; the timed step
runs, through the engine API:

  1. hash group-by aggregation (SUM/COUNT per key) — CDNA4 kernels with
     LDS pre-aggregation, plus the cross-rank partial-merge shuffle
     (RCCL all-to-all over xGMI) when N>1,
  2. an inner hash join of the aggregate with a dimension table
     (broadcast join),
  3. a filter+projection on the joined result.

Usage:  python bench.py [--gpus N] [--steps K] [--warmup W] [--rows R]
The driver launches N>1 via torch.distributed.run (one rank per GPU).
"""
import argparse
import json
import os
import sys
import time

import numpy as np
import torch

ROWS_PER_GPU_DEFAULT = 125_000_000  # 1e9 / 8 GPUs
N_GROUPS = 1_000_000
DIM_ROWS = 1_000_000


def main() -> None:
    parser = argparse.ArgumentParser()
    parser.add_argument("--gpus", type=int, default=1)
    parser.add_argument("--steps", type=int, default=5)
    parser.add_argument("--warmup", type=int, default=2)
    parser.add_argument("--rows", type=int, default=ROWS_PER_GPU_DEFAULT,
                        help="rows per GPU (weak scaling)")
    args = parser.parse_args()

    world = int(os.environ.get("WORLD_SIZE", "1"))
    rank = int(os.environ.get("RANK", "0"))

    from fugue_amd.hip.execution_engine import HipExecutionEngine
    from fugue_amd.hip.frame import DeviceColumn, HipDataFrame
    from fugue_amd.collections.partition import PartitionSpec
    from fugue_amd.column.expressions import col
    from fugue_amd.column import functions as f
    import pyarrow as pa
    from fugue_amd.schema import Schema

    engine = HipExecutionEngine()
    device = torch.device(engine.device)

    n = args.rows
    gen = torch.Generator(device=device)
    gen.manual_seed(42 + rank)
    keys = torch.randint(0, N_GROUPS, (n,), dtype=torch.int64, device=device,
                         generator=gen)
    vals = torch.rand(n, dtype=torch.float64, device=device, generator=gen)
    fact = HipDataFrame.from_columns(
        {
            "k": DeviceColumn(keys, None, pa.int64()),
            "v": DeviceColumn(vals, None, pa.float64()),
        },
        Schema("k:long,v:double"),
        engine.device,
    )
    dim_gen = torch.Generator(device=device)
    dim_gen.manual_seed(7)  # identical dims on every rank (broadcast table)
    dim_k = torch.arange(0, DIM_ROWS, dtype=torch.int64, device=device)
    dim_w = torch.rand(DIM_ROWS, dtype=torch.float64, device=device,
                       generator=dim_gen)
    dims = HipDataFrame.from_columns(
        {
            "k": DeviceColumn(dim_k, None, pa.int64()),
            "w": DeviceColumn(dim_w, None, pa.float64()),
        },
        Schema("k:long,w:double"),
        engine.device,
    )
    dims.metadata["broadcasted"] = True  # every rank holds the full table

    spec = PartitionSpec(by=["k"])
    agg_cols = [
        f.sum(col("v")).alias("s"),
        f.count(col("v")).alias("n"),
    ]

    def step() -> int:
        agg = engine.aggregate(fact, spec, agg_cols)
        joined = engine.join(agg, dims, how="inner")
        res = engine.filter(joined, col("s") > col("w"))
        return res.count()

    def sync() -> None:
        if torch.cuda.is_available():
            torch.cuda.synchronize()
        engine.comm.barrier()

    for _ in range(args.warmup):
        step()
    sync()
    t0 = time.perf_counter()
    out_count = 0
    for _ in range(args.steps):
        out_count = step()
    sync()
    elapsed = time.perf_counter() - t0
    # MAX over ranks
    if engine.comm.is_distributed:
        import torch.distributed as dist

        t = torch.tensor([elapsed], dtype=torch.float64)
        if engine.comm.backend == "nccl":
            t = t.to(device)
        dist.all_reduce(t, op=dist.ReduceOp.MAX)
        elapsed = float(t.cpu().item())

    total_rows = n * world
    rows_per_sec = total_rows * args.steps / elapsed
    if rank == 0:
        print(
            json.dumps(
                dict(
                    metric="rows_per_sec_groupby_join",
                    value=rows_per_sec,
                    unit="rows/s",
                    n_gpus=world,
                    steps=args.steps,
                    warmup=args.warmup,
                    ms_per_step=elapsed / args.steps * 1000.0,
                    higher_is_better=True,
                    scaling="weak",
                    # measured comparator: the reference's pandas backend
                    # does this step at 3.46M rows/s on the host CPU
                    # (BASELINE.md "Measured comparator")
                    vs_baseline=round(rows_per_sec / 3.46e6, 1),
                    dtype="int64+fp64",
                    data="synthetic",
                    config=dict(
                        model="groupby(sum,count)+broadcast-join+filter",
                        global_rows=total_rows,
                        rows_per_gpu=n,
                        n_groups=N_GROUPS,
                        dim_rows=DIM_ROWS,
                        parallelism=f"dp{world}",
                        out_rows=out_count,
                    ),
                )
            ),
            flush=True,
        )


if __name__ == "__main__":
    main()
