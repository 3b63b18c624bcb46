"""Interleaved A/B for group-by variants: alternates configs within one
process (several rounds) so box clock drift cancels; reports per-config
median.  Env knobs are re-read per call, so flipping os.environ between
steps selects the variant."""
import os
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import numpy as np
import torch

CONFIGS = [
    dict(FUGUE_GB_PARTS="1024"),
    dict(FUGUE_GB_PARTS="1024", FUGUE_GB_SCATTER_CHUNK="8192"),
    dict(FUGUE_GB_PARTS="1024", FUGUE_GB_SCATTER_CHUNK="32768"),
    dict(FUGUE_GB_PARTS="1024", FUGUE_GB_SCATTER_CHUNK="65536"),
    dict(FUGUE_GB_PARTS="1024", FUGUE_GB_AGG_CHUNK="8192"),
    dict(FUGUE_GB_PARTS="1024", FUGUE_GB_AGG_CHUNK="32768"),
    dict(FUGUE_GB_PARTS="2048"),
    dict(FUGUE_GB_PARTS="2048", FUGUE_GB_SCATTER_CHUNK="32768"),
    dict(FUGUE_GB_PARTS="512", FUGUE_SC_ILP="4"),
    dict(FUGUE_GB_PARTS="1024", FUGUE_GB_ILP="1"),
]
KNOBS = (
    "FUGUE_GB_PARTS", "FUGUE_GB_AGG_CHUNK", "FUGUE_GB_SCATTER_CHUNK",
    "FUGUE_SC_ILP", "FUGUE_GB_ILP",
)


def main() -> None:
    rows = int(sys.argv[1]) if len(sys.argv) > 1 else 125_000_000
    rounds = int(sys.argv[2]) if len(sys.argv) > 2 else 5
    import pyarrow as pa

    from fugue_amd.collections.partition import PartitionSpec
    from fugue_amd.column import functions as f
    from fugue_amd.column.expressions import col
    from fugue_amd.hip.execution_engine import HipExecutionEngine
    from fugue_amd.hip.frame import DeviceColumn, HipDataFrame
    from fugue_amd.schema import Schema

    engine = HipExecutionEngine()
    device = torch.device(engine.device)
    gen = torch.Generator(device=device)
    gen.manual_seed(42)
    keys = torch.randint(0, 1_000_000, (rows,), dtype=torch.int64,
                         device=device, generator=gen)
    vals = torch.rand(rows, dtype=torch.float64, device=device,
                      generator=gen)
    fact = HipDataFrame.from_columns(
        {"k": DeviceColumn(keys, None, pa.int64()),
         "v": DeviceColumn(vals, None, pa.float64())},
        Schema("k:long,v:double"), engine.device,
    )
    spec = PartitionSpec(by=["k"])
    agg_cols = [f.sum(col("v")).alias("s"), f.count(col("v")).alias("n")]

    def set_cfg(cfg):
        for k in KNOBS:
            os.environ.pop(k, None)
        os.environ.update(cfg)

    def time_steps(k):
        torch.cuda.synchronize()
        t0 = time.perf_counter()
        for _ in range(k):
            engine.aggregate(fact, spec, agg_cols)
        torch.cuda.synchronize()
        return (time.perf_counter() - t0) / k * 1000

    # warmup all configs once (allocs, JIT)
    for cfg in CONFIGS:
        set_cfg(cfg)
        time_steps(2)
    results = {i: [] for i in range(len(CONFIGS))}
    for r in range(rounds):
        for i, cfg in enumerate(CONFIGS):
            set_cfg(cfg)
            results[i].append(time_steps(5))
    for i, cfg in enumerate(CONFIGS):
        med = sorted(results[i])[len(results[i]) // 2]
        tag = ",".join(f"{k.split('_')[-1]}={v}" for k, v in cfg.items())
        print(f"{tag:40s} median={med:.3f} ms  all="
              f"{[round(x,2) for x in results[i]]}", flush=True)


if __name__ == "__main__":
    main()
