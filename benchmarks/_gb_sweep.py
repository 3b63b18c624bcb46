"""Chunk-size sweep for the partitioned groupby kernels (GPU diagnostic).

Times `ext.gb_aggregate_partitioned` directly on the bench's shape
(125M rows, 1M groups, 1 fp64 SUM) across runtime scatter/agg chunk sizes.
Run under gpurun; prints one line per config.
"""
import os
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import torch

from fugue_amd.hip.ext import get_ext


def main() -> None:
    ext = get_ext()
    dev = torch.device("cuda:0")
    n = 125_000_000
    g = torch.Generator(device=dev).manual_seed(1)
    keys = torch.randint(0, 1_000_000, (n,), device=dev, generator=g)
    vals = torch.rand((1, n), device=dev, dtype=torch.float64, generator=g)
    ops = torch.tensor([0], dtype=torch.int32, device=dev)  # SUM
    tsize = 1 << 22  # 4.2M slots for ~1M groups

    def run(sc, ag, iters=6):
        torch.cuda.synchronize()
        # warmup
        ext.gb_aggregate_partitioned(keys, vals, ops, 512, tsize, sc, ag)
        torch.cuda.synchronize()
        t0 = time.perf_counter()
        for _ in range(iters):
            ext.gb_aggregate_partitioned(keys, vals, ops, 512, tsize, sc, ag)
        torch.cuda.synchronize()
        return (time.perf_counter() - t0) / iters * 1000

    base = run(0, 0)
    print(f"baseline (compiled defaults): {base:.3f} ms", flush=True)
    for ag in (4096, 8192, 16384, 32768):
        t = run(0, ag)
        print(f"scatter=default agg_chunk={ag:6d}: {t:.3f} ms", flush=True)
    for sc, ag in ((4096, 16384), (4096, 32768), (2048, 16384)):
        t = run(sc, ag)
        print(f"scatter_chunk={sc:6d} agg_chunk={ag:6d}: {t:.3f} ms", flush=True)


if __name__ == "__main__":
    main()
