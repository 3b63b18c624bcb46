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

    def run(sc, ag, nt, narrow, iters=8):
        torch.cuda.synchronize()
        # warmup
        ext.gb_aggregate_partitioned(keys, vals, ops, 512, tsize, sc, ag, nt, narrow)
        torch.cuda.synchronize()
        t0 = time.perf_counter()
        for _ in range(iters):
            ext.gb_aggregate_partitioned(keys, vals, ops, 512, tsize, sc, ag, nt, narrow)
        torch.cuda.synchronize()
        return (time.perf_counter() - t0) / iters * 1000

    # correctness cross-check narrow vs not
    k0, a0, c0, _ = ext.gb_aggregate_partitioned(keys, vals, ops, 512, tsize, 0, 0, 0, 0)
    k1, a1, c1, _ = ext.gb_aggregate_partitioned(keys, vals, ops, 512, tsize, 0, 0, 0, 1)
    m0 = k0 != -9223372036854775808
    m1 = k1 != -9223372036854775808
    kk0, kk1 = k0[m0], k1[m1]
    aa0, aa1 = a0[0][m0], a1[0][m1]
    s0 = torch.argsort(kk0); s1 = torch.argsort(kk1)
    assert torch.equal(kk0[s0], kk1[s1]), "keys mismatch"
    assert torch.allclose(aa0[s0], aa1[s1]), "aggs mismatch"
    k2, a2, c2, _ = ext.gb_aggregate_partitioned(keys, vals, ops, 512, tsize, 0, 0, 0, -1)
    m2 = k2 != -9223372036854775808
    kk2, aa2 = k2[m2], a2[0][m2]
    s2 = torch.argsort(kk2)
    assert torch.equal(kk0[s0], kk2[s2]), "auto keys mismatch"
    assert torch.allclose(aa0[s0], aa2[s2]), "auto aggs mismatch"
    print("narrow correctness OK (explicit + auto)", flush=True)
    import os as _os

    for ilp, narrow in ((2, 1), (1, 1), (2, 1), (1, 1), (2, 0), (1, 0)):
        if ilp == 1:
            _os.environ["FUGUE_GB_ILP"] = "1"
        else:
            _os.environ.pop("FUGUE_GB_ILP", None)
        t = run(0, 0, 0, narrow)
        print(f"ilp={ilp} narrow={narrow}: {t:.3f} ms", flush=True)
    _os.environ.pop("FUGUE_GB_ILP", None)


if __name__ == "__main__":
    main()
