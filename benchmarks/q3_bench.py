"""TPC-H-Q3-like FugueSQL benchmark (BASELINE config #4).

Synthetic customer/orders/lineitem of TPC-H shape (integer keys, int-day
dates, string market segment); the 3-way hash join + group-by aggregate +
top-10 runs through FugueSQL on the MI355X engine (plan lowering →
device kernels).  SF=1 ≈ 6M lineitem rows (TPC-H row-count ratios).

Usage: python benchmarks/q3_bench.py [--sf N] [--steps K] [--warmup W]
Multi-GPU: launch via torch.distributed.run (one rank per GPU; weak
scaling — each rank generates SF worth of data).
"""
import argparse
import json
import os
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import numpy as np
import torch

Q3 = """
SELECT orderkey, SUM(extendedprice * (1 - discount)) AS revenue,
       orderdate, shippriority
FROM customer INNER JOIN orders ON customer.custkey = orders.custkey
     INNER JOIN lineitem ON orders.orderkey = lineitem.orderkey
WHERE mktsegment = 'BUILDING' AND orderdate < 9204 AND shipdate > 9204
GROUP BY orderkey, orderdate, shippriority
ORDER BY revenue DESC
LIMIT 10
"""

SEGMENTS = ["AUTOMOBILE", "BUILDING", "FURNITURE", "MACHINERY", "HOUSEHOLD"]


def gen_tables(sf: float, device: str, rank: int):
    import pyarrow as pa

    from fugue_amd.hip.frame import DeviceColumn, HipDataFrame, StringDeviceColumn
    from fugue_amd.schema import Schema

    n_cust = int(150_000 * sf)
    n_ord = int(1_500_000 * sf)
    n_li = int(6_000_000 * sf)
    dev = torch.device(device)
    gen = torch.Generator(device=dev)
    gen.manual_seed(13 + rank)

    seg_codes = torch.randint(0, 5, (n_cust,), device=dev, generator=gen)
    seg_lengths = torch.tensor(
        [len(s) for s in SEGMENTS], dtype=torch.int64, device=dev
    )
    # build the segment string column from codes
    lens = seg_lengths.index_select(0, seg_codes)
    offsets = torch.zeros(n_cust + 1, dtype=torch.int64, device=dev)
    torch.cumsum(lens, 0, out=offsets[1:])
    seg_bytes_np = np.concatenate(
        [np.frombuffer(s.encode(), dtype=np.uint8) for s in SEGMENTS]
    )
    seg_offs = np.concatenate(([0], np.cumsum([len(s) for s in SEGMENTS])))
    # gather bytes per row on host once (cheap relative to joins) — codes→bytes
    codes_np = seg_codes.cpu().numpy()
    bytes_np = np.concatenate(
        [seg_bytes_np[seg_offs[c] : seg_offs[c + 1]] for c in codes_np]
    ) if n_cust > 0 else np.empty(0, dtype=np.uint8)
    seg_col = StringDeviceColumn(
        offsets, torch.from_numpy(bytes_np).to(dev), None
    )
    customer = HipDataFrame.from_columns(
        {
            "custkey": DeviceColumn(
                torch.arange(n_cust, dtype=torch.int64, device=dev),
                None,
                pa.int64(),
            ),
            "mktsegment": seg_col,
        },
        Schema("custkey:long,mktsegment:str"),
        device,
    )
    orders = HipDataFrame.from_columns(
        {
            "orderkey": DeviceColumn(
                torch.arange(n_ord, dtype=torch.int64, device=dev),
                None,
                pa.int64(),
            ),
            "custkey": DeviceColumn(
                torch.randint(0, max(n_cust, 1), (n_ord,), device=dev,
                              generator=gen),
                None,
                pa.int64(),
            ),
            "orderdate": DeviceColumn(
                torch.randint(8766, 11192, (n_ord,), device=dev, generator=gen),
                None,
                pa.int64(),
            ),
            "shippriority": DeviceColumn(
                torch.zeros(n_ord, dtype=torch.int64, device=dev),
                None,
                pa.int64(),
            ),
        },
        Schema("orderkey:long,custkey:long,orderdate:long,shippriority:long"),
        device,
    )
    lineitem = HipDataFrame.from_columns(
        {
            "orderkey": DeviceColumn(
                torch.randint(0, max(n_ord, 1), (n_li,), device=dev,
                              generator=gen),
                None,
                pa.int64(),
            ),
            "extendedprice": DeviceColumn(
                torch.rand(n_li, dtype=torch.float64, device=dev,
                           generator=gen) * 100000,
                None,
                pa.float64(),
            ),
            "discount": DeviceColumn(
                torch.rand(n_li, dtype=torch.float64, device=dev,
                           generator=gen) * 0.1,
                None,
                pa.float64(),
            ),
            "shipdate": DeviceColumn(
                torch.randint(8766, 11192, (n_li,), device=dev, generator=gen),
                None,
                pa.int64(),
            ),
        },
        Schema(
            "orderkey:long,extendedprice:double,discount:double,shipdate:long"
        ),
        device,
    )
    return customer, orders, lineitem, n_cust + n_ord + n_li


def main() -> None:
    parser = argparse.ArgumentParser()
    parser.add_argument("--sf", type=float, default=10.0)
    parser.add_argument("--steps", type=int, default=3)
    parser.add_argument("--warmup", type=int, default=1)
    args = parser.parse_args()
    world = int(os.environ.get("WORLD_SIZE", "1"))
    rank = int(os.environ.get("RANK", "0"))

    import fugue_amd.api as fa
    from fugue_amd.hip.execution_engine import HipExecutionEngine

    engine = HipExecutionEngine()
    customer, orders, lineitem, total_rows = gen_tables(
        args.sf, engine.device, rank
    )

    def step():
        return fa.fugue_sql(
            Q3,
            customer=customer,
            orders=orders,
            lineitem=lineitem,
            engine=engine,
            as_fugue=True,
        )

    def sync():
        if torch.cuda.is_available():
            torch.cuda.synchronize()
        engine.comm.barrier()

    for _ in range(args.warmup):
        step()
    sync()
    t0 = time.perf_counter()
    res = None
    for _ in range(args.steps):
        res = step()
    sync()
    elapsed = time.perf_counter() - t0
    if engine.comm.is_distributed:
        import torch.distributed as dist

        t = torch.tensor([elapsed], dtype=torch.float64)
        if engine.comm.backend == "nccl":
            t = t.to(torch.device(engine.device))
        dist.all_reduce(t, op=dist.ReduceOp.MAX)
        elapsed = float(t.cpu().item())
    rows_per_sec = total_rows * world * args.steps / elapsed
    if rank == 0:
        print(
            json.dumps(
                dict(
                    metric="rows_per_sec_tpch_q3_like",
                    value=rows_per_sec,
                    unit="rows/s",
                    n_gpus=world,
                    steps=args.steps,
                    warmup=args.warmup,
                    ms_per_step=elapsed / args.steps * 1000.0,
                    higher_is_better=True,
                    scaling="weak",
                    vs_baseline=round(rows_per_sec / 1.63e6, 1),
                    dtype="int64+fp64+str",
                    data="synthetic",
                    config=dict(
                        model="fuguesql-3way-join-groupby-top10",
                        sf_per_gpu=args.sf,
                        rows_per_gpu=total_rows,
                        parallelism=f"dp{world}",
                        top1=res.as_array()[0] if res is not None and res.count() else None,
                    ),
                )
            ),
            flush=True,
        )

    # orderly distributed teardown: a rank exiting while peers still
    # hold gloo/NCCL state can SIGABRT in the transport destructor
    if world > 1:
        import torch.distributed as dist

        dist.barrier()
        dist.destroy_process_group()


if __name__ == "__main__":
    main()
