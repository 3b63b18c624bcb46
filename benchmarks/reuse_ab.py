"""Interleaved A/B of FUGUE_GB_LAYOUT_REUSE on the flagship and q3
pipelines (same box/process)."""
import importlib.util
import os
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import torch


def run_ab(step, label, rounds=5, k=8):
    def timed(n):
        torch.cuda.synchronize()
        t0 = time.perf_counter()
        for _ in range(n):
            step()
        torch.cuda.synchronize()
        return (time.perf_counter() - t0) / n * 1000

    for v in ("1", "0"):
        os.environ["FUGUE_GB_LAYOUT_REUSE"] = v
        timed(3)
    res = {"1": [], "0": []}
    for _ in range(rounds):
        for v in ("1", "0"):
            os.environ["FUGUE_GB_LAYOUT_REUSE"] = v
            res[v].append(timed(k))
    for v in ("1", "0"):
        med = sorted(res[v])[len(res[v]) // 2]
        print(f"{label} REUSE={v}: median {med:.3f} ms "
              f"all={[round(x,2) for x in res[v]]}", flush=True)


def main():
    import pyarrow as pa

    import fugue_amd.api as fa
    from fugue_amd.hip.execution_engine import HipExecutionEngine
    from fugue_amd.hip.frame import DeviceColumn, HipDataFrame
    from fugue_amd.schema import Schema

    engine = HipExecutionEngine()
    device = torch.device(engine.device)
    gen = torch.Generator(device=device)
    gen.manual_seed(42)
    n = 125_000_000
    fact = HipDataFrame.from_columns(
        {"k": DeviceColumn(torch.randint(0, 1_000_000, (n,),
                                         dtype=torch.int64, device=device,
                                         generator=gen), None, pa.int64()),
         "v": DeviceColumn(torch.rand(n, dtype=torch.float64, device=device,
                                      generator=gen), None, pa.float64())},
        Schema("k:long,v:double"), engine.device)
    dims = HipDataFrame.from_columns(
        {"k": DeviceColumn(torch.arange(0, 1_000_000, dtype=torch.int64,
                                        device=device), None, pa.int64()),
         "w": DeviceColumn(torch.rand(1_000_000, dtype=torch.float64,
                                      device=device, generator=gen),
                           None, pa.float64())},
        Schema("k:long,w:double"), engine.device)

    def scale(df: HipDataFrame) -> HipDataFrame:
        v = df.col("v")
        return HipDataFrame.from_columns(
            {"k": df.col("k"),
             "v": DeviceColumn(v.data * 1.000001, v.valid, pa.float64())},
            Schema("k:long,v:double"), df.device)

    SQL = ("t = TRANSFORM fact USING scale SCHEMA k:long,v:double\n"
           "agg = SELECT k, SUM(v) AS s, COUNT(v) AS n FROM t GROUP BY k\n"
           "res = SELECT agg.k, s, n, w FROM agg INNER JOIN dims "
           "ON agg.k = dims.k WHERE s > w\nYIELD DATAFRAME AS result\n")

    def flag_step():
        fa.fugue_sql(SQL, fact=fact, dims=dims, scale=scale, engine=engine,
                     as_fugue=True)

    run_ab(flag_step, "flagship")

    spec = importlib.util.spec_from_file_location(
        "q3b", os.path.join(os.path.dirname(__file__), "q3_bench.py"))
    q3 = importlib.util.module_from_spec(spec)
    spec.loader.exec_module(q3)
    customer, orders, lineitem, _ = q3.gen_tables(10.0, engine.device, 0)

    def q3_step():
        fa.fugue_sql(q3.Q3, customer=customer, orders=orders,
                     lineitem=lineitem, engine=engine, as_fugue=True)

    run_ab(q3_step, "q3")


if __name__ == "__main__":
    main()
