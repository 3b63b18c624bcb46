"""Trace every torch op issued during one benchmark step, with the
fugue_amd call site — maps library-kernel residue in rocprof profiles
(elementwise/reduce/copyBuffer/fill) back to source lines.

Usage: python benchmarks/op_trace.py [q3|flagship] [rows]
"""
import collections
import os
import sys
import traceback

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import torch
from torch.utils._python_dispatch import TorchDispatchMode


class OpLogger(TorchDispatchMode):
    def __init__(self):
        super().__init__()
        self.counts = collections.Counter()

    def __torch_dispatch__(self, func, types, args=(), kwargs=None):
        site = "?"
        for fr in reversed(traceback.extract_stack()):
            if "/fugue_amd/" in fr.filename and "op_trace" not in fr.filename:
                site = f"{fr.filename.split('/fugue_amd/')[-1]}:{fr.lineno}"
                break
        self.counts[(str(func), site)] += 1
        return func(*args, **(kwargs or {}))


def main():
    which = sys.argv[1] if len(sys.argv) > 1 else "q3"
    rows = int(sys.argv[2]) if len(sys.argv) > 2 else 5_000_000

    if which == "q3":
        sys.argv = ["q3_bench.py"]
        import importlib.util
        spec = importlib.util.spec_from_file_location(
            "q3b", os.path.join(os.path.dirname(__file__), "q3_bench.py"))
        q3 = importlib.util.module_from_spec(spec)
        spec.loader.exec_module(q3)
        import fugue_amd.api as fa
        from fugue_amd.hip.execution_engine import HipExecutionEngine
        engine = HipExecutionEngine()
        sf = rows / 7_650_000
        customer, orders, lineitem, _ = q3.gen_tables(sf, engine.device, 0)

        def step():
            return fa.fugue_sql(q3.Q3, customer=customer, orders=orders,
                                lineitem=lineitem, engine=engine,
                                as_fugue=True)
        step()  # warmup
        log = OpLogger()
        with log:
            step()
    else:
        import pyarrow as pa
        import fugue_amd.api as fa
        from fugue_amd.hip.execution_engine import HipExecutionEngine
        from fugue_amd.hip.frame import DeviceColumn, HipDataFrame
        from fugue_amd.schema import Schema
        SQL = (
            "t = TRANSFORM fact USING scale SCHEMA k:long,v:double\n"
            "agg = SELECT k, SUM(v) AS s, COUNT(v) AS n FROM t GROUP BY k\n"
            "res = SELECT agg.k, s, n, w FROM agg INNER JOIN dims "
            "ON agg.k = dims.k WHERE s > w\n"
            "YIELD DATAFRAME AS result\n"
        )
        engine = HipExecutionEngine()
        device = torch.device(engine.device)
        gen = torch.Generator(device=device); gen.manual_seed(0)
        keys = torch.randint(0, 1_000_000, (rows,), dtype=torch.int64,
                             device=device, generator=gen)
        vals = torch.rand(rows, dtype=torch.float64, device=device,
                          generator=gen)
        fact = HipDataFrame.from_columns(
            {"k": DeviceColumn(keys, None, pa.int64()),
             "v": DeviceColumn(vals, None, pa.float64())},
            Schema("k:long,v:double"), engine.device)
        dk = torch.arange(0, 1_000_000, dtype=torch.int64, device=device)
        dw = torch.rand(1_000_000, dtype=torch.float64, device=device,
                        generator=gen)
        dims = HipDataFrame.from_columns(
            {"k": DeviceColumn(dk, None, pa.int64()),
             "w": DeviceColumn(dw, None, pa.float64())},
            Schema("k:long,w:double"), engine.device)

        def scale(df: HipDataFrame) -> HipDataFrame:
            v = df.col("v")
            return HipDataFrame.from_columns(
                {"k": df.col("k"),
                 "v": DeviceColumn(v.data * 1.000001, v.valid, pa.float64())},
                Schema("k:long,v:double"), df.device)

        def step():
            return fa.fugue_sql(SQL, fact=fact, dims=dims, scale=scale,
                                engine=engine, as_fugue=True)
        step()
        log = OpLogger()
        with log:
            step()

    print(f"== torch ops in ONE {which} step ==")
    total = sum(log.counts.values())
    for (op, site), c in log.counts.most_common(60):
        print(f"{c:5d}  {op:45s} {site}")
    print(f"total dispatched ops: {total}")


if __name__ == "__main__":
    main()
