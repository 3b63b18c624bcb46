"""Host-side cProfile of the flagship FugueSQL step at full size (GPU).
Identifies per-step host overhead (full step minus engine_ops)."""
import cProfile, io, os, pstats, sys, time
sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import pyarrow as pa
import torch

import fugue_amd.api as fa
from fugue_amd.hip.execution_engine import HipExecutionEngine
from fugue_amd.hip.frame import DeviceColumn, HipDataFrame
from fugue_amd.schema import Schema

SQL = """
t = TRANSFORM fact USING scale SCHEMA k:long,v:double
agg = SELECT k, SUM(v) AS s, COUNT(v) AS n FROM t GROUP BY k
res = SELECT agg.k, s, n, w FROM agg INNER JOIN dims ON agg.k = dims.k
      WHERE s > w
YIELD DATAFRAME AS result
"""

def main():
    n = int(sys.argv[1]) if len(sys.argv) > 1 else 125_000_000
    steps = int(sys.argv[2]) if len(sys.argv) > 2 else 30
    engine = HipExecutionEngine()
    device = torch.device(engine.device)
    gen = torch.Generator(device=device); gen.manual_seed(42)
    keys = torch.randint(0, 1_000_000, (n,), dtype=torch.int64, device=device, generator=gen)
    vals = torch.rand(n, dtype=torch.float64, device=device, generator=gen)
    fact = HipDataFrame.from_columns(
        {"k": DeviceColumn(keys, None, pa.int64()),
         "v": DeviceColumn(vals, None, pa.float64())},
        Schema("k:long,v:double"), engine.device)
    dim_k = torch.arange(0, 1_000_000, dtype=torch.int64, device=device)
    dim_w = torch.rand(1_000_000, dtype=torch.float64, device=device, generator=gen)
    dims = HipDataFrame.from_columns(
        {"k": DeviceColumn(dim_k, None, pa.int64()),
         "w": DeviceColumn(dim_w, None, pa.float64())},
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

    def sync():
        if torch.cuda.is_available():
            torch.cuda.synchronize()

    for _ in range(5):
        step()
    sync()
    t0 = time.perf_counter()
    for _ in range(steps):
        step()
    sync()
    print(f"unprofiled: {(time.perf_counter()-t0)/steps*1000:.3f} ms/step")

    prof = cProfile.Profile(); prof.enable()
    for _ in range(steps):
        step()
    prof.disable()
    sync()
    s = io.StringIO()
    ps = pstats.Stats(prof, stream=s).sort_stats("cumulative")
    ps.print_stats(70)
    print(s.getvalue()[:14000])

if __name__ == "__main__":
    main()
