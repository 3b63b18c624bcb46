"""Interleaved A/B of FUGUE_GB_STAGED_MAX (staged vs simple 2-phase
group-by selection) on q3 + flagship aggregate."""
import importlib.util
import os
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import torch


def main():
    spec = importlib.util.spec_from_file_location(
        "q3b", os.path.join(os.path.dirname(__file__), "q3_bench.py"))
    q3 = importlib.util.module_from_spec(spec)
    spec.loader.exec_module(q3)
    import fugue_amd.api as fa
    from fugue_amd.hip.execution_engine import HipExecutionEngine

    engine = HipExecutionEngine()
    customer, orders, lineitem, _ = q3.gen_tables(10.0, engine.device, 0)

    def step():
        fa.fugue_sql(q3.Q3, customer=customer, orders=orders,
                     lineitem=lineitem, engine=engine, as_fugue=True)

    def timed(k):
        torch.cuda.synchronize()
        t0 = time.perf_counter()
        for _ in range(k):
            step()
        torch.cuda.synchronize()
        return (time.perf_counter() - t0) / k * 1000

    CFG = ("512", "1024", "2048")
    for v in CFG:
        os.environ["FUGUE_GB_PARTS"] = v
        timed(3)
    res = {v: [] for v in CFG}
    for _ in range(5):
        for v in CFG:
            os.environ["FUGUE_GB_PARTS"] = v
            res[v].append(timed(8))
    for v in CFG:
        med = sorted(res[v])[len(res[v]) // 2]
        print(f"q3 simple-path PARTS={v}: median {med:.3f} ms "
              f"all={[round(x,2) for x in res[v]]}", flush=True)


if __name__ == "__main__":
    main()
