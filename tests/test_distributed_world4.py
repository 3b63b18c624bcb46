"""World-4 gloo tests of distributed edge shapes the round-end 8-GPU
bench will hit: uneven shards, empty shards on some ranks, and
string-keyed shuffles."""
import os
import pickle
from typing import Any, Callable, Dict

import numpy as np
import pandas as pd
import torch.multiprocessing as mp

WORLD = 4


def _run_worker(rank: int, world: int, port: int, fn_bytes: bytes, out_q) -> None:
    os.environ["MASTER_ADDR"] = "127.0.0.1"
    os.environ["MASTER_PORT"] = str(port)
    os.environ["WORLD_SIZE"] = str(world)
    os.environ["RANK"] = str(rank)
    os.environ["LOCAL_RANK"] = str(rank)
    try:
        import torch.distributed as dist

        dist.init_process_group("gloo", rank=rank, world_size=world)
        from fugue_amd.parallel.comm import reset_communicator

        reset_communicator()
        fn = pickle.loads(fn_bytes)
        res = fn(rank)
        out_q.put((rank, "ok", res))
    except Exception:  # pragma: no cover
        import traceback

        out_q.put((rank, "error", traceback.format_exc()))
    finally:
        try:
            import torch.distributed as dist

            if dist.is_initialized():
                dist.destroy_process_group()
        except Exception:
            pass


def run_world4(fn: Callable[[int], Any], port: int) -> Dict[int, Any]:
    ctx = mp.get_context("spawn")
    q = ctx.Queue()
    procs = [
        ctx.Process(
            target=_run_worker, args=(r, WORLD, port, pickle.dumps(fn), q)
        )
        for r in range(WORLD)
    ]
    for p in procs:
        p.start()
    results: Dict[int, Any] = {}
    try:
        for _ in range(WORLD):
            rank, status, res = q.get(timeout=240)
            if status == "error":
                raise RuntimeError(f"rank {rank} failed:\n{res}")
            results[rank] = res
    finally:
        for p in procs:
            p.join(timeout=30)
            if p.is_alive():
                p.terminate()
    return results


def _uneven_job(rank: int):
    """Each rank holds a different row count (rank 2 none at all); the
    distributed groupby must still be globally exact."""
    import fugue_amd.api as fa
    from fugue_amd.column import functions as f
    from fugue_amd.column.expressions import col
    from fugue_amd.hip.execution_engine import HipExecutionEngine

    from fugue_amd import PandasDataFrame
    from fugue_amd.hip.frame import HipDataFrame

    e = HipExecutionEngine()
    n = 0 if rank == 2 else 1000 * (rank + 1) + 7
    rng = np.random.default_rng(100 + rank)
    pdf = pd.DataFrame(
        dict(
            k=rng.integers(0, 37, n).astype("int64"),
            v=np.arange(n, dtype="f8"),
        )
    )
    # per-rank shard (host pandas inputs are treated as replicated)
    shard = HipDataFrame(
        PandasDataFrame(pdf, "k:long,v:double").as_arrow(),
        "k:long,v:double",
        device=e.device,
    )
    res = fa.aggregate(
        shard, partition_by="k", engine=e,
        s=f.sum(col("v")), c=f.count(col("v")), as_fugue=True,
    )
    local = e._gather_all(res)
    got = local.as_pandas().sort_values("k").reset_index(drop=True)
    return dict(
        got=got.to_dict("list"),
        mine=pdf.to_dict("list"),
    )


def test_uneven_and_empty_shards():
    results = run_world4(_uneven_job, 29611)
    # rebuild the global input from every rank's shard and compare
    frames = [pd.DataFrame(results[r]["mine"]) for r in range(WORLD)]
    full = pd.concat(frames, ignore_index=True)
    exp = (
        full.groupby("k", as_index=False)
        .agg(s=("v", "sum"), c=("v", "count"))
        .sort_values("k")
        .reset_index(drop=True)
    )
    for r in range(WORLD):
        got = pd.DataFrame(results[r]["got"])
        assert got["k"].tolist() == exp["k"].tolist()
        np.testing.assert_allclose(got["s"], exp["s"], rtol=1e-12)
        assert got["c"].tolist() == exp["c"].tolist()


def _string_job(rank: int):
    """String-keyed groupby + join across ranks (string hash shuffle)."""
    import fugue_amd.api as fa
    from fugue_amd.column import functions as f
    from fugue_amd.column.expressions import col
    from fugue_amd.hip.execution_engine import HipExecutionEngine

    from fugue_amd import PandasDataFrame
    from fugue_amd.hip.frame import HipDataFrame

    e = HipExecutionEngine()
    cats = np.array(["apple", "banana", "cherry", "date", "elder", ""])
    rng = np.random.default_rng(7 + rank)
    n = 4000
    pdf = pd.DataFrame(
        dict(k=cats[rng.integers(0, len(cats), n)], v=rng.random(n))
    )
    shard = HipDataFrame(
        PandasDataFrame(pdf, "k:str,v:double").as_arrow(),
        "k:str,v:double",
        device=e.device,
    )
    agg = fa.aggregate(
        shard, partition_by="k", engine=e, s=f.sum(col("v")), as_fugue=True
    )
    # join with a small string-keyed dim sharded across ranks
    dim_all = pd.DataFrame(dict(k=cats, w=np.arange(len(cats), dtype="f8")))
    mine = dim_all.iloc[rank::4].reset_index(drop=True)
    dim_shard = HipDataFrame(
        PandasDataFrame(mine, "k:str,w:double").as_arrow(),
        "k:str,w:double",
        device=e.device,
    )
    joined = fa.join(agg, dim_shard, how="inner", engine=e, as_fugue=True)
    local = e._gather_all(joined)
    return dict(
        got=local.as_pandas().sort_values("k").reset_index(drop=True).to_dict("list"),
        mine=pdf.to_dict("list"),
    )


def test_string_key_shuffle():
    results = run_world4(_string_job, 29617)
    frames = [pd.DataFrame(results[r]["mine"]) for r in range(WORLD)]
    full = pd.concat(frames, ignore_index=True)
    cats = ["apple", "banana", "cherry", "date", "elder", ""]
    dim = pd.DataFrame(dict(k=cats, w=np.arange(len(cats), dtype="f8")))
    exp = (
        full.groupby("k", as_index=False)
        .agg(s=("v", "sum"))
        .merge(dim, on="k")
        .sort_values("k")
        .reset_index(drop=True)
    )
    for r in range(WORLD):
        got = pd.DataFrame(results[r]["got"])
        assert got["k"].tolist() == exp["k"].tolist()
        np.testing.assert_allclose(got["s"], exp["s"], rtol=1e-9)
        np.testing.assert_allclose(got["w"], exp["w"], rtol=1e-12)
