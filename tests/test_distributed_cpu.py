"""Multi-process distributed-engine tests on CPU (gloo, world_size=2).

These exercise the exact orchestration the 8-GPU RCCL path runs
(shuffle, partial-aggregate merge, broadcast join, repartition) with the
kernel calls replaced by their CPU equivalents (``fugue_amd/hip/ops.py``
dispatches on tensor device).
"""
import os
import pickle
from typing import Any, Callable, Dict, List

import numpy as np
import pandas as pd
import pytest
import torch
import torch.multiprocessing as mp

WORLD = 2


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
    except Exception as e:  # pragma: no cover
        import traceback

        out_q.put((rank, "error", traceback.format_exc()))
    finally:
        try:
            import torch.distributed as dist

            if dist.is_initialized():
                dist.destroy_process_group()
        except Exception:
            pass


def run_distributed(fn: Callable[[int], Any], port: int) -> Dict[int, Any]:
    ctx = mp.get_context("spawn")
    q = ctx.Queue()
    procs = [
        ctx.Process(target=_run_worker, args=(r, WORLD, port, pickle.dumps(fn), q))
        for r in range(WORLD)
    ]
    for p in procs:
        p.start()
    results: Dict[int, Any] = {}
    try:
        for _ in range(WORLD):
            rank, status, res = q.get(timeout=180)
            if status == "error":
                raise RuntimeError(f"rank {rank} failed:\n{res}")
            results[rank] = res
    finally:
        for p in procs:
            p.join(timeout=30)
            if p.is_alive():
                p.terminate()
    return results


def _agg_job(rank: int):
    import fugue_amd.api as fa
    from fugue_amd.column.expressions import col
    from fugue_amd.column import functions as f
    from fugue_amd.hip.execution_engine import HipExecutionEngine

    e = HipExecutionEngine()
    assert e.is_distributed and e.world_size == WORLD
    rng = np.random.default_rng(0)  # same data on both ranks (driver-style)
    pdf = pd.DataFrame(dict(k=rng.integers(0, 50, 10000), v=rng.random(10000)))
    res = fa.aggregate(
        pdf, partition_by="k", engine=e,
        s=f.sum(col("v")), n=f.count(col("v")), as_fugue=True,
    )
    # gather for verification
    local = e._gather_all(res)  # noqa
    return local.as_pandas().sort_values("k").reset_index(drop=True).to_dict("list")


def test_distributed_aggregate():
    results = run_distributed(_agg_job, 29511)
    rng = np.random.default_rng(0)
    pdf = pd.DataFrame(dict(k=rng.integers(0, 50, 10000), v=rng.random(10000)))
    expected = (
        pdf.groupby("k", as_index=False)
        .agg(s=("v", "sum"), n=("v", "count"))
        .sort_values("k")
        .reset_index(drop=True)
    )
    for rank, got in results.items():
        assert got["k"] == expected["k"].tolist()
        np.testing.assert_allclose(got["s"], expected["s"].values, rtol=1e-9)
        assert got["n"] == expected["n"].tolist()


def _join_job(rank: int):
    import fugue_amd.api as fa
    from fugue_amd.hip.execution_engine import HipExecutionEngine

    e = HipExecutionEngine()
    rng = np.random.default_rng(1)
    left = pd.DataFrame(dict(k=rng.integers(0, 100, 5000), a=rng.random(5000)))
    right = pd.DataFrame(dict(k=np.arange(0, 150, 2), b=np.arange(75).astype("f8")))
    j = fa.join(left, right, how="inner", engine=e, as_fugue=True)
    local = e._gather_all(j)
    return sorted(map(tuple, local.as_array()))


def test_distributed_join():
    results = run_distributed(_join_job, 29513)
    rng = np.random.default_rng(1)
    left = pd.DataFrame(dict(k=rng.integers(0, 100, 5000), a=rng.random(5000)))
    right = pd.DataFrame(dict(k=np.arange(0, 150, 2), b=np.arange(75).astype("f8")))
    expected = sorted(map(tuple, left.merge(right, on="k").values.tolist()))
    for rank, got in results.items():
        assert len(got) == len(expected)
        got_r = [tuple(round(float(x), 9) for x in row) for row in got]
        exp_r = [tuple(round(float(x), 9) for x in row) for row in expected]
        assert sorted(got_r) == sorted(exp_r)


def _repartition_job(rank: int):
    from fugue_amd.collections.partition import PartitionSpec
    from fugue_amd.hip.execution_engine import HipExecutionEngine

    e = HipExecutionEngine()
    rng = np.random.default_rng(2)
    pdf = pd.DataFrame(dict(k=rng.integers(0, 20, 1000), v=np.arange(1000.0)))
    d = e.to_df(pdf)  # sharded
    local_before = d.count()
    shuffled = e.repartition(d, PartitionSpec(algo="hash", by=["k"]))
    keys_here = sorted(set(r[0] for r in shuffled.as_array()))
    total = e.comm.allreduce_sum(shuffled.count())
    even = e.repartition(d, PartitionSpec(algo="even", num=WORLD))
    even_count = even.count()
    return dict(
        local_before=local_before,
        keys_here=keys_here,
        total=total,
        even_count=even_count,
    )


def test_distributed_repartition():
    results = run_distributed(_repartition_job, 29515)
    # all 1000 rows preserved
    assert results[0]["total"] == 1000
    # hash partition: key sets disjoint across ranks
    k0 = set(results[0]["keys_here"])
    k1 = set(results[1]["keys_here"])
    assert k0.isdisjoint(k1)
    # even: both ranks have 500
    assert results[0]["even_count"] == 500
    assert results[1]["even_count"] == 500


def _transform_job(rank: int):
    import fugue_amd.api as fa
    from fugue_amd.hip.execution_engine import HipExecutionEngine

    e = HipExecutionEngine()
    rng = np.random.default_rng(3)
    pdf = pd.DataFrame(dict(g=rng.integers(0, 10, 2000), v=rng.random(2000)))

    # schema: g:long,total:double
    def per_group(df: pd.DataFrame) -> pd.DataFrame:
        return pd.DataFrame(dict(g=[df["g"].iloc[0]], total=[df["v"].sum()]))

    res = fa.transform(pdf, per_group, partition=dict(by=["g"]), engine=e, as_fugue=True)
    local = e._gather_all(res)
    return local.as_pandas().sort_values("g").reset_index(drop=True).to_dict("list")


def test_distributed_transform():
    results = run_distributed(_transform_job, 29517)
    rng = np.random.default_rng(3)
    pdf = pd.DataFrame(dict(g=rng.integers(0, 10, 2000), v=rng.random(2000)))
    expected = pdf.groupby("g", as_index=False).agg(total=("v", "sum"))
    for rank, got in results.items():
        assert got["g"] == expected["g"].tolist()
        np.testing.assert_allclose(got["total"], expected["total"].values, rtol=1e-9)


def _q3_job(rank: int):
    import sys

    sys.path.insert(0, os.path.join(os.path.dirname(__file__), ".."))
    from benchmarks.q3_bench import Q3, gen_tables
    import fugue_amd.api as fa
    from fugue_amd.hip.execution_engine import HipExecutionEngine

    e = HipExecutionEngine()
    c, o, l, _ = gen_tables(0.005, "cpu", 0)
    # pandas input → to_df shards it per rank (driver-replicated input)
    res = fa.fugue_sql(
        Q3,
        customer=c.as_pandas(),
        orders=o.as_pandas(),
        lineitem=l.as_pandas(),
        engine=e,
        as_fugue=True,
    )
    local = e._gather_all(res)
    return sorted(map(tuple, local.as_array()))


def test_distributed_q3():
    """The 3-way-join+groupby FugueSQL pipeline, world_size=2.  Tables are
    generated identically on both ranks (rank arg pinned), then sharded by
    to_df — the result must equal the single-process answer."""
    results = run_distributed(_q3_job, 29519)
    import sys

    sys.path.insert(0, os.path.join(os.path.dirname(__file__), ".."))
    from benchmarks.q3_bench import Q3, gen_tables
    import fugue_amd.api as fa
    from fugue_amd.hip.execution_engine import HipExecutionEngine

    e = HipExecutionEngine()
    customer, orders, lineitem, _ = gen_tables(0.005, e.device, 0)
    expected = fa.fugue_sql(
        Q3, customer=customer, orders=orders, lineitem=lineitem,
        engine=e, as_fugue=True,
    )
    exp = sorted(map(tuple, expected.as_array()))
    for rank, got in results.items():
        assert len(got) == len(exp)
        for g, x in zip(sorted(got), exp):
            assert g[0] == x[0] and abs(g[1] - x[1]) < 1e-6


def _io_job(rank: int):
    import tempfile

    import fugue_amd.api as fa
    from fugue_amd.hip.execution_engine import HipExecutionEngine

    e = HipExecutionEngine()
    pdf = pd.DataFrame(dict(a=np.arange(100), b=np.arange(100) * 1.5))
    d = e.to_df(pdf)  # sharded
    path = os.path.join(os.environ["FUGUE_TEST_TMP"], "out")
    e.save_df(d, path)
    back = e.load_df(path)
    total = e.comm.allreduce_sum(back.count())
    local = e._gather_all(back) if hasattr(back, "as_arrow") else back
    s = sorted(r[0] for r in local.as_array())
    # csv part-file round trip
    cpath = os.path.join(os.environ["FUGUE_TEST_TMP"], "out.csv")
    e.save_df(d, cpath, format_hint="csv")
    cback = e.load_df(cpath)
    ctotal = e.comm.allreduce_sum(cback.count())
    return dict(total=total, keys=s, ctotal=ctotal)


def test_distributed_parquet_parts(tmp_path):
    os.environ["FUGUE_TEST_TMP"] = str(tmp_path)
    try:
        results = run_distributed(_io_job, 29521)
        assert results[0]["total"] == 100
        assert results[0]["keys"] == list(range(100))
        assert results[0]["ctotal"] == 100
        files = os.listdir(os.path.join(str(tmp_path), "out"))
        assert len([f for f in files if f.startswith("part-")]) == 2
    finally:
        os.environ.pop("FUGUE_TEST_TMP", None)


def _empty_shard_job(rank: int):
    import fugue_amd.api as fa
    from fugue_amd.column.expressions import col
    from fugue_amd.column import functions as f
    from fugue_amd.hip.execution_engine import HipExecutionEngine

    e = HipExecutionEngine()
    # rank 1's shard is emptied by the filter (values 50..99 live there)
    pdf = pd.DataFrame(dict(k=np.arange(100) % 7, v=np.arange(100.0)))
    d = e.to_df(pdf)
    filtered = e.filter(d, col("v") < 50)
    agg = fa.aggregate(
        filtered, partition_by="k", engine=e, s=f.sum(col("v")), as_fugue=True
    )
    local = e._gather_all(agg)
    j = fa.join(
        agg,
        pd.DataFrame(dict(k=np.arange(7), w=np.arange(7.0))),
        how="inner",
        engine=e,
        as_fugue=True,
    )
    jn = e.comm.allreduce_sum(j.count())
    return dict(rows=sorted(map(tuple, local.as_array())), join_count=jn)


def test_distributed_empty_shard():
    results = run_distributed(_empty_shard_job, 29523)
    pdf = pd.DataFrame(dict(k=np.arange(100) % 7, v=np.arange(100.0)))
    sub = pdf[pdf.v < 50]
    expected = sorted(
        map(
            tuple,
            sub.groupby("k", as_index=False).agg(s=("v", "sum")).values.tolist(),
        )
    )
    for rank, got in results.items():
        assert [tuple(map(float, r)) for r in got["rows"]] == [
            tuple(map(float, r)) for r in expected
        ]
        assert got["join_count"] == 7


def _firstlast_cd_job(rank: int):
    import fugue_amd.api as fa
    from fugue_amd.column import functions as f
    from fugue_amd.column.expressions import col
    from fugue_amd.hip.execution_engine import HipExecutionEngine

    e = HipExecutionEngine()
    rng = np.random.default_rng(3)
    pdf = pd.DataFrame(
        dict(
            k=rng.integers(0, 20, 4000),
            v=rng.integers(0, 5, 4000).astype("f8"),
        )
    )
    res = fa.aggregate(
        pdf, partition_by="k", engine=e,
        cd=f.count_distinct(col("v")),
        fv=f.first(col("v")),
        as_fugue=True,
    )
    local = e._gather_all(res)
    return local.as_pandas().sort_values("k").reset_index(drop=True).to_dict("list")


def test_distributed_count_distinct_first():
    results = run_distributed(_firstlast_cd_job, 29525)
    rng = np.random.default_rng(3)
    pdf = pd.DataFrame(
        dict(
            k=rng.integers(0, 20, 4000),
            v=rng.integers(0, 5, 4000).astype("f8"),
        )
    )
    expected = (
        pdf.groupby("k", as_index=False)
        .agg(cd=("v", "nunique"))
        .sort_values("k")
        .reset_index(drop=True)
    )
    member_vals = pdf.groupby("k")["v"].agg(lambda s: set(s)).to_dict()
    for rank, got in results.items():
        assert got["k"] == expected["k"].tolist()
        assert got["cd"] == expected["cd"].tolist()
        for k, fv in zip(got["k"], got["fv"]):
            assert fv in member_vals[k]


def _rowops_job(rank: int):
    from fugue_amd.hip.execution_engine import HipExecutionEngine

    e = HipExecutionEngine()
    rng = np.random.default_rng(5)
    pdf = pd.DataFrame(
        dict(
            k=rng.integers(0, 30, 3000).astype("f8"),
            v=rng.random(3000),
        )
    )
    pdf.loc[pdf.index % 7 == 0, "k"] = np.nan
    d = e.to_df(pdf)
    out = {}
    out["dropna"] = e._gather_all(e.dropna(d)).as_pandas().shape[0]
    out["fillna"] = float(
        e._gather_all(e.fillna(d, 0.0)).as_pandas()["k"].sum()
    )
    out["distinct"] = e._gather_all(
        e.distinct(e._select_columns_by_names(d, ["k"]))
        if hasattr(e, "_select_columns_by_names")
        else e.distinct(d)
    ).as_pandas().shape[0]
    t = e._gather_all(e.take(d, 5, presort="v desc")).as_pandas()
    out["take"] = t["v"].tolist()
    s = e._gather_all(e.sample(d, frac=0.5, seed=3)).as_pandas().shape[0]
    out["sample_n"] = s
    return out


def test_distributed_row_ops():
    results = run_distributed(_rowops_job, 29527)
    rng = np.random.default_rng(5)
    pdf = pd.DataFrame(
        dict(
            k=rng.integers(0, 30, 3000).astype("f8"),
            v=rng.random(3000),
        )
    )
    pdf.loc[pdf.index % 7 == 0, "k"] = np.nan
    exp_drop = pdf.dropna().shape[0]
    exp_fill = float(pdf["k"].fillna(0.0).sum())
    exp_take = pdf.nlargest(5, "v")["v"].tolist()
    for rank, got in results.items():
        assert got["dropna"] == exp_drop
        assert abs(got["fillna"] - exp_fill) < 1e-6
        assert got["take"] == exp_take
        # bernoulli sample: expect roughly half (loose bounds)
        assert 1000 < got["sample_n"] < 2000


def _setops_job(rank: int):
    from fugue_amd.hip.execution_engine import HipExecutionEngine

    e = HipExecutionEngine()
    rng = np.random.default_rng(6)
    a = pd.DataFrame(dict(x=rng.integers(0, 200, 1000)))
    b = pd.DataFrame(dict(x=rng.integers(100, 300, 1000)))
    da, db = e.to_df(a), e.to_df(b)
    out = {}
    out["union"] = sorted(
        e._gather_all(e.union(da, db, distinct=True)).as_pandas()["x"].tolist()
    )
    out["sub"] = sorted(
        e._gather_all(e.subtract(da, db, distinct=True)).as_pandas()["x"].tolist()
    )
    out["inter"] = sorted(
        e._gather_all(e.intersect(da, db, distinct=True)).as_pandas()["x"].tolist()
    )
    return out


def test_distributed_set_ops():
    results = run_distributed(_setops_job, 29529)
    rng = np.random.default_rng(6)
    a = set(rng.integers(0, 200, 1000).tolist())
    b = set(rng.integers(100, 300, 1000).tolist())
    for rank, got in results.items():
        assert got["union"] == sorted(a | b)
        assert got["sub"] == sorted(a - b)
        assert got["inter"] == sorted(a & b)


def _global_agg_job(rank: int):
    from fugue_amd.column import functions as f
    from fugue_amd.column.expressions import col
    from fugue_amd.hip.execution_engine import HipExecutionEngine

    e = HipExecutionEngine()
    pdf = pd.DataFrame(dict(v=np.arange(100, dtype="f8")))
    d = e.to_df(pdf)  # sharded
    r = e.aggregate(
        d, None,
        [f.sum(col("v")).alias("s"), f.count(col("v")).alias("c"),
         f.min(col("v")).alias("mn"), f.avg(col("v")).alias("av")],
    )
    local = e._gather_all(r)
    return local.as_array()


def test_distributed_global_aggregate():
    results = run_distributed(_global_agg_job, 29535)
    for rank, rows in results.items():
        assert len(rows) == 1
        s, c, mn, av = rows[0]
        assert s == sum(range(100)) and c == 100 and mn == 0.0
        assert abs(av - 49.5) < 1e-9


def _sum_distinct_job(rank: int):
    import fugue_amd.api as fa
    from fugue_amd.column import functions as f
    from fugue_amd.column.expressions import _UnaryAggFuncExpr, col
    from fugue_amd.hip.execution_engine import HipExecutionEngine

    e = HipExecutionEngine()
    rng = np.random.default_rng(13)
    pdf = pd.DataFrame(
        dict(
            k=rng.integers(0, 12, 3000),
            v=rng.integers(0, 7, 3000).astype("f8"),
        )
    )
    res = fa.aggregate(
        pdf, partition_by="k", engine=e,
        sd=_UnaryAggFuncExpr("SUM", col("v"), arg_distinct=True),
        tot=f.sum(col("v")),
        as_fugue=True,
    )
    local = e._gather_all(res)
    return (
        local.as_pandas().sort_values("k").reset_index(drop=True).to_dict("list")
    )


def test_distributed_sum_distinct():
    """SUM DISTINCT through the dedupe decomposition across 2 ranks
    (each rank holds a copy of the frame → sharded by the engine)."""
    results = run_distributed(_sum_distinct_job, 29531)
    rng = np.random.default_rng(13)
    pdf = pd.DataFrame(
        dict(
            k=rng.integers(0, 12, 3000),
            v=rng.integers(0, 7, 3000).astype("f8"),
        )
    )
    expected = (
        pdf.groupby("k", as_index=False)
        .agg(
            sd=("v", lambda s: s.drop_duplicates().sum()),
            tot=("v", "sum"),
        )
        .sort_values("k")
        .reset_index(drop=True)
    )
    for rank, got in results.items():
        assert got["k"] == expected["k"].tolist()
        assert got["sd"] == expected["sd"].tolist()
        assert got["tot"] == pytest.approx(expected["tot"].tolist())
