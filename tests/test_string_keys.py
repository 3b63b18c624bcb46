"""String group/join keys on the MI355X engine (CPU tensors here; the
same code paths run the string-hash HIP kernels on device — see
tests/test_hip_gpu.py for the device run)."""
import numpy as np
import pandas as pd
import pytest

import fugue_amd.api as fa
from fugue_amd.column.expressions import col
from fugue_amd.column import functions as f
from fugue_amd.hip.execution_engine import HipExecutionEngine


@pytest.fixture(scope="module")
def engine():
    return HipExecutionEngine()


def test_string_groupby(engine):
    rng = np.random.default_rng(0)
    cats = np.array(["apple", "banana", "cherry", "date", ""])
    pdf = pd.DataFrame(
        dict(k=cats[rng.integers(0, 5, 5000)], v=rng.random(5000))
    )
    res = fa.aggregate(
        pdf, partition_by="k", engine=engine,
        s=f.sum(col("v")), n=f.count(col("v")), as_fugue=True,
    )
    got = res.as_pandas().sort_values("k").reset_index(drop=True)
    exp = (
        pdf.groupby("k", as_index=False)
        .agg(s=("v", "sum"), n=("v", "count"))
        .sort_values("k")
        .reset_index(drop=True)
    )
    assert got["k"].tolist() == exp["k"].tolist()
    np.testing.assert_allclose(got["s"], exp["s"], rtol=1e-9)
    assert got["n"].tolist() == exp["n"].tolist()


def test_string_groupby_with_nulls(engine):
    pdf = pd.DataFrame(
        dict(k=["a", None, "a", None, "b"], v=[1.0, 2.0, 3.0, 4.0, 5.0])
    )
    res = fa.aggregate(
        pdf, partition_by="k", engine=engine, s=f.sum(col("v")), as_fugue=True
    )
    got = {r[0]: r[1] for r in res.as_array()}
    assert got["a"] == 4.0
    assert got["b"] == 5.0
    assert got[None] == 6.0


def test_mixed_string_int_keys(engine):
    rng = np.random.default_rng(1)
    pdf = pd.DataFrame(
        dict(
            s=np.array(["x", "y"])[rng.integers(0, 2, 1000)],
            g=rng.integers(0, 3, 1000),
            v=rng.random(1000),
        )
    )
    res = fa.aggregate(
        pdf, partition_by=["s", "g"], engine=engine,
        total=f.sum(col("v")), as_fugue=True,
    )
    got = (
        res.as_pandas().sort_values(["s", "g"]).reset_index(drop=True)
    )
    exp = (
        pdf.groupby(["s", "g"], as_index=False)
        .agg(total=("v", "sum"))
        .sort_values(["s", "g"])
        .reset_index(drop=True)
    )
    assert got["s"].tolist() == exp["s"].tolist()
    assert got["g"].tolist() == exp["g"].tolist()
    np.testing.assert_allclose(got["total"], exp["total"], rtol=1e-9)


@pytest.mark.parametrize("how", ["inner", "left_outer", "semi", "anti", "full_outer"])
def test_string_join(engine, how):
    left = pd.DataFrame(
        dict(k=["a", "b", "c", "a", None], x=[1.0, 2.0, 3.0, 4.0, 5.0])
    )
    right = pd.DataFrame(dict(k=["a", "c", "d"], y=[10.0, 30.0, 40.0]))
    exp = fa.join(left, right, how=how, engine="native")
    got = fa.join(left, right, how=how, engine=engine, as_fugue=True)
    from fugue_amd.dataframe.pandas_dataframe import PandasDataFrame
    from fugue_amd.dataframe.utils import _df_eq

    assert _df_eq(got.as_local_bounded(), PandasDataFrame(exp), throw=True)


def test_string_shuffle_sql(engine):
    rng = np.random.default_rng(2)
    pdf = pd.DataFrame(
        dict(
            name=np.array(["aa", "bb", "cc"])[rng.integers(0, 3, 2000)],
            v=rng.random(2000),
        )
    )
    res = fa.fugue_sql(
        "SELECT name, SUM(v) AS s FROM t GROUP BY name",
        t=pdf,
        engine=engine,
        as_fugue=True,
    )
    got = res.as_pandas().sort_values("name").reset_index(drop=True)
    exp = pdf.groupby("name", as_index=False).agg(s=("v", "sum"))
    np.testing.assert_allclose(got["s"], exp["s"], rtol=1e-9)


def test_datetime_keys_and_values(engine):
    import datetime

    pdf = pd.DataFrame(
        dict(
            ts=pd.to_datetime(
                ["2024-01-01", "2024-01-02", "2024-01-01", "2024-01-03"]
            ),
            v=[1.0, 2.0, 3.0, 4.0],
        )
    )
    res = fa.aggregate(
        pdf, partition_by="ts", engine=engine, s=f.sum(col("v")), as_fugue=True
    )
    got = res.as_pandas().sort_values("ts").reset_index(drop=True)
    exp = pdf.groupby("ts", as_index=False).agg(s=("v", "sum"))
    assert got["ts"].tolist() == exp["ts"].tolist()
    np.testing.assert_allclose(got["s"], exp["s"])
    # datetime round-trip
    d = engine.to_df(pdf)
    back = d.as_pandas()
    pd.testing.assert_frame_equal(back, pdf)
    # join on datetime keys
    dims = pd.DataFrame(
        dict(ts=pd.to_datetime(["2024-01-01", "2024-01-03"]), w=[10.0, 20.0])
    )
    j = fa.join(pdf, dims, how="inner", engine=engine, as_fugue=True)
    assert j.count() == 3


def test_map_num_partitions(engine):
    from typing import Any, List

    pdf = pd.DataFrame(dict(x=np.arange(10)))

    def count_rows(rows: List[List[Any]]) -> List[List[Any]]:
        return [[len(rows)]]

    res = fa.transform(
        pdf, count_rows, schema="n:long", partition=dict(num=5), engine=engine
    )
    counts = sorted(r for r in pd.DataFrame(res)["n"].tolist())
    assert sum(counts) == 10
    assert len(counts) == 5


def test_device_like_patterns():
    import pandas as pd

    from fugue_amd.column import functions as F
    from fugue_amd.column.expressions import col
    from fugue_amd.hip.execution_engine import HipExecutionEngine

    e = HipExecutionEngine()
    df = pd.DataFrame(
        dict(
            k=list(range(8)),
            s=["apple", "banana", "grape", "applet", "", "nap", None, "app"],
        )
    )
    d = e.to_df(df)
    cases = [
        ("app%", [0, 3, 7]),
        ("%ap", [5]),
        ("%ap%", [0, 2, 3, 5, 7]),
        ("app", [7]),
        ("%", [0, 1, 2, 3, 4, 5, 7]),
        ("", [4]),
        ("%zz%", []),
    ]
    for pat, exp in cases:
        r = e.filter(d, F.like(col("s"), pat)).as_pandas()
        assert sorted(r["k"].tolist()) == exp, pat
    # NOT LIKE excludes nulls (three-valued logic)
    r = e.filter(d, ~F.like(col("s"), "app%")).as_pandas()
    assert sorted(r["k"].tolist()) == [1, 2, 4, 5]


def test_sql_like_on_engine():
    import pandas as pd

    from fugue_amd.hip.execution_engine import HipExecutionEngine
    from fugue_amd.sql.executor import parse_select
    from fugue_amd.sql.planner import execute_plan

    e = HipExecutionEngine()
    df = pd.DataFrame(dict(k=[0, 1, 2], s=["foo", "bar", None]))
    d = e.to_df(df)
    r = execute_plan(
        parse_select("SELECT k FROM a WHERE s LIKE 'f%'"), dict(a=d), e
    ).as_pandas()
    assert r["k"].tolist() == [0]
    r2 = execute_plan(
        parse_select("SELECT k FROM a WHERE s NOT LIKE 'f%'"), dict(a=d), e
    ).as_pandas()
    assert r2["k"].tolist() == [1]


def _device_group_sum(df):
    import pyarrow as pa
    import torch

    from fugue_amd.hip.frame import DeviceColumn, HipDataFrame
    from fugue_amd.schema import Schema

    k = df.col("k").data[:1]
    s = df.col("v").data.sum().reshape(1)
    return HipDataFrame.from_columns(
        {
            "k": DeviceColumn(k, None, pa.int64()),
            "s": DeviceColumn(s.to(torch.float64), None, pa.float64()),
        },
        Schema("k:long,s:double"),
        df.device,
    )


def test_device_resident_udf():
    """HipDataFrame-annotated transformers get the HBM shard directly
    (fugue_polars pattern; reference fugue_polars/registry.py:24)."""
    import pandas as pd

    from fugue_amd.hip.execution_engine import HipExecutionEngine
    from fugue_amd.hip.frame import HipDataFrame
    from fugue_amd.workflow import transform

    _device_group_sum.__annotations__ = {
        "df": HipDataFrame,
        "return": HipDataFrame,
    }
    e = HipExecutionEngine()
    pdf = pd.DataFrame(dict(k=[1, 1, 2, 2, 2], v=[1.0, 2.0, 3.0, 4.0, 5.0]))
    res = transform(
        pdf,
        _device_group_sum,
        schema="k:long,s:double",
        partition=dict(by=["k"]),
        engine=e,
    )
    r = res if isinstance(res, pd.DataFrame) else res.as_pandas()
    r = r.sort_values("k").reset_index(drop=True)
    assert r["s"].tolist() == [3.0, 12.0]
