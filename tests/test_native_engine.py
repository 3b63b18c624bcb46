import pandas as pd
import pytest

import fugue_amd.api as fa
from fugue_amd import ArrayDataFrame, PandasDataFrame
from fugue_amd.collections.partition import PartitionSpec
from fugue_amd.column.expressions import col, lit
from fugue_amd.column.sql import SelectColumns
from fugue_amd.column import functions as f
from fugue_amd.dataframe.dataframes import DataFrames
from fugue_amd.dataframe.utils import _df_eq
from fugue_amd.execution import NativeExecutionEngine


@pytest.fixture
def engine():
    return NativeExecutionEngine()


def test_joins(engine):
    a = engine.to_df([[1, "x"], [2, "y"], [3, "z"]], "k:long,a:str")
    b = engine.to_df([[2, 20.0], [3, 30.0], [4, 40.0]], "k:long,b:double")
    assert _df_eq(
        engine.join(a, b, "inner"),
        [[2, "y", 20.0], [3, "z", 30.0]],
        "k:long,a:str,b:double",
        throw=True,
    )
    assert engine.join(a, b, "left_outer").count() == 3
    assert engine.join(a, b, "right_outer").count() == 3
    assert engine.join(a, b, "full_outer").count() == 4
    assert _df_eq(
        engine.join(a, b, "semi"), [[2, "y"], [3, "z"]], "k:long,a:str"
    )
    assert _df_eq(engine.join(a, b, "anti"), [[1, "x"]], "k:long,a:str")
    c = engine.to_df([[9.0]], "c:double")
    assert engine.join(a, c, "cross").count() == 3


def test_set_ops(engine):
    a = engine.to_df([[1], [2], [2]], "x:long")
    b = engine.to_df([[2], [3]], "x:long")
    assert sorted(engine.union(a, b).as_array()) == [[1], [2], [3]]
    assert sorted(engine.union(a, b, distinct=False).as_array()) == [
        [1], [2], [2], [2], [3]
    ]
    assert sorted(engine.subtract(a, b).as_array()) == [[1]]
    assert sorted(engine.intersect(a, b).as_array()) == [[2]]
    assert sorted(engine.distinct(a).as_array()) == [[1], [2]]


def test_dropna_fillna_sample_take(engine):
    a = engine.to_df([[1, None], [None, 2.0], [3, 4.0]], "x:double,y:double")
    assert engine.dropna(a).count() == 1
    assert engine.dropna(a, how="all").count() == 3
    assert engine.dropna(a, subset=["y"]).count() == 2
    filled = engine.fillna(a, 0)
    assert [[1.0, 0.0], [0.0, 2.0], [3.0, 4.0]] == filled.as_array()
    assert engine.sample(a, n=2, seed=0).count() == 2
    assert engine.sample(a, frac=0.5, seed=0).count() in (1, 2)
    t = engine.take(a, 1, presort="x desc")
    assert t.as_array() == [[3.0, 4.0]]
    grouped = engine.to_df(
        [[1, 10], [1, 20], [2, 30], [2, 5]], "g:long,v:long"
    )
    t2 = engine.take(
        grouped, 1, presort="v", partition_spec=PartitionSpec(by=["g"])
    )
    assert sorted(t2.as_array()) == [[1, 10], [2, 5]]


def test_select_filter_assign_aggregate(engine):
    a = engine.to_df([[1, 2.0], [2, 3.0], [3, 4.0]], "x:long,y:double")
    r = engine.select(
        a, SelectColumns(col("x"), (col("y") * 2).alias("y2"))
    )
    assert r.as_array() == [[1, 4.0], [2, 6.0], [3, 8.0]]
    r2 = engine.filter(a, col("x") > 1)
    assert r2.count() == 2
    r3 = engine.assign(a, [lit(1).alias("z")])
    assert r3.schema.names == ["x", "y", "z"]
    r4 = engine.aggregate(a, None, [f.sum(col("y")).alias("s")])
    assert r4.as_array() == [[9.0]]
    r5 = engine.aggregate(
        engine.to_df([[1, 1.0], [1, 2.0], [2, 3.0]], "k:long,v:double"),
        PartitionSpec(by=["k"]),
        [f.sum(col("v")).alias("s"), f.count(col("v")).alias("n")],
    )
    assert sorted(r5.as_array()) == [[1, 3.0, 2], [2, 3.0, 1]]


def test_map_engine_partitions(engine):
    def mapper(cursor, df):
        pdf = df.as_pandas()
        return PandasDataFrame(
            pd.DataFrame(
                dict(g=[cursor.key_value_array[0]], n=[len(pdf)])
            ),
            "g:long,n:long",
        )

    a = engine.to_df([[1, 1], [1, 2], [2, 3]], "g:long,v:long")
    res = engine.map_engine.map_dataframe(
        a, mapper, "g:long,n:long", PartitionSpec(by=["g"])
    )
    assert sorted(res.as_array()) == [[1, 2], [2, 1]]


def test_zip_comap(engine):
    a = engine.to_df([[1, "a"], [2, "b"]], "k:long,x:str")
    b = engine.to_df([[1, 10.0], [1, 20.0], [3, 30.0]], "k:long,y:double")
    z = engine.zip(DataFrames(a, b), how="inner")
    assert z.metadata["serialized"]

    def comap(cursor, dfs):
        assert len(dfs) == 2
        n1 = dfs[0].count()
        n2 = dfs[1].count()
        return ArrayDataFrame(
            [[cursor.key_value_array[0], n1, n2]], "k:long,n1:long,n2:long"
        )

    res = engine.comap(z, comap, "k:long,n1:long,n2:long", PartitionSpec())
    assert sorted(res.as_array()) == [[1, 1, 2]]


def test_engine_context(engine):
    with engine.as_context():
        assert fa.get_context_engine() is engine


def test_fa_eager():
    df1 = pd.DataFrame(dict(a=[1, 2, 3]))
    res = fa.filter(df1, col("a") > 1)
    assert isinstance(res, pd.DataFrame)
    assert len(res) == 2
    assert fa.count(df1) == 3
    j = fa.inner_join(
        pd.DataFrame(dict(k=[1, 2], x=[1, 2])),
        pd.DataFrame(dict(k=[2, 3], y=[5, 6])),
    )
    assert len(j) == 1


def test_engine_api_eager_ops(tmp_path):
    """fa.* eager API parity sweep (reference fugue_test
    execution_suite.test_engine_api)."""
    import os

    import fugue_amd.api as fa
    from fugue_amd.column import functions as f
    from fugue_amd.column.expressions import col, lit

    a = pd.DataFrame(dict(k=[1, 1, 2], v=[1.0, 2.0, 3.0]))
    b = pd.DataFrame(dict(k=[1, 3], w=[10.0, 30.0]))

    assert fa.get_current_parallelism() >= 1
    r = fa.select(a, col("k"), (col("v") * 2).alias("v2"), engine="native")
    assert r["v2"].tolist() == [2.0, 4.0, 6.0]
    r = fa.filter(a, col("v") > 1.5, engine="native")
    assert r["v"].tolist() == [2.0, 3.0]
    r = fa.assign(a, z=col("v") + 1, engine="native")
    assert r["z"].tolist() == [2.0, 3.0, 4.0]
    r = fa.aggregate(a, partition_by="k", s=f.sum(col("v")), engine="native")
    assert sorted(r["s"].tolist()) == [3.0, 3.0]
    r = fa.inner_join(a, b, engine="native")
    assert r["w"].tolist() == [10.0, 10.0]
    r = fa.union(a[["k"]], b[["k"]], distinct=True, engine="native")
    assert sorted(r["k"].tolist()) == [1, 2, 3]
    r = fa.subtract(a[["k"]], b[["k"]], engine="native")
    assert r["k"].tolist() == [2]
    r = fa.intersect(a[["k"]], b[["k"]], engine="native")
    assert r["k"].tolist() == [1]
    r = fa.distinct(a[["k"]], engine="native")
    assert sorted(r["k"].tolist()) == [1, 2]
    r = fa.take(a, 1, presort="v desc", engine="native")
    assert r["v"].tolist() == [3.0]
    p = os.path.join(str(tmp_path), "t.parquet")
    fa.save(a, p, engine="native")
    back = fa.load(p, engine="native")
    assert len(back) == 3
