"""Reusable conformance suites: any engine proves conformance by
subclassing.

Reference parity: ``fugue_test/execution_suite.py`` (ExecutionEngineTests),
``fugue_test/builtin_suite.py`` (BuiltInTests) and
``fugue_test/dataframe_suite.py`` (DataFrameTests) — the reference ships
these as a pytest plugin so third-party engines can verify themselves
(SURVEY.md §4); same pattern here.
"""
import os
import pickle
import tempfile
from typing import Any, Callable, Dict, Iterable, List

import numpy as np
import pandas as pd
import pytest

import fugue_amd.api as fa
from fugue_amd import ArrayDataFrame, DataFrame, PandasDataFrame
from fugue_amd.collections.partition import PartitionSpec
from fugue_amd.column.expressions import col, lit
from fugue_amd.column import functions as f
from fugue_amd.column.sql import SelectColumns
from fugue_amd.dataframe.dataframes import DataFrames
from fugue_amd.dataframe.utils import _df_eq
from fugue_amd.execution.execution_engine import ExecutionEngine
from fugue_amd.workflow import FugueWorkflow, transform


class ExecutionEngineTestSuite:
    """Every ExecutionEngine method; subclass and implement
    ``make_engine``."""

    @classmethod
    def make_engine(cls) -> ExecutionEngine:  # pragma: no cover
        raise NotImplementedError

    @pytest.fixture(autouse=True)
    def _engine(self):
        self.engine = self.make_engine()
        yield

    def df(self, data, schema) -> DataFrame:
        return self.engine.to_df(ArrayDataFrame(data, schema))

    # --- to_df ---------------------------------------------------------- #
    def test_to_df(self):
        e = self.engine
        d = e.to_df([[1, "a"]], "x:long,y:str")
        assert d.schema == "x:long,y:str"
        assert d.as_array() == [[1, "a"]]
        d2 = e.to_df(pd.DataFrame(dict(x=[1], y=["a"])))
        assert d2.schema == "x:long,y:str"

    # --- map ------------------------------------------------------------ #
    def test_map_no_partition(self):
        def m(cursor, data):
            return ArrayDataFrame(
                [[sum(r[0] for r in data.as_array())]], "s:long"
            )

        d = self.df([[1], [2], [3]], "x:long")
        res = self.engine.map_engine.map_dataframe(
            d, m, "s:long", PartitionSpec()
        )
        total = sum(r[0] for r in res.as_array())
        assert total == 6

    def test_map_with_keys(self):
        def m(cursor, data):
            rows = data.as_array()
            return ArrayDataFrame(
                [[cursor.key_value_array[0], len(rows)]], "g:long,n:long"
            )

        d = self.df([[1, 10], [1, 20], [2, 30]], "g:long,v:long")
        res = self.engine.map_engine.map_dataframe(
            d, m, "g:long,n:long", PartitionSpec(by=["g"])
        )
        assert sorted(res.as_array()) == [[1, 2], [2, 1]]

    def test_map_with_presort(self):
        def m(cursor, data):
            first = data.as_array()[0]
            return ArrayDataFrame([first], "g:long,v:long")

        d = self.df([[1, 20], [1, 10], [2, 30]], "g:long,v:long")
        res = self.engine.map_engine.map_dataframe(
            d, m, "g:long,v:long", PartitionSpec(by=["g"], presort="v desc")
        )
        assert sorted(res.as_array()) == [[1, 20], [2, 30]]

    # --- joins ----------------------------------------------------------- #
    def _join_case(self, how, expected, on=None):
        a = self.df([[1, "x"], [2, "y"], [3, "z"]], "k:long,a:str")
        b = self.df([[2, 20.0], [3, 30.0], [4, 40.0]], "k:long,b:double")
        res = self.engine.join(a, b, how=how, on=on)
        schema = (
            "k:long,a:str,b:double"
            if how not in ("semi", "anti")
            else "k:long,a:str"
        )
        assert _df_eq(res, expected, schema, throw=True)

    def test_join_inner(self):
        self._join_case("inner", [[2, "y", 20.0], [3, "z", 30.0]])

    def test_join_left(self):
        self._join_case(
            "left_outer",
            [[1, "x", None], [2, "y", 20.0], [3, "z", 30.0]],
        )

    def test_join_right(self):
        self._join_case(
            "right_outer",
            [[2, "y", 20.0], [3, "z", 30.0], [4, None, 40.0]],
        )

    def test_join_full(self):
        self._join_case(
            "full_outer",
            [
                [1, "x", None],
                [2, "y", 20.0],
                [3, "z", 30.0],
                [4, None, 40.0],
            ],
        )

    def test_join_semi_anti(self):
        self._join_case("semi", [[2, "y"], [3, "z"]])
        self._join_case("anti", [[1, "x"]])

    def test_join_cross(self):
        a = self.df([[1], [2]], "x:long")
        b = self.df([[10.0]], "y:double")
        res = self.engine.join(a, b, how="cross")
        assert _df_eq(res, [[1, 10.0], [2, 10.0]], "x:long,y:double", throw=True)

    def test_join_null_keys(self):
        a = self.engine.to_df(
            PandasDataFrame(
                pd.DataFrame(dict(k=[1.0, None, 2.0], a=[1, 2, 3])),
                "k:double,a:long",
            )
        )
        b = self.engine.to_df(
            PandasDataFrame(
                pd.DataFrame(dict(k=[1.0, None], b=[10, 20])), "k:double,b:long"
            )
        )
        res = self.engine.join(a, b, how="inner")
        # SQL semantics: null keys never match
        assert _df_eq(res, [[1.0, 1, 10]], "k:double,a:long,b:long", throw=True)

    # --- set ops ---------------------------------------------------------- #
    def test_set_ops(self):
        a = self.df([[1], [2], [2]], "x:long")
        b = self.df([[2], [3]], "x:long")
        assert sorted(
            self.engine.union(a, b).as_array()
        ) == [[1], [2], [3]]
        assert sorted(
            self.engine.union(a, b, distinct=False).as_array()
        ) == [[1], [2], [2], [2], [3]]
        assert sorted(self.engine.subtract(a, b).as_array()) == [[1]]
        assert sorted(self.engine.intersect(a, b).as_array()) == [[2]]
        assert sorted(self.engine.distinct(a).as_array()) == [[1], [2]]

    # --- row ops ----------------------------------------------------------- #
    def test_dropna_fillna(self):
        a = self.engine.to_df(
            PandasDataFrame(
                pd.DataFrame(dict(x=[1.0, None, 3.0], y=[None, 2.0, 4.0])),
                "x:double,y:double",
            )
        )
        assert self.engine.dropna(a).count() == 1
        assert self.engine.dropna(a, how="all").count() == 3
        assert self.engine.dropna(a, thresh=1).count() == 3
        assert self.engine.dropna(a, subset=["x"]).count() == 2
        filled = self.engine.fillna(a, 0)
        arr = filled.as_array()
        assert sorted(arr) == [[0.0, 2.0], [1.0, 0.0], [3.0, 4.0]]

    def test_sample(self):
        a = self.df([[i] for i in range(100)], "x:long")
        s = self.engine.sample(a, frac=0.5, seed=0)
        assert 20 <= s.count() <= 80
        s2 = self.engine.sample(a, n=10, seed=0)
        assert s2.count() == 10

    def test_take(self):
        a = self.df([[1, 10], [1, 5], [2, 8], [2, 9]], "g:long,v:long")
        t = self.engine.take(a, 1, presort="v desc")
        assert t.count() == 1
        assert t.as_array()[0][1] == 10
        t2 = self.engine.take(
            a, 1, presort="v", partition_spec=PartitionSpec(by=["g"])
        )
        assert sorted(t2.as_array()) == [[1, 5], [2, 8]]

    # --- select/aggregate --------------------------------------------------- #
    def test_select(self):
        a = self.df([[1, 2.0], [2, 3.0]], "x:long,y:double")
        r = self.engine.select(
            a, SelectColumns(col("x"), (col("y") * 2).alias("y2"))
        )
        assert _df_eq(r, [[1, 4.0], [2, 6.0]], "x:long,y2:double", throw=True)
        r2 = self.engine.filter(a, col("x") > 1)
        assert r2.as_array() == [[2, 3.0]]
        r3 = self.engine.assign(a, [lit("v").alias("z")])
        assert r3.schema.names == ["x", "y", "z"]

    def test_aggregate(self):
        a = self.df(
            [[1, 1.0], [1, 2.0], [2, 5.0]], "k:long,v:double"
        )
        r = self.engine.aggregate(
            a,
            PartitionSpec(by=["k"]),
            [
                f.sum(col("v")).alias("s"),
                f.count(col("v")).alias("n"),
                f.min(col("v")).alias("mn"),
                f.max(col("v")).alias("mx"),
                f.avg(col("v")).alias("av"),
            ],
        )
        rows = sorted(r.as_array())
        assert rows[0][0] == 1 and abs(rows[0][1] - 3.0) < 1e-9
        assert rows[0][2] == 2
        assert abs(rows[0][3] - 1.0) < 1e-9 and abs(rows[0][4] - 2.0) < 1e-9
        assert abs(rows[0][5] - 1.5) < 1e-9
        assert rows[1][0] == 2 and rows[1][2] == 1

    # --- persist / broadcast / repartition ---------------------------------- #
    def test_persist_broadcast(self):
        a = self.df([[1]], "x:long")
        assert self.engine.persist(a).as_array() == [[1]]
        assert self.engine.broadcast(a).as_array() == [[1]]

    def test_repartition(self):
        a = self.df([[i, i % 3] for i in range(30)], "x:long,g:long")
        r = self.engine.repartition(a, PartitionSpec(algo="hash", by=["g"]))
        assert r.count() == 30 or self.engine.is_distributed

    # --- zip / comap --------------------------------------------------------- #
    def test_zip_comap(self):
        a = self.df([[1, "a"], [2, "b"]], "k:long,x:str")
        b = self.df([[1, 10.0], [1, 20.0], [3, 30.0]], "k:long,y:double")
        z = self.engine.zip(DataFrames(a, b), how="inner")

        def cm(cursor, dfs):
            return ArrayDataFrame(
                [[cursor.key_value_array[0], dfs[0].count(), dfs[1].count()]],
                "k:long,n1:long,n2:long",
            )

        res = self.engine.comap(z, cm, "k:long,n1:long,n2:long", PartitionSpec())
        assert sorted(res.as_array()) == [[1, 1, 2]]

    def test_zip_left(self):
        a = self.df([[1, "a"], [2, "b"]], "k:long,x:str")
        b = self.df([[1, 10.0]], "k:long,y:double")
        z = self.engine.zip(DataFrames(a, b), how="left_outer")

        def cm(cursor, dfs):
            return ArrayDataFrame(
                [[cursor.key_value_array[0], dfs[1].count()]], "k:long,n2:long"
            )

        res = self.engine.comap(z, cm, "k:long,n2:long", PartitionSpec())
        assert sorted(res.as_array()) == [[1, 1], [2, 0]]

    # --- IO -------------------------------------------------------------------- #
    def test_load_save(self):
        with tempfile.TemporaryDirectory() as tmp:
            a = self.df([[1, "a"], [2, "b"]], "x:long,y:str")
            for fmt in ("parquet", "csv", "json"):
                path = os.path.join(tmp, f"f.{fmt}")
                kwargs = dict(header=True) if fmt == "csv" else {}
                self.engine.save_df(a, path, **kwargs)
                back = self.engine.load_df(
                    path,
                    columns="x:long,y:str" if fmt != "parquet" else None,
                    **(dict(header=True) if fmt == "csv" else {}),
                )
                assert _df_eq(
                    self.engine.to_df(back),
                    [[1, "a"], [2, "b"]],
                    "x:long,y:str",
                    throw=True,
                ), fmt

    def test_load_glob_and_single_save(self):
        """Glob loads + force_single saves incl. headerless CSV
        (reference execution_suite test_load_*_folder,
        test_save_single_and_load_csv_no_header)."""
        with tempfile.TemporaryDirectory() as tmp:
            a = self.df([[1, "a"], [2, "b"]], "x:long,y:str")
            b = self.df([[3, "c"]], "x:long,y:str")
            p1 = os.path.join(tmp, "part1.parquet")
            p2 = os.path.join(tmp, "part2.parquet")
            self.engine.save_df(a, p1, force_single=True)
            self.engine.save_df(b, p2, force_single=True)
            back = self.engine.to_df(
                self.engine.load_df(os.path.join(tmp, "*.parquet"))
            )
            assert sorted(back.as_array()) == [[1, "a"], [2, "b"], [3, "c"]]
            # headerless single csv round-trip (schema supplied on load)
            pc = os.path.join(tmp, "single.csv")
            self.engine.save_df(a, pc, force_single=True, header=False)
            back2 = self.engine.to_df(
                self.engine.load_df(pc, columns="x:long,y:str", header=False)
            )
            assert sorted(back2.as_array()) == [[1, "a"], [2, "b"]]

    # --- sql facet --------------------------------------------------------------- #
    def test_sql_select(self):
        from fugue_amd.collections.sql import StructuredRawSQL

        a = self.df([[1, 2.0], [1, 3.0], [2, 4.0]], "k:long,v:double")
        stmt = StructuredRawSQL(
            [
                (False, "SELECT k, SUM(v) AS s FROM "),
                (True, "t"),
                (False, " GROUP BY k"),
            ]
        )
        res = self.engine.sql_engine.select(DataFrames(t=a), stmt)
        assert sorted(res.as_array()) == [[1, 5.0], [2, 4.0]]


class BuiltInWorkflowTestSuite:
    """End-to-end workflow tests on an engine (reference BuiltInTests)."""

    @classmethod
    def make_engine(cls) -> ExecutionEngine:  # pragma: no cover
        raise NotImplementedError

    @pytest.fixture(autouse=True)
    def _engine(self):
        self.engine = self.make_engine()
        yield

    def run_dag(self, dag: FugueWorkflow):
        return dag.run(self.engine)

    def test_create_show_yield(self):
        dag = FugueWorkflow()
        a = dag.df([[1, "a"]], "x:long,y:str")
        a.yield_dataframe_as("r")
        res = self.run_dag(dag)
        assert res["r"].as_array() == [[1, "a"]]

    def test_transformer_styles(self):
        # pandas in/out
        def t1(df: pd.DataFrame) -> pd.DataFrame:
            df["z"] = df["x"] * 2
            return df

        # iterable dicts
        def t2(rows: Iterable[Dict[str, Any]]) -> Iterable[Dict[str, Any]]:
            for r in rows:
                r["z"] = r["x"] + 1
                yield r

        # list of lists
        def t3(rows: List[List[Any]]) -> List[List[Any]]:
            return [[r[0], r[0] * 10] for r in rows]

        dag = FugueWorkflow()
        a = dag.df([[1], [2]], "x:long")
        a.transform(t1, schema="*,z:long").yield_dataframe_as("r1")
        a.transform(t2, schema="*,z:long").yield_dataframe_as("r2")
        a.transform(t3, schema="x:long,z:long").yield_dataframe_as("r3")
        res = self.run_dag(dag)
        assert sorted(res["r1"].as_array()) == [[1, 2], [2, 4]]
        assert sorted(res["r2"].as_array()) == [[1, 2], [2, 3]]
        assert sorted(res["r3"].as_array()) == [[1, 10], [2, 20]]

    def test_transform_api(self):
        pdf = pd.DataFrame(dict(g=["a", "a", "b"], v=[1, 2, 3]))

        # schema: g:str,s:long
        def summ(df: pd.DataFrame) -> pd.DataFrame:
            return pd.DataFrame(dict(g=[df["g"].iloc[0]], s=[df["v"].sum()]))

        res = transform(
            pdf, summ, partition=dict(by=["g"]), engine=self.engine
        )
        if isinstance(res, pd.DataFrame):
            rows = sorted(res.values.tolist())
        else:
            rows = sorted(fa.as_fugue_df(res).as_array())
        assert rows == [["a", 3], ["b", 3]]

    def test_workflow_select_sql(self):
        dag = FugueWorkflow()
        a = dag.df([[1, 2.0], [1, 4.0], [2, 6.0]], "k:long,v:double")
        r = dag.select("SELECT k, SUM(v) AS s FROM", a, "GROUP BY k")
        r.yield_dataframe_as("r")
        res = self.run_dag(dag)
        assert sorted(res["r"].as_array()) == [[1, 6.0], [2, 6.0]]

    def test_checkpoint_and_persist(self):
        with tempfile.TemporaryDirectory() as tmp:
            dag = FugueWorkflow()
            a = dag.df([[1]], "x:long").persist()
            a.strong_checkpoint().yield_dataframe_as("r")
            res = dag.run(
                self.engine,
                {"fugue.workflow.checkpoint.path": tmp},
            )
            assert res["r"].as_array() == [[1]]

    def test_callbacks(self):
        collected = []

        def cb(n: int) -> None:
            collected.append(n)

        def worker(df: pd.DataFrame, callback: Callable) -> pd.DataFrame:
            callback(len(df))
            return df

        dag = FugueWorkflow()
        a = dag.df([[1], [2]], "x:long")
        a.transform(worker, schema="*", callback=cb).yield_dataframe_as("r")
        self.run_dag(dag)
        assert sum(collected) == 2

    def test_ignore_errors(self):
        def bad(df: pd.DataFrame) -> pd.DataFrame:
            if df["g"].iloc[0] == 0:
                raise ValueError("x")
            return df

        dag = FugueWorkflow()
        a = dag.df([[0, 1], [1, 2]], "g:long,v:long")
        a.partition(by=["g"]).transform(
            bad, schema="*", ignore_errors=[ValueError]
        ).yield_dataframe_as("r")
        res = self.run_dag(dag)
        assert res["r"].as_array() == [[1, 2]]

    def test_out_transform(self):
        side: List[int] = []

        def sink(rows: List[List[Any]]) -> None:
            side.append(len(rows))

        dag = FugueWorkflow()
        a = dag.df([[1], [2]], "x:long")
        a.out_transform(sink)
        self.run_dag(dag)
        assert sum(side) == 2

    def test_join_union_in_workflow(self):
        dag = FugueWorkflow()
        a = dag.df([[1, "x"]], "k:long,a:str")
        b = dag.df([[1, 5.0]], "k:long,b:double")
        a.inner_join(b).yield_dataframe_as("j")
        a.union(a, distinct=False).yield_dataframe_as("u")
        res = self.run_dag(dag)
        assert res["j"].as_array() == [[1, "x", 5.0]]
        assert res["u"].count() == 2


class DataFrameTestSuite:
    """Frame semantics per backend type (reference DataFrameTests)."""

    @classmethod
    def make_df(cls, data: Any, schema: Any) -> DataFrame:  # pragma: no cover
        raise NotImplementedError

    def test_init_and_basic(self):
        df = self.make_df([[1, "a"], [2, None]], "x:long,y:str")
        assert df.schema == "x:long,y:str"
        assert df.count() == 2
        assert not df.empty
        assert df.peek_array() == [1, "a"]
        assert df.peek_dict() == dict(x=1, y="a")

    def test_conversions(self):
        df = self.make_df([[1, "a"]], "x:long,y:str")
        assert df.as_array() == [[1, "a"]]
        assert df.as_dicts() == [dict(x=1, y="a")]
        assert list(df.as_dict_iterable()) == [dict(x=1, y="a")]
        assert len(df.as_pandas()) == 1
        assert df.as_arrow().num_rows == 1

    def test_ops(self):
        df = self.make_df([[1, "a", 2.0]], "x:long,y:str,z:double")
        assert df.drop(["y"]).schema == "x:long,z:double"
        assert df[["z", "x"]].schema == "z:double,x:long"
        assert df.rename({"x": "xx"}).schema == "xx:long,y:str,z:double"
        assert df.head(1).count() == 1
        altered = df.alter_columns("x:double")
        assert altered.schema == "x:double,y:str,z:double"

    def test_nulls(self):
        df = self.make_df([[None, None]], "x:double,y:str")
        row = df.as_array(type_safe=True)[0]
        assert row[0] is None and row[1] is None

    def test_empty(self):
        df = self.make_df([], "x:long")
        assert df.empty
        from fugue_amd.exceptions import FugueDataFrameEmptyError

        with pytest.raises(FugueDataFrameEmptyError):
            df.peek_array()

    def test_rename_invalid(self):
        df = self.make_df([[1]], "x:long")
        with pytest.raises(Exception):
            df.rename({"nope": "y"})

    def test_alter_columns_invalid(self):
        df = self.make_df([[1]], "x:long")
        with pytest.raises(Exception):
            df.alter_columns("nope:str")

    def test_show(self):
        df = self.make_df([[1, "a"], [2, None]], "x:long,y:str")
        df.show()
        df.show(1, with_count=True, title="t")

    supports_nested = True

    def test_nested_types(self):
        if not self.supports_nested:
            pytest.skip("backend frame holds flat columns only")
        df = self.make_df(
            [[[1, 2], dict(a=1)], [[3], dict(a=2)]],
            "x:[long],y:{a:long}",
        )
        arr = df.as_array()
        assert list(arr[0][0]) == [1, 2]

    def test_binary_type(self):
        if not self.supports_nested:
            pytest.skip("backend frame holds flat columns only")
        df = self.make_df([[b"ab"], [b"c"]], "x:bytes")
        got = df.as_array()
        assert bytes(got[0][0]) == b"ab" and bytes(got[1][0]) == b"c"


class _SelectTopHelper:
    @staticmethod
    def select_top(cursor, data):
        return ArrayDataFrame([cursor.row], cursor.row_schema)


def _suite_select_top(cursor, data):
    return ArrayDataFrame([cursor.row], cursor.row_schema)


class ExecutionEngineEdgeCaseTests:
    """Tricky semantics from the reference suite (null partition keys,
    NaT datetimes, nested/binary columns through map)."""

    @classmethod
    def make_engine(cls) -> ExecutionEngine:  # pragma: no cover
        raise NotImplementedError

    @pytest.fixture(autouse=True)
    def _engine(self):
        self.engine = self.make_engine()
        yield

    def test_map_with_null_keys(self):
        e = self.engine
        o = e.to_df(
            ArrayDataFrame(
                [[1.0, None, 1], [1.0, None, 0], [None, None, 2]],
                "a:double,b:double,c:int",
            )
        )
        c = e.map_engine.map_dataframe(
            o, _suite_select_top, o.schema, PartitionSpec(by=["a", "b"], presort="c")
        )
        assert _df_eq(
            c,
            [[1.0, None, 0], [None, None, 2]],
            "a:double,b:double,c:int",
            throw=True,
        )

    def test_map_with_nat(self):
        import datetime as _dt

        dt = _dt.datetime(2024, 5, 6, 7, 8, 9)
        e = self.engine
        o = e.to_df(
            ArrayDataFrame(
                [
                    [dt, 2, 1.0],
                    [None, 2, None],
                    [None, 1, None],
                    [dt, 5, 1.0],
                    [None, 4, None],
                ],
                "a:datetime,b:int,c:double",
            )
        )
        c = e.map_engine.map_dataframe(
            o, _suite_select_top, o.schema, PartitionSpec(by=["a", "c"], presort="b DESC")
        )
        assert _df_eq(
            c,
            [[None, 4, None], [dt, 5, 1.0]],
            "a:datetime,b:int,c:double",
            throw=True,
        )

    def test_map_nested_and_binary(self):
        import pickle

        e = self.engine
        # nested list column round-trips through map
        o = e.to_df(ArrayDataFrame([[3, [1, 2]]], "a:long,b:[int]"))
        c = e.map_engine.map_dataframe(
            o, _suite_select_top, o.schema, PartitionSpec(by=["a"])
        )
        assert c.as_array() == [[3, [1, 2]]]
        # binary column
        def bmap(cursor, data):
            rows = [[r[0] + b"x"] for r in data.as_array()]
            return ArrayDataFrame(rows, "a:bytes")

        o2 = e.to_df(ArrayDataFrame([[b"a"], [b"b"]], "a:bytes"))
        c2 = e.map_engine.map_dataframe(o2, bmap, "a:bytes", PartitionSpec())
        assert sorted(c2.as_array()) == [[b"ax"], [b"bx"]]

    def test_take_na_first(self):
        e = self.engine
        o = e.to_df(
            ArrayDataFrame([[1.0], [None], [3.0]], "a:double")
        )
        t = e.take(o, 1, presort="a", na_position="first")
        assert t.as_array()[0][0] is None

    def test_aggregate_with_nulls(self):
        e = self.engine
        o = e.to_df(
            ArrayDataFrame(
                [[1, 1.0], [1, None], [2, None]], "k:long,v:double"
            )
        )
        r = e.aggregate(
            o,
            PartitionSpec(by=["k"]),
            [f.sum(col("v")).alias("s"), f.count(col("v")).alias("n")],
        )
        rows = sorted(r.as_array())
        assert rows[0][0] == 1 and rows[0][2] == 1
        assert rows[1][0] == 2 and rows[1][2] == 0

    def test_first_last_count_distinct(self):
        e = self.engine
        o = e.to_df(
            ArrayDataFrame(
                [
                    [1, 5.0, "a"],
                    [1, None, "b"],
                    [2, 1.0, "c"],
                    [2, 2.0, "c"],
                    [2, 1.0, "d"],
                    [3, None, None],
                ],
                "k:long,v:double,s:str",
            )
        )
        r = e.aggregate(
            o,
            PartitionSpec(by=["k"]),
            [
                f.first(col("v")).alias("fv"),
                f.last(col("v")).alias("lv"),
                f.count_distinct(col("s")).alias("cd"),
            ],
        )
        rows = sorted(r.as_array(), key=lambda x: x[0])
        assert rows[0][0] == 1 and rows[0][1] == 5.0 and rows[0][2] == 5.0
        assert rows[0][3] == 2
        assert rows[1][0] == 2 and rows[1][3] == 2
        # group values {1.0, 2.0}: first/last must be members of the group
        assert rows[1][1] in (1.0, 2.0) and rows[1][2] in (1.0, 2.0)
        assert rows[2][0] == 3 and rows[2][1] is None and rows[2][2] is None
        assert rows[2][3] == 0

    def test_first_on_string_column(self):
        e = self.engine
        o = e.to_df(
            ArrayDataFrame(
                [[1, "x"], [1, "y"], [2, None]], "k:long,s:str"
            )
        )
        r = e.aggregate(
            o, PartitionSpec(by=["k"]), [f.first(col("s")).alias("fs")]
        )
        rows = sorted(r.as_array(), key=lambda x: x[0])
        assert rows[0][1] in ("x", "y")
        assert rows[1][1] is None

    def test_case_when_select(self):
        from fugue_amd.column.expressions import lit
        from fugue_amd.column.sql import SelectColumns

        e = self.engine
        o = e.to_df(
            ArrayDataFrame(
                [[1, 5.0], [2, None], [3, 1.0]], "k:long,v:double"
            )
        )
        r = e.select(
            o,
            SelectColumns(
                col("k"),
                f.case_when(
                    (col("v") > 2.0, lit("hi")),
                    (col("v") > 0.0, lit("lo")),
                    else_=lit("na"),
                ).alias("c"),
            ),
        )
        rows = sorted(r.as_array(), key=lambda x: x[0])
        assert [x[1] for x in rows] == ["hi", "na", "lo"]

    def test_like_filter(self):
        from fugue_amd.column import functions as ff

        e = self.engine
        o = e.to_df(
            ArrayDataFrame(
                [
                    [0, "apple"],
                    [1, "banana"],
                    [2, "applet"],
                    [3, None],
                    [4, "nap"],
                ],
                "k:long,s:str",
            )
        )
        r = e.filter(o, ff.like(col("s"), "app%"))
        assert sorted(x[0] for x in r.as_array()) == [0, 2]
        r2 = e.filter(o, ff.like(col("s"), "%ap%"))
        assert sorted(x[0] for x in r2.as_array()) == [0, 2, 4]
        r3 = e.filter(o, ~ff.like(col("s"), "app%"))
        assert sorted(x[0] for x in r3.as_array()) == [1, 4]


    def test_any_column_name(self):
        """Underscored/numbered column names flow through filter/select."""
        e = self.engine
        o = e.to_df(ArrayDataFrame([[1, "x"], [2, "y"]], "a_b:long,c1:str"))
        r = e.filter(o, col("a_b") > 1)
        assert r.as_array() == [[2, "y"]]


    def test_global_aggregate(self):
        e = self.engine
        o = e.to_df(
            ArrayDataFrame(
                [[1.0, 1], [2.0, 2], [None, 3], [4.0, 4]], "v:double,w:long"
            )
        )
        r = e.aggregate(
            o,
            None,
            [
                f.sum(col("v")).alias("s"),
                f.min(col("v")).alias("mn"),
                f.max(col("w")).alias("mx"),
                f.avg(col("v")).alias("av"),
                f.count(col("v")).alias("c"),
            ],
        )
        row = r.as_array()[0]
        assert row[0] == 7.0 and row[1] == 1.0 and row[2] == 4.0
        assert abs(row[3] - 7.0 / 3) < 1e-9 and row[4] == 3


class BagTestSuite:
    """Bag conformance (reference parity: ``fugue_test/bag_suite.py``)."""

    @classmethod
    def make_bag(cls, data):  # pragma: no cover
        from fugue_amd.bag.array_bag import ArrayBag

        return ArrayBag(data)

    def test_init_basic(self):
        b = self.make_bag([2, 1, "a", None])
        assert b.count() == 4
        assert not b.empty
        assert self.make_bag([]).empty

    def test_peek(self):
        b = self.make_bag([5])
        assert b.peek() == 5

    def test_as_array_special_values(self):
        data = [1, "x", None, 2.5, b"bytes", [1, 2]]
        b = self.make_bag(data)
        assert list(b.as_array()) == data
        assert list(b.as_array_iterable()) == data

    def test_head(self):
        b = self.make_bag(list(range(10)))
        arr = b.as_array()
        assert arr[:3] == [0, 1, 2]

    def test_show(self):
        self.make_bag([1, 2]).show()

