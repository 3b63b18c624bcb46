"""Case-by-case port of the reference acceptance suite
``fugue_test/execution_suite.py`` (ExecutionEngineTests, 42 cases).

Each test keeps the reference method name so conformance can be checked
line by line; the bodies are re-expressed around this module's helpers
(``edf``/``deq``/engine-context fixture).  Cases the MI355X engine
intentionally deviates on carry a note in their docstring.
"""
import copy
import os
import pickle
from datetime import datetime
from typing import Any

import pandas as pd
import pytest
from pytest import raises

import fugue_amd.api as fa
from fugue_amd import ArrayDataFrame, DataFrame, PandasDataFrame
from fugue_amd.collections.partition import PartitionSpec
from fugue_amd.column import functions as ff
from fugue_amd.column.expressions import all_cols, col, lit
from fugue_amd.dataframe.dataframes import DataFrames
from fugue_amd.dataframe.utils import _df_eq
from fugue_amd.exceptions import FugueInvalidOperation
from fugue_amd.execution.execution_engine import ExecutionEngine
from fugue_amd.execution.native_execution_engine import NativeExecutionEngine


def select_top(cursor, data):
    """Per-partition head(1) after presort — the reference's shared map
    function."""
    return ArrayDataFrame([cursor.row], cursor.row_schema)


class _Pickled:
    """An object round-tripped through a bytes column."""

    def __init__(self, data=None):
        self.data = data


def _append_x(cursor, df):
    rows = df.as_array(type_safe=True)
    out = []
    for r in rows:
        obj = pickle.loads(r[0])
        obj.data += "x"
        out.append([pickle.dumps(obj)])
    return ArrayDataFrame(out, df.schema)


class ExecutionEngineConformance:
    """Subclass with ``make_engine`` to run all 42 reference cases."""

    @classmethod
    def make_engine(cls) -> ExecutionEngine:  # pragma: no cover
        raise NotImplementedError

    @pytest.fixture(autouse=True)
    def _ctx(self, tmpdir):
        self.engine = self.make_engine()
        self.tmpdir = str(tmpdir)
        with fa.engine_context(self.engine):
            yield

    # helpers ----------------------------------------------------------- #
    def edf(self, data, schema) -> DataFrame:
        return fa.as_fugue_engine_df(self.engine, data, schema)

    def deq(self, actual, data, schema=None, **kw) -> bool:
        kw.setdefault("throw", True)
        if not isinstance(actual, DataFrame):
            actual = fa.as_fugue_df(actual)
        return _df_eq(actual, data, schema, **kw)

    # ------------------------------------------------------------------- #
    def test_init(self):
        assert self.engine.log is not None
        assert copy.copy(self.engine) is self.engine
        assert copy.deepcopy(self.engine) is self.engine

    def test_get_parallelism(self):
        assert fa.get_current_parallelism() == 1

    def test_to_df_general(self):
        e = self.engine
        exp = ArrayDataFrame([[1.1, 2.2], [3.3, 4.4]], "a:double,b:double")
        for src in (
            exp,
            fa.as_fugue_engine_df(
                e, [[1.1, 2.2], [3.3, 4.4]], "a:double,b:double"
            ),
            pd.DataFrame([[1.1, 2.2], [3.3, 4.4]], columns=["a", "b"]),
        ):
            self.deq(fa.as_fugue_engine_df(e, src), exp)
        # string → datetime coercion
        self.deq(
            fa.as_fugue_engine_df(e, [["2020-01-01"]], "a:datetime"),
            [[datetime(2020, 1, 1)]],
            "a:datetime",
        )
        # empty pandas input keeps the schema
        pdf = pd.DataFrame([[0.1, "a"]], columns=["a", "b"])
        self.deq(
            fa.as_fugue_engine_df(e, pdf[pdf.a < 0]),
            ArrayDataFrame([], "a:double,b:str"),
        )

    # filter / select / assign / aggregate ------------------------------- #
    _AB = ([[1, 2], [None, 2], [None, 1], [3, 4], [None, 4]],
           "a:double,b:int")

    def test_filter(self):
        a = ArrayDataFrame(*self._AB)
        self.deq(fa.filter(a, col("a").not_null(), as_fugue=True),
                 [[1, 2], [3, 4]], "a:double,b:int")
        self.deq(
            fa.filter(a, col("a").not_null() & (col("b") < 3), as_fugue=True),
            [[1, 2]], "a:double,b:int",
        )
        self.deq(fa.filter(a, col("a") + col("b") == 3, as_fugue=True),
                 [[1, 2]], "a:double,b:int")

    def test_select(self):
        a = ArrayDataFrame(*self._AB)
        self.deq(
            fa.select(a, col("b"), (col("b") + 1).alias("c").cast(str),
                      as_fugue=True),
            [[2, "3"], [2, "3"], [1, "2"], [4, "5"], [4, "5"]],
            "b:int,c:str",
        )
        self.deq(
            fa.select(a, col("b"), (col("b") + 1).alias("c").cast(str),
                      distinct=True, as_fugue=True),
            [[2, "3"], [1, "2"], [4, "5"]],
            "b:int,c:str",
        )
        self.deq(
            fa.select(a, all_cols(), where=col("a") + col("b") == 3,
                      as_fugue=True),
            [[1, 2]], "a:double,b:int",
        )
        self.deq(
            fa.select(a, col("a"), ff.sum(col("b")).cast(float).alias("b"),
                      as_fugue=True),
            [[1, 2], [3, 4], [None, 7]], "a:double,b:double",
        )
        col_b = ff.sum(col("b"))
        self.deq(
            fa.select(a, col("a"), col_b.cast(float).alias("c"),
                      having=(col_b >= 7) | (col("a") == 1), as_fugue=True),
            [[1, 2], [None, 7]], "a:double,c:double",
        )
        self.deq(
            fa.select(a, col("a"), lit(1, "o").cast(str),
                      col_b.cast(float).alias("c"),
                      having=(col_b >= 7) | (col("a") == 1), as_fugue=True),
            [[1, "1", 2], [None, "1", 7]], "a:double,o:str,c:double",
        )

    def test_assign(self):
        a = ArrayDataFrame(*self._AB)
        self.deq(
            fa.assign(a, x=1, b=col("b").cast(str),
                      c=(col("b") + 1).cast(int), as_fugue=True),
            [[1, "2", 1, 3], [None, "2", 1, 3], [None, "1", 1, 2],
             [3, "4", 1, 5], [None, "4", 1, 5]],
            "a:double,b:str,x:long,c:long",
        )

    def test_aggregate(self):
        a = ArrayDataFrame(*self._AB)
        self.deq(
            fa.aggregate(a, b=ff.max(col("b")),
                         c=(ff.max(col("b")) * 2).cast("int32").alias("c"),
                         as_fugue=True),
            [[4, 8]], "b:int,c:int",
        )
        self.deq(
            fa.aggregate(a, "a", b=ff.max(col("b")),
                         c=(ff.max(col("b")) * 2).cast("int32").alias("c"),
                         as_fugue=True),
            [[None, 4, 8], [1, 2, 4], [3, 4, 8]],
            "a:double,b:int,c:int",
        )
        with raises(ValueError):
            fa.aggregate(a, "a", b=ff.max(col("b")), x=1)
        with raises(ValueError):
            fa.aggregate(a, "a")

    # map ---------------------------------------------------------------- #
    def test_map(self):
        def noop(cursor, data):
            return data

        def on_init(partition_no, data):
            assert partition_no >= 0
            data.peek_array()

        e = self.engine
        o = ArrayDataFrame(*self._AB)
        a = self.edf(*self._AB)
        m = e.map_engine.map_dataframe
        self.deq(m(a, noop, a.schema, PartitionSpec()), o)
        self.deq(m(a, noop, a.schema, PartitionSpec(by=["a"], presort="b")), o)
        self.deq(
            m(a, select_top, a.schema, PartitionSpec(by=["a"], presort="b")),
            [[None, 1], [1, 2], [3, 4]], "a:double,b:int",
        )
        self.deq(
            m(a, select_top, a.schema,
              PartitionSpec(partition_by=["a"], presort="b DESC")),
            [[None, 4], [1, 2], [3, 4]], "a:double,b:int",
        )
        self.deq(
            m(a, select_top, a.schema,
              PartitionSpec(partition_by=["a"], presort="b DESC",
                            num_partitions=3),
              on_init=on_init),
            [[None, 4], [1, 2], [3, 4]], "a:double,b:int",
        )

    def test_map_with_special_values(self):
        e = self.engine
        m = e.map_engine.map_dataframe
        # multiple keys with nulls
        o = ArrayDataFrame(
            [[1, None, 1], [1, None, 0], [None, None, 2]],
            "a:double,b:double,c:int",
        )
        self.deq(
            m(o, select_top, o.schema,
              PartitionSpec(by=["a", "b"], presort="c")),
            [[1, None, 0], [None, None, 2]], "a:double,b:double,c:int",
        )
        # datetime keys incl. NaT output
        dt = datetime.now()
        o = ArrayDataFrame(
            [[dt, 2, 1], [None, 2, None], [None, 1, None], [dt, 5, 1],
             [None, 4, None]],
            "a:datetime,b:int,c:double",
        )
        c = m(o, select_top, o.schema,
              PartitionSpec(by=["a", "c"], presort="b DESC"))
        self.deq(c, [[None, 4, None], [dt, 5, 1]], "a:datetime,b:int,c:double")

        def with_nat(cursor, data):
            df = data.as_pandas()
            df["nat"] = pd.NaT
            return PandasDataFrame(df, data.schema + "nat:datetime")

        self.deq(
            m(c, with_nat, "a:datetime,b:int,c:double,nat:datetime",
              PartitionSpec()),
            [[None, 4, None, None], [dt, 5, 1, None]],
            "a:datetime,b:int,c:double,nat:datetime",
        )
        # list-typed column passes through
        o = ArrayDataFrame([[dt, [1, 2]]], "a:datetime,b:[int]")
        self.deq(m(o, select_top, o.schema, PartitionSpec(by=["a"])), o,
                 check_order=True)

    def test_map_with_dict_col(self):
        e = self.engine
        m = e.map_engine.map_dataframe
        dt = datetime.now()
        o = PandasDataFrame([[dt, dict(a=1)]], "a:datetime,b:{a:long}")
        self.deq(m(o, select_top, o.schema, PartitionSpec(by=["a"])), o,
                 no_pandas=True, check_order=True)

        def drop_dict(cursor, data):
            return data[["a"]]

        self.deq(
            m(o, drop_dict, "a:datetime", PartitionSpec(by=["a"])),
            PandasDataFrame([[dt]], "a:datetime"),
            no_pandas=True, check_order=True,
        )

        def add_dict(cursor, data):
            return PandasDataFrame([[dt, dict(a=1)]], "a:datetime,b:{a:long}")

        self.deq(
            m(m(o, drop_dict, "a:datetime", PartitionSpec(by=["a"])),
              add_dict, "a:datetime,b:{a:long}", PartitionSpec(by=["a"])),
            o, no_pandas=True, check_order=True,
        )

    def test_map_with_binary(self):
        e = self.engine
        o = ArrayDataFrame(
            [[pickle.dumps(_Pickled("a"))], [pickle.dumps(_Pickled("b"))]],
            "a:bytes",
        )
        c = e.map_engine.map_dataframe(o, _append_x, o.schema, PartitionSpec())
        self.deq(
            ArrayDataFrame(
                [[pickle.dumps(_Pickled("ax"))],
                 [pickle.dumps(_Pickled("bx"))]],
                "a:bytes",
            ),
            c, no_pandas=True, check_order=False,
        )

    # joins -------------------------------------------------------------- #
    def test_join_multiple(self):
        a = self.edf([[1, 2], [3, 4]], "a:int,b:int")
        b = self.edf([[1, 20], [3, 40]], "a:int,c:int")
        c = self.edf([[1, 200], [3, 400]], "a:int,d:int")
        self.deq(fa.inner_join(a, b, c, as_fugue=True),
                 [[1, 2, 20, 200], [3, 4, 40, 400]],
                 "a:int,b:int,c:int,d:int")

    def test__join_cross(self):
        a = self.edf([[1, 2], [3, 4]], "a:int,b:int")
        b = self.edf([[6], [7]], "c:int")
        self.deq(fa.join(a, b, how="Cross", as_fugue=True),
                 [[1, 2, 6], [1, 2, 7], [3, 4, 6], [3, 4, 7]],
                 "a:int,b:int,c:int")
        self.deq(fa.cross_join(a, self.edf([], "c:int"), as_fugue=True),
                 [], "a:int,b:int,c:int")
        self.deq(
            fa.join(self.edf([], "a:int,b:int"), self.edf([], "c:int"),
                    how="Cross", as_fugue=True),
            [], "a:int,b:int,c:int",
        )

    def test__join_inner(self):
        a = self.edf([[1, 2], [3, 4]], "a:int,b:int")
        b = self.edf([[6, 1], [2, 7]], "c:int,a:int")
        self.deq(fa.join(a, b, how="INNER", on=["a"], as_fugue=True),
                 [[1, 2, 6]], "a:int,b:int,c:int")
        self.deq(fa.inner_join(b, a, as_fugue=True),
                 [[6, 1, 2]], "c:int,a:int,b:int")
        self.deq(
            fa.join(self.edf([], "a:int,b:int"), self.edf([], "c:int,a:int"),
                    how="INNER", on=["a"], as_fugue=True),
            [], "a:int,b:int,c:int",
        )

    def test__join_outer(self):
        self.deq(
            fa.left_outer_join(self.edf([], "a:int,b:int"),
                               self.edf([], "c:str,a:int"), as_fugue=True),
            [], "a:int,b:int,c:str",
        )
        self.deq(
            fa.right_outer_join(self.edf([], "a:int,b:str"),
                                self.edf([], "c:int,a:int"), as_fugue=True),
            [], "a:int,b:str,c:int",
        )
        self.deq(
            fa.full_outer_join(self.edf([], "a:int,b:str"),
                               self.edf([], "c:str,a:int"), as_fugue=True),
            [], "a:int,b:str,c:str",
        )
        a = self.edf([[1, "2"], [3, "4"]], "a:int,b:str")
        b = self.edf([["6", 1], ["2", 7]], "c:str,a:int")
        self.deq(fa.join(a, b, how="left_OUTER", on=["a"], as_fugue=True),
                 [[1, "2", "6"], [3, "4", None]], "a:int,b:str,c:str")
        self.deq(fa.join(b, a, how="left_outer", on=["a"], as_fugue=True),
                 [["6", 1, "2"], ["2", 7, None]], "c:str,a:int,b:str")
        b2 = self.edf([[6, 1], [2, 7]], "c:double,a:int")
        self.deq(fa.join(a, b2, how="left_OUTER", on=["a"], as_fugue=True),
                 [[1, "2", 6.0], [3, "4", None]], "a:int,b:str,c:double")
        self.deq(fa.join(b2, a, how="left_outer", on=["a"], as_fugue=True),
                 [[6.0, 1, "2"], [2.0, 7, None]], "c:double,a:int,b:str")
        self.deq(fa.join(a, b, how="right_outer", on=["a"], as_fugue=True),
                 [[1, "2", "6"], [7, None, "2"]], "a:int,b:str,c:str")
        self.deq(fa.join(a, b, how="full_outer", on=["a"], as_fugue=True),
                 [[1, "2", "6"], [3, "4", None], [7, None, "2"]],
                 "a:int,b:str,c:str")

    def test__join_outer_pandas_incompatible(self):
        # int and bool columns that pandas would silently upcast
        a = self.edf([[1, "2"], [3, "4"]], "a:int,b:str")
        b = self.edf([[6, 1], [2, 7]], "c:int,a:int")
        self.deq(fa.join(a, b, how="left_OUTER", on=["a"], as_fugue=True),
                 [[1, "2", 6], [3, "4", None]], "a:int,b:str,c:int")
        self.deq(fa.join(b, a, how="left_outer", on=["a"], as_fugue=True),
                 [[6, 1, "2"], [2, 7, None]], "c:int,a:int,b:str")
        b2 = self.edf([[True, 1], [False, 7]], "c:bool,a:int")
        self.deq(fa.join(a, b2, how="left_OUTER", on=["a"], as_fugue=True),
                 [[1, "2", True], [3, "4", None]], "a:int,b:str,c:bool")
        self.deq(fa.join(b2, a, how="left_outer", on=["a"], as_fugue=True),
                 [[True, 1, "2"], [False, 7, None]], "c:bool,a:int,b:str")

    def test__join_semi(self):
        a = self.edf([[1, 2], [3, 4]], "a:int,b:int")
        b = self.edf([[6, 1], [2, 7]], "c:int,a:int")
        self.deq(fa.join(a, b, how="semi", on=["a"], as_fugue=True),
                 [[1, 2]], "a:int,b:int")
        self.deq(fa.semi_join(b, a, as_fugue=True), [[6, 1]], "c:int,a:int")
        self.deq(
            fa.join(a, self.edf([], "c:int,a:int"), how="semi", on=["a"],
                    as_fugue=True),
            [], "a:int,b:int",
        )
        self.deq(
            fa.join(self.edf([], "a:int,b:int"), self.edf([], "c:int,a:int"),
                    how="semi", on=["a"], as_fugue=True),
            [], "a:int,b:int",
        )

    def test__join_anti(self):
        a = self.edf([[1, 2], [3, 4]], "a:int,b:int")
        b = self.edf([[6, 1], [2, 7]], "c:int,a:int")
        self.deq(fa.join(a, b, how="anti", on=["a"], as_fugue=True),
                 [[3, 4]], "a:int,b:int")
        self.deq(fa.anti_join(b, a, as_fugue=True), [[2, 7]], "c:int,a:int")
        self.deq(
            fa.join(a, self.edf([], "c:int,a:int"), how="anti", on=["a"],
                    as_fugue=True),
            [[1, 2], [3, 4]], "a:int,b:int",
        )
        self.deq(
            fa.join(self.edf([], "a:int,b:int"), self.edf([], "c:int,a:int"),
                    how="anti", on=["a"], as_fugue=True),
            [], "a:int,b:int",
        )

    def test__join_with_null_keys(self):
        # SQL semantics: null keys never match
        a = self.edf([[1, 2, 3], [4, None, 6]], "a:double,b:double,c:int")
        b = self.edf([[1, 2, 33], [4, None, 63]], "a:double,b:double,d:int")
        self.deq(fa.join(a, b, how="INNER", as_fugue=True),
                 [[1, 2, 3, 33]], "a:double,b:double,c:int,d:int")

    # set ops ------------------------------------------------------------ #
    def test_union(self):
        a = self.edf([[1, 2, 3], [4, None, 6]], "a:double,b:double,c:int")
        b = self.edf([[1, 2, 33], [4, None, 6]], "a:double,b:double,c:int")
        self.deq(fa.union(a, b, as_fugue=True),
                 [[1, 2, 3], [4, None, 6], [1, 2, 33]],
                 "a:double,b:double,c:int")
        c = fa.union(a, b, distinct=False, as_fugue=True)
        self.deq(c,
                 [[1, 2, 3], [4, None, 6], [1, 2, 33], [4, None, 6]],
                 "a:double,b:double,c:int")
        self.deq(fa.union(a, b, c, distinct=False, as_fugue=True),
                 [[1, 2, 3], [4, None, 6], [1, 2, 33], [4, None, 6]] * 2,
                 "a:double,b:double,c:int")

    def test_subtract(self):
        a = self.edf([[1, 2, 3], [1, 2, 3], [4, None, 6]],
                     "a:double,b:double,c:int")
        b = self.edf([[1, 2, 33], [4, None, 6]], "a:double,b:double,c:int")
        self.deq(fa.subtract(a, b, as_fugue=True),
                 [[1, 2, 3]], "a:double,b:double,c:int")
        x = self.edf([[1, 2, 33]], "a:double,b:double,c:int")
        y = self.edf([[4, None, 6]], "a:double,b:double,c:int")
        self.deq(fa.subtract(a, x, y, as_fugue=True),
                 [[1, 2, 3]], "a:double,b:double,c:int")

    def test_intersect(self):
        a = self.edf([[1, 2, 3], [4, None, 6], [4, None, 6]],
                     "a:double,b:double,c:int")
        b = self.edf([[1, 2, 33], [4, None, 6], [4, None, 6], [4, None, 6]],
                     "a:double,b:double,c:int")
        self.deq(fa.intersect(a, b, as_fugue=True),
                 [[4, None, 6]], "a:double,b:double,c:int")
        x = self.edf([[1, 2, 33]], "a:double,b:double,c:int")
        y = self.edf([[4, None, 6], [4, None, 6], [4, None, 6]],
                     "a:double,b:double,c:int")
        self.deq(fa.intersect(a, x, y, as_fugue=True),
                 [], "a:double,b:double,c:int")

    def test_distinct(self):
        a = self.edf([[4, None, 6], [1, 2, 3], [4, None, 6]],
                     "a:double,b:double,c:int")
        self.deq(fa.distinct(a, as_fugue=True),
                 [[4, None, 6], [1, 2, 3]], "a:double,b:double,c:int")

    # null handling / sampling / take ------------------------------------ #
    def test_dropna(self):
        a = self.edf([[4, None, 6], [1, 2, 3], [4, None, None]],
                     "a:double,b:double,c:double")
        sch = "a:double,b:double,c:double"
        self.deq(fa.dropna(a, as_fugue=True), [[1, 2, 3]], sch)
        self.deq(fa.dropna(a, how="all", as_fugue=True),
                 [[4, None, 6], [1, 2, 3], [4, None, None]], sch)
        self.deq(fa.dropna(a, how="any", thresh=2, as_fugue=True),
                 [[4, None, 6], [1, 2, 3]], sch)
        self.deq(fa.dropna(a, how="any", subset=["a", "c"], as_fugue=True),
                 [[4, None, 6], [1, 2, 3]], sch)
        self.deq(
            fa.dropna(a, how="any", thresh=1, subset=["a", "c"],
                      as_fugue=True),
            [[4, None, 6], [1, 2, 3], [4, None, None]], sch,
        )

    def test_fillna(self):
        a = self.edf([[4, None, 6], [1, 2, 3], [4, None, None]],
                     "a:double,b:double,c:double")
        sch = "a:double,b:double,c:double"
        self.deq(fa.fillna(a, value=1, as_fugue=True),
                 [[4, 1, 6], [1, 2, 3], [4, 1, 1]], sch)
        d = fa.fillna(a, {"b": 99, "c": -99}, as_fugue=True)
        self.deq(d, [[4, 99, 6], [1, 2, 3], [4, 99, -99]], sch)
        self.deq(fa.fillna(a, value=-99, subset=["c"], as_fugue=True),
                 [[4, None, 6], [1, 2, 3], [4, None, -99]], sch)
        # mapping value ignores subset
        self.deq(fa.fillna(a, {"b": 99, "c": -99}, subset=["c"],
                           as_fugue=True), d)
        raises(ValueError, lambda: fa.fillna(a, {"b": None, "c": "99"}))
        raises(ValueError, lambda: fa.fillna(a, None))

    def test_sample(self):
        a = self.edf([[x] for x in range(100)], "a:int")
        with raises(ValueError):
            fa.sample(a)
        with raises(ValueError):
            fa.sample(a, n=90, frac=0.9)
        f_ = fa.sample(a, frac=0.8, replace=False, as_fugue=True)
        g = fa.sample(a, frac=0.8, replace=True, as_fugue=True)
        h = fa.sample(a, frac=0.8, seed=1, as_fugue=True)
        h2 = fa.sample(a, frac=0.8, seed=1, as_fugue=True)
        i = fa.sample(a, frac=0.8, seed=2, as_fugue=True)
        assert not self.deq(f_, g, throw=False)
        self.deq(h, h2)
        assert not self.deq(h, i, throw=False)
        assert abs(len(i.as_array()) - 80) < 10

    def test_sample_n(self):
        a = self.edf([[x] for x in range(100)], "a:int")
        b = fa.sample(a, n=90, replace=False, as_fugue=True)
        c = fa.sample(a, n=90, replace=True, as_fugue=True)
        d = fa.sample(a, n=90, seed=1, as_fugue=True)
        d2 = fa.sample(a, n=90, seed=1, as_fugue=True)
        g = fa.sample(a, n=90, seed=2, as_fugue=True)
        assert not self.deq(b, c, throw=False)
        self.deq(d, d2)
        assert not self.deq(d, g, throw=False)
        assert abs(len(g.as_array()) - 90) < 2

    def test_take(self):
        sch = "a:str,b:int,c:long"
        a = self.edf(
            [["a", 2, 3], ["a", 3, 4], ["b", 1, 2], ["b", 2, 2],
             [None, 4, 2], [None, 2, 1]],
            sch,
        )
        self.deq(fa.take(a, n=1, presort="b desc", as_fugue=True),
                 [[None, 4, 2]], sch)
        self.deq(
            fa.take(a, n=2, presort="a desc", na_position="first",
                    as_fugue=True),
            [[None, 4, 2], [None, 2, 1]], sch,
        )
        self.deq(
            fa.take(a, n=1, presort="a asc, b desc",
                    partition=dict(by=["a"], presort="b DESC,c DESC"),
                    as_fugue=True),
            [["a", 3, 4], ["b", 2, 2], [None, 4, 2]], sch,
        )
        self.deq(
            fa.take(a, n=1, presort=None,
                    partition=dict(by=["c"], presort="b ASC"), as_fugue=True),
            [["a", 2, 3], ["a", 3, 4], ["b", 1, 2], [None, 2, 1]], sch,
        )
        self.deq(
            fa.take(a, n=2, presort="a desc", na_position="last",
                    as_fugue=True),
            [["b", 1, 2], ["b", 2, 2]], sch,
        )
        self.deq(
            fa.take(a, n=2, presort="a", na_position="first", as_fugue=True),
            [[None, 4, 2], [None, 2, 1]], sch,
        )
        a = self.edf([["a", 2, 3], [None, 4, 2], [None, 2, 1]], sch)
        i = fa.take(a, n=1, partition="a", presort=None, as_fugue=True)
        assert (
            self.deq(i, [["a", 2, 3], [None, 4, 2]], sch, throw=False)
            or self.deq(i, [["a", 2, 3], [None, 2, 1]], sch, throw=False)
        )
        self.deq(
            fa.take(a, n=2, partition="a", presort=None, as_fugue=True),
            [["a", 2, 3], [None, 4, 2], [None, 2, 1]], sch,
        )
        raises(ValueError, lambda: fa.take(a, n=0.5, presort=None))

    # zip / comap --------------------------------------------------------- #
    def test_comap(self):
        ps = PartitionSpec(presort="b,c")
        e = self.engine
        a = self.edf([[1, 2], [3, 4], [1, 5]], "a:int,b:int")
        b = self.edf([[6, 1], [2, 7]], "c:int,a:int")
        with raises(FugueInvalidOperation):
            e.zip(DataFrames([a, b]),
                  partition_spec=PartitionSpec(by=["a"]), how="cross")
        with raises(NotImplementedError):
            e.zip(DataFrames([a, b]),
                  partition_spec=PartitionSpec(by=["a"]), how="left_anti")
        z1 = fa.persist(e.zip(DataFrames([a, b])))
        z2 = fa.persist(e.zip(DataFrames([a, b]), partition_spec=ps,
                              how="left_outer"))
        z3 = fa.persist(e.zip(DataFrames([b, a]), partition_spec=ps,
                              how="right_outer"))
        z4 = fa.persist(e.zip(DataFrames([a, b]), partition_spec=ps,
                              how="cross"))
        z5 = fa.persist(e.zip(DataFrames([a, b]), partition_spec=ps,
                              how="full_outer"))

        def comap(cursor, dfs):
            assert not dfs.has_key
            v = ",".join(k + str(d.count()) for k, d in dfs.items())
            keys = (cursor.key_value_array if not dfs[0].empty
                    else dfs[1][["a"]].peek_array())
            if len(keys) == 0:
                return ArrayDataFrame([[v]], "v:str")
            return ArrayDataFrame([keys + [v]], cursor.key_schema + "v:str")

        def on_init(partition_no, dfs):
            assert not dfs.has_key
            assert partition_no >= 0
            assert len(dfs) > 0

        self.deq(
            e.comap(z1, comap, "a:int,v:str", PartitionSpec(),
                    on_init=on_init),
            [[1, "_02,_11"]], "a:int,v:str",
        )
        self.deq(e.comap(z2, comap, "a:int,v:str", PartitionSpec()),
                 [[1, "_02,_11"], [3, "_01,_10"]], "a:int,v:str")
        self.deq(e.comap(z3, comap, "a:int,v:str", PartitionSpec()),
                 [[1, "_01,_12"], [3, "_00,_11"]], "a:int,v:str")
        self.deq(e.comap(z4, comap, "v:str", PartitionSpec()),
                 [["_03,_12"]], "v:str")
        self.deq(e.comap(z5, comap, "a:int,v:str", PartitionSpec()),
                 [[1, "_02,_11"], [3, "_01,_10"], [7, "_00,_11"]],
                 "a:int,v:str")

    def test_comap_with_key(self):
        e = self.engine
        a = self.edf([[1, 2], [3, 4], [1, 5]], "a:int,b:int")
        b = self.edf([[6, 1], [2, 7]], "c:int,a:int")
        c = self.edf([[6, 1]], "c:int,a:int")
        z1 = fa.persist(e.zip(DataFrames(x=a, y=b)))
        z2 = fa.persist(e.zip(DataFrames(x=a, y=b, z=b)))
        z3 = fa.persist(e.zip(DataFrames(z=c),
                              partition_spec=PartitionSpec(by=["a"])))

        def comap(cursor, dfs):
            assert dfs.has_key
            v = ",".join(k + str(d.count()) for k, d in dfs.items())
            return ArrayDataFrame([cursor.key_value_array + [v]],
                                  cursor.key_schema + "v:str")

        def on_init(partition_no, dfs):
            assert dfs.has_key
            assert partition_no >= 0
            assert len(dfs) > 0

        self.deq(
            e.comap(z1, comap, "a:int,v:str", PartitionSpec(),
                    on_init=on_init),
            [[1, "x2,y1"]], "a:int,v:str",
        )
        self.deq(
            e.comap(z2, comap, "a:int,v:str", PartitionSpec(),
                    on_init=on_init),
            [[1, "x2,y1,z1"]], "a:int,v:str",
        )
        self.deq(
            e.comap(z3, comap, "a:int,v:str", PartitionSpec(),
                    on_init=on_init),
            [[1, "z1"]], "a:int,v:str",
        )

    # IO ------------------------------------------------------------------ #
    def _path(self, *parts):
        return os.path.join(self.tmpdir, *parts)

    def test_save_single_and_load_parquet(self):
        b = ArrayDataFrame([[6, 1], [2, 7]], "c:int,a:long")
        path = self._path("a", "b")
        os.makedirs(path, exist_ok=True)
        fa.save(b, path, format_hint="parquet", force_single=True)
        assert os.path.isfile(path)
        c = fa.load(path, format_hint="parquet", columns=["a", "c"],
                    as_fugue=True)
        self.deq(c, [[1, 6], [7, 2]], "a:long,c:int")
        b = ArrayDataFrame([[60, 1], [20, 7]], "c:int,a:long")
        fa.save(b, path, format_hint="parquet", mode="overwrite")
        c = fa.load(path, format_hint="parquet", columns=["a", "c"],
                    as_fugue=True)
        self.deq(c, [[1, 60], [7, 20]], "a:long,c:int")

    def test_save_and_load_parquet(self):
        b = ArrayDataFrame([[6, 1], [2, 7]], "c:int,a:long")
        path = self._path("a", "b")
        fa.save(b, path, format_hint="parquet")
        c = fa.load(path, format_hint="parquet", columns=["a", "c"],
                    as_fugue=True)
        self.deq(c, [[1, 6], [7, 2]], "a:long,c:int")

    def test_load_parquet_folder(self):
        native = NativeExecutionEngine()
        path = self._path("a", "b")
        fa.save(ArrayDataFrame([[6, 1]], "c:int,a:long"),
                os.path.join(path, "a.parquet"), engine=native)
        fa.save(ArrayDataFrame([[2, 7], [4, 8]], "c:int,a:long"),
                os.path.join(path, "b.parquet"), engine=native)
        open(os.path.join(path, "_SUCCESS"), "w").close()
        c = fa.load(path, format_hint="parquet", columns=["a", "c"],
                    as_fugue=True)
        self.deq(c, [[1, 6], [7, 2], [8, 4]], "a:long,c:int")

    def test_load_parquet_files(self):
        native = NativeExecutionEngine()
        path = self._path("a", "b")
        f1, f2 = os.path.join(path, "a.parquet"), os.path.join(path, "b.parquet")
        fa.save(ArrayDataFrame([[6, 1]], "c:int,a:long"), f1, engine=native)
        fa.save(ArrayDataFrame([[2, 7], [4, 8]], "c:int,a:long"), f2,
                engine=native)
        c = fa.load([f1, f2], format_hint="parquet", columns=["a", "c"],
                    as_fugue=True)
        self.deq(c, [[1, 6], [7, 2], [8, 4]], "a:long,c:int")

    def test_save_single_and_load_csv(self):
        b = ArrayDataFrame([[6.1, 1.1], [2.1, 7.1]], "c:double,a:double")
        path = self._path("a", "b")
        os.makedirs(path, exist_ok=True)
        fa.save(b, path, format_hint="csv", header=True, force_single=True)
        assert os.path.isfile(path)
        c = fa.load(path, format_hint="csv", header=True, infer_schema=False,
                    as_fugue=True)
        self.deq(c, [["6.1", "1.1"], ["2.1", "7.1"]], "c:str,a:str")
        c = fa.load(path, format_hint="csv", header=True, infer_schema=True,
                    as_fugue=True)
        self.deq(c, [[6.1, 1.1], [2.1, 7.1]], "c:double,a:double")
        with raises(ValueError):
            fa.load(path, format_hint="csv", header=True, infer_schema=True,
                    columns="c:str,a:str", as_fugue=True)
        c = fa.load(path, format_hint="csv", header=True, infer_schema=False,
                    columns=["a", "c"], as_fugue=True)
        self.deq(c, [["1.1", "6.1"], ["7.1", "2.1"]], "a:str,c:str")
        c = fa.load(path, format_hint="csv", header=True, infer_schema=False,
                    columns="a:double,c:double", as_fugue=True)
        self.deq(c, [[1.1, 6.1], [7.1, 2.1]], "a:double,c:double")
        b = ArrayDataFrame([[60.1, 1.1], [20.1, 7.1]], "c:double,a:double")
        fa.save(b, path, format_hint="csv", header=True, mode="overwrite")
        c = fa.load(path, format_hint="csv", header=True, infer_schema=False,
                    columns=["a", "c"], as_fugue=True)
        self.deq(c, [["1.1", "60.1"], ["7.1", "20.1"]], "a:str,c:str")

    def test_save_single_and_load_csv_no_header(self):
        b = ArrayDataFrame([[6.1, 1.1], [2.1, 7.1]], "c:double,a:double")
        path = self._path("a", "b")
        os.makedirs(path, exist_ok=True)
        fa.save(b, path, format_hint="csv", header=False, force_single=True)
        assert os.path.isfile(path)
        with raises(ValueError):
            fa.load(path, format_hint="csv", header=False,
                    infer_schema=False, as_fugue=True)
        c = fa.load(path, format_hint="csv", header=False,
                    infer_schema=False, columns=["c", "a"], as_fugue=True)
        self.deq(c, [["6.1", "1.1"], ["2.1", "7.1"]], "c:str,a:str")
        c = fa.load(path, format_hint="csv", header=False, infer_schema=True,
                    columns=["c", "a"], as_fugue=True)
        self.deq(c, [[6.1, 1.1], [2.1, 7.1]], "c:double,a:double")
        with raises(ValueError):
            fa.load(path, format_hint="csv", header=False, infer_schema=True,
                    columns="c:double,a:double", as_fugue=True)
        c = fa.load(path, format_hint="csv", header=False,
                    infer_schema=False, columns="c:double,a:str",
                    as_fugue=True)
        self.deq(c, [[6.1, "1.1"], [2.1, "7.1"]], "c:double,a:str")

    def test_save_and_load_csv(self):
        b = ArrayDataFrame([[6.1, 1.1], [2.1, 7.1]], "c:double,a:double")
        path = self._path("a", "b")
        fa.save(b, path, format_hint="csv", header=True)
        c = fa.load(path, format_hint="csv", header=True, infer_schema=True,
                    columns=["a", "c"], as_fugue=True)
        self.deq(c, [[1.1, 6.1], [7.1, 2.1]], "a:double,c:double")

    def test_load_csv_folder(self):
        native = NativeExecutionEngine()
        path = self._path("a", "b")
        fa.save(ArrayDataFrame([[6.1, 1.1]], "c:double,a:double"),
                os.path.join(path, "a.csv"), format_hint="csv", header=True,
                engine=native)
        fa.save(ArrayDataFrame([[2.1, 7.1], [4.1, 8.1]], "c:double,a:double"),
                os.path.join(path, "b.csv"), format_hint="csv", header=True,
                engine=native)
        open(os.path.join(path, "_SUCCESS"), "w").close()
        c = fa.load(path, format_hint="csv", header=True, infer_schema=True,
                    columns=["a", "c"], as_fugue=True)
        self.deq(c, [[1.1, 6.1], [7.1, 2.1], [8.1, 4.1]],
                 "a:double,c:double")

    def test_save_single_and_load_json(self):
        b = ArrayDataFrame([[6, 1], [2, 7]], "c:int,a:long")
        path = self._path("a", "b")
        os.makedirs(path, exist_ok=True)
        fa.save(b, path, format_hint="json", force_single=True)
        assert os.path.isfile(path)
        c = fa.load(path, format_hint="json", columns=["a", "c"],
                    as_fugue=True)
        self.deq(c, [[1, 6], [7, 2]], "a:long,c:long")
        b = ArrayDataFrame([[60, 1], [20, 7]], "c:long,a:long")
        fa.save(b, path, format_hint="json", mode="overwrite")
        c = fa.load(path, format_hint="json", columns=["a", "c"],
                    as_fugue=True)
        self.deq(c, [[1, 60], [7, 20]], "a:long,c:long")

    def test_save_and_load_json(self):
        e = self.engine
        b = ArrayDataFrame([[6, 1], [3, 4], [2, 7], [4, 8], [6, 7]],
                           "c:int,a:long")
        path = self._path("a", "b")
        fa.save(
            e.repartition(fa.as_fugue_engine_df(e, b), PartitionSpec(num=2)),
            path, format_hint="json",
        )
        c = fa.load(path, format_hint="json", columns=["a", "c"],
                    as_fugue=True)
        self.deq(c, [[1, 6], [7, 2], [4, 3], [8, 4], [7, 6]],
                 "a:long,c:long")

    def test_load_json_folder(self):
        native = NativeExecutionEngine()
        path = self._path("a", "b")
        fa.save(ArrayDataFrame([[6, 1], [3, 4]], "c:int,a:long"),
                os.path.join(path, "a.json"), format_hint="json",
                engine=native)
        fa.save(ArrayDataFrame([[2, 7], [4, 8]], "c:int,a:long"),
                os.path.join(path, "b.json"), format_hint="json",
                engine=native)
        open(os.path.join(path, "_SUCCESS"), "w").close()
        c = fa.load(path, format_hint="json", columns=["a", "c"],
                    as_fugue=True)
        self.deq(c, [[1, 6], [7, 2], [8, 4], [4, 3]], "a:long,c:long")

    # engine api ---------------------------------------------------------- #
    #: engines whose native frame type IS a fugue DataFrame (the MI355X
    #: engine: the HBM-resident HipDataFrame is both the native form and
    #: the fugue frame — there is no external frame library underneath)
    #: set this True; the reference backends all wrap external types.
    native_is_fugue = False

    def test_engine_api(self):
        with fa.engine_context(self.engine):
            df1 = fa.as_fugue_df([[0, 1], [2, 3]], schema="a:long,b:long")
            df1 = fa.repartition(df1, {"num": 2})
            df1 = fa.get_native_as_df(fa.broadcast(df1))
            df2 = pd.DataFrame([[0, 1], [2, 3]], columns=["a", "b"])
            df3 = fa.union(df1, df2, as_fugue=False)
            assert fa.is_df(df3)
            if not self.native_is_fugue:
                assert not isinstance(df3, DataFrame)
            df4 = fa.union(df1, df2, as_fugue=True)
            assert isinstance(df4, DataFrame)
            self.deq(df4, fa.as_fugue_df(fa.as_pandas(df3)))
