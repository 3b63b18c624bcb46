"""Case-by-case port of the reference acceptance suite
``fugue_test/dataframe_suite.py`` (DataFrameTests, 24 cases) and
``fugue_test/bag_suite.py`` (BagTests, 6 cases).  Reference method names
kept; bodies re-expressed on this package's ``fa.*`` functional API.

Frame types that don't support nested/map data set ``supports_nested``
/ ``supports_map`` False with a reason (the MI355X device frame stores
flat columns; nested data stays on host frames).
"""
import copy
from datetime import date, datetime
from typing import Any

import numpy as np
import pandas as pd
import pytest
from pytest import raises

import fugue_amd.api as fa
from fugue_amd import ArrowDataFrame, DataFrame
from fugue_amd.dataframe.utils import _df_eq
from fugue_amd.exceptions import (
    FugueDataFrameOperationError,
    FugueDatasetEmptyError,
)


class DataFrameConformance:
    """Subclass with ``df`` returning the frame type under test."""

    #: frame types without nested/struct/map storage may opt out
    supports_nested = True
    supports_map = True
    #: True when the native form is itself a fugue DataFrame
    native_is_fugue = False

    def df(self, data: Any = None, schema: Any = None) -> Any:  # pragma: no cover
        raise NotImplementedError

    def df_eq(self, df: Any, *args: Any, **kwargs: Any) -> bool:
        kwargs.setdefault("throw", True)
        if not isinstance(df, DataFrame):
            df = fa.as_fugue_df(df)
        return _df_eq(df, *args, **kwargs)

    # ------------------------------------------------------------------ #
    def test_native(self):
        df = self.df([[1]], "a:int")
        assert fa.is_df(df)
        fdf = fa.as_fugue_df(df)
        assert isinstance(fdf, DataFrame)
        assert fa.is_df(fdf)
        ndf = fa.get_native_as_df(fdf)
        assert fa.is_df(ndf)
        if not self.native_is_fugue:
            assert not isinstance(ndf, DataFrame)
        ndf2 = fa.get_native_as_df(ndf)
        if not self.native_is_fugue:
            assert ndf2 is ndf

    def test_peek(self):
        df = self.df([], "x:str,y:double")
        raises(FugueDatasetEmptyError, lambda: fa.peek_array(df))
        raises(FugueDatasetEmptyError, lambda: fa.peek_dict(df))

        df = self.df([["a", 1.0], ["b", 2.0]], "x:str,y:double")
        assert not fa.is_bounded(df) or 2 == fa.count(df)
        assert not fa.is_empty(df)
        assert ["a", 1.0] == fa.peek_array(df)
        assert dict(x="a", y=1.0) == fa.peek_dict(df)

    def test_as_pandas(self):
        df = self.df([["a", 1.0], ["b", 2.0]], "x:str,y:double")
        pdf = fa.as_pandas(df)
        assert [["a", 1.0], ["b", 2.0]] == pdf.values.tolist()

        df = self.df([], "x:str,y:double")
        pdf = fa.as_pandas(df)
        assert [] == pdf.values.tolist()
        assert fa.is_local(pdf)

    def test_as_local(self):
        with raises(NotImplementedError):
            fa.as_local(10)
        with raises(NotImplementedError):
            fa.as_local_bounded(10)

        df = self.df([["a", 1.0], ["b", 2.0]], "x:str,y:double")
        ldf = fa.as_local(df)
        assert fa.is_local(ldf)
        lbdf = fa.as_local_bounded(df)
        assert fa.is_local(lbdf) and fa.is_bounded(lbdf)

        fdf = fa.as_fugue_df(df)
        fdf.reset_metadata({"a": 1})
        ldf = fa.as_local(fdf)
        assert ldf.metadata == {"a": 1}
        lbdf = fa.as_local_bounded(fdf)
        assert fa.is_local(lbdf) and fa.is_bounded(lbdf)
        assert ldf.metadata == {"a": 1}

    def test_drop_columns(self):
        df = fa.drop_columns(self.df([], "a:str,b:int"), ["a"])
        assert fa.get_schema(df) == "b:int"
        raises(FugueDataFrameOperationError, lambda: fa.drop_columns(df, ["b"]))
        raises(FugueDataFrameOperationError, lambda: fa.drop_columns(df, ["x"]))

        df = fa.drop_columns(self.df([["a", 1]], "a:str,b:int"), ["a"])
        assert fa.get_schema(df) == "b:int"
        raises(FugueDataFrameOperationError, lambda: fa.drop_columns(df, ["b"]))
        raises(FugueDataFrameOperationError, lambda: fa.drop_columns(df, ["x"]))
        assert [[1]] == fa.as_array(df, type_safe=True)

    def test_select(self):
        df = fa.select_columns(self.df([], "a:str,b:int"), ["b"])
        assert fa.get_schema(df) == "b:int"
        assert fa.get_column_names(df) == ["b"]
        raises(FugueDataFrameOperationError, lambda: fa.select_columns(df, []))
        raises(FugueDataFrameOperationError, lambda: fa.select_columns(df, ["a"]))

        df = fa.select_columns(self.df([["a", 1]], "a:str,b:int"), ["b"])
        assert fa.get_schema(df) == "b:int"
        raises(FugueDataFrameOperationError, lambda: fa.select_columns(df, ["a"]))
        assert [[1]] == fa.as_array(df, type_safe=True)

        df = self.df([["a", 1, 2]], "a:str,b:int,c:int")
        # (the reference writes the expectation with a reordered schema
        # string and relies on triad's order-insensitive compare; the
        # assertion here is the same: selection follows requested order)
        self.df_eq(
            fa.as_fugue_df(fa.select_columns(df, ["c", "a"])),
            [[2, "a"]],
            "c:int,a:str",
        )

    def test_rename(self):
        for data in [[["a", 1]], []]:
            df = self.df(data, "a:str,b:int")
            df2 = fa.rename(df, columns=dict(a="aa"))
            assert fa.get_schema(df) == "a:str,b:int"
            self.df_eq(fa.as_fugue_df(df2), data, "aa:str,b:int")

        for data in [[["a", 1]], []]:
            df = self.df(data, "a:str,b:int")
            df3 = fa.rename(df, columns={})
            assert fa.get_schema(df3) == "a:str,b:int"
            self.df_eq(fa.as_fugue_df(df3), data, "a:str,b:int")

    def test_rename_invalid(self):
        df = self.df([["a", 1]], "a:str,b:int")
        raises(
            FugueDataFrameOperationError,
            lambda: fa.rename(df, columns=dict(aa="ab")),
        )

    def test_as_array(self):
        for func in [
            lambda df, *a, **k: fa.as_array(df, *a, **k, type_safe=True),
            lambda df, *a, **k: list(
                fa.as_array_iterable(df, *a, **k, type_safe=True)
            ),
        ]:
            df = self.df([], "a:str,b:int")
            assert [] == func(df)

            df = self.df([["a", 1]], "a:str,b:int")
            assert [["a", 1]] == func(df)
            df = self.df([["a", 1]], "a:str,b:int")
            assert [["a", 1]] == func(df, ["a", "b"])
            df = self.df([["a", 1]], "a:str,b:int")
            assert [[1, "a"]] == func(df, ["b", "a"])

            for v in [1.0, np.float64(1.0)]:
                df = self.df([[v, 1]], "a:double,b:int")
                d = func(df)
                assert [[1.0, 1]] == d
                assert isinstance(d[0][0], float)
                assert isinstance(d[0][1], int)

    def test_as_array_special_values(self):
        for func in [
            lambda df, *a, **k: fa.as_array(df, *a, **k, type_safe=True),
            lambda df, *a, **k: list(
                fa.as_array_iterable(df, *a, **k, type_safe=True)
            ),
        ]:
            df = self.df([[pd.Timestamp("2020-01-01"), 1]], "a:datetime,b:int")
            data = func(df)
            assert [[datetime(2020, 1, 1), 1]] == data
            assert isinstance(data[0][0], datetime)
            assert isinstance(data[0][1], int)

            df = self.df([[pd.NaT, 1]], "a:datetime,b:int")
            assert [[None, 1]] == func(df)

            df = self.df([[float("nan"), 1]], "a:double,b:int")
            assert [[None, 1]] == func(df)

            df = self.df([[float("inf"), 1]], "a:double,b:int")
            assert [[float("inf"), 1]] == func(df)

    def test_as_dict_iterable(self):
        df = self.df([[pd.NaT, 1]], "a:datetime,b:int")
        assert [dict(a=None, b=1)] == list(fa.as_dict_iterable(df))
        df = self.df([[pd.NaT, 1]], "a:datetime,b:int")
        assert [dict(b=1)] == list(fa.as_dict_iterable(df, ["b"]))
        df = self.df([[pd.Timestamp("2020-01-01"), 1]], "a:datetime,b:int")
        assert [dict(a=datetime(2020, 1, 1), b=1)] == list(
            fa.as_dict_iterable(df)
        )
        df = self.df([[pd.Timestamp("2020-01-01"), 1]], "a:datetime,b:int")
        assert [dict(b=1)] == list(fa.as_dict_iterable(df, ["b"]))

    def test_as_dicts(self):
        df = self.df([[pd.NaT, 1]], "a:datetime,b:int")
        assert [dict(a=None, b=1)] == fa.as_dicts(df)
        df = self.df([[pd.NaT, 1]], "a:datetime,b:int")
        assert [dict(b=1)] == fa.as_dicts(df, ["b"])
        df = self.df([[pd.Timestamp("2020-01-01"), 1]], "a:datetime,b:int")
        assert [dict(a=datetime(2020, 1, 1), b=1)] == fa.as_dicts(df)
        df = self.df([[pd.Timestamp("2020-01-01"), 1]], "a:datetime,b:int")
        assert [dict(b=1)] == fa.as_dicts(df, ["b"])

    def test_list_type(self):
        if not self.supports_nested:
            pytest.skip("frame type stores flat columns only")
        data = [[[30, 40]]]
        df = self.df(data, "a:[int]")
        assert data == fa.as_array(df, type_safe=True)

    def test_struct_type(self):
        if not self.supports_nested:
            pytest.skip("frame type stores flat columns only")
        data = [[{"a": 1}], [{"a": 2}]]
        df = self.df(data, "x:{a:int}")
        assert data == fa.as_array(df, type_safe=True)

    def test_map_type(self):
        if not self.supports_map:
            pytest.skip("frame type does not support map columns")
        data = [[[("a", 1), ("b", 3)]], [[("b", 2)]]]
        df = self.df(data, "x:<str,int>")
        assert data == fa.as_array(df, type_safe=True)

    def test_deep_nested_types(self):
        if not self.supports_nested:
            pytest.skip("frame type stores flat columns only")
        data = [[dict(a="1", b=[3, 4], d=1.0)], [dict(b=[30, 40])]]
        df = self.df(data, "a:{a:str,b:[int]}")
        a = fa.as_array(df, type_safe=True)
        assert [[dict(a="1", b=[3, 4])], [dict(a=None, b=[30, 40])]] == a

        data = [[[dict(b=[30, 40])]]]
        df = self.df(data, "a:[{a:str,b:[int]}]")
        a = fa.as_array(df, type_safe=True)
        assert [[[dict(a=None, b=[30, 40])]]] == a

    def test_binary_type(self):
        data = [[b"\x01\x05"]]
        df = self.df(data, "a:bytes")
        assert data == fa.as_array(df, type_safe=True)

    def test_as_arrow(self):
        df = self.df([], "a:int,b:int")
        assert [] == list(ArrowDataFrame(fa.as_arrow(df)).as_dict_iterable())
        assert fa.is_local(fa.as_arrow(df))
        df = self.df([[pd.NaT, 1]], "a:datetime,b:int")
        assert [dict(a=None, b=1)] == list(
            ArrowDataFrame(fa.as_arrow(df)).as_dict_iterable()
        )
        df = self.df([[pd.Timestamp("2020-01-01"), 1]], "a:datetime,b:int")
        assert [dict(a=datetime(2020, 1, 1), b=1)] == list(
            ArrowDataFrame(fa.as_arrow(df)).as_dict_iterable()
        )
        if self.supports_nested:
            data = [[[float("nan"), 2.0]]]
            df = self.df(data, "a:[float]")
            assert [[[None, 2.0]]] == ArrowDataFrame(fa.as_arrow(df)).as_array()
            data = [[dict(b=True)]]
            df = self.df(data, "a:{b:bool}")
            assert data == ArrowDataFrame(fa.as_arrow(df)).as_array()
            data = [[[dict(b=[30, 40])]]]
            df = self.df(data, "a:[{b:[long]}]")
            assert data == ArrowDataFrame(fa.as_arrow(df)).as_array()

    def test_head(self):
        df = self.df([], "a:str,b:int")
        assert [] == fa.as_array(fa.head(df, 1))
        assert [] == fa.as_array(fa.head(df, 1, ["b"]))
        df = self.df([["a", 1]], "a:str,b:int")
        if fa.is_bounded(df):
            assert [["a", 1]] == fa.as_array(fa.head(df, 1))
        assert [[1, "a"]] == fa.as_array(fa.head(df, 1, ["b", "a"]))
        assert [] == fa.as_array(fa.head(df, 0))

        df = self.df([[0, 1], [0, 2], [1, 1], [1, 3]], "a:int,b:int")
        assert 2 == fa.count(fa.head(df, 2))
        df = self.df([[0, 1], [0, 2], [1, 1], [1, 3]], "a:int,b:int")
        assert 4 == fa.count(fa.head(df, 10))
        h = fa.head(df, 10)
        assert fa.is_local(h) and fa.is_bounded(h)

    def test_show(self):
        df = self.df([["a", 1]], "a:str,b:int")
        fa.show(df)

    def test_alter_columns(self):
        # empty
        df = self.df([], "a:str,b:int")
        ndf = fa.alter_columns(df, "a:str,b:str")
        assert [] == fa.as_array(ndf, type_safe=True)
        assert fa.get_schema(ndf) == "a:str,b:str"

        # no change
        df = self.df([["a", 1], ["c", None]], "a:str,b:int")
        ndf = fa.alter_columns(df, "b:int,a:str", as_fugue=True)
        assert [["a", 1], ["c", None]] == fa.as_array(ndf, type_safe=True)
        assert fa.get_schema(ndf) == "a:str,b:int"

        # bool -> str
        df = self.df([["a", True], ["b", False], ["c", None]], "a:str,b:bool")
        ndf = fa.alter_columns(df, "b:str", as_fugue=True)
        actual = fa.as_array(ndf, type_safe=True)
        expected1 = [["a", "True"], ["b", "False"], ["c", None]]
        expected2 = [["a", "true"], ["b", "false"], ["c", None]]
        assert expected1 == actual or expected2 == actual
        assert fa.get_schema(ndf) == "a:str,b:str"

        # int -> str
        df = self.df([["a", 1], ["c", None]], "a:str,b:int")
        ndf = fa.alter_columns(df, "b:str", as_fugue=True)
        arr = fa.as_array(ndf, type_safe=True)
        assert [["a", "1"], ["c", None]] == arr or [
            ["a", "1.0"],
            ["c", None],
        ] == arr
        assert fa.get_schema(ndf) == "a:str,b:str"

        # int -> double
        df = self.df([["a", 1], ["c", None]], "a:str,b:int")
        ndf = fa.alter_columns(df, "b:double", as_fugue=True)
        assert [["a", 1], ["c", None]] == fa.as_array(ndf, type_safe=True)
        assert fa.get_schema(ndf) == "a:str,b:double"

        # double -> str
        df = self.df([["a", 1.1], ["b", None]], "a:str,b:double")
        data = fa.as_array(
            fa.alter_columns(df, "b:str", as_fugue=True), type_safe=True
        )
        assert [["a", "1.1"], ["b", None]] == data

        # double -> int
        df = self.df([["a", 1.0], ["b", None]], "a:str,b:double")
        data = fa.as_array(
            fa.alter_columns(df, "b:int", as_fugue=True), type_safe=True
        )
        assert [["a", 1], ["b", None]] == data

        # date -> str
        df = self.df(
            [["a", date(2020, 1, 1)], ["b", date(2020, 1, 2)], ["c", None]],
            "a:str,b:date",
        )
        data = fa.as_array(
            fa.alter_columns(df, "b:str", as_fugue=True), type_safe=True
        )
        assert [["a", "2020-01-01"], ["b", "2020-01-02"], ["c", None]] == data

        # datetime -> str
        df = self.df(
            [
                ["a", datetime(2020, 1, 1, 3, 4, 5)],
                ["b", datetime(2020, 1, 2, 16, 7, 8)],
                ["c", None],
            ],
            "a:str,b:datetime",
        )
        data = fa.as_array(
            fa.alter_columns(df, "b:str", as_fugue=True), type_safe=True
        )
        assert [
            ["a", "2020-01-01 03:04:05"],
            ["b", "2020-01-02 16:07:08"],
            ["c", None],
        ] == data

        # str -> bool
        df = self.df([["a", "trUe"], ["b", "False"], ["c", None]], "a:str,b:str")
        ndf = fa.alter_columns(df, "b:bool,a:str", as_fugue=True)
        assert [["a", True], ["b", False], ["c", None]] == fa.as_array(
            ndf, type_safe=True
        )
        assert fa.get_schema(ndf) == "a:str,b:bool"

        # str -> int
        df = self.df([["a", "1"]], "a:str,b:str")
        ndf = fa.alter_columns(df, "b:int,a:str")
        assert [["a", 1]] == fa.as_array(ndf, type_safe=True)
        assert fa.get_schema(ndf) == "a:str,b:int"

        # str -> double
        df = self.df([["a", "1.1"], ["b", "2"], ["c", None]], "a:str,b:str")
        ndf = fa.alter_columns(df, "b:double", as_fugue=True)
        assert [["a", 1.1], ["b", 2.0], ["c", None]] == fa.as_array(
            ndf, type_safe=True
        )
        assert fa.get_schema(ndf) == "a:str,b:double"

        # str -> date
        df = self.df(
            [["1", "2020-01-01"], ["2", "2020-01-02"], ["3", None]],
            "a:str,b:str",
        )
        ndf = fa.alter_columns(df, "b:date,a:int", as_fugue=True)
        assert [
            [1, date(2020, 1, 1)],
            [2, date(2020, 1, 2)],
            [3, None],
        ] == fa.as_array(ndf, type_safe=True)
        assert fa.get_schema(ndf) == "a:int,b:date"

        # str -> datetime
        df = self.df(
            [
                ["1", "2020-01-01 01:02:03"],
                ["2", "2020-01-02 01:02:03"],
                ["3", None],
            ],
            "a:str,b:str",
        )
        ndf = fa.alter_columns(df, "b:datetime,a:int", as_fugue=True)
        assert [
            [1, datetime(2020, 1, 1, 1, 2, 3)],
            [2, datetime(2020, 1, 2, 1, 2, 3)],
            [3, None],
        ] == fa.as_array(ndf, type_safe=True)
        assert fa.get_schema(ndf) == "a:int,b:datetime"

    def test_alter_columns_invalid(self):
        with raises(Exception):
            df = self.df(
                [["1", "x"], ["2", "y"], ["3", None]],
                "a:str,b:str",
            )
            ndf = fa.alter_columns(df, "b:int")
            fa.show(ndf)


class NativeDataFrameConformance(DataFrameConformance):
    """Extra cases for frame types wrapping an external native object
    (reference ``DataFrameTests.NativeTests``)."""

    def to_native_df(self, pdf: pd.DataFrame) -> Any:  # pragma: no cover
        raise NotImplementedError

    def test_get_column_names(self):
        df = self.to_native_df(pd.DataFrame([[0, 1, 2]], columns=["0", "1", "2"]))
        assert fa.get_column_names(df) == ["0", "1", "2"]

    def test_rename_any_names(self):
        pdf = self.to_native_df(pd.DataFrame([[0, 1, 2]], columns=["a", "b", "c"]))
        df = fa.rename(pdf, {})
        assert fa.get_column_names(df) == ["a", "b", "c"]

        pdf = self.to_native_df(pd.DataFrame([[0, 1, 2]], columns=["0", "1", "2"]))
        df = fa.rename(pdf, {"0": "_0", "1": "_1", "2": "_2"})
        assert fa.get_column_names(df) == ["_0", "_1", "_2"]


class BagConformance:
    """Subclass with ``bg`` returning the bag type under test
    (reference ``fugue_test/bag_suite.py``, 6 cases)."""

    def bg(self, data: Any = None) -> Any:  # pragma: no cover
        raise NotImplementedError

    def test_init_basic(self):
        raises(Exception, lambda: self.bg())
        bg = self.bg([])
        assert bg.empty
        assert copy.copy(bg) is bg
        assert copy.deepcopy(bg) is bg

    def test_peek(self):
        bg = self.bg([])
        raises(FugueDatasetEmptyError, lambda: bg.peek())

        bg = self.bg(["x"])
        assert not bg.is_bounded or 1 == bg.count()
        assert not bg.empty
        assert "x" == bg.peek()

    def test_as_array(self):
        bg = self.bg([2, 1, "a"])
        assert set([1, 2, "a"]) == set(bg.as_array())

    def test_as_array_special_values(self):
        bg = self.bg([2, None, "a"])
        assert set([None, 2, "a"]) == set(bg.as_array())

        bg = self.bg([np.float16(0.1)])
        assert set([np.float16(0.1)]) == set(bg.as_array())

    def test_head(self):
        bg = self.bg([])
        assert [] == bg.head(0).as_array()
        assert [] == bg.head(1).as_array()
        bg = self.bg([["a", 1]])
        if bg.is_bounded:
            assert [["a", 1]] == bg.head(1).as_array()
        assert [] == bg.head(0).as_array()

        bg = self.bg([1, 2, 3, 4])
        assert 2 == bg.head(2).count()
        bg = self.bg([1, 2, 3, 4])
        assert 4 == bg.head(10).count()
        h = bg.head(10)
        assert h.is_local and h.is_bounded

    def test_show(self):
        bg = self.bg(["a", 1])
        bg.show()
        bg.show(n=0)
        bg.show(n=1)
        bg.show(n=2)
        bg.show(title="title")
        bg.metadata["m"] = 1
        bg.show()
