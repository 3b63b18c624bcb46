import pandas as pd
import pyarrow as pa
import pytest

from fugue_amd import (
    ArrayDataFrame,
    ArrowDataFrame,
    DataFrames,
    IterableDataFrame,
    LocalDataFrameIterableDataFrame,
    PandasDataFrame,
    as_fugue_df,
)
from fugue_amd.exceptions import FugueDataFrameEmptyError


FRAME_TYPES = [ArrayDataFrame, PandasDataFrame, ArrowDataFrame]


@pytest.mark.parametrize("cls", FRAME_TYPES)
def test_basic_roundtrip(cls):
    df = cls([[1, "a"], [2, None]], "x:long,y:str")
    assert df.schema == "x:long,y:str"
    assert df.count() == 2
    assert not df.empty
    assert df.peek_array() == [1, "a"]
    assert df.as_array() == [[1, "a"], [2, None]]
    assert df.as_dicts() == [dict(x=1, y="a"), dict(x=2, y=None)]
    pdf = df.as_pandas()
    assert len(pdf) == 2
    t = df.as_arrow()
    assert t.num_rows == 2
    assert t.schema == pa.schema([("x", pa.int64()), ("y", pa.string())])


@pytest.mark.parametrize("cls", FRAME_TYPES)
def test_empty(cls):
    df = cls([], "x:long,y:str")
    assert df.empty
    assert df.count() == 0
    with pytest.raises(FugueDataFrameEmptyError):
        df.peek_array()


@pytest.mark.parametrize("cls", FRAME_TYPES)
def test_ops(cls):
    df = cls([[1, "a", 1.5], [2, "b", 2.5]], "x:long,y:str,z:double")
    assert df.drop(["y"]).as_array() == [[1, 1.5], [2, 2.5]]
    assert df[["z", "x"]].as_array() == [[1.5, 1], [2.5, 2]]
    assert df.rename({"x": "xx"}).schema == "xx:long,y:str,z:double"
    assert df.head(1).as_array() == [[1, "a", 1.5]]
    altered = df.alter_columns("x:double")
    assert altered.schema == "x:double,y:str,z:double"
    assert altered.as_array()[0][0] == 1.0


def test_pandas_df():
    pdf = pd.DataFrame(dict(a=[1, 2], b=["x", "y"]))
    df = PandasDataFrame(pdf)
    assert df.schema == "a:long,b:str"
    df2 = PandasDataFrame(pdf, "a:int,b:str")
    assert df2.schema == "a:int,b:str"
    assert df2.as_array() == [[1, "x"], [2, "y"]]


def test_iterable_df():
    df = IterableDataFrame(iter([[1, "a"], [2, "b"]]), "x:long,y:str")
    assert not df.is_bounded
    assert df.peek_array() == [1, "a"]
    assert df.as_array() == [[1, "a"], [2, "b"]]
    # consumed now
    df2 = IterableDataFrame(iter([]), "x:long,y:str")
    assert df2.empty


def test_local_df_iterable_df():
    frames = [
        PandasDataFrame(pd.DataFrame(dict(a=[1], b=["x"]))),
        PandasDataFrame(pd.DataFrame(dict(a=[2], b=["y"]))),
    ]
    df = LocalDataFrameIterableDataFrame(iter(frames))
    assert df.schema == "a:long,b:str"
    assert df.as_array() == [[1, "x"], [2, "y"]]


def test_as_fugue_df():
    assert isinstance(as_fugue_df(pd.DataFrame(dict(a=[1]))), PandasDataFrame)
    assert isinstance(
        as_fugue_df(pa.table({"a": [1]})), ArrowDataFrame
    )
    assert isinstance(as_fugue_df([[1]], schema="a:int"), ArrayDataFrame)


def test_dataframes():
    d1 = ArrayDataFrame([[1]], "a:int")
    d2 = ArrayDataFrame([[2]], "b:int")
    dfs = DataFrames(d1, d2)
    assert not dfs.has_key
    assert dfs[0] is d1 and dfs[1] is d2
    named = DataFrames(x=d1, y=d2)
    assert named.has_key
    assert named["x"] is d1
    with pytest.raises(ValueError):
        named._append(d1)


def test_nulls_nan():
    df = PandasDataFrame(
        pd.DataFrame(dict(a=[1.0, None], b=["x", None])), "a:double,b:str"
    )
    arr = df.as_array(type_safe=True)
    assert arr[1][0] is None
    assert arr[1][1] is None
