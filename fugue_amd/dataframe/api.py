"""Functional API over DataFrames (reference parity: ``fugue/dataframe/api.py``)."""
from typing import Any, Dict, Iterable, List, Optional, Tuple

import pandas as pd
import pyarrow as pa

from fugue_amd.dataframe.dataframe import AnyDataFrame, DataFrame, as_fugue_df
from fugue_amd.schema import Schema


def is_df(df: Any) -> bool:
    if isinstance(df, DataFrame):
        return True
    return isinstance(df, (pd.DataFrame, pa.Table))


def get_native_as_df(df: AnyDataFrame) -> AnyDataFrame:
    if isinstance(df, DataFrame):
        return df.native_as_df()
    return df


def get_schema(df: AnyDataFrame) -> Schema:
    return as_fugue_df(df).schema


def as_pandas(df: AnyDataFrame) -> pd.DataFrame:
    return as_fugue_df(df).as_pandas()


def as_arrow(df: AnyDataFrame) -> pa.Table:
    return as_fugue_df(df).as_arrow()


def as_array(
    df: AnyDataFrame, columns: Optional[List[str]] = None, type_safe: bool = False
) -> List[Any]:
    return as_fugue_df(df).as_array(columns, type_safe=type_safe)


def as_array_iterable(
    df: AnyDataFrame, columns: Optional[List[str]] = None, type_safe: bool = False
) -> Iterable[Any]:
    return as_fugue_df(df).as_array_iterable(columns, type_safe=type_safe)


def as_dicts(df: AnyDataFrame, columns: Optional[List[str]] = None) -> List[Dict[str, Any]]:
    return as_fugue_df(df).as_dicts(columns)


def as_dict_iterable(
    df: AnyDataFrame, columns: Optional[List[str]] = None
) -> Iterable[Dict[str, Any]]:
    return as_fugue_df(df).as_dict_iterable(columns)


def peek_array(df: AnyDataFrame) -> List[Any]:
    return as_fugue_df(df).peek_array()


def peek_dict(df: AnyDataFrame) -> Dict[str, Any]:
    return as_fugue_df(df).peek_dict()


def head(
    df: AnyDataFrame,
    n: int,
    columns: Optional[List[str]] = None,
    as_fugue: bool = False,
) -> AnyDataFrame:
    res = as_fugue_df(df).head(n, columns)
    return res if as_fugue else res.native_as_df()


def alter_columns(df: AnyDataFrame, columns: Any, as_fugue: bool = False) -> AnyDataFrame:
    res = as_fugue_df(df).alter_columns(columns)
    return res if as_fugue else res.native_as_df()


def drop_columns(df: AnyDataFrame, columns: List[str], as_fugue: bool = False) -> AnyDataFrame:
    res = as_fugue_df(df).drop(columns)
    return res if as_fugue else res.native_as_df()


def select_columns(df: AnyDataFrame, columns: List[Any], as_fugue: bool = False) -> AnyDataFrame:
    res = as_fugue_df(df)[columns]
    return res if as_fugue else res.native_as_df()


def get_column_names(df: AnyDataFrame) -> List[Any]:
    if isinstance(df, pd.DataFrame):
        return list(df.columns)
    if isinstance(df, pa.Table):
        return list(df.schema.names)
    return as_fugue_df(df).columns


def rename(df: AnyDataFrame, columns: Dict[str, Any], as_fugue: bool = False) -> AnyDataFrame:
    if len(columns) == 0:
        return as_fugue_df(df) if as_fugue else df
    res = as_fugue_df(df).rename({str(k): str(v) for k, v in columns.items()})
    return res if as_fugue else res.native_as_df()


def normalize_column_names(df: AnyDataFrame) -> Tuple[AnyDataFrame, Dict[str, Any]]:
    names = get_column_names(df)
    new_names: Dict[str, Any] = {}
    inverse: Dict[str, Any] = {}
    for i, n in enumerate(names):
        sn = str(n)
        if not sn.isidentifier():
            nn = f"_{i}"
            new_names[sn] = nn
            inverse[nn] = n
    if len(new_names) == 0:
        return df, {}
    return rename(df, new_names), inverse
