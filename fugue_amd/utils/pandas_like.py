"""Pandas interop: schema inference, casting, and relational helpers.

Replaces the reference's dependency on ``triad.utils.pandas_like.PandasLikeUtils``
(used by its Native and Dask engines, ``fugue/execution/native_execution_engine.py:206``)
with a fresh implementation on pandas+pyarrow.
"""
from typing import Any, Callable, Dict, Iterable, List, Optional

import numpy as np
import pandas as pd
import pyarrow as pa

from fugue_amd.schema import Schema


def pandas_to_schema(df: pd.DataFrame) -> pa.Schema:
    if len(df.columns) == 0:
        return pa.schema([])
    for c in df.columns:
        if not isinstance(c, str):
            raise ValueError(f"column name {c!r} is not a string")
    if len(df) == 0:
        # infer object columns as string
        fields = []
        for c in df.columns:
            dt = df[c].dtype
            if dt == np.dtype("object"):
                fields.append(pa.field(c, pa.string()))
            else:
                fields.append(pa.field(c, pa.from_numpy_dtype(dt)))
        return pa.schema(fields)
    return pa.Schema.from_pandas(df, preserve_index=False)


def _cast_series(s: pd.Series, tp: pa.DataType) -> pd.Series:
    arr = pa.Array.from_pandas(s)
    if arr.type != tp:
        if pa.types.is_string(tp) and (
            pa.types.is_timestamp(arr.type) or pa.types.is_date(arr.type)
        ):
            # arrow's timestamp->string keeps trailing fractional zeros
            # ("03:04:05.000000"); the contract is python str(datetime)
            vals = [
                None if v is None else str(v)
                for v in arr.to_pylist()
            ]
            return pd.Series(vals, dtype=object)
        arr = arr.cast(tp, safe=False)
    return arr.to_pandas()


def cast_pandas(df: pd.DataFrame, schema: Schema) -> pd.DataFrame:
    """Cast/reorder columns of ``df`` to match ``schema`` using arrow casting
    rules (NaN→NULL preserved for nullable types)."""
    cols = {}
    for f in schema.fields:
        if f.name not in df.columns:
            raise ValueError(f"column {f.name} missing from dataframe")
        s = df[f.name]
        cur = None
        try:
            cur = pa.from_numpy_dtype(s.dtype) if s.dtype != np.dtype("object") else None
        except Exception:
            cur = None
        if cur is not None and cur == f.type:
            cols[f.name] = s.reset_index(drop=True)
        else:
            cols[f.name] = _cast_series(s.reset_index(drop=True), f.type)
    return pd.DataFrame(cols, columns=schema.names)


def enforce_type(df: pd.DataFrame, schema: Schema) -> pd.DataFrame:
    return cast_pandas(df, schema)


def pandas_join(
    df1: pd.DataFrame,
    df2: pd.DataFrame,
    how: str,
    on: List[str],
) -> pd.DataFrame:
    """Join with fugue's 9 join types. ``how`` is normalized (see
    ``fugue_amd.dataframe.utils.parse_join_type``).

    SQL null semantics: null keys never match (pandas ``merge`` would
    treat NaN keys as equal, so null-key rows are split out first)."""
    if how == "cross":
        d1 = df1.assign(__fugue_cross__=1)
        d2 = df2.assign(__fugue_cross__=1)
        res = d1.merge(d2, on="__fugue_cross__").drop(columns=["__fugue_cross__"])
        return res.reset_index(drop=True)
    null1 = df1[on].isna().any(axis=1)
    null2 = df2[on].isna().any(axis=1)
    d1, d1n = df1[~null1], df1[null1]
    d2, d2n = df2[~null2], df2[null2]
    if how in ("semi", "left_semi"):
        keys = d2[on].drop_duplicates()
        return d1.merge(keys, on=on, how="inner").reset_index(drop=True)
    if how in ("anti", "left_anti"):
        keys = d2[on].drop_duplicates().assign(__fugue_anti__=1)
        res = d1.merge(keys, on=on, how="left")
        res = res[res["__fugue_anti__"].isna()].drop(columns=["__fugue_anti__"])
        return pd.concat([res, d1n], ignore_index=True)
    pd_how = {
        "inner": "inner",
        "left_outer": "left",
        "right_outer": "right",
        "full_outer": "outer",
    }[how]
    res = d1.merge(d2, on=on, how=pd_how)
    parts = [res]
    if how in ("left_outer", "full_outer") and len(d1n) > 0:
        parts.append(d1n)
    if how in ("right_outer", "full_outer") and len(d2n) > 0:
        parts.append(d2n)
    if len(parts) > 1:
        res = pd.concat(parts, ignore_index=True)
    return res.reset_index(drop=True)


def pandas_union(df1: pd.DataFrame, df2: pd.DataFrame, unique: bool) -> pd.DataFrame:
    res = pd.concat([df1, df2[df1.columns]], ignore_index=True)
    if unique:
        res = drop_duplicates(res)
    return res.reset_index(drop=True)


def drop_duplicates(df: pd.DataFrame) -> pd.DataFrame:
    try:
        return df.drop_duplicates(ignore_index=True)
    except TypeError:  # unhashable (e.g. list) columns
        hashable = df.applymap(lambda x: str(x))
        return df[~hashable.duplicated()].reset_index(drop=True)


def pandas_intersect(df1: pd.DataFrame, df2: pd.DataFrame, unique: bool) -> pd.DataFrame:
    d2 = drop_duplicates(df2[df1.columns])
    res = df1.merge(d2, on=list(df1.columns), how="inner")
    if unique:
        res = drop_duplicates(res)
    return res.reset_index(drop=True)


def pandas_except(df1: pd.DataFrame, df2: pd.DataFrame, unique: bool) -> pd.DataFrame:
    d2 = drop_duplicates(df2[df1.columns]).assign(__fugue_exc__=1)
    res = df1.merge(d2, on=list(df1.columns), how="left")
    res = res[res["__fugue_exc__"].isna()].drop(columns=["__fugue_exc__"])
    if unique:
        res = drop_duplicates(res)
    return res.reset_index(drop=True)


def safe_groupby_apply(
    df: pd.DataFrame,
    cols: List[str],
    func: Callable[[pd.DataFrame], pd.DataFrame],
) -> pd.DataFrame:
    """Group by ``cols`` treating NULLs as a normal group (matching the
    reference's semantics), apply ``func`` per group, concat results."""
    if len(cols) == 0:
        return func(df.reset_index(drop=True))
    keys, starts = _group_starts(df, cols)
    results: List[pd.DataFrame] = []
    sorted_df = df.iloc[keys].reset_index(drop=True)
    bounds = list(starts) + [len(sorted_df)]
    for i in range(len(bounds) - 1):
        sub = sorted_df.iloc[bounds[i] : bounds[i + 1]].reset_index(drop=True)
        results.append(func(sub))
    if len(results) == 0:
        return df.head(0)
    return pd.concat(results, ignore_index=True)


def _group_starts(df: pd.DataFrame, cols: List[str]):
    """Stable-sort rows by ``cols`` (nulls grouped) and return the ordering
    index plus the start offset of each group."""
    idx = df.reset_index(drop=True).sort_values(
        cols, kind="stable", na_position="last"
    ).index.to_numpy()
    sorted_keys = df.iloc[idx][cols]
    n = len(sorted_keys)
    if n == 0:
        return idx, np.array([], dtype=np.int64)
    arr = sorted_keys.astype(object).to_numpy()
    change = np.ones(n, dtype=bool)
    if n > 1:
        prev = arr[:-1]
        cur = arr[1:]
        same = np.ones(n - 1, dtype=bool)
        for j in range(arr.shape[1]):
            a, b = prev[:, j], cur[:, j]
            eq = np.array(
                [
                    (bool(pd.isna(x)) and bool(pd.isna(y)))
                    or (
                        not bool(pd.isna(x))
                        and not bool(pd.isna(y))
                        and x == y
                    )
                    for x, y in zip(a, b)
                ]
            )
            same &= eq
        change[1:] = ~same
    return idx, np.nonzero(change)[0]


def _is_nan(x: Any) -> bool:
    try:
        return isinstance(x, float) and np.isnan(x)
    except Exception:
        return False
