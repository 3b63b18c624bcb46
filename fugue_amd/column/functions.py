"""Aggregation and scalar column functions.

Reference parity: ``fugue/column/functions.py``.
"""
from typing import Any, Optional

import pyarrow as pa

from fugue_amd.column.expressions import (
    ColumnExpr,
    _FuncExpr,
    _is_agg,
    _to_col,
    _UnaryAggFuncExpr,
)
from fugue_amd.schema import Schema


def coalesce(*args: Any) -> ColumnExpr:
    return _FuncExpr("COALESCE", *[_to_col(x) for x in args])


def min(col: ColumnExpr) -> ColumnExpr:  # pylint: disable=redefined-builtin
    assert isinstance(col, ColumnExpr)
    return _UnaryAggFuncExpr("MIN", col)


def max(col: ColumnExpr) -> ColumnExpr:  # pylint: disable=redefined-builtin
    assert isinstance(col, ColumnExpr)
    return _UnaryAggFuncExpr("MAX", col)


def count(col: ColumnExpr) -> ColumnExpr:
    assert isinstance(col, ColumnExpr)
    return _UnaryAggFuncExpr("COUNT", col)


def count_distinct(col: ColumnExpr) -> ColumnExpr:
    assert isinstance(col, ColumnExpr)
    return _UnaryAggFuncExpr("COUNT", col, arg_distinct=True)


def avg(col: ColumnExpr) -> ColumnExpr:
    assert isinstance(col, ColumnExpr)
    return _UnaryAggFuncExpr("AVG", col)


def sum(col: ColumnExpr) -> ColumnExpr:  # pylint: disable=redefined-builtin
    assert isinstance(col, ColumnExpr)
    return _UnaryAggFuncExpr("SUM", col)


def first(col: ColumnExpr) -> ColumnExpr:
    assert isinstance(col, ColumnExpr)
    return _UnaryAggFuncExpr("FIRST", col)


def last(col: ColumnExpr) -> ColumnExpr:
    assert isinstance(col, ColumnExpr)
    return _UnaryAggFuncExpr("LAST", col)


def is_agg(column: Any) -> bool:
    return _is_agg(column)
