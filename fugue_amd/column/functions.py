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


def case_when(*whens: Any, else_: Any = None) -> ColumnExpr:
    """``CASE WHEN c1 THEN v1 [WHEN c2 THEN v2 ...] ELSE e END``.

    ``whens`` are ``(condition, value)`` pairs. Encoded as a ``_FuncExpr``
    with flattened args ``[c1, v1, c2, v2, ..., else]`` so alias/cast/
    mention machinery applies unchanged (reference parity:
    QPD's CASE WHEN support used by ``fugue/execution`` SQL lowering)."""
    assert len(whens) > 0, "case_when requires at least one (cond, value) pair"
    flat = []
    for pair in whens:
        cond, value = pair
        flat.append(_to_col(cond))
        flat.append(_to_col(value))
    flat.append(_to_col(else_))
    return _FuncExpr("CASE_WHEN", *flat)


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


def like(col: ColumnExpr, pattern: str) -> ColumnExpr:
    """SQL ``LIKE`` predicate (``%`` any run, ``_`` any single char).
    Negation via ``~like(...)``."""
    assert isinstance(col, ColumnExpr)
    assert isinstance(pattern, str)
    return _FuncExpr("LIKE", col, _to_col(pattern))


def is_agg(column: Any) -> bool:
    return _is_agg(column)
