"""Column-expression interpreter over pandas.

This is the MI355X framework's replacement for the reference's
SQL-generation path (``fugue/execution/execution_engine.py:736-939``
compiles select/filter/assign/aggregate to SQL text and routes through a
SQL engine): here the expression tree is evaluated directly against the
frame — on pandas for the CPU engine, and on device columns for the HIP
engine (see ``fugue_amd/hip``).
"""
from typing import Any, Dict, List, Optional

import numpy as np
import pandas as pd
import pyarrow as pa

from fugue_amd.column.expressions import (
    ColumnExpr,
    _BinaryOpExpr,
    _FuncExpr,
    _LiteralColumnExpr,
    _NamedColumnExpr,
    _NotOpExpr,
    _UnaryAggFuncExpr,
    _UnaryOpExpr,
    _WildcardExpr,
    _is_agg,
)
from fugue_amd.column.sql import SelectColumns
from fugue_amd.schema import Schema
from fugue_amd.utils.pandas_like import cast_pandas


def _cast_series(s: pd.Series, tp: pa.DataType) -> pd.Series:
    arr = pa.Array.from_pandas(s)
    if arr.type != tp:
        arr = arr.cast(tp, safe=False)
    return arr.to_pandas()


def eval_scalar_expr(expr: ColumnExpr, df: pd.DataFrame) -> pd.Series:
    """Evaluate a non-aggregate expression to a Series aligned with df."""
    res = _eval(expr, df)
    if not isinstance(res, pd.Series):
        res = pd.Series([res] * len(df), index=df.index)
    if expr.as_type is not None:
        res = pd.Series(
            _cast_series(res.reset_index(drop=True), expr.as_type).values,
            index=df.index,
        )
    return res


def _eval(expr: ColumnExpr, df: pd.DataFrame) -> Any:
    if isinstance(expr, _LiteralColumnExpr):
        return expr.value
    if isinstance(expr, _NamedColumnExpr):
        return df[expr.name]
    if isinstance(expr, _NotOpExpr):
        v = _eval(expr.col, df)
        return ~v.astype("boolean") if isinstance(v, pd.Series) else (not v)
    if isinstance(expr, _UnaryOpExpr):
        v = _eval(expr.col, df)
        if expr.op == "IS_NULL":
            return v.isna() if isinstance(v, pd.Series) else v is None
        if expr.op == "NOT_NULL":
            return v.notna() if isinstance(v, pd.Series) else v is not None
        if expr.op == "-":
            return -v
        raise NotImplementedError(f"unary op {expr.op}")
    if isinstance(expr, _BinaryOpExpr):
        left = _eval(expr.left, df)
        right = _eval(expr.right, df)
        op = expr.op
        if op == "&" or op == "|":
            lb = left.astype("boolean") if isinstance(left, pd.Series) else left
            rb = right.astype("boolean") if isinstance(right, pd.Series) else right
            return (lb & rb) if op == "&" else (lb | rb)
        if op == "+":
            return left + right
        if op == "-":
            return left - right
        if op == "*":
            return left * right
        if op == "/":
            return left / right
        if op == "==":
            return left == right
        if op == "!=":
            return left != right
        if op == "<":
            return left < right
        if op == "<=":
            return left <= right
        if op == ">":
            return left > right
        if op == ">=":
            return left >= right
        raise NotImplementedError(f"binary op {op}")
    if isinstance(expr, _FuncExpr) and not isinstance(expr, _UnaryAggFuncExpr):
        fname = expr.func.upper()
        if fname == "COALESCE":
            out: Optional[pd.Series] = None
            for a in expr.args:
                v = _eval(a, df)
                s = (
                    v
                    if isinstance(v, pd.Series)
                    else pd.Series([v] * len(df), index=df.index)
                )
                out = s if out is None else out.where(out.notna(), s)
            return out
        if fname == "LIKE":
            import re as _re

            v = _eval(expr.args[0], df)
            pat = _eval(expr.args[1], df)
            if isinstance(pat, pd.Series):
                pat = pat.iloc[0]
            regex = "^" + _re.escape(str(pat)).replace("%", ".*").replace(
                "_", "."
            ) + "$"
            s = v if isinstance(v, pd.Series) else pd.Series([v] * len(df), index=df.index)
            res = s.astype(str).str.match(regex).astype("boolean")
            # SQL three-valued logic: NULL LIKE p → NULL (so NOT LIKE
            # also excludes nulls)
            return res.mask(s.isna())
        if fname == "CASE_WHEN":
            args = expr.args
            n = len(df)
            idx = df.index

            def _as_series(v: Any) -> pd.Series:
                return v if isinstance(v, pd.Series) else pd.Series([v] * n, index=idx)

            out = _as_series(_eval(args[-1], df))  # ELSE
            # apply WHEN branches in reverse so the FIRST match wins
            for i in range(len(args) - 2, 0, -2):
                cond = _as_series(_eval(args[i - 1], df)).astype("boolean").fillna(False)
                val = _as_series(_eval(args[i], df))
                out = val.where(cond.to_numpy(dtype=bool), out)
            return out
        raise NotImplementedError(f"function {expr.func}")
    raise NotImplementedError(f"can't evaluate {expr}")


def eval_filter(df: pd.DataFrame, condition: ColumnExpr) -> pd.DataFrame:
    mask = eval_scalar_expr(condition, df)
    mask = mask.astype("boolean").fillna(False)
    return df[mask.to_numpy(dtype=bool)].reset_index(drop=True)


_AGG_MAP = {
    "MIN": "min",
    "MAX": "max",
    "SUM": "sum",
    "AVG": "mean",
    "COUNT": "count",
    "FIRST": "first",
    "LAST": "last",
}


def _eval_agg_series(expr: _UnaryAggFuncExpr, df: pd.DataFrame) -> Any:
    """Evaluate one aggregation over the whole frame, returning a scalar."""
    fname = expr.func.upper()
    argexpr = expr.args[0]
    if isinstance(argexpr, _WildcardExpr) or (
        isinstance(argexpr, _NamedColumnExpr) and argexpr.name == "*"
    ):
        if fname == "COUNT":
            return len(df)
        raise NotImplementedError(f"{fname}(*)")
    s = eval_scalar_expr(argexpr, df)
    if expr.is_distinct:
        if fname == "COUNT":
            return s.dropna().nunique()
        if fname in ("SUM", "AVG", "MIN", "MAX"):
            s = s.drop_duplicates()
        else:
            raise NotImplementedError(f"DISTINCT not supported for {fname}")
    if fname == "COUNT":
        return int(s.notna().sum())
    if fname == "MIN":
        return s.min() if s.notna().any() else None
    if fname == "MAX":
        return s.max() if s.notna().any() else None
    if fname == "SUM":
        return s.sum() if s.notna().any() else None
    if fname == "AVG":
        return s.mean() if s.notna().any() else None
    if fname == "FIRST":
        nn = s.dropna()
        return nn.iloc[0] if len(nn) > 0 else None
    if fname == "LAST":
        nn = s.dropna()
        return nn.iloc[-1] if len(nn) > 0 else None
    raise NotImplementedError(f"aggregation {fname}")


def eval_select(
    df: pd.DataFrame,
    input_schema: Schema,
    columns: SelectColumns,
    where: Optional[ColumnExpr] = None,
    having: Optional[ColumnExpr] = None,
) -> pd.DataFrame:
    """Full select semantics: optional where → projection/aggregation →
    optional having → optional distinct."""
    cols = columns.replace_wildcard(input_schema)
    if where is not None:
        df = eval_filter(df, where)
    if not cols.has_agg:
        out: Dict[str, pd.Series] = {}
        for c in cols.all_cols:
            name = c.output_name if c.output_name != "" else c.name
            out[name] = eval_scalar_expr(c, df).reset_index(drop=True)
        res = pd.DataFrame(out)
    else:
        group_names = [k.output_name or k.name for k in cols.group_keys]
        if len(cols.group_keys) == 0:
            row: Dict[str, Any] = {}
            for c in cols.all_cols:
                row[c.output_name] = _eval_one_agg_cell(c, df)
            res = pd.DataFrame([row])
        else:
            # evaluate group key expressions, then aggregate per group
            keydf = pd.DataFrame(
                {
                    (k.output_name or k.name): eval_scalar_expr(k, df).reset_index(
                        drop=True
                    )
                    for k in cols.group_keys
                }
            )
            work = df.reset_index(drop=True)
            rows: List[Dict[str, Any]] = []
            grouped = keydf.groupby(group_names, dropna=False, sort=False)
            for key_vals, idx in grouped.indices.items():
                sub = work.iloc[idx]
                if len(group_names) == 1 and not isinstance(key_vals, tuple):
                    key_vals = (key_vals,)
                row = dict(zip(group_names, key_vals))
                for c in cols.all_cols:
                    name = c.output_name
                    if name in row:
                        continue
                    row[name] = _eval_one_agg_cell(c, sub)
                if having is not None and _is_agg(having):
                    # HAVING with aggregate terms (possibly absent from
                    # the select list): evaluate per group
                    row["__fugue_having"] = bool(
                        _eval_one_agg_cell(having, sub)
                    )
                rows.append(row)
            names = [c.output_name for c in cols.all_cols]
            if having is not None and _is_agg(having):
                names = names + ["__fugue_having"]
            res = pd.DataFrame(rows, columns=names) if rows else pd.DataFrame(
                columns=names
            )
            if having is not None and _is_agg(having):
                res = res[res["__fugue_having"]].drop(
                    columns=["__fugue_having"]
                )
                having = None
        if having is not None:
            res = eval_filter(res, having)
    if cols.is_distinct:
        from fugue_amd.utils.pandas_like import drop_duplicates

        res = drop_duplicates(res)
    return res.reset_index(drop=True)


def _eval_one_agg_cell(c: ColumnExpr, df: pd.DataFrame) -> Any:
    """Evaluate a (possibly compound) aggregate expression to a scalar."""
    if isinstance(c, _UnaryAggFuncExpr):
        return _apply_cast(_eval_agg_series(c, df), c)
    if isinstance(c, _BinaryOpExpr):
        left = _eval_one_agg_cell(c.left, df) if _is_agg(c.left) else _scalar(c.left, df)
        right = (
            _eval_one_agg_cell(c.right, df) if _is_agg(c.right) else _scalar(c.right, df)
        )
        op = c.op
        try:
            if op == "+":
                v = left + right
            elif op == "-":
                v = left - right
            elif op == "*":
                v = left * right
            elif op == "/":
                v = left / right
            elif op == "==":
                v = left == right
            elif op == "!=":
                v = left != right
            elif op == "<":
                v = left < right
            elif op == "<=":
                v = left <= right
            elif op == ">":
                v = left > right
            elif op == ">=":
                v = left >= right
            elif op == "&":
                v = bool(left) and bool(right)
            elif op == "|":
                v = bool(left) or bool(right)
            else:
                raise NotImplementedError(f"agg op {op}")
        except TypeError:
            v = None
        return _apply_cast(v, c)
    if isinstance(c, _LiteralColumnExpr):
        return c.value
    if isinstance(c, _NamedColumnExpr):
        # a group key evaluated elsewhere; fallback to first value
        return df[c.name].iloc[0] if len(df) > 0 else None
    raise NotImplementedError(f"can't aggregate {c}")


def _scalar(c: ColumnExpr, df: pd.DataFrame) -> Any:
    if isinstance(c, _LiteralColumnExpr):
        return c.value
    s = eval_scalar_expr(c, df)
    return s.iloc[0] if len(s) > 0 else None


def _apply_cast(v: Any, c: ColumnExpr) -> Any:
    if c.as_type is None or v is None:
        return v
    try:
        return pa.scalar(v).cast(c.as_type).as_py()
    except Exception:
        return v
