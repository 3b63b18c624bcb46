"""SQL SELECT executor over pandas frames.

This replaces the reference's ``qpd`` dependency (SQL-on-pandas used by
``QPDPandasEngine``, ``fugue/execution/native_execution_engine.py:42``):
a hand-written recursive-descent parser producing a small AST, evaluated
directly against pandas.  The same AST is consumed by the HIP SQL engine
(``fugue_amd/hip``) which maps the relational nodes to device kernels.

Supported: SELECT [DISTINCT] exprs, FROM with aliases and subqueries,
INNER/LEFT/RIGHT/FULL/CROSS JOIN ... ON equi-conjunctions (non-equi
residuals become post-filters), WHERE, GROUP BY, HAVING, ORDER BY,
LIMIT, UNION [ALL], INTERSECT, EXCEPT, CASE WHEN, IN, BETWEEN, LIKE,
CAST, COALESCE, and the standard aggregates (incl. COUNT(DISTINCT x)).
"""
import datetime
import re
from typing import Any, Callable, Dict, List, Optional, Tuple, Union

import numpy as np
import pandas as pd

from fugue_amd.schema import Schema
from fugue_amd.sql._tokenizer import Token, TokenStream, tokenize

_AGG_FUNCS = {"SUM", "MIN", "MAX", "AVG", "MEAN", "COUNT", "FIRST", "LAST"}

_KEYWORDS_STOP = {
    "FROM", "WHERE", "GROUP", "HAVING", "ORDER", "LIMIT", "UNION", "INTERSECT",
    "EXCEPT", "JOIN", "INNER", "LEFT", "RIGHT", "FULL", "CROSS", "ON", "AS",
    "BY", "ASC", "DESC", "AND", "OR", "NOT", "THEN", "ELSE", "END", "WHEN",
    "USING", "SEMI", "ANTI", "OUTER", "DISTINCT", "ALL",
}


# --------------------------------------------------------------------- #
# AST                                                                    #
# --------------------------------------------------------------------- #
class Expr:
    def eval(self, ctx: "Scope") -> Any:
        raise NotImplementedError

    @property
    def is_agg(self) -> bool:
        return False

    def default_name(self) -> str:
        return ""


class Star(Expr):
    def __init__(self, qualifier: Optional[str] = None):
        self.qualifier = qualifier


class ColRef(Expr):
    def __init__(self, name: str, qualifier: Optional[str] = None):
        self.name = name
        self.qualifier = qualifier

    def eval(self, ctx: "Scope") -> Any:
        return ctx.resolve(self.name, self.qualifier)

    def default_name(self) -> str:
        return self.name

    def __repr__(self):
        return f"ColRef({self.qualifier}.{self.name})" if self.qualifier else f"ColRef({self.name})"


class Lit(Expr):
    def __init__(self, value: Any):
        self.value = value

    def eval(self, ctx: "Scope") -> Any:
        return self.value


class BinOp(Expr):
    def __init__(self, op: str, left: Expr, right: Expr):
        self.op = op
        self.left = left
        self.right = right

    @property
    def is_agg(self) -> bool:
        return self.left.is_agg or self.right.is_agg

    def eval(self, ctx: "Scope") -> Any:
        op = self.op
        if op == "AND":
            lv = _as_bool(self.left.eval(ctx))
            rv = _as_bool(self.right.eval(ctx))
            return lv & rv
        if op == "OR":
            lv = _as_bool(self.left.eval(ctx))
            rv = _as_bool(self.right.eval(ctx))
            return lv | rv
        lv = self.left.eval(ctx)
        rv = self.right.eval(ctx)
        lv, rv = _coerce_pair(lv, rv)
        if op == "+":
            return lv + rv
        if op == "-":
            return lv - rv
        if op == "*":
            return lv * rv
        if op == "/":
            return lv / rv
        if op == "%":
            return lv % rv
        if op == "||":
            return lv.astype(str) + rv.astype(str) if isinstance(lv, pd.Series) else str(lv) + str(rv)
        if op in ("=", "=="):
            return lv == rv
        if op in ("<>", "!="):
            return lv != rv
        if op == "<":
            return lv < rv
        if op == "<=":
            return lv <= rv
        if op == ">":
            return lv > rv
        if op == ">=":
            return lv >= rv
        raise NotImplementedError(f"operator {op}")


class UnOp(Expr):
    def __init__(self, op: str, operand: Expr):
        self.op = op
        self.operand = operand

    @property
    def is_agg(self) -> bool:
        return self.operand.is_agg

    def eval(self, ctx: "Scope") -> Any:
        v = self.operand.eval(ctx)
        if self.op == "-":
            return -v
        if self.op == "NOT":
            b = _as_bool(v)
            return ~b if isinstance(b, pd.Series) else (not b)
        if self.op == "ISNULL":
            return v.isna() if isinstance(v, pd.Series) else v is None
        if self.op == "NOTNULL":
            return v.notna() if isinstance(v, pd.Series) else v is not None
        raise NotImplementedError(f"unary {self.op}")


class FuncCall(Expr):
    def __init__(self, name: str, args: List[Expr], distinct: bool = False):
        self.name = name.upper()
        self.args = args
        self.distinct = distinct

    @property
    def is_agg(self) -> bool:
        return self.name in _AGG_FUNCS or any(a.is_agg for a in self.args)

    def default_name(self) -> str:
        return self.name.lower()

    def eval(self, ctx: "Scope") -> Any:
        name = self.name
        if name in _AGG_FUNCS:
            return self._eval_agg(ctx)
        args = [a.eval(ctx) for a in self.args]
        return _eval_scalar_func(name, args, ctx)

    def _eval_agg(self, ctx: "Scope") -> Any:
        name = self.name
        if name == "COUNT":
            if len(self.args) == 1 and isinstance(self.args[0], Star):
                return len(ctx.frame)
            s = _as_series(self.args[0].eval(ctx), ctx)
            if self.distinct:
                return s.dropna().nunique()
            return int(s.notna().sum())
        s = _as_series(self.args[0].eval(ctx), ctx)
        nn = s.dropna()
        if self.distinct:
            nn = nn.drop_duplicates()
        if len(nn) == 0:
            return None
        if name == "SUM":
            return nn.sum()
        if name in ("AVG", "MEAN"):
            return nn.mean()
        if name == "MIN":
            return nn.min()
        if name == "MAX":
            return nn.max()
        if name == "FIRST":
            return nn.iloc[0]
        if name == "LAST":
            return nn.iloc[-1]
        raise NotImplementedError(name)


class SubqueryExpr(Expr):
    """Uncorrelated scalar subquery in expression position."""

    def __init__(self, stmt: "SelectStmt"):
        self.stmt = stmt

    @property
    def is_agg(self) -> bool:
        return False

    def eval(self, ctx: "Scope") -> Any:
        res = _execute_select(self.stmt, ctx.tables)
        if res.shape[1] != 1:
            raise SyntaxError("scalar subquery must return one column")
        if len(res) == 0:
            return None
        if len(res) > 1:
            raise ValueError("scalar subquery returned more than one row")
        return res.iloc[0, 0]


class InSubquery(Expr):
    """``expr [NOT] IN (SELECT ...)`` (uncorrelated)."""

    def __init__(self, expr: Expr, stmt: "SelectStmt", negate: bool):
        self.expr = expr
        self.stmt = stmt
        self.negate = negate

    @property
    def is_agg(self) -> bool:
        return self.expr.is_agg

    def eval(self, ctx: "Scope") -> Any:
        res = _execute_select(self.stmt, ctx.tables)
        if res.shape[1] != 1:
            raise SyntaxError("IN subquery must return one column")
        values = set(res.iloc[:, 0].dropna().tolist())
        v = self.expr.eval(ctx)
        s = v if isinstance(v, pd.Series) else pd.Series([v] * len(ctx.frame))
        m = s.isin(values)
        m = m.mask(s.isna())  # NULL IN (...) -> NULL
        return (~m.astype("boolean")) if self.negate else m.astype("boolean")


class ExistsSubquery(Expr):
    """``EXISTS (SELECT ...)`` (uncorrelated)."""

    def __init__(self, stmt: "SelectStmt"):
        self.stmt = stmt

    @property
    def is_agg(self) -> bool:
        return False

    def eval(self, ctx: "Scope") -> Any:
        res = _execute_select(self.stmt, ctx.tables)
        return len(res) > 0


class WindowFunc(Expr):
    """``func(args) OVER (PARTITION BY ... ORDER BY ...)``.

    Implemented in the executor itself (the reference delegates window
    functions to its SQL backends, e.g. duckdb/spark).  With ORDER BY,
    aggregate windows are cumulative (ROWS UNBOUNDED PRECEDING..CURRENT
    ROW); without, they span the whole partition."""

    def __init__(
        self,
        func: "FuncCall",
        partition_by: List[Expr],
        order_by: List["OrderItem"],
    ):
        self.func = func
        self.partition_by = partition_by
        self.order_by = order_by

    @property
    def is_agg(self) -> bool:
        return False

    def default_name(self) -> str:
        return self.func.name.lower()

    def eval(self, ctx: "Scope") -> Any:
        import numpy as np

        frame = ctx.frame
        n = len(frame)
        if n == 0:
            return pd.Series([], dtype="float64")
        work = pd.DataFrame(index=pd.RangeIndex(n))
        pcols: List[str] = []
        for i, e in enumerate(self.partition_by):
            v = e.eval(ctx)
            work[f"__p{i}"] = (
                v.reset_index(drop=True)
                if isinstance(v, pd.Series)
                else pd.Series([v] * n)
            )
            pcols.append(f"__p{i}")
        if not pcols:
            work["__p0"] = 0
            pcols = ["__p0"]
        ocols: List[str] = []
        oasc: List[bool] = []
        for i, o in enumerate(self.order_by):
            v = o.expr.eval(ctx)
            work[f"__o{i}"] = (
                v.reset_index(drop=True)
                if isinstance(v, pd.Series)
                else pd.Series([v] * n)
            )
            ocols.append(f"__o{i}")
            oasc.append(o.asc)
        fname = self.func.name
        args = self.func.args
        val: Optional[pd.Series] = None
        if len(args) > 0 and not isinstance(args[0], Star):
            v = args[0].eval(ctx)
            val = (
                v.reset_index(drop=True)
                if isinstance(v, pd.Series)
                else pd.Series([v] * n)
            )
            work["__v"] = val
        sorted_w = work.sort_values(
            pcols + ocols,
            ascending=[True] * len(pcols) + oasc,
            kind="stable",
        )
        g = sorted_w.groupby(pcols, sort=False, dropna=False)
        rn = g.cumcount() + 1
        if fname == "ROW_NUMBER":
            res = rn
        elif fname in ("RANK", "DENSE_RANK"):
            if not ocols:
                raise SyntaxError(f"{fname} requires ORDER BY")
            prev = sorted_w[pcols + ocols].shift(1)
            same_part = (
                (sorted_w[pcols] == prev[pcols]) | (sorted_w[pcols].isna() & prev[pcols].isna())
            ).all(axis=1)
            same_ord = (
                (sorted_w[ocols].values == prev[ocols].values)
                | (sorted_w[ocols].isna().values & prev[ocols].isna().values)
            ).all(axis=1)
            new_peer = ~(same_part.to_numpy() & same_ord)
            if fname == "RANK":
                res = rn.where(pd.Series(new_peer, index=rn.index))
                res = res.groupby(
                    [sorted_w[c] for c in pcols], sort=False, dropna=False
                ).ffill()
            else:
                res = (
                    pd.Series(new_peer.astype("int64"), index=sorted_w.index)
                    .groupby(
                        [sorted_w[c] for c in pcols], sort=False, dropna=False
                    )
                    .cumsum()
                )
        elif fname in ("LAG", "LEAD"):
            if val is None:
                raise SyntaxError(f"{fname} requires an argument")
            k = 1
            default = None
            if len(args) > 1 and isinstance(args[1], Lit):
                k = int(args[1].value)
            if len(args) > 2 and isinstance(args[2], Lit):
                default = args[2].value
            shift = k if fname == "LAG" else -k
            res = g["__v"].shift(shift)
            if default is not None:
                res = res.fillna(default)
        elif fname in ("FIRST_VALUE", "LAST_VALUE", "FIRST", "LAST"):
            if val is None:
                raise SyntaxError(f"{fname} requires an argument")
            which = "first" if fname in ("FIRST_VALUE", "FIRST") else "last"
            res = g["__v"].transform(which)
        elif fname in ("SUM", "MIN", "MAX", "AVG", "MEAN", "COUNT"):
            if fname == "COUNT" and val is None:
                res = g.cumcount() + 1 if ocols else g["__p0" if "__p0" in sorted_w else pcols[0]].transform("size")
            else:
                if val is None:
                    raise SyntaxError(f"{fname} requires an argument")
                if ocols:
                    gv = g["__v"]
                    if fname == "SUM":
                        res = gv.cumsum()
                    elif fname == "MIN":
                        res = gv.cummin()
                    elif fname == "MAX":
                        res = gv.cummax()
                    elif fname in ("AVG", "MEAN"):
                        res = gv.expanding().mean().reset_index(
                            level=list(range(len(pcols))), drop=True
                        )
                    else:  # COUNT(expr)
                        res = gv.expanding().count().reset_index(
                            level=list(range(len(pcols))), drop=True
                        )
                else:
                    m = {"SUM": "sum", "MIN": "min", "MAX": "max",
                         "AVG": "mean", "MEAN": "mean", "COUNT": "count"}
                    res = g["__v"].transform(m[fname])
        else:
            raise NotImplementedError(f"window function {fname}")
        return pd.Series(res, index=sorted_w.index).sort_index()


class Case(Expr):
    def __init__(self, whens: List[Tuple[Expr, Expr]], else_: Optional[Expr]):
        self.whens = whens
        self.else_ = else_

    @property
    def is_agg(self) -> bool:
        return any(c.is_agg or v.is_agg for c, v in self.whens) or (
            self.else_ is not None and self.else_.is_agg
        )

    def eval(self, ctx: "Scope") -> Any:
        n = len(ctx.frame)
        result = pd.Series([None] * n, index=ctx.frame.index, dtype=object)
        assigned = pd.Series([False] * n, index=ctx.frame.index)
        for cond, val in self.whens:
            c = _as_bool(_as_series(cond.eval(ctx), ctx)).fillna(False)
            sel = c & ~assigned
            v = _as_series(val.eval(ctx), ctx)
            result[sel] = v[sel]
            assigned = assigned | c
        if self.else_ is not None:
            v = _as_series(self.else_.eval(ctx), ctx)
            result[~assigned] = v[~assigned]
        return result.infer_objects()


class InList(Expr):
    def __init__(self, expr: Expr, values: List[Expr], negate: bool):
        self.expr = expr
        self.values = values
        self.negate = negate

    def eval(self, ctx: "Scope") -> Any:
        v = self.expr.eval(ctx)
        vals = [x.eval(ctx) for x in self.values]
        res = v.isin(vals) if isinstance(v, pd.Series) else v in vals
        if self.negate:
            res = ~res if isinstance(res, pd.Series) else not res
        return res


class Between(Expr):
    def __init__(self, expr: Expr, low: Expr, high: Expr, negate: bool):
        self.expr = expr
        self.low = low
        self.high = high
        self.negate = negate

    def eval(self, ctx: "Scope") -> Any:
        v = self.expr.eval(ctx)
        lo, hi = self.low.eval(ctx), self.high.eval(ctx)
        v2, lo = _coerce_pair(v, lo)
        v2, hi = _coerce_pair(v2, hi)
        res = (v2 >= lo) & (v2 <= hi)
        if self.negate:
            res = ~res
        return res


class Like(Expr):
    def __init__(self, expr: Expr, pattern: str, negate: bool):
        self.expr = expr
        self.pattern = pattern
        self.negate = negate

    def eval(self, ctx: "Scope") -> Any:
        regex = "^" + re.escape(self.pattern).replace("%", ".*").replace("_", ".") + "$"
        # re.escape escapes % and _? it escapes neither (word chars no), but escapes nothing for % _
        v = self.expr.eval(ctx)
        res = v.astype(str).str.match(regex)
        if self.negate:
            res = ~res
        return res


class Cast(Expr):
    def __init__(self, expr: Expr, type_name: str):
        self.expr = expr
        self.type_name = type_name.lower()

    @property
    def is_agg(self) -> bool:
        return self.expr.is_agg

    def default_name(self) -> str:
        return self.expr.default_name()

    def eval(self, ctx: "Scope") -> Any:
        v = self.expr.eval(ctx)
        return _cast_value(v, self.type_name)


# --------------------------------------------------------------------- #
# scope / helpers                                                        #
# --------------------------------------------------------------------- #
class Scope:
    """Column resolution over an internal frame whose columns are
    ``alias␟name`` pairs.  ``tables`` carries the statement's source
    tables so uncorrelated subquery expressions can execute."""

    SEP = "␟"

    def __init__(self, frame: pd.DataFrame, tables: Optional[Dict[str, pd.DataFrame]] = None):
        self.frame = frame
        self.tables = tables or {}

    def resolve(self, name: str, qualifier: Optional[str]) -> pd.Series:
        if qualifier is not None:
            key = qualifier + self.SEP + name
            if key in self.frame.columns:
                return self.frame[key]
            raise KeyError(f"column {qualifier}.{name} not found")
        matches = [
            c
            for c in self.frame.columns
            if c == name or c.split(self.SEP)[-1] == name
        ]
        if len(matches) == 0:
            raise KeyError(f"column {name} not found")
        if len(set(matches)) > 1:
            # ambiguous only if values differ; prefer exact match
            exact = [c for c in matches if c == name]
            if exact:
                return self.frame[exact[0]]
            raise KeyError(f"column {name} is ambiguous: {matches}")
        return self.frame[matches[0]]

    def all_columns(self, qualifier: Optional[str] = None) -> List[str]:
        if qualifier is None:
            return list(self.frame.columns)
        return [
            c for c in self.frame.columns if c.startswith(qualifier + self.SEP)
        ]


def _as_series(v: Any, ctx: Scope) -> pd.Series:
    if isinstance(v, pd.Series):
        return v
    return pd.Series([v] * len(ctx.frame), index=ctx.frame.index)


def _as_bool(v: Any) -> Any:
    if isinstance(v, pd.Series):
        if v.dtype == bool:
            return v
        return v.astype("boolean").fillna(False).astype(bool)
    return bool(v)


def _coerce_pair(lv: Any, rv: Any) -> Tuple[Any, Any]:
    """Handle string-literal vs datetime-column comparisons."""

    def _dt(x):
        return isinstance(x, pd.Series) and pd.api.types.is_datetime64_any_dtype(x)

    if _dt(lv) and isinstance(rv, str):
        return lv, pd.Timestamp(rv)
    if _dt(rv) and isinstance(lv, str):
        return pd.Timestamp(lv), rv
    return lv, rv


def _cast_value(v: Any, tp: str) -> Any:
    if tp in ("int", "integer", "bigint", "long", "int64", "int32", "smallint", "tinyint"):
        if isinstance(v, pd.Series):
            return pd.to_numeric(v, errors="coerce").astype("Int64")
        return int(v) if v is not None else None
    if tp in ("double", "float", "real", "float64", "float32", "decimal"):
        if isinstance(v, pd.Series):
            return pd.to_numeric(v, errors="coerce").astype(float)
        return float(v) if v is not None else None
    if tp in ("str", "string", "varchar", "text"):
        if isinstance(v, pd.Series):
            return v.astype(str).where(v.notna(), None)
        return str(v) if v is not None else None
    if tp in ("bool", "boolean"):
        if isinstance(v, pd.Series):
            return v.astype("boolean")
        return bool(v) if v is not None else None
    if tp in ("date", "datetime", "timestamp"):
        if isinstance(v, pd.Series):
            return pd.to_datetime(v)
        return pd.Timestamp(v)
    raise NotImplementedError(f"CAST to {tp}")


def _eval_scalar_func(name: str, args: List[Any], ctx: Scope) -> Any:
    if name == "COALESCE":
        out = None
        for a in args:
            s = _as_series(a, ctx)
            out = s if out is None else out.where(out.notna(), s)
        return out
    if name == "ABS":
        return args[0].abs() if isinstance(args[0], pd.Series) else abs(args[0])
    if name in ("UPPER", "UCASE"):
        return args[0].str.upper() if isinstance(args[0], pd.Series) else args[0].upper()
    if name in ("LOWER", "LCASE"):
        return args[0].str.lower() if isinstance(args[0], pd.Series) else args[0].lower()
    if name == "LENGTH":
        return args[0].str.len() if isinstance(args[0], pd.Series) else len(args[0])
    if name == "ROUND":
        nd = int(args[1]) if len(args) > 1 else 0
        return args[0].round(nd) if isinstance(args[0], pd.Series) else round(args[0], nd)
    if name == "FLOOR":
        return np.floor(args[0])
    if name == "CEIL" or name == "CEILING":
        return np.ceil(args[0])
    if name == "SQRT":
        return np.sqrt(args[0])
    if name == "CONCAT":
        out = None
        for a in args:
            s = a.astype(str) if isinstance(a, pd.Series) else str(a)
            out = s if out is None else out + s
        return out
    if name == "SUBSTRING" or name == "SUBSTR":
        start = int(args[1]) - 1
        n = int(args[2]) if len(args) > 2 else None
        s = args[0]
        if isinstance(s, pd.Series):
            return s.str.slice(start, None if n is None else start + n)
        return s[start : None if n is None else start + n]
    if name == "DATE":
        return pd.to_datetime(args[0])
    if name == "YEAR":
        return args[0].dt.year if isinstance(args[0], pd.Series) else args[0].year
    if name == "MONTH":
        return args[0].dt.month if isinstance(args[0], pd.Series) else args[0].month
    if name == "DAY":
        return args[0].dt.day if isinstance(args[0], pd.Series) else args[0].day
    raise NotImplementedError(f"function {name}")


# --------------------------------------------------------------------- #
# parser                                                                 #
# --------------------------------------------------------------------- #
class FromItem:
    def __init__(self, table: Optional[str], subquery: Optional["SelectStmt"], alias: str):
        self.table = table
        self.subquery = subquery
        self.alias = alias


class JoinClause:
    def __init__(self, how: str, item: FromItem, on: Optional[Expr], using: Optional[List[str]]):
        self.how = how
        self.item = item
        self.on = on
        self.using = using


class OrderItem:
    def __init__(self, expr: Expr, asc: bool):
        self.expr = expr
        self.asc = asc


class SelectStmt:
    def __init__(self):
        self.distinct = False
        self.columns: List[Tuple[Expr, Optional[str]]] = []
        self.from_item: Optional[FromItem] = None
        self.joins: List[JoinClause] = []
        self.where: Optional[Expr] = None
        self.group_by: List[Expr] = []
        self.having: Optional[Expr] = None
        self.order_by: List[OrderItem] = []
        self.limit: Optional[int] = None
        self.set_ops: List[Tuple[str, bool, "SelectStmt"]] = []  # (op, all, stmt)


def parse_select(sql: str) -> SelectStmt:
    """Parse a SELECT into its AST, memoized by statement text: the AST
    is read-only downstream (planner/executor never mutate it), so
    repeated identical queries skip the parse entirely."""
    hit = _SELECT_CACHE.get(sql)
    if hit is not None:
        return hit
    stmt = _parse_select_uncached(sql)
    if len(_SELECT_CACHE) > 256:
        _SELECT_CACHE.clear()
    _SELECT_CACHE[sql] = stmt
    return stmt


_SELECT_CACHE: Dict[str, SelectStmt] = {}


def _parse_select_uncached(sql: str) -> SelectStmt:
    ts = TokenStream(tokenize(sql))
    stmt = _parse_select_stmt(ts)
    if not ts.eof and not ts.match_punct(";"):
        t = ts.peek()
        raise SyntaxError(f"unexpected token {t.value!r} at {t.pos}")
    return stmt


def _parse_select_stmt(ts: TokenStream) -> SelectStmt:
    stmt = _parse_select_core(ts)
    while ts.match_kw("UNION", "INTERSECT", "EXCEPT"):
        op = ts.next().upper
        all_ = ts.take_kw("ALL")
        rhs = _parse_select_core(ts)
        stmt.set_ops.append((op, all_, rhs))
    # trailing ORDER BY / LIMIT of the combined statement
    if ts.match_kw("ORDER"):
        _parse_order_limit(ts, stmt)
    return stmt


def _parse_select_core(ts: TokenStream) -> SelectStmt:
    if ts.take_punct("("):
        stmt = _parse_select_stmt(ts)
        ts.expect_punct(")")
        return stmt
    ts.expect_kw("SELECT")
    stmt = SelectStmt()
    if ts.take_kw("DISTINCT"):
        stmt.distinct = True
    else:
        ts.take_kw("ALL")
    while True:
        expr = _parse_expr(ts)
        alias: Optional[str] = None
        if ts.take_kw("AS"):
            alias = ts.next().value
        elif (
            ts.peek() is not None
            and ts.peek().kind == "NAME"
            and ts.peek().upper not in _KEYWORDS_STOP
        ):
            alias = ts.next().value
        stmt.columns.append((expr, alias))
        if not ts.take_punct(","):
            break
    if ts.take_kw("FROM"):
        stmt.from_item = _parse_from_item(ts)
        while True:
            how = None
            if ts.take_kw("CROSS"):
                ts.expect_kw("JOIN")
                how = "cross"
            elif ts.take_kw("INNER"):
                ts.expect_kw("JOIN")
                how = "inner"
            elif ts.match_kw("LEFT", "RIGHT", "FULL"):
                d = ts.next().upper
                if ts.take_kw("SEMI"):
                    how = "semi"
                elif ts.take_kw("ANTI"):
                    how = "anti"
                else:
                    ts.take_kw("OUTER")
                    how = {"LEFT": "left", "RIGHT": "right", "FULL": "outer"}[d]
                ts.expect_kw("JOIN")
            elif ts.take_kw("JOIN"):
                how = "inner"
            else:
                break
            item = _parse_from_item(ts)
            on: Optional[Expr] = None
            using: Optional[List[str]] = None
            if ts.take_kw("ON"):
                on = _parse_expr(ts)
            elif ts.take_kw("USING"):
                ts.expect_punct("(")
                using = []
                while True:
                    using.append(ts.next().value)
                    if not ts.take_punct(","):
                        break
                ts.expect_punct(")")
            stmt.joins.append(JoinClause(how, item, on, using))
    if ts.take_kw("WHERE"):
        stmt.where = _parse_expr(ts)
    if ts.take_kw("GROUP"):
        ts.expect_kw("BY")
        while True:
            stmt.group_by.append(_parse_expr(ts))
            if not ts.take_punct(","):
                break
    if ts.take_kw("HAVING"):
        stmt.having = _parse_expr(ts)
    _parse_order_limit(ts, stmt)
    return stmt


def _parse_order_limit(ts: TokenStream, stmt: SelectStmt) -> None:
    if ts.take_kw("ORDER"):
        ts.expect_kw("BY")
        while True:
            e = _parse_expr(ts)
            asc = True
            if ts.take_kw("DESC"):
                asc = False
            else:
                ts.take_kw("ASC")
            if ts.take_kw("NULLS"):
                ts.next()  # FIRST/LAST — ignored (pandas default)
            stmt.order_by.append(OrderItem(e, asc))
            if not ts.take_punct(","):
                break
    if ts.take_kw("LIMIT"):
        stmt.limit = int(ts.next().value)


def _parse_from_item(ts: TokenStream) -> FromItem:
    if ts.take_punct("("):
        sub = _parse_select_stmt(ts)
        ts.expect_punct(")")
        alias = None
        if ts.take_kw("AS"):
            alias = ts.next().value
        elif ts.peek() is not None and ts.peek().kind == "NAME" and ts.peek().upper not in _KEYWORDS_STOP:
            alias = ts.next().value
        return FromItem(None, sub, alias or "_subq")
    name = ts.next().value
    alias = name
    if ts.take_kw("AS"):
        alias = ts.next().value
    elif (
        ts.peek() is not None
        and ts.peek().kind == "NAME"
        and ts.peek().upper not in _KEYWORDS_STOP
    ):
        alias = ts.next().value
    return FromItem(name, None, alias)


# precedence: OR < AND < NOT < comparison < add < mul < unary
def _parse_expr(ts: TokenStream) -> Expr:
    return _parse_or(ts)


def _parse_or(ts: TokenStream) -> Expr:
    left = _parse_and(ts)
    while ts.take_kw("OR"):
        left = BinOp("OR", left, _parse_and(ts))
    return left


def _parse_and(ts: TokenStream) -> Expr:
    left = _parse_not(ts)
    while ts.take_kw("AND"):
        left = BinOp("AND", left, _parse_not(ts))
    return left


def _parse_not(ts: TokenStream) -> Expr:
    if ts.take_kw("NOT"):
        return UnOp("NOT", _parse_not(ts))
    return _parse_comparison(ts)


def _parse_comparison(ts: TokenStream) -> Expr:
    left = _parse_additive(ts)
    t = ts.peek()
    if t is not None and t.kind == "OP" and t.value in ("=", "==", "<>", "!=", "<", "<=", ">", ">="):
        ts.next()
        right = _parse_additive(ts)
        return BinOp(t.value, left, right)
    if ts.match_kw("IS"):
        ts.next()
        negate = ts.take_kw("NOT")
        ts.expect_kw("NULL")
        return UnOp("NOTNULL" if negate else "ISNULL", left)
    negate = False
    if ts.match_kw("NOT") and ts.peek(1) is not None and ts.peek(1).upper in ("IN", "BETWEEN", "LIKE"):
        ts.next()
        negate = True
    if ts.take_kw("IN"):
        ts.expect_punct("(")
        if ts.match_kw("SELECT"):
            sub = _parse_select_stmt(ts)
            ts.expect_punct(")")
            return InSubquery(left, sub, negate)
        vals: List[Expr] = []
        while True:
            vals.append(_parse_expr(ts))
            if not ts.take_punct(","):
                break
        ts.expect_punct(")")
        return InList(left, vals, negate)
    if ts.take_kw("BETWEEN"):
        lo = _parse_additive(ts)
        ts.expect_kw("AND")
        hi = _parse_additive(ts)
        return Between(left, lo, hi, negate)
    if ts.take_kw("LIKE"):
        pat = ts.next()
        if pat.kind != "STRING":
            raise SyntaxError("LIKE requires a string literal")
        return Like(left, pat.value[1:-1], negate)
    return left


def _parse_additive(ts: TokenStream) -> Expr:
    left = _parse_multiplicative(ts)
    while True:
        t = ts.peek()
        if t is not None and t.kind == "OP" and t.value in ("+", "-", "||"):
            ts.next()
            left = BinOp(t.value, left, _parse_multiplicative(ts))
        else:
            return left


def _parse_multiplicative(ts: TokenStream) -> Expr:
    left = _parse_unary(ts)
    while True:
        t = ts.peek()
        if t is not None and t.kind == "OP" and t.value in ("*", "/", "%"):
            ts.next()
            left = BinOp(t.value, left, _parse_unary(ts))
        else:
            return left


def _parse_unary(ts: TokenStream) -> Expr:
    t = ts.peek()
    if t is not None and t.kind == "OP" and t.value == "-":
        ts.next()
        return UnOp("-", _parse_unary(ts))
    if t is not None and t.kind == "OP" and t.value == "+":
        ts.next()
        return _parse_unary(ts)
    return _parse_primary(ts)


def _parse_primary(ts: TokenStream) -> Expr:
    t = ts.peek()
    if t is None:
        raise SyntaxError("unexpected end of expression")
    if t.kind == "NUMBER":
        ts.next()
        if "." in t.value:
            return Lit(float(t.value))
        return Lit(int(t.value))
    if t.kind == "STRING":
        ts.next()
        return Lit(t.value[1:-1].replace("''", "'"))
    if ts.take_punct("("):
        if ts.match_kw("SELECT"):
            sub = _parse_select_stmt(ts)
            ts.expect_punct(")")
            return SubqueryExpr(sub)
        e = _parse_expr(ts)
        ts.expect_punct(")")
        return e
    if t.kind == "OP" and t.value == "*":
        ts.next()
        return Star()
    if t.kind == "NAME":
        up = t.upper
        if up == "CASE":
            ts.next()
            whens: List[Tuple[Expr, Expr]] = []
            base: Optional[Expr] = None
            if not ts.match_kw("WHEN"):
                base = _parse_expr(ts)
            while ts.take_kw("WHEN"):
                cond = _parse_expr(ts)
                ts.expect_kw("THEN")
                val = _parse_expr(ts)
                if base is not None:
                    cond = BinOp("=", base, cond)
                whens.append((cond, val))
            else_: Optional[Expr] = None
            if ts.take_kw("ELSE"):
                else_ = _parse_expr(ts)
            ts.expect_kw("END")
            return Case(whens, else_)
        if up == "CAST":
            ts.next()
            ts.expect_punct("(")
            e = _parse_expr(ts)
            ts.expect_kw("AS")
            tp = ts.next().value
            # allow parameterized types like decimal(10,2)
            if ts.take_punct("("):
                while not ts.take_punct(")"):
                    ts.next()
            ts.expect_punct(")")
            return Cast(e, tp)
        if up in ("DATE", "TIMESTAMP") and ts.peek(1) is not None and ts.peek(1).kind == "STRING":
            ts.next()
            s = ts.next().value[1:-1]
            return Lit(pd.Timestamp(s))
        if up == "EXISTS" and ts.peek(1) is not None and ts.peek(1).value == "(":
            ts.next()
            ts.expect_punct("(")
            sub = _parse_select_stmt(ts)
            ts.expect_punct(")")
            return ExistsSubquery(sub)
        if up == "NULL":
            ts.next()
            return Lit(None)
        if up == "TRUE":
            ts.next()
            return Lit(True)
        if up == "FALSE":
            ts.next()
            return Lit(False)
        # function call?
        if ts.peek(1) is not None and ts.peek(1).kind == "PUNCT" and ts.peek(1).value == "(":
            name = ts.next().value
            ts.expect_punct("(")
            distinct = ts.take_kw("DISTINCT")
            args: List[Expr] = []
            if not ts.match_punct(")"):
                while True:
                    args.append(_parse_expr(ts))
                    if not ts.take_punct(","):
                        break
            ts.expect_punct(")")
            fc = FuncCall(name, args, distinct=distinct)
            if ts.take_kw("OVER"):
                ts.expect_punct("(")
                partition_by: List[Expr] = []
                order_by: List[OrderItem] = []
                if ts.take_kw("PARTITION"):
                    ts.expect_kw("BY")
                    while True:
                        partition_by.append(_parse_expr(ts))
                        if not ts.take_punct(","):
                            break
                if ts.take_kw("ORDER"):
                    ts.expect_kw("BY")
                    while True:
                        e = _parse_expr(ts)
                        asc = True
                        if ts.take_kw("DESC"):
                            asc = False
                        elif ts.take_kw("ASC"):
                            pass
                        order_by.append(OrderItem(e, asc))
                        if not ts.take_punct(","):
                            break
                nt = ts.peek()
                if nt is not None and nt.kind == "NAME" and nt.upper in (
                    "ROWS", "RANGE", "GROUPS",
                ):
                    raise NotImplementedError(
                        "explicit window frame specs are not supported"
                    )
                ts.expect_punct(")")
                return WindowFunc(fc, partition_by, order_by)
            return fc
        # qualified / unqualified column, or alias.*
        name = ts.next().value
        if ts.match_punct(".") and ts.peek(1) is not None:
            ts.next()
            nxt = ts.next()
            if nxt.kind == "OP" and nxt.value == "*":
                return Star(qualifier=name)
            return ColRef(nxt.value, qualifier=name)
        return ColRef(name)
    raise SyntaxError(f"unexpected token {t.value!r} at {t.pos}")


# --------------------------------------------------------------------- #
# executor                                                               #
# --------------------------------------------------------------------- #
SEP = Scope.SEP


def _internal_frame(df: pd.DataFrame, alias: str) -> pd.DataFrame:
    out = df.copy()
    out.columns = [alias + SEP + str(c) for c in df.columns]
    return out


def _resolve_from(item: FromItem, tables: Dict[str, pd.DataFrame]) -> pd.DataFrame:
    if item.subquery is not None:
        sub = _execute_select(item.subquery, tables)
        return _internal_frame(sub, item.alias)
    key = item.table
    if key not in tables:
        raise KeyError(f"table {key} not found; available: {list(tables)}")
    return _internal_frame(tables[key], item.alias)


def _extract_equi_keys(
    on: Expr, left_cols: List[str], right_cols: List[str]
) -> Tuple[List[str], List[str], Optional[Expr]]:
    """Split an ON condition into equi-join key pairs + residual filter."""
    lkeys: List[str] = []
    rkeys: List[str] = []
    residual: Optional[Expr] = None

    def _col_key(e: Expr, cols: List[str]) -> Optional[str]:
        if not isinstance(e, ColRef):
            return None
        for c in cols:
            alias, _, name = c.partition(SEP)
            if e.name == name and (e.qualifier is None or e.qualifier == alias):
                return c
        return None

    def _walk(e: Expr) -> Optional[Expr]:
        nonlocal lkeys, rkeys
        if isinstance(e, BinOp) and e.op == "AND":
            l = _walk(e.left)
            r = _walk(e.right)
            if l is None:
                return r
            if r is None:
                return l
            return BinOp("AND", l, r)
        if isinstance(e, BinOp) and e.op in ("=", "=="):
            lk = _col_key(e.left, left_cols)
            rk = _col_key(e.right, right_cols)
            if lk is not None and rk is not None:
                lkeys.append(lk)
                rkeys.append(rk)
                return None
            lk2 = _col_key(e.right, left_cols)
            rk2 = _col_key(e.left, right_cols)
            if lk2 is not None and rk2 is not None:
                lkeys.append(lk2)
                rkeys.append(rk2)
                return None
        return e

    residual = _walk(on)
    return lkeys, rkeys, residual


def _apply_join(
    left: pd.DataFrame, join: JoinClause, tables: Dict[str, pd.DataFrame]
) -> pd.DataFrame:
    right = _resolve_from(join.item, tables)
    how = join.how
    if how == "cross":
        lf = left.assign(__x__=1)
        rf = right.assign(__x__=1)
        return lf.merge(rf, on="__x__").drop(columns="__x__")
    lkeys: List[str] = []
    rkeys: List[str] = []
    residual: Optional[Expr] = None
    if join.using is not None:
        for name in join.using:
            lmatch = [c for c in left.columns if c.split(SEP)[-1] == name]
            rmatch = [c for c in right.columns if c.split(SEP)[-1] == name]
            if not lmatch or not rmatch:
                raise KeyError(f"USING column {name} not found")
            lkeys.append(lmatch[0])
            rkeys.append(rmatch[0])
    elif join.on is not None:
        lkeys, rkeys, residual = _extract_equi_keys(
            join.on, list(left.columns), list(right.columns)
        )
        if len(lkeys) == 0:
            raise NotImplementedError(
                "JOIN requires at least one equi condition in ON"
            )
    else:
        raise SyntaxError("JOIN requires ON or USING")
    if how in ("semi", "anti"):
        keys = right[rkeys].drop_duplicates()
        keys.columns = lkeys
        if how == "semi":
            return left.merge(keys, on=lkeys, how="inner")
        marked = keys.assign(__anti__=1)
        res = left.merge(marked, on=lkeys, how="left")
        return res[res["__anti__"].isna()].drop(columns="__anti__")
    pd_how = {"inner": "inner", "left": "left", "right": "right", "outer": "outer"}[how]
    res = left.merge(right, left_on=lkeys, right_on=rkeys, how=pd_how)
    if residual is not None:
        mask = _as_bool(_as_series(residual.eval(Scope(res)), Scope(res)))
        if how == "inner":
            res = res[mask]
        else:  # residual on outer joins: approximate by post-filter on matched rows
            res = res[mask | res[rkeys[0]].isna() | res[lkeys[0]].isna()]
    return res.reset_index(drop=True)


def _output_name(expr: Expr, alias: Optional[str], idx: int) -> str:
    if alias is not None:
        return alias
    d = expr.default_name()
    if d != "":
        return d
    return f"_col{idx}"


def _execute_select(stmt: SelectStmt, tables: Dict[str, pd.DataFrame]) -> pd.DataFrame:
    res = _execute_core(stmt, tables)
    for op, all_, rhs in stmt.set_ops:
        other = _execute_core(rhs, tables)
        other.columns = res.columns
        if op == "UNION":
            res = pd.concat([res, other], ignore_index=True)
            if not all_:
                res = res.drop_duplicates(ignore_index=True)
        elif op == "INTERSECT":
            res = res.merge(other.drop_duplicates(), on=list(res.columns), how="inner")
            res = res.drop_duplicates(ignore_index=True)
        elif op == "EXCEPT":
            marked = other.drop_duplicates().assign(__exc__=1)
            res = res.merge(marked, on=list(res.columns), how="left")
            res = res[res["__exc__"].isna()].drop(columns="__exc__")
            res = res.drop_duplicates(ignore_index=True)
    if len(stmt.set_ops) > 0 and len(stmt.order_by) > 0:
        res = _apply_order_limit_plain(res, stmt)
    return res.reset_index(drop=True)


def _apply_order_limit_plain(res: pd.DataFrame, stmt: SelectStmt) -> pd.DataFrame:
    # order by output column names after set ops
    names = [
        o.expr.name if isinstance(o.expr, ColRef) else None for o in stmt.order_by
    ]
    if any(n is None for n in names):
        raise NotImplementedError("ORDER BY after set ops must use column names")
    res = res.sort_values(names, ascending=[o.asc for o in stmt.order_by])
    if stmt.limit is not None:
        res = res.head(stmt.limit)
    return res


def _execute_core(stmt: SelectStmt, tables: Dict[str, pd.DataFrame]) -> pd.DataFrame:
    # FROM + JOINs
    if stmt.from_item is None:
        frame = pd.DataFrame({"__dummy__": [0]})
    else:
        frame = _resolve_from(stmt.from_item, tables)
        for j in stmt.joins:
            frame = _apply_join(frame, j, tables)
    scope = Scope(frame, tables)
    # WHERE
    if stmt.where is not None:
        mask = _as_bool(_as_series(stmt.where.eval(scope), scope))
        frame = frame[mask.to_numpy(dtype=bool)].reset_index(drop=True)
        scope = Scope(frame, tables)
    has_agg = any(c[0].is_agg for c in stmt.columns) or len(stmt.group_by) > 0
    if not has_agg:
        out: Dict[str, Any] = {}
        idx = 0
        for expr, alias in stmt.columns:
            if isinstance(expr, Star):
                for c in scope.all_columns(expr.qualifier):
                    out[c.split(SEP)[-1]] = frame[c].reset_index(drop=True)
                continue
            v = expr.eval(scope)
            s = v if isinstance(v, pd.Series) else pd.Series([v] * len(frame))
            out[_output_name(expr, alias, idx)] = s.reset_index(drop=True)
            idx += 1
        res = pd.DataFrame(out) if out else pd.DataFrame(index=range(len(frame)))
    else:
        res = _execute_groupby(stmt, frame, scope)
    # HAVING handled inside groupby; DISTINCT
    if stmt.distinct:
        res = res.drop_duplicates(ignore_index=True)
    # ORDER BY / LIMIT (on output columns or original exprs)
    if len(stmt.order_by) > 0:
        sort_cols: List[str] = []
        tmp = res.copy()
        drop_cols = []
        for i, o in enumerate(stmt.order_by):
            if isinstance(o.expr, ColRef) and o.expr.qualifier is None and o.expr.name in res.columns:
                sort_cols.append(o.expr.name)
            elif isinstance(o.expr, Lit) and isinstance(o.expr.value, int):
                sort_cols.append(res.columns[o.expr.value - 1])
            else:
                if has_agg:
                    raise NotImplementedError(
                        "ORDER BY expression must be an output column for aggregates"
                    )
                cname = f"__ord{i}__"
                tmp[cname] = _as_series(o.expr.eval(scope), scope).reset_index(drop=True)
                sort_cols.append(cname)
                drop_cols.append(cname)
        tmp = tmp.sort_values(sort_cols, ascending=[o.asc for o in stmt.order_by], kind="stable")
        res = tmp.drop(columns=drop_cols).reset_index(drop=True)
    if stmt.limit is not None:
        res = res.head(stmt.limit).reset_index(drop=True)
    return res


def _execute_groupby(stmt: SelectStmt, frame: pd.DataFrame, scope: Scope) -> pd.DataFrame:
    out_names: List[str] = []
    for i, (expr, alias) in enumerate(stmt.columns):
        if isinstance(expr, Star):
            raise NotImplementedError("* can't be used with GROUP BY/aggregates")
        out_names.append(_output_name(expr, alias, i))
    if len(stmt.group_by) == 0:
        row: Dict[str, Any] = {}
        for (expr, alias), name in zip(stmt.columns, out_names):
            row[name] = expr.eval(scope)
        res = pd.DataFrame([row], columns=out_names)
        if stmt.having is not None:
            hscope = Scope(res)
            mask = _as_bool(_as_series(stmt.having.eval(hscope), hscope))
            res = res[mask.to_numpy(dtype=bool)]
        return res.reset_index(drop=True)
    # evaluate group keys
    keycols: List[pd.Series] = []
    keynames: List[str] = []
    alias_by_id: Dict[int, str] = {}
    for i, g in enumerate(stmt.group_by):
        if isinstance(g, Lit) and isinstance(g.value, int):
            # positional group by
            expr = stmt.columns[g.value - 1][0]
            name = out_names[g.value - 1]
        else:
            expr = g
            # find matching output column
            name = None
            for (ce, alias), on in zip(stmt.columns, out_names):
                if _expr_eq(ce, g):
                    name = on
                    break
            if name is None:
                name = f"__g{i}__"
        keycols.append(_as_series(expr.eval(scope), scope).reset_index(drop=True))
        keynames.append(name)
    work = frame.reset_index(drop=True)
    keydf = pd.DataFrame({n: s for n, s in zip(keynames, keycols)})
    rows: List[Dict[str, Any]] = []
    grouped = keydf.groupby(keynames, dropna=False, sort=False)
    for key_vals, idx in grouped.indices.items():
        sub = work.iloc[idx]
        subscope = Scope(sub)
        if not isinstance(key_vals, tuple):
            key_vals = (key_vals,)
        row = dict(zip(keynames, key_vals))
        for (expr, alias), name in zip(stmt.columns, out_names):
            if name in row:
                continue
            row[name] = expr.eval(subscope)
        if stmt.having is not None:
            hrow = dict(row)
            hframe = pd.DataFrame([hrow])
            # having may reference aggregates not in output; evaluate on sub
            hv = _eval_having(stmt.having, subscope, hframe)
            if not hv:
                continue
        rows.append(row)
    return pd.DataFrame(rows, columns=out_names) if rows else pd.DataFrame(
        columns=out_names
    )


def _eval_having(expr: Expr, subscope: Scope, row_frame: pd.DataFrame) -> bool:
    v = expr.eval(subscope)
    if isinstance(v, pd.Series):
        v = bool(v.iloc[0]) if len(v) else False
    return bool(v)


def _expr_eq(a: Expr, b: Expr) -> bool:
    if isinstance(a, ColRef) and isinstance(b, ColRef):
        return a.name == b.name and (
            a.qualifier == b.qualifier or a.qualifier is None or b.qualifier is None
        )
    return False


def run_sql_on_pandas(
    sql: str,
    dfs: Dict[str, pd.DataFrame],
    schemas: Optional[Dict[str, Schema]] = None,
) -> Tuple[pd.DataFrame, Optional[Schema]]:
    """Execute one SELECT statement against named pandas frames.
    Returns (result, inferred schema or None)."""
    stmt = parse_select(sql)
    res = _execute_select(stmt, dfs)
    return res.reset_index(drop=True), None
