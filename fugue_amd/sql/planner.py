"""SQL plan → ExecutionEngine lowering.

Takes the parsed ``SelectStmt`` AST (``fugue_amd/sql/executor.py``) and
executes it through ``ExecutionEngine`` relational ops (join / filter /
select / aggregate / union / take), so a FugueSQL SELECT runs on the HIP
engine's device kernels with the distributed shuffle — instead of
gathering to pandas.  Raises :class:`UnsupportedPlan` for shapes outside
the lowering (CASE/LIKE, non-equi joins, correlated subqueries, ...); the
SQL facet then falls back to the pandas executor.
"""
from typing import Any, Dict, List, Optional, Tuple

from fugue_amd.column import functions as F
from fugue_amd.column.expressions import ColumnExpr, col, lit
from fugue_amd.column.sql import SelectColumns
from fugue_amd.dataframe.dataframe import DataFrame
from fugue_amd.schema import Schema
from fugue_amd.sql import executor as X


class UnsupportedPlan(Exception):
    pass


def _to_column_expr(e: X.Expr, schema: Schema, alias_map: Dict[str, str]) -> ColumnExpr:
    if isinstance(e, X.ColRef):
        name = e.name
        if name not in schema:
            raise UnsupportedPlan(f"column {name} not in {schema}")
        return col(name)
    if isinstance(e, X.Lit):
        v = e.value
        if v is None or isinstance(v, (bool, int, float, str)):
            return lit(v)
        raise UnsupportedPlan(f"literal {v!r}")
    if isinstance(e, X.BinOp):
        l = _to_column_expr(e.left, schema, alias_map)
        r = _to_column_expr(e.right, schema, alias_map)
        op = e.op
        if op == "AND":
            return l & r
        if op == "OR":
            return l | r
        m = {
            "+": lambda: l + r,
            "-": lambda: l - r,
            "*": lambda: l * r,
            "/": lambda: l / r,
            "=": lambda: l == r,
            "==": lambda: l == r,
            "<>": lambda: l != r,
            "!=": lambda: l != r,
            "<": lambda: l < r,
            "<=": lambda: l <= r,
            ">": lambda: l > r,
            ">=": lambda: l >= r,
        }
        if op not in m:
            raise UnsupportedPlan(f"operator {op}")
        return m[op]()
    if isinstance(e, X.UnOp):
        inner = _to_column_expr(e.operand, schema, alias_map)
        if e.op == "-":
            return -inner
        if e.op == "NOT":
            return ~inner
        if e.op == "ISNULL":
            return inner.is_null()
        if e.op == "NOTNULL":
            return inner.not_null()
        raise UnsupportedPlan(f"unary {e.op}")
    if isinstance(e, X.Between):
        inner = _to_column_expr(e.expr, schema, alias_map)
        lo = _to_column_expr(e.low, schema, alias_map)
        hi = _to_column_expr(e.high, schema, alias_map)
        res = (inner >= lo) & (inner <= hi)
        return ~res if e.negate else res
    if isinstance(e, X.InList):
        inner = _to_column_expr(e.expr, schema, alias_map)
        res: Optional[ColumnExpr] = None
        for v in e.values:
            term = inner == _to_column_expr(v, schema, alias_map)
            res = term if res is None else (res | term)
        if res is None:
            raise UnsupportedPlan("empty IN list")
        return ~res if e.negate else res
    if isinstance(e, X.Cast):
        inner = _to_column_expr(e.expr, schema, alias_map)
        try:
            return inner.cast(e.type_name)
        except Exception:
            raise UnsupportedPlan(f"cast to {e.type_name}")
    if isinstance(e, X.FuncCall):
        name = e.name
        if name in ("SUM", "MIN", "MAX", "AVG", "COUNT", "FIRST", "LAST"):
            if len(e.args) != 1:
                raise UnsupportedPlan(f"{name} with {len(e.args)} args")
            arg = e.args[0]
            if isinstance(arg, X.Star):
                if name != "COUNT":
                    raise UnsupportedPlan(f"{name}(*)")
                inner = col("*")
            else:
                inner = _to_column_expr(arg, schema, alias_map)
            fn = {
                "SUM": F.sum,
                "MIN": F.min,
                "MAX": F.max,
                "AVG": F.avg,
                "COUNT": F.count_distinct if e.distinct else F.count,
                "FIRST": F.first,
                "LAST": F.last,
            }[name]
            if e.distinct and name in ("FIRST", "LAST"):
                # FIRST/LAST over a deduplicated argument has no device
                # decomposition and silently dropping the qualifier can
                # change the answer — route to the host executor instead
                raise UnsupportedPlan(f"{name}(DISTINCT ...)")
            res = fn(inner)
            if e.distinct and name in ("SUM", "AVG", "MIN", "MAX"):
                from fugue_amd.column.expressions import _UnaryAggFuncExpr

                res = _UnaryAggFuncExpr(name, inner, arg_distinct=True)
            return res
        if name == "COALESCE":
            return F.coalesce(
                *[_to_column_expr(a, schema, alias_map) for a in e.args]
            )
        raise UnsupportedPlan(f"function {name}")
    if isinstance(e, X.Like):
        inner = _to_column_expr(e.expr, schema, alias_map)
        res = F.like(inner, e.pattern)
        return ~res if e.negate else res
    if isinstance(e, X.Case):
        whens = [
            (
                _to_column_expr(c, schema, alias_map),
                _to_column_expr(v, schema, alias_map),
            )
            for c, v in e.whens
        ]
        else_ = (
            _to_column_expr(e.else_, schema, alias_map)
            if e.else_ is not None
            else lit(None)
        )
        return F.case_when(*whens, else_=else_)
    raise UnsupportedPlan(f"expression {type(e).__name__}")


_JOIN_MAP = {
    "inner": "inner",
    "left": "left_outer",
    "right": "right_outer",
    "outer": "full_outer",
    "cross": "cross",
    "semi": "semi",
    "anti": "anti",
}


def _extract_on_names(on: X.Expr) -> List[Tuple[str, str]]:
    """ON must be a conjunction of colA = colB equalities."""
    pairs: List[Tuple[str, str]] = []

    def _walk(e: X.Expr) -> None:
        if isinstance(e, X.BinOp) and e.op == "AND":
            _walk(e.left)
            _walk(e.right)
            return
        if (
            isinstance(e, X.BinOp)
            and e.op in ("=", "==")
            and isinstance(e.left, X.ColRef)
            and isinstance(e.right, X.ColRef)
        ):
            pairs.append((e.left.name, e.right.name))
            return
        raise UnsupportedPlan("non-equi join condition")

    _walk(on)
    return pairs


def execute_plan(
    stmt: X.SelectStmt, tables: Dict[str, DataFrame], engine: Any
) -> DataFrame:
    """Execute the statement on the engine; raises UnsupportedPlan."""
    res = _execute_core(stmt, tables, engine)
    for op, all_, rhs in stmt.set_ops:
        other = _execute_core(rhs, tables, engine)
        if op == "UNION":
            res = engine.union(res, other, distinct=not all_)
        elif op == "INTERSECT":
            res = engine.intersect(res, other, distinct=True)
        elif op == "EXCEPT":
            res = engine.subtract(res, other, distinct=True)
    if stmt.order_by or stmt.limit is not None:
        res = _order_limit(res, stmt, engine)
    return res


def _order_limit(res: DataFrame, stmt: X.SelectStmt, engine: Any) -> DataFrame:
    for o in stmt.order_by:
        if not isinstance(o.expr, X.ColRef):
            raise UnsupportedPlan("ORDER BY must use output columns")
    if stmt.limit is not None and stmt.order_by:
        presort = ",".join(
            f"{o.expr.name} {'asc' if o.asc else 'desc'}" for o in stmt.order_by
        )
        return engine.take(res, stmt.limit, presort=presort)
    # full sort (no limit) or limit without order: do it locally (results
    # at this point are post-projection; a full global sort is a later
    # optimization)
    local = res.as_local_bounded() if hasattr(res, "as_local_bounded") else res
    pdf = local.as_pandas()
    if stmt.order_by:
        pdf = pdf.sort_values(
            [o.expr.name for o in stmt.order_by],
            ascending=[o.asc for o in stmt.order_by],
        ).reset_index(drop=True)
    if stmt.limit is not None:
        pdf = pdf.head(stmt.limit)
    from fugue_amd.dataframe.pandas_dataframe import PandasDataFrame

    return engine.to_df(PandasDataFrame(pdf, res.schema))


def _resolve_from(item: X.FromItem, tables: Dict[str, DataFrame], engine: Any) -> DataFrame:
    if item.subquery is not None:
        return execute_plan(item.subquery, tables, engine)
    if item.table not in tables:
        raise UnsupportedPlan(f"table {item.table} not found")
    return engine.to_df(tables[item.table])


def _split_conjuncts(e: Optional[X.Expr]) -> List[X.Expr]:
    if e is None:
        return []
    if isinstance(e, X.BinOp) and e.op == "AND":
        return _split_conjuncts(e.left) + _split_conjuncts(e.right)
    return [e]


def _mentioned_cols(e: X.Expr) -> List[str]:
    out: List[str] = []

    def _walk(x: X.Expr) -> None:
        if isinstance(x, X.ColRef):
            out.append(x.name)
        elif isinstance(x, X.BinOp):
            _walk(x.left)
            _walk(x.right)
        elif isinstance(x, X.UnOp):
            _walk(x.operand)
        elif isinstance(x, X.Between):
            _walk(x.expr)
            _walk(x.low)
            _walk(x.high)
        elif isinstance(x, X.InList):
            _walk(x.expr)
            for v in x.values:
                _walk(v)
        elif isinstance(x, X.Cast):
            _walk(x.expr)
        elif isinstance(x, X.FuncCall):
            for a in x.args:
                _walk(a)
        elif isinstance(x, X.Like):
            _walk(x.expr)
        elif isinstance(x, X.Case):
            for cnd, val in x.whens:
                _walk(cnd)
                _walk(val)
            if x.else_ is not None:
                _walk(x.else_)

    _walk(e)
    return out


def _needed_columns(
    stmt: X.SelectStmt, where: Optional[X.Expr]
) -> Optional[set]:
    """All column names the statement can reference; None when a ``*``
    makes every column live.  ``where`` is the residual (post-pushdown)
    predicate — columns only used by already-pushed filters are dead."""
    needed: set = set()
    for e, _alias in stmt.columns:
        if isinstance(e, X.Star):
            return None
        needed.update(_mentioned_cols(e))
    for e in (where, stmt.having):
        if e is not None:
            needed.update(_mentioned_cols(e))
    for e in stmt.group_by:
        if not isinstance(e, X.Star):
            needed.update(_mentioned_cols(e))
    for o in stmt.order_by:
        needed.update(_mentioned_cols(o.expr))
    for j in stmt.joins:
        if j.on is not None:
            needed.update(_mentioned_cols(j.on))
        if j.using is not None:
            needed.update(j.using)
    return needed


def _execute_core(
    stmt: X.SelectStmt, tables: Dict[str, DataFrame], engine: Any
) -> DataFrame:
    if stmt.from_item is None:
        raise UnsupportedPlan("SELECT without FROM")
    res = _resolve_from(stmt.from_item, tables, engine)
    # predicate pushdown: when ALL joins are inner, single-table conjuncts
    # of WHERE filter their source table before the joins
    join_inputs: List[DataFrame] = []
    residual_where: Optional[X.Expr] = stmt.where
    pushed: Dict[int, List[X.Expr]] = {}
    all_inner = all(j.how == "inner" for j in stmt.joins)
    if stmt.joins and all_inner and stmt.where is not None:
        frames: List[DataFrame] = [res] + [
            _resolve_from(j.item, tables, engine) for j in stmt.joins
        ]
        remaining: List[X.Expr] = []
        for conj in _split_conjuncts(stmt.where):
            cols = set(_mentioned_cols(conj))
            target = None
            for idx, fr in enumerate(frames):
                if cols <= set(fr.schema.names):
                    # unique owner only (ambiguous cols stay post-join)
                    owners = [
                        i
                        for i, f2 in enumerate(frames)
                        if cols <= set(f2.schema.names)
                    ]
                    if len(owners) == 1:
                        target = idx
                    break
            if target is None:
                remaining.append(conj)
            else:
                pushed.setdefault(target, []).append(conj)
        res = frames[0]
        join_inputs = frames[1:]
        residual_where = None
        for conj in remaining:
            residual_where = (
                conj
                if residual_where is None
                else X.BinOp("AND", residual_where, conj)
            )
    # projection pushdown fused with the pushed filters: each base table
    # keeps only the columns referenced downstream, and a pushed filter
    # runs as select(keep, where=...) so the compaction never gathers
    # columns (strings especially) that the filter output drops
    if stmt.joins:
        frames2: List[DataFrame] = [res] + [
            join_inputs[i]
            if i < len(join_inputs)
            else _resolve_from(stmt.joins[i].item, tables, engine)
            for i in range(len(stmt.joins))
        ]
        needed = _needed_columns(stmt, residual_where)
        pruned: List[DataFrame] = []
        for idx, fr in enumerate(frames2):
            conjs = pushed.get(idx, [])
            where_ce = None
            for conj in conjs:
                ce = _to_column_expr(conj, fr.schema, {})
                where_ce = ce if where_ce is None else (where_ce & ce)
            if needed is not None:
                keep = [n for n in fr.schema.names if n in needed]
            else:
                keep = list(fr.schema.names)
            if len(keep) == 0:
                keep = [fr.schema.names[0]]
            if where_ce is not None:
                fr = engine.select(
                    fr,
                    SelectColumns(*[col(n) for n in keep]),
                    where=where_ce,
                )
            elif len(keep) < len(fr.schema.names):
                fr = engine.select(
                    fr, SelectColumns(*[col(n) for n in keep])
                )
            pruned.append(fr)
        frames2 = pruned
        res = frames2[0]
        join_inputs = frames2[1:]
    for ji, j in enumerate(stmt.joins):
        right = (
            join_inputs[ji]
            if ji < len(join_inputs)
            else _resolve_from(j.item, tables, engine)
        )
        how = _JOIN_MAP.get(j.how)
        if how is None:
            raise UnsupportedPlan(f"join {j.how}")
        if j.using is not None:
            keys = list(j.using)
        elif j.on is not None:
            pairs = _extract_on_names(j.on)
            keys = []
            for a, b in pairs:
                if a in res.schema and b in right.schema:
                    if a != b:
                        raise UnsupportedPlan(
                            "join keys with different names need rename"
                        )
                    keys.append(a)
                elif b in res.schema and a in right.schema:
                    if a != b:
                        raise UnsupportedPlan(
                            "join keys with different names need rename"
                        )
                    keys.append(a)
                else:
                    raise UnsupportedPlan("can't resolve join key sides")
        elif how == "cross":
            keys = []
        else:
            raise UnsupportedPlan("JOIN without ON/USING")
        common = [n for n in res.schema.names if n in right.schema]
        if how != "cross" and set(keys) != set(common):
            raise UnsupportedPlan(
                f"join keys {keys} != common columns {common} "
                "(fugue joins use all common columns)"
            )
        res = engine.join(res, right, how=how, on=keys if keys else None)
    # select columns
    schema = res.schema
    where_expr: Optional[ColumnExpr] = None
    if residual_where is not None:
        where_expr = _to_column_expr(residual_where, schema, {})
    cols: List[ColumnExpr] = []
    from fugue_amd.column.expressions import all_cols

    for i, (e, alias) in enumerate(stmt.columns):
        if isinstance(e, X.Star):
            if e.qualifier is not None:
                raise UnsupportedPlan("qualified * in select")
            cols.append(all_cols())
            continue
        ce = _to_column_expr(e, schema, {})
        name = alias or e.default_name() or f"_col{i}"
        if isinstance(e, X.ColRef) and alias is None:
            cols.append(ce)
        else:
            cols.append(ce.alias(name))
    having_expr: Optional[ColumnExpr] = None
    sc = SelectColumns(*cols, arg_distinct=stmt.distinct)
    # GROUP BY validation: keys must match the non-agg select columns
    if stmt.group_by:
        if not sc.has_agg:
            raise UnsupportedPlan("GROUP BY without aggregates")
        group_names = set()
        for g in stmt.group_by:
            if isinstance(g, X.ColRef):
                group_names.add(g.name)
            elif isinstance(g, X.Lit) and isinstance(g.value, int):
                e0 = stmt.columns[g.value - 1][0]
                if isinstance(e0, X.ColRef):
                    group_names.add(e0.name)
                else:
                    raise UnsupportedPlan("positional GROUP BY on expression")
            else:
                raise UnsupportedPlan("GROUP BY expression")
        sel_keys = {k.output_name or k.name for k in sc.group_keys}
        if group_names != sel_keys:
            raise UnsupportedPlan(
                f"GROUP BY {group_names} != selected keys {sel_keys}"
            )
    if stmt.having is not None:
        # having references output aggregate aliases
        out_schema_names = [c.output_name for c in sc.all_cols]
        having_expr = _to_column_expr(
            stmt.having, Schema([(n, "double") for n in out_schema_names if n]), {}
        )
    return engine.select(res, sc, where=where_expr, having=having_expr)
