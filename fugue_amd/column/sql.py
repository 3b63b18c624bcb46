"""SelectColumns (validated select expression sets with group-key
inference) and SQLExpressionGenerator (expression tree → SQL text).

Reference parity: ``fugue/column/sql.py:38,233``.
"""
from typing import Any, Callable, Dict, Iterable, List, Optional, Tuple

import pyarrow as pa

from fugue_amd.column.expressions import (
    ColumnExpr,
    _BinaryOpExpr,
    _FuncExpr,
    _LiteralColumnExpr,
    _NamedColumnExpr,
    _UnaryAggFuncExpr,
    _UnaryOpExpr,
    _WildcardExpr,
    _is_agg,
    col,
)
from fugue_amd.schema import Schema, _type_to_expression
from fugue_amd.utils.hash import to_uuid


class SelectColumns:
    def __init__(self, *cols: ColumnExpr, arg_distinct: bool = False):
        self._distinct = arg_distinct
        from fugue_amd.column.expressions import all_cols as _all_cols

        # normalize col("*") into the canonical wildcard expression
        self._cols = [
            _all_cols()
            if isinstance(c, _NamedColumnExpr)
            and c.name == "*"
            and c.as_name == ""
            else c.infer_alias()
            for c in cols
        ]
        if len(self._cols) == 0:
            raise ValueError("select columns can't be empty")
        self._literals = [
            c for c in self._cols if isinstance(c, _LiteralColumnExpr)
        ]
        self._simple_cols = [
            c for c in self._cols if isinstance(c, _NamedColumnExpr) and c.name != "*"
        ]
        self._wildcards = [c for c in self._cols if isinstance(c, _WildcardExpr)]
        if len(self._wildcards) > 1:
            raise ValueError("at most one * is allowed in select")
        others = [
            c
            for c in self._cols
            if not isinstance(c, (_LiteralColumnExpr, _NamedColumnExpr, _WildcardExpr))
            or (isinstance(c, _NamedColumnExpr) and c.name == "*")
        ]
        self._agg_funcs = [c for c in others if _is_agg(c)]
        self._non_agg_funcs = [
            c for c in others if not _is_agg(c) and not isinstance(c, _WildcardExpr)
        ]
        self._has_agg = len(self._agg_funcs) > 0
        if self._has_agg and len(self._wildcards) > 0:
            raise ValueError("* can't be used with aggregations")
        # group keys = simple cols + non-agg funcs' mentioned columns
        self._group_keys: List[ColumnExpr] = []
        if self._has_agg:
            self._group_keys.extend(self._simple_cols)
            for c in self._non_agg_funcs:
                self._group_keys.append(c)
        names = [c.output_name for c in self._cols if c.output_name != ""]
        if len(names) != len(set(names)):
            raise ValueError(f"duplicated output names in {names}")
        if len(self._wildcards) > 0:
            # plain unaliased columns next to * duplicate its output
            for c in self._cols:
                if (
                    isinstance(c, _NamedColumnExpr)
                    and c.name != "*"
                    and c.as_name == ""
                ):
                    raise ValueError(
                        f"with *, column {c} must have an alias"
                    )

    def __str__(self):
        return ", ".join(str(c) for c in self._cols)

    def __uuid__(self):
        return to_uuid(self._distinct, [c.__uuid__() for c in self._cols])

    @property
    def is_distinct(self) -> bool:
        return self._distinct

    def replace_wildcard(self, schema: Schema) -> "SelectColumns":
        def _get_cols() -> Iterable[ColumnExpr]:
            for c in self._cols:
                if isinstance(c, _WildcardExpr):
                    yield from [col(n) for n in schema.names]
                else:
                    yield c

        return SelectColumns(*list(_get_cols()), arg_distinct=self._distinct)

    def assert_all_with_names(self) -> "SelectColumns":
        for c in self._cols:
            if isinstance(c, _WildcardExpr):
                continue
            if c.output_name == "":
                raise ValueError(f"{c} does not have an output name")
        return self

    def assert_no_wildcard(self) -> "SelectColumns":
        if len(self._wildcards) > 0:
            raise ValueError("wildcard is not allowed here")
        return self

    def assert_no_agg(self) -> "SelectColumns":
        if self._has_agg:
            raise ValueError("aggregation is not allowed here")
        return self

    @property
    def all_cols(self) -> List[ColumnExpr]:
        return self._cols

    @property
    def literals(self) -> List[ColumnExpr]:
        return self._literals

    @property
    def simple_cols(self) -> List[ColumnExpr]:
        return self._simple_cols

    @property
    def non_agg_funcs(self) -> List[ColumnExpr]:
        return self._non_agg_funcs

    @property
    def agg_funcs(self) -> List[ColumnExpr]:
        return self._agg_funcs

    @property
    def group_keys(self) -> List[ColumnExpr]:
        return self._group_keys

    @property
    def has_agg(self) -> bool:
        return self._has_agg

    @property
    def has_literals(self) -> bool:
        return len(self._literals) > 0

    @property
    def simple(self) -> bool:
        return len(self._simple_cols) + len(self._wildcards) == len(self._cols)

    def infer_schema(self, schema: Schema) -> Optional[Schema]:
        """Best-effort output schema inference; None if any type unknown."""
        fields = []
        for c in self.replace_wildcard(schema).all_cols:
            tp = c.infer_type(schema)
            if tp is None or c.output_name == "":
                return None
            fields.append(pa.field(c.output_name, tp))
        return Schema(fields)


class SQLExpressionGenerator:
    """Compile column expression trees to SQL text (single-table SELECT)."""

    def __init__(self, enable_cast: bool = True):
        self._enable_cast = enable_cast
        self._func_handler: Dict[str, Callable[[_FuncExpr], Iterable[str]]] = {}

    def where(self, condition: ColumnExpr, table: str) -> str:
        if _is_agg(condition):
            raise ValueError("aggregation is not allowed in where")
        cond = self.generate(condition)
        return f"SELECT * FROM {table} WHERE {cond}"

    def select(
        self,
        columns: SelectColumns,
        table: str,
        where: Optional[ColumnExpr] = None,
        having: Optional[ColumnExpr] = None,
    ) -> str:
        w = ""
        if where is not None:
            if _is_agg(where):
                raise ValueError("aggregation is not allowed in where")
            w = " WHERE " + self.generate(where)
        distinct = "DISTINCT " if columns.is_distinct else ""
        if not columns.has_agg:
            expr = ", ".join(self.generate(c) for c in columns.all_cols)
            return f"SELECT {distinct}{expr} FROM {table}{w}"
        columns.assert_all_with_names()
        h = ""
        if having is not None:
            h = " HAVING " + self.generate(having)
        expr = ", ".join(self.generate(c) for c in columns.all_cols)
        if len(columns.group_keys) == 0:
            return f"SELECT {distinct}{expr} FROM {table}{w}{h}"
        keys = ", ".join(self.generate_no_alias(k) for k in columns.group_keys)
        return f"SELECT {distinct}{expr} FROM {table}{w} GROUP BY {keys}{h}"

    def generate(self, expr: ColumnExpr) -> str:
        return "".join(self._generate(expr))

    def generate_no_alias(self, expr: ColumnExpr) -> str:
        return "".join(self._generate(expr, with_alias=False))

    def add_func_handler(
        self, name: str, handler: Callable[[_FuncExpr], Iterable[str]]
    ) -> "SQLExpressionGenerator":
        self._func_handler[name] = handler
        return self

    def correct_select_schema(
        self, input_schema: Schema, select: SelectColumns, output_schema: Schema
    ) -> Optional[Schema]:
        """Compute the subset of output columns whose types need a cast to
        match the expected output schema (engines apply it after select)."""
        cols = select.replace_wildcard(input_schema).assert_all_with_names()
        fields: List[pa.Field] = []
        for c in cols.all_cols:
            tp = c.infer_type(input_schema)
            if tp is not None and tp == output_schema[c.output_name].type:
                continue
            fields.append(output_schema[c.output_name])
        if len(fields) == 0:
            return None
        return Schema(fields)

    def type_to_expr(self, data_type: pa.DataType) -> str:
        return _type_to_expression(data_type)

    def _generate(self, expr: ColumnExpr, with_alias: bool = True) -> Iterable[str]:
        body = list(self._body(expr))
        if self._enable_cast and expr.as_type is not None:
            body = ["CAST("] + body + [f" AS {self.type_to_expr(expr.as_type)})"]
        yield from body
        if with_alias and expr.as_name != "":
            yield f" AS {expr.as_name}"
        elif (
            with_alias
            and expr.as_type is not None
            and expr.name != ""
            and not isinstance(expr, _WildcardExpr)
        ):
            yield f" AS {expr.name}"

    def _body(self, expr: ColumnExpr) -> Iterable[str]:
        if isinstance(expr, _LiteralColumnExpr):
            yield expr.body_str
        elif isinstance(expr, _WildcardExpr):
            yield "*"
        elif isinstance(expr, _NamedColumnExpr):
            yield expr.name
        elif isinstance(expr, _FuncExpr):
            if expr.func in self._func_handler:
                yield from self._func_handler[expr.func](expr)
            else:
                yield expr.func + "("
                if expr.is_distinct:
                    yield "DISTINCT "
                first = True
                for a in expr.args:
                    if not first:
                        yield ", "
                    yield from self._generate(a, with_alias=False)
                    first = False
                yield ")"
        elif isinstance(expr, _BinaryOpExpr):
            op = {"==": "=", "!=": "<>", "&": " AND ", "|": " OR "}.get(
                expr.op, expr.op
            )
            yield "("
            yield from self._generate(expr.left, with_alias=False)
            if op in ("=", "<>", "<", ">", "<=", ">=", "+", "-", "*", "/"):
                yield f" {op} "
            else:
                yield op
            yield from self._generate(expr.right, with_alias=False)
            yield ")"
        elif isinstance(expr, _UnaryOpExpr):
            if expr.op == "IS_NULL":
                yield from self._generate(expr.col, with_alias=False)
                yield " IS NULL"
            elif expr.op == "NOT_NULL":
                yield from self._generate(expr.col, with_alias=False)
                yield " IS NOT NULL"
            elif expr.op == "~":
                yield "NOT "
                yield from self._generate(expr.col, with_alias=False)
            else:
                yield expr.op
                yield from self._generate(expr.col, with_alias=False)
        else:
            raise NotImplementedError(f"can't generate SQL for {expr}")
