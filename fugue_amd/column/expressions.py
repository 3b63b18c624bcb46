"""Typed column expression tree: ``col("a") + lit(1)``, comparisons,
logical ops, cast, alias, distinct.

Reference parity: ``fugue/column/expressions.py``.  New implementation; the
tree is consumed both by the SQL generator (``fugue_amd/column/sql.py``)
and by the pandas/HIP interpreters (``fugue_amd/column/interpreter.py``).
"""
from typing import Any, Dict, Iterable, List, Optional, Union

import pyarrow as pa

from fugue_amd.schema import Schema, to_pa_datatype
from fugue_amd.utils.hash import to_uuid


class ColumnExpr:
    def __init__(self):
        self._as_name = ""
        self._as_type: Optional[pa.DataType] = None

    @property
    def name(self) -> str:
        return ""

    @property
    def as_name(self) -> str:
        return self._as_name

    @property
    def as_type(self) -> Optional[pa.DataType]:
        return self._as_type

    @property
    def output_name(self) -> str:
        return self.as_name if self.as_name != "" else self.name

    def alias(self, as_name: str) -> "ColumnExpr":
        raise NotImplementedError

    def infer_alias(self) -> "ColumnExpr":
        return self

    def cast(self, data_type: Any) -> "ColumnExpr":
        raise NotImplementedError

    def infer_type(self, schema: Schema) -> Optional[pa.DataType]:
        return self.as_type

    def infer_alias(self) -> "ColumnExpr":
        """Infer the output name from underlying columns when unnamed
        (reference ``expressions.py:111``)."""
        return self

    def __str__(self) -> str:
        res = self.body_str
        if self.as_type is not None:
            from fugue_amd.schema import _type_to_expression

            res = f"CAST({res} AS {_type_to_expression(self.as_type)})"
        if self.as_name != "":
            res = res + " AS " + self.as_name
        return res

    @property
    def body_str(self) -> str:
        raise NotImplementedError

    def is_null(self) -> "ColumnExpr":
        return _UnaryOpExpr("IS_NULL", self)

    def not_null(self) -> "ColumnExpr":
        return _UnaryOpExpr("NOT_NULL", self)

    def __neg__(self) -> "ColumnExpr":
        return _UnaryOpExpr("-", self)

    def __pos__(self) -> "ColumnExpr":
        return self

    def __invert__(self) -> "ColumnExpr":
        return _NotOpExpr("~", self)

    def __add__(self, other: Any) -> "ColumnExpr":
        return _BinaryOpExpr("+", self, other)

    def __radd__(self, other: Any) -> "ColumnExpr":
        return _BinaryOpExpr("+", _to_col(other), self)

    def __sub__(self, other: Any) -> "ColumnExpr":
        return _BinaryOpExpr("-", self, other)

    def __rsub__(self, other: Any) -> "ColumnExpr":
        return _BinaryOpExpr("-", _to_col(other), self)

    def __mul__(self, other: Any) -> "ColumnExpr":
        return _BinaryOpExpr("*", self, other)

    def __rmul__(self, other: Any) -> "ColumnExpr":
        return _BinaryOpExpr("*", _to_col(other), self)

    def __truediv__(self, other: Any) -> "ColumnExpr":
        return _BinaryOpExpr("/", self, other)

    def __rtruediv__(self, other: Any) -> "ColumnExpr":
        return _BinaryOpExpr("/", _to_col(other), self)

    def __and__(self, other: Any) -> "ColumnExpr":
        return _BoolBinaryOpExpr("&", self, other)

    def __rand__(self, other: Any) -> "ColumnExpr":
        return _BoolBinaryOpExpr("&", _to_col(other), self)

    def __or__(self, other: Any) -> "ColumnExpr":
        return _BoolBinaryOpExpr("|", self, other)

    def __ror__(self, other: Any) -> "ColumnExpr":
        return _BoolBinaryOpExpr("|", _to_col(other), self)

    def __lt__(self, other: Any) -> "ColumnExpr":
        return _BoolBinaryOpExpr("<", self, other)

    def __gt__(self, other: Any) -> "ColumnExpr":
        return _BoolBinaryOpExpr(">", self, other)

    def __le__(self, other: Any) -> "ColumnExpr":
        return _BoolBinaryOpExpr("<=", self, other)

    def __ge__(self, other: Any) -> "ColumnExpr":
        return _BoolBinaryOpExpr(">=", self, other)

    def __eq__(self, other: Any) -> "ColumnExpr":  # type: ignore
        return _BoolBinaryOpExpr("==", self, other)

    def __ne__(self, other: Any) -> "ColumnExpr":  # type: ignore
        return _BoolBinaryOpExpr("!=", self, other)

    def __uuid__(self) -> str:
        return to_uuid(
            str(type(self)),
            self.as_name,
            str(self.as_type),
            *self._uuid_keys(),
        )

    def _uuid_keys(self) -> List[Any]:
        return []


def lit(obj: Any, alias: str = "") -> ColumnExpr:
    e = _LiteralColumnExpr(obj)
    return e if alias == "" else e.alias(alias)


def null() -> ColumnExpr:
    return lit(None)


def col(obj: Union[str, ColumnExpr], alias: str = "") -> ColumnExpr:
    if isinstance(obj, ColumnExpr):
        return obj if alias == "" else obj.alias(alias)
    if isinstance(obj, str):
        e: ColumnExpr = _NamedColumnExpr(obj)
        return e if alias == "" else e.alias(alias)
    raise ValueError(f"{obj} can't be converted to a column expression")


def all_cols() -> ColumnExpr:
    return _WildcardExpr()


def function(name: str, *args: Any, arg_distinct: bool = False, **kwargs: Any) -> ColumnExpr:
    return _FuncExpr(name, *args, arg_distinct=arg_distinct, **kwargs)


def _get_column_mentions(column: ColumnExpr) -> Iterable[str]:
    if isinstance(column, _NamedColumnExpr):
        yield column.name
    elif isinstance(column, _FuncExpr):
        for a in column.args:
            yield from _get_column_mentions(a)
        for a in column.kwargs.values():
            yield from _get_column_mentions(a)
    elif isinstance(column, _BinaryOpExpr):
        yield from _get_column_mentions(column.left)
        yield from _get_column_mentions(column.right)
    elif isinstance(column, _UnaryOpExpr):
        yield from _get_column_mentions(column.col)


def _to_col(obj: Any) -> ColumnExpr:
    if isinstance(obj, ColumnExpr):
        return obj
    return lit(obj)


class _NamedColumnExpr(ColumnExpr):
    def __init__(self, name: Any):
        self._name = name
        super().__init__()

    @property
    def body_str(self) -> str:
        return self.name

    @property
    def name(self) -> str:
        return self._name

    @property
    def wildcard(self) -> bool:
        return self.name == "*"

    def alias(self, as_name: str) -> ColumnExpr:
        other = _NamedColumnExpr(self.name)
        other._as_name = as_name
        other._as_type = self.as_type
        return other

    def cast(self, data_type: Any) -> ColumnExpr:
        other = _NamedColumnExpr(self.name)
        other._as_name = self.as_name
        other._as_type = None if data_type is None else to_pa_datatype(data_type)
        return other

    def infer_alias(self) -> ColumnExpr:
        if self.as_type is not None and self.as_name == "":
            return self.alias(self.name)
        return self

    def infer_type(self, schema: Schema) -> Optional[pa.DataType]:
        if self.as_type is not None:
            return self.as_type
        if self.name in schema:
            return schema[self.name].type
        return None

    def _uuid_keys(self) -> List[Any]:
        return [self.name]


class _WildcardExpr(ColumnExpr):
    @property
    def body_str(self) -> str:
        return "*"

    @property
    def name(self) -> str:
        return "*"

    @property
    def output_name(self) -> str:
        return ""

    def __uuid__(self) -> str:
        return to_uuid("_WildcardExpr")


class _LiteralColumnExpr(ColumnExpr):
    _VALID_TYPES = (int, bool, float, str)

    def __init__(self, value: Any):
        if value is not None and not isinstance(value, self._VALID_TYPES):
            raise NotImplementedError(f"{value} is not a valid literal")
        self._value = value
        super().__init__()

    @property
    def body_str(self) -> str:
        if self.value is None:
            return "NULL"
        if isinstance(self.value, str):
            body = self.value.translate(
                str.maketrans({"\\": r"\\", "'": r"\'"})
            )
            return f"'{body}'"
        if isinstance(self.value, bool):
            return "TRUE" if self.value else "FALSE"
        return str(self.value)

    @property
    def value(self) -> Any:
        return self._value

    def alias(self, as_name: str) -> ColumnExpr:
        other = _LiteralColumnExpr(self.value)
        other._as_name = as_name
        other._as_type = self.as_type
        return other

    def cast(self, data_type: Any) -> ColumnExpr:
        other = _LiteralColumnExpr(self.value)
        other._as_name = self.as_name
        other._as_type = None if data_type is None else to_pa_datatype(data_type)
        return other

    def infer_type(self, schema: Schema) -> Optional[pa.DataType]:
        if self.as_type is not None:
            return self.as_type
        if self.value is None:
            return None
        if isinstance(self.value, bool):
            return pa.bool_()
        if isinstance(self.value, int):
            return pa.int64()
        if isinstance(self.value, float):
            return pa.float64()
        if isinstance(self.value, str):
            return pa.string()
        return None

    def _uuid_keys(self) -> List[Any]:
        return [repr(self.value)]


class _UnaryOpExpr(ColumnExpr):
    def __init__(self, op: str, column: ColumnExpr):
        self._op = op
        self._col = column
        super().__init__()

    @property
    def op(self) -> str:
        return self._op

    @property
    def col(self) -> ColumnExpr:
        return self._col

    @property
    def name(self) -> str:
        return self.col.name

    @property
    def body_str(self) -> str:
        if self.op == "IS_NULL":
            return f"{self.col.body_str} IS NULL"
        if self.op == "NOT_NULL":
            return f"{self.col.body_str} IS NOT NULL"
        return f"{self.op}({self.col.body_str})"

    def infer_alias(self) -> ColumnExpr:
        if self.output_name != "":
            return self
        inner = self.col.infer_alias().output_name
        return self.alias(inner) if inner != "" else self

    def alias(self, as_name: str) -> ColumnExpr:
        other = type(self)(self.op, self.col)
        other._as_name = as_name
        other._as_type = self.as_type
        return other

    def cast(self, data_type: Any) -> ColumnExpr:
        other = type(self)(self.op, self.col)
        other._as_name = self.as_name
        other._as_type = None if data_type is None else to_pa_datatype(data_type)
        return other

    def infer_alias(self) -> ColumnExpr:
        if self.as_name == "" and self.name != "":
            return self.alias(self.name)
        return self

    def infer_type(self, schema: Schema) -> Optional[pa.DataType]:
        if self.as_type is not None:
            return self.as_type
        if self.op in ("IS_NULL", "NOT_NULL"):
            return pa.bool_()
        return self.col.infer_type(schema)

    def _uuid_keys(self) -> List[Any]:
        return [self.op, self.col.__uuid__()]


class _NotOpExpr(_UnaryOpExpr):
    @property
    def body_str(self) -> str:
        return f"NOT {self.col.body_str}"

    def infer_type(self, schema: Schema) -> Optional[pa.DataType]:
        return self.as_type or pa.bool_()


class _BinaryOpExpr(ColumnExpr):
    def __init__(self, op: str, left: Any, right: Any):
        self._op = op
        self._left = _to_col(left)
        self._right = _to_col(right)
        super().__init__()

    @property
    def op(self) -> str:
        return self._op

    @property
    def left(self) -> ColumnExpr:
        return self._left

    @property
    def right(self) -> ColumnExpr:
        return self._right

    @property
    def body_str(self) -> str:
        sql_op = "=" if self.op == "==" else ("<>" if self.op == "!=" else self.op)
        return f"({self.left.body_str} {sql_op} {self.right.body_str})"

    def alias(self, as_name: str) -> ColumnExpr:
        other = type(self)(self.op, self.left, self.right)
        other._as_name = as_name
        other._as_type = self.as_type
        return other

    def cast(self, data_type: Any) -> ColumnExpr:
        other = type(self)(self.op, self.left, self.right)
        other._as_name = self.as_name
        other._as_type = None if data_type is None else to_pa_datatype(data_type)
        return other

    def infer_type(self, schema: Schema) -> Optional[pa.DataType]:
        if self.as_type is not None:
            return self.as_type
        lt = self.left.infer_type(schema)
        rt = self.right.infer_type(schema)
        if lt is None or rt is None:
            return None
        if self.op == "/":
            return pa.float64()
        if pa.types.is_floating(lt) or pa.types.is_floating(rt):
            return pa.float64()
        if pa.types.is_integer(lt) and pa.types.is_integer(rt):
            return pa.int64() if (lt == pa.int64() or rt == pa.int64()) else lt
        if lt == rt:
            return lt
        return None

    def _uuid_keys(self) -> List[Any]:
        return [self.op, self.left.__uuid__(), self.right.__uuid__()]


class _BoolBinaryOpExpr(_BinaryOpExpr):
    def infer_type(self, schema: Schema) -> Optional[pa.DataType]:
        return self.as_type or pa.bool_()


class _FuncExpr(ColumnExpr):
    def __init__(
        self,
        func: str,
        *args: Any,
        arg_distinct: bool = False,
        **kwargs: Any,
    ):
        self._func = func
        self._args = [_to_col(a) for a in args]
        self._kwargs = {k: _to_col(v) for k, v in kwargs.items()}
        self._is_distinct = arg_distinct
        super().__init__()

    @property
    def func(self) -> str:
        return self._func

    @property
    def is_distinct(self) -> bool:
        return self._is_distinct

    @property
    def args(self) -> List[ColumnExpr]:
        return self._args

    @property
    def kwargs(self) -> Dict[str, ColumnExpr]:
        return self._kwargs

    @property
    def body_str(self) -> str:
        inner = ", ".join(a.body_str for a in self.args)
        distinct = "DISTINCT " if self.is_distinct else ""
        return f"{self.func}({distinct}{inner})"

    def alias(self, as_name: str) -> ColumnExpr:
        other = self._copy()
        other._as_name = as_name
        other._as_type = self.as_type
        return other

    def cast(self, data_type: Any) -> ColumnExpr:
        other = self._copy()
        other._as_name = self.as_name
        other._as_type = None if data_type is None else to_pa_datatype(data_type)
        return other

    def _copy(self) -> "_FuncExpr":
        return _FuncExpr(
            self.func, *self.args, arg_distinct=self.is_distinct, **self.kwargs
        )

    def _uuid_keys(self) -> List[Any]:
        return [
            self.func,
            self.is_distinct,
            [a.__uuid__() for a in self.args],
            {k: v.__uuid__() for k, v in self.kwargs.items()},
        ]


class _UnaryAggFuncExpr(_FuncExpr):
    def __init__(self, func: str, col: ColumnExpr, arg_distinct: bool = False):
        super().__init__(func, col, arg_distinct=arg_distinct)

    def _copy(self) -> "_FuncExpr":
        return _UnaryAggFuncExpr(
            self.func, self.args[0], arg_distinct=self.is_distinct
        )

    def infer_alias(self) -> ColumnExpr:
        if self.output_name != "":
            return self
        inner = self.args[0]
        name = (
            inner.infer_alias().output_name
            if isinstance(inner, ColumnExpr)
            else ""
        )
        return self.alias(name) if name != "" else self

    def infer_type(self, schema: Schema) -> Optional[pa.DataType]:
        if self.as_type is not None:
            return self.as_type
        f = self.func.upper()
        if f in ("COUNT", "COUNT_DISTINCT"):
            return pa.int64()
        if f == "AVG":
            return pa.float64()
        if f in ("MIN", "MAX", "FIRST", "LAST", "SUM"):
            arg = self.args[0]
            return arg.infer_type(schema) if isinstance(arg, ColumnExpr) else None
        return None


def _is_agg(column: Any) -> bool:
    if isinstance(column, _UnaryAggFuncExpr):
        return True
    if isinstance(column, _BinaryOpExpr):
        return _is_agg(column.left) or _is_agg(column.right)
    if isinstance(column, _UnaryOpExpr):
        return _is_agg(column.col)
    if isinstance(column, _FuncExpr):
        return any(_is_agg(a) for a in column.args) or any(
            _is_agg(v) for v in column.kwargs.values()
        )
    return False
