"""Schema expression language: ``"a:int,b:str"`` ⇄ ``pyarrow.Schema``.

This replaces the ``triad.collections.schema.Schema`` dependency of the
reference (see ``/root/reference`` usage documented in SURVEY.md §1 — the
schema expression language is used by every layer).  The implementation is
new: a small tokenizer for the type expression grammar plus a thin ordered
wrapper over ``pyarrow`` fields.

Grammar (informal)::

    schema   := field ("," field)*
    field    := name ":" type
    type     := simple | "[" type "]" | "<" type ">"        (list)
              | "{" schema "}"                               (struct)
              | "<" type "," type ">"                        (map)
              | "decimal(p,s)" | "timestamp(unit[,tz])" | "binary"
"""
from typing import Any, Dict, Iterable, List, Optional, Tuple, Union

import numpy as np
import pandas as pd
import pyarrow as pa

__all__ = [
    "Schema",
    "expression_to_schema",
    "schema_to_expression",
    "to_pa_datatype",
]

_SIMPLE_TYPES: Dict[str, pa.DataType] = {
    "null": pa.null(),
    "bool": pa.bool_(),
    "boolean": pa.bool_(),
    "int8": pa.int8(),
    "byte": pa.int8(),
    "tinyint": pa.int8(),
    "int16": pa.int16(),
    "short": pa.int16(),
    "smallint": pa.int16(),
    "int": pa.int32(),
    "int32": pa.int32(),
    "integer": pa.int32(),
    "long": pa.int64(),
    "int64": pa.int64(),
    "bigint": pa.int64(),
    "uint8": pa.uint8(),
    "ubyte": pa.uint8(),
    "uint16": pa.uint16(),
    "ushort": pa.uint16(),
    "uint32": pa.uint32(),
    "uint": pa.uint32(),
    "uint64": pa.uint64(),
    "ulong": pa.uint64(),
    "float16": pa.float16(),
    "float": pa.float32(),
    "float32": pa.float32(),
    "double": pa.float64(),
    "float64": pa.float64(),
    "str": pa.string(),
    "string": pa.string(),
    "bytes": pa.binary(),
    "binary": pa.binary(),
    "date": pa.date32(),
    "datetime": pa.timestamp("us"),
    "timestamp": pa.timestamp("us"),
}

_TYPE_TO_EXPR: Dict[pa.DataType, str] = {
    pa.null(): "null",
    pa.bool_(): "bool",
    pa.int8(): "byte",
    pa.int16(): "short",
    pa.int32(): "int",
    pa.int64(): "long",
    pa.uint8(): "uint8",
    pa.uint16(): "uint16",
    pa.uint32(): "uint32",
    pa.uint64(): "uint64",
    pa.float16(): "float16",
    pa.float32(): "float",
    pa.float64(): "double",
    pa.string(): "str",
    pa.large_string(): "str",
    pa.binary(): "bytes",
    pa.date32(): "date",
    pa.timestamp("us"): "datetime",
}


def _quote_name(name: str) -> str:
    """Backtick-quote a column name when it isn't a plain identifier."""
    if name != "" and all(c.isalnum() or c == "_" for c in name):
        return name
    return "`" + name + "`"


def _type_to_expression(tp: pa.DataType) -> str:
    if tp in _TYPE_TO_EXPR:
        return _TYPE_TO_EXPR[tp]
    if pa.types.is_timestamp(tp):
        if tp.tz is None:
            return f"timestamp({tp.unit})"
        return f"timestamp({tp.unit},{tp.tz})"
    if pa.types.is_decimal(tp):
        return f"decimal({tp.precision},{tp.scale})"
    if pa.types.is_list(tp) or pa.types.is_large_list(tp):
        return "[" + _type_to_expression(tp.value_type) + "]"
    if pa.types.is_map(tp):
        return (
            "<"
            + _type_to_expression(tp.key_type)
            + ","
            + _type_to_expression(tp.item_type)
            + ">"
        )
    if pa.types.is_struct(tp):
        inner = ",".join(
            _quote_name(f.name) + ":" + _type_to_expression(f.type) for f in tp
        )
        return "{" + inner + "}"
    raise ValueError(f"can't convert {tp} to schema expression")


class _Tokenizer:
    def __init__(self, expr: str):
        self.expr = expr
        self.pos = 0

    def peek(self) -> str:
        return self.expr[self.pos] if self.pos < len(self.expr) else ""

    def next(self) -> str:
        ch = self.peek()
        self.pos += 1
        return ch

    def skip_ws(self) -> None:
        while self.peek() != "" and self.peek() in " \t\n":
            self.pos += 1

    def read_name(self) -> str:
        self.skip_ws()
        if self.peek() == "`":
            # backtick-quoted name: any characters until the closing tick
            self.next()
            start = self.pos
            while self.peek() != "" and self.peek() != "`":
                self.pos += 1
            if self.peek() != "`":
                raise SyntaxError(f"unclosed ` at {start} in {self.expr!r}")
            name = self.expr[start : self.pos]
            self.next()
            if name == "":
                raise SyntaxError(f"empty quoted name at {start}")
            return name
        start = self.pos
        while self.peek() and (self.peek().isalnum() or self.peek() == "_"):
            self.pos += 1
        name = self.expr[start : self.pos]
        if name == "":
            raise SyntaxError(f"expected name at {start} in {self.expr!r}")
        return name

    def expect(self, ch: str) -> None:
        self.skip_ws()
        got = self.next()
        if got != ch:
            raise SyntaxError(
                f"expected {ch!r} got {got!r} at {self.pos - 1} in {self.expr!r}"
            )

    def read_type(self) -> pa.DataType:
        self.skip_ws()
        ch = self.peek()
        if ch == "[":
            self.next()
            inner = self.read_type()
            self.expect("]")
            return pa.list_(inner)
        if ch == "{":
            self.next()
            fields = self._read_fields("}")
            return pa.struct(fields)
        if ch == "<":
            self.next()
            k = self.read_type()
            self.skip_ws()
            if self.peek() == ",":
                self.next()
                v = self.read_type()
                self.expect(">")
                return pa.map_(k, v)
            self.expect(">")
            return pa.list_(k)
        # word type, possibly parameterized
        start = self.pos
        while self.peek() and (self.peek().isalnum() or self.peek() == "_"):
            self.pos += 1
        word = self.expr[start : self.pos].lower()
        self.skip_ws()
        if self.peek() == "(":
            self.next()
            args: List[str] = []
            cur = ""
            while True:
                ch = self.next()
                if ch == "" or ch == ")":
                    args.append(cur.strip())
                    break
                if ch == ",":
                    args.append(cur.strip())
                    cur = ""
                else:
                    cur += ch
            if word == "decimal":
                return pa.decimal128(int(args[0]), int(args[1]) if len(args) > 1 else 0)
            if word in ("timestamp", "datetime"):
                unit = args[0] if args and args[0] else "us"
                tz = args[1] if len(args) > 1 and args[1] else None
                return pa.timestamp(unit, tz)
            raise SyntaxError(f"unknown parameterized type {word!r}")
        if word not in _SIMPLE_TYPES:
            raise SyntaxError(f"unknown type {word!r} in {self.expr!r}")
        return _SIMPLE_TYPES[word]

    def _read_fields(self, closing: str) -> List[pa.Field]:
        fields: List[pa.Field] = []
        while True:
            self.skip_ws()
            if self.peek() == closing:
                self.next()
                return fields
            name = self.read_name()
            self.expect(":")
            tp = self.read_type()
            fields.append(pa.field(name, tp))
            self.skip_ws()
            if self.peek() == ",":
                self.next()
            elif self.peek() == closing:
                self.next()
                return fields
            elif self.peek() == "":
                if closing == "":
                    return fields
                raise SyntaxError(f"unclosed {closing!r} in {self.expr!r}")


def expression_to_schema(expr: str) -> pa.Schema:
    """Parse a schema expression like ``a:int,b:[str],c:{x:long}``."""
    tk = _Tokenizer(expr)
    fields: List[pa.Field] = []
    while True:
        tk.skip_ws()
        if tk.peek() == "":
            break
        name = tk.read_name()
        tk.expect(":")
        tp = tk.read_type()
        fields.append(pa.field(name, tp))
        tk.skip_ws()
        if tk.peek() == ",":
            tk.next()
        elif tk.peek() == "":
            break
        else:
            raise SyntaxError(f"unexpected {tk.peek()!r} in {expr!r}")
    if len(fields) == 0:
        raise SyntaxError(f"empty schema expression {expr!r}")
    names = [f.name for f in fields]
    if len(set(names)) != len(names):
        raise SyntaxError(f"duplicate field names in {expr!r}")
    return pa.schema(fields)


def schema_to_expression(schema: pa.Schema) -> str:
    return ",".join(
        _quote_name(f.name) + ":" + _type_to_expression(f.type) for f in schema
    )


def to_pa_datatype(obj: Any) -> pa.DataType:
    """Convert a type-ish object (string expression, python/numpy/pandas type,
    or pyarrow type) to a ``pa.DataType``."""
    if isinstance(obj, pa.DataType):
        return obj
    if isinstance(obj, pa.Field):
        return obj.type
    if isinstance(obj, str):
        tk = _Tokenizer(obj)
        tp = tk.read_type()
        tk.skip_ws()
        if tk.peek() != "":
            raise SyntaxError(f"trailing input in type expression {obj!r}")
        return tp
    if obj is int:
        return pa.int64()
    if obj is float:
        return pa.float64()
    if obj is str:
        return pa.string()
    if obj is bool:
        return pa.bool_()
    if obj is bytes:
        return pa.binary()
    import datetime

    if obj is datetime.datetime:
        return pa.timestamp("us")
    if obj is datetime.date:
        return pa.date32()
    if isinstance(obj, type) and issubclass(obj, np.generic):
        return pa.from_numpy_dtype(obj)
    if isinstance(obj, np.dtype):
        return pa.from_numpy_dtype(obj)
    try:
        return pa.from_numpy_dtype(np.dtype(obj))
    except Exception:
        raise ValueError(f"can't convert {obj!r} to pyarrow data type")


class Schema:
    """Ordered, immutable-ish schema: a list of named, typed fields.

    Accepts: expression strings, ``pa.Schema``/``pa.Field`` lists,
    ``(name, type)`` pair lists, dicts, pandas DataFrames, other ``Schema``
    instances, or a mix (varargs are concatenated).
    """

    def __init__(self, *args: Any, **kwargs: Any):
        fields: List[pa.Field] = []
        for a in args:
            fields.extend(self._parse(a))
        for k, v in kwargs.items():
            fields.append(pa.field(k, to_pa_datatype(v)))
        names = [f.name for f in fields]
        if len(set(names)) != len(names):
            raise SchemaError(f"duplicate columns in {names}")
        for n in names:
            # whitespace-only names are legal when quoted (`` ` ``);
            # only truly empty names are invalid
            if n == "":
                raise SchemaError("empty column name")
        self._schema = pa.schema(fields)
        self._index = {f.name: i for i, f in enumerate(fields)}

    # --- construction helpers -------------------------------------------------
    def _parse(self, obj: Any) -> List[pa.Field]:
        if obj is None:
            return []
        if isinstance(obj, Schema):
            return list(obj.fields)
        if isinstance(obj, pa.Schema):
            return list(obj)
        if isinstance(obj, pa.Field):
            return [obj]
        if isinstance(obj, str):
            if obj.strip() == "":
                return []
            return list(expression_to_schema(obj))
        if isinstance(obj, pd.DataFrame):
            from fugue_amd.utils.pandas_like import pandas_to_schema

            return list(pandas_to_schema(obj))
        if isinstance(obj, tuple) and len(obj) == 2 and isinstance(obj[0], str):
            return [pa.field(obj[0], to_pa_datatype(obj[1]))]
        if isinstance(obj, dict):
            return [pa.field(k, to_pa_datatype(v)) for k, v in obj.items()]
        if isinstance(obj, Iterable):
            res: List[pa.Field] = []
            for x in obj:
                res.extend(self._parse(x))
            return res
        raise SchemaError(f"can't parse schema from {obj!r}")

    # --- basic properties -----------------------------------------------------
    @property
    def names(self) -> List[str]:
        return list(self._schema.names)

    @property
    def fields(self) -> List[pa.Field]:
        return list(self._schema)

    @property
    def types(self) -> List[pa.DataType]:
        return [f.type for f in self._schema]

    @property
    def pa_schema(self) -> pa.Schema:
        return self._schema

    @property
    def pandas_dtype(self) -> Dict[str, Any]:
        return {f.name: f.type.to_pandas_dtype() for f in self._schema}

    def __len__(self) -> int:
        return len(self._schema)

    def __iter__(self):
        return iter(self.names)

    def items(self) -> Iterable[Tuple[str, pa.DataType]]:
        return [(f.name, f.type) for f in self._schema]

    def __getitem__(self, key: Union[str, int]) -> pa.Field:
        if isinstance(key, int):
            return self._schema.field(key)
        if key in self._index:
            return self._schema.field(self._index[key])
        raise SchemaError(f"{key} not in schema {self}")

    def get_value_type(self, key: Union[str, int]) -> pa.DataType:
        return self[key].type

    def index_of_key(self, key: str) -> int:
        if key not in self._index:
            raise SchemaError(f"{key} not in schema {self}")
        return self._index[key]

    def __contains__(self, key: Any) -> bool:
        if key is None:
            return False
        if isinstance(key, str):
            if ":" in key:
                try:
                    other = Schema(key)
                except Exception:
                    return False
                return all(self.__contains__(f) for f in other.fields)
            return key in self._index
        if isinstance(key, pa.Field):
            return key.name in self._index and self[key.name].type == key.type
        if isinstance(key, Schema):
            return all(self.__contains__(f) for f in key.fields)
        if isinstance(key, (list, tuple)):
            return all(self.__contains__(k) for k in key)
        return False

    def __eq__(self, other: Any) -> bool:
        if other is None:
            return False
        if isinstance(other, Schema):
            return self._schema.equals(other._schema)
        if isinstance(other, str):
            try:
                return self._schema.equals(Schema(other)._schema)
            except Exception:
                return False
        if isinstance(other, pa.Schema):
            return self._schema.equals(other)
        try:
            return self._schema.equals(Schema(other)._schema)
        except Exception:
            return False

    def __ne__(self, other: Any) -> bool:
        return not self.__eq__(other)

    def __hash__(self) -> int:
        return hash(str(self))

    def __repr__(self) -> str:
        return str(self)

    def __str__(self) -> str:
        return schema_to_expression(self._schema)

    def __uuid__(self) -> str:
        from fugue_amd.utils.hash import to_uuid

        return to_uuid(str(self))

    # --- transformations ------------------------------------------------------
    def __add__(self, other: Any) -> "Schema":
        return Schema(self, other)

    def __sub__(self, other: Any) -> "Schema":
        return self.exclude(other, require_type_match=True)

    def assert_not_empty(self) -> "Schema":
        if len(self) == 0:
            raise SchemaError("schema is empty")
        return self

    def copy(self) -> "Schema":
        return Schema(self)

    def extract(
        self,
        obj: Any,
        ignore_key_mismatch: bool = False,
        require_type_match: bool = True,
        ignore_type_mismatch: bool = False,
    ) -> "Schema":
        """Subset of this schema by names / fields / sub-schema, keeping the
        requested order."""
        keys = self._as_keys(obj)
        fields: List[pa.Field] = []
        for k in keys:
            if isinstance(k, pa.Field):
                if k.name not in self._index:
                    if ignore_key_mismatch:
                        continue
                    raise SchemaError(f"{k.name} not in {self}")
                mine = self[k.name]
                if require_type_match and not ignore_type_mismatch and mine.type != k.type:
                    raise SchemaError(
                        f"type mismatch for {k.name}: {mine.type} vs {k.type}"
                    )
                fields.append(mine)
            else:
                if k not in self._index:
                    if ignore_key_mismatch:
                        continue
                    raise SchemaError(f"{k} not in {self}")
                fields.append(self[k])
        return Schema(fields)

    def exclude(
        self,
        obj: Any,
        require_type_match: bool = False,
        ignore_type_mismatch: bool = True,
    ) -> "Schema":
        keys = self._as_keys(obj)
        to_remove = set()
        for k in keys:
            if isinstance(k, pa.Field):
                if k.name not in self._index:
                    continue
                mine = self[k.name]
                if mine.type != k.type:
                    if require_type_match and not ignore_type_mismatch:
                        raise SchemaError(
                            f"type mismatch for {k.name}: {mine.type} vs {k.type}"
                        )
                    if not ignore_type_mismatch:
                        continue
                    if require_type_match:
                        continue
                    to_remove.add(k.name)
                else:
                    to_remove.add(k.name)
            else:
                to_remove.add(k)
        return Schema([f for f in self.fields if f.name not in to_remove])

    def intersect(self, other: Any) -> "Schema":
        other_s = other if isinstance(other, Schema) else Schema(other)
        return Schema([f for f in self.fields if f.name in other_s._index])

    def union(self, other: Any) -> "Schema":
        other_s = other if isinstance(other, Schema) else Schema(other)
        extra = [f for f in other_s.fields if f.name not in self._index]
        return Schema(list(self.fields) + extra)

    def rename(self, columns: Dict[str, str]) -> "Schema":
        for k in columns:
            if k not in self._index:
                raise SchemaError(f"{k} not in schema {self}")
        new_names = [columns.get(f.name, f.name) for f in self.fields]
        if len(set(new_names)) != len(new_names):
            raise SchemaError(f"rename causes duplicates: {new_names}")
        return Schema(
            [pa.field(n, f.type) for n, f in zip(new_names, self.fields)]
        )

    def alter(self, subschema: Any) -> "Schema":
        """Change types of a subset of columns (by name)."""
        if subschema is None:
            return self.copy()
        sub = subschema if isinstance(subschema, Schema) else Schema(subschema)
        for f in sub.fields:
            if f.name not in self._index:
                raise SchemaError(f"{f.name} not in schema {self}")
        return Schema(
            [
                pa.field(f.name, sub[f.name].type) if f.name in sub._index else f
                for f in self.fields
            ]
        )

    def transform(self, *args: Any) -> "Schema":
        """Schema transformation expressions (triad parity), e.g.::

            schema.transform("*")            # identity
            schema.transform("*,c:int")      # append column
            schema.transform("*-b")          # remove column b
            schema.transform("*~b,c")        # remove b and c if they exist
            schema.transform(lambda s: ...)  # callable form
        """
        result_fields: List[pa.Field] = []
        for a in args:
            if callable(a):
                result_fields.extend(Schema(a(self)).fields)
                continue
            if isinstance(a, Schema):
                result_fields.extend(a.fields)
                continue
            if not isinstance(a, str):
                result_fields.extend(self._parse(a))
                continue
            for token in self._split_top_level(a):
                token = token.strip()
                if token == "":
                    continue
                if token.startswith("*"):
                    rest = token[1:]
                    removed: List[str] = []
                    soft_removed: List[str] = []
                    while rest != "":
                        if rest[0] == "-":
                            rest = rest[1:]
                            name, rest = self._take_name(rest)
                            removed.append(name)
                        elif rest[0] == "~":
                            rest = rest[1:]
                            name, rest = self._take_name(rest)
                            soft_removed.append(name)
                        else:
                            raise SchemaError(
                                f"invalid transform expression {token!r}"
                            )
                    for r in removed:
                        if r not in self._index:
                            raise SchemaError(f"{r} not in schema {self}")
                    drop = set(removed) | {
                        s for s in soft_removed if s in self._index
                    }
                    result_fields.extend(
                        f for f in self.fields if f.name not in drop
                    )
                elif token.startswith("-") or token.startswith("~"):
                    hard = token[0] == "-"
                    names = [
                        t.strip() for t in token[1:].split("+") if t.strip() != ""
                    ]
                    keep = list(result_fields)
                    for n in names:
                        if hard and n not in [f.name for f in keep]:
                            raise SchemaError(f"{n} not in schema")
                        keep = [f for f in keep if f.name != n]
                    result_fields = keep
                else:
                    result_fields.extend(list(expression_to_schema(token)))
        return Schema(result_fields)

    @staticmethod
    def _take_name(s: str) -> Tuple[str, str]:
        i = 0
        while i < len(s) and (s[i].isalnum() or s[i] == "_"):
            i += 1
        if i == 0:
            raise SchemaError(f"expected name in {s!r}")
        return s[:i], s[i:]

    @staticmethod
    def _split_top_level(s: str) -> List[str]:
        parts: List[str] = []
        depth = 0
        cur = ""
        for ch in s:
            if ch in "[{<(":
                depth += 1
            elif ch in "]}>)":
                depth -= 1
            if ch == "," and depth == 0:
                parts.append(cur)
                cur = ""
            else:
                cur += ch
        parts.append(cur)
        return parts

    def _as_keys(self, obj: Any) -> List[Any]:
        if obj is None:
            return []
        if isinstance(obj, Schema):
            return list(obj.fields)
        if isinstance(obj, pa.Schema):
            return list(obj)
        if isinstance(obj, pa.Field):
            return [obj]
        if isinstance(obj, str):
            if ":" in obj:
                return list(expression_to_schema(obj))
            return [x.strip() for x in obj.split(",") if x.strip() != ""]
        if isinstance(obj, (list, tuple, set)):
            res: List[Any] = []
            for x in obj:
                res.extend(self._as_keys(x))
            return res
        raise SchemaError(f"can't interpret {obj!r} as schema keys")

    # --- conversion -----------------------------------------------------------
    def create_empty_pandas(self) -> pd.DataFrame:
        return pa.Table.from_batches([], schema=self._schema).to_pandas()

    def create_empty_arrow(self) -> pa.Table:
        return pa.Table.from_batches([], schema=self._schema)


class SchemaError(ValueError):
    pass
