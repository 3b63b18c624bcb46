"""Partition specification language.

Reference parity: ``fugue/collections/partition.py`` — algo ∈ {default,
hash, rand, even, coarse}, ``num`` expressions with ``ROWCOUNT`` /
``CONCURRENCY`` keywords, partition_by keys, presort.  New implementation.
"""
import json
from typing import Any, Callable, Dict, Iterable, Iterator, List, Optional, Tuple

from fugue_amd.schema import Schema
from fugue_amd.utils.hash import to_uuid
from fugue_amd.utils.params import ParamDict

_VALID_ALGOS = ("", "default", "hash", "rand", "even", "coarse")

# canonical field names for the spec-dict aliases users may pass
_FIELD_ALIASES = {"by": "partition_by", "num": "num_partitions"}


def _spec_shorthand(a: Any) -> Optional[Dict[str, Any]]:
    """Interpret the one-positional-argument shorthands
    (``PartitionSpec(4)``, ``PartitionSpec("a")``, ``PartitionSpec(["a"])``,
    ``PartitionSpec("per_row")``); returns None when ``a`` is a full spec
    (dict / JSON string / PartitionSpec) that the merge path handles."""
    if isinstance(a, bool):
        return None
    if isinstance(a, int):
        return {"num_partitions": str(a)}
    if isinstance(a, (list, tuple)):
        return {"partition_by": list(a)}
    if isinstance(a, str) and not a.startswith("{"):
        if a.lower() == "per_row":
            return {"algo": "even", "num_partitions": "ROWCOUNT"}
        return {"partition_by": [a]} if a != "" else {}
    return None


def _merge_spec_sources(
    args: Tuple[Any, ...], kwargs: Dict[str, Any]
) -> Dict[str, Any]:
    """Fold positional spec sources (PartitionSpec / dict / JSON string)
    left-to-right, then keyword overrides, normalizing field aliases."""
    merged: Dict[str, Any] = {}

    def absorb(u: Dict[str, Any]) -> None:
        for k, v in u.items():
            merged[_FIELD_ALIASES.get(k, k)] = v

    for a in args:
        if a is None:
            continue
        if isinstance(a, PartitionSpec):
            absorb(a.jsondict)
        elif isinstance(a, dict):
            absorb(a)
        elif isinstance(a, str):
            absorb(json.loads(a))
        else:
            raise TypeError(f"{a} is not supported by PartitionSpec")
    absorb(kwargs)
    return merged


def parse_presort_exp(presort: Any) -> Dict[str, bool]:
    """Parse ``"b desc, c asc"`` (or pair lists) into an ordered
    {column: ascending} dict."""
    res: Dict[str, bool] = {}
    if presort is None:
        return res
    if isinstance(presort, dict):
        return dict(presort)
    pairs: List[Tuple[str, bool]] = []
    if isinstance(presort, str):
        s = presort.strip()
        if s == "":
            return res
        for part in s.split(","):
            tokens = part.strip().split()
            if len(tokens) == 1:
                pairs.append((tokens[0], True))
            elif len(tokens) == 2:
                direction = tokens[1].lower()
                if direction == "asc":
                    pairs.append((tokens[0], True))
                elif direction == "desc":
                    pairs.append((tokens[0], False))
                else:
                    raise SyntaxError(f"invalid presort expression {presort!r}")
            else:
                raise SyntaxError(f"invalid presort expression {presort!r}")
    elif isinstance(presort, list):
        for p in presort:
            if isinstance(p, str):
                pairs.append((p, True))
            elif isinstance(p, tuple) and len(p) == 2:
                pairs.append((str(p[0]), bool(p[1])))
            else:
                raise SyntaxError(f"invalid presort expression {presort!r}")
    else:
        raise SyntaxError(f"invalid presort expression {presort!r}")
    for k, v in pairs:
        if k in res:
            raise SyntaxError(f"duplicated presort key {k}")
        res[k] = v
    return res


class PartitionSpec:
    """Partition specification: algo + num expression + partition keys +
    presort.  See reference docs for the user-facing semantics."""

    def __init__(self, *args: Any, **kwargs: Any):
        p: Optional[Dict[str, Any]] = (
            _spec_shorthand(args[0])
            if len(args) == 1 and len(kwargs) == 0
            else None
        )
        if not p:  # no shorthand matched (or it was empty): merge sources
            p = _merge_spec_sources(args, kwargs)
        self._init_fields(p)

    def _init_fields(self, p: Dict[str, Any]) -> None:
        self._num_partitions = str(p.get("num_partitions", "0"))
        self._algo = str(p.get("algo", "")).lower()
        if self._algo not in _VALID_ALGOS:
            raise SyntaxError(f"invalid partition algo {self._algo!r}")
        by = p.get("partition_by", [])
        self._partition_by: List[str] = [by] if isinstance(by, str) else list(by)
        if len(self._partition_by) != len(set(self._partition_by)):
            raise SyntaxError(f"{self._partition_by} has duplicated keys")
        self._presort = parse_presort_exp(p.get("presort", None))
        overlap = set(self._presort) & set(self._partition_by)
        if overlap:
            raise SyntaxError(
                f"partition by overlaps with presort: "
                f"{self._partition_by}, {list(self._presort)}"
            )

    def __repr__(self) -> str:
        return (
            f"PartitionSpec(num='{self._num_partitions}', "
            f"by={self._partition_by}, presort='{self.presort_expr}')"
        )

    def __eq__(self, other: Any) -> bool:
        if other is self:
            return True
        if not isinstance(other, PartitionSpec):
            other = PartitionSpec(other)
        return self.jsondict == other.jsondict

    @property
    def empty(self) -> bool:
        return (
            self._num_partitions == "0"
            and self._algo == ""
            and len(self._partition_by) == 0
            and len(self._presort) == 0
        )

    @property
    def num_partitions(self) -> str:
        return self._num_partitions

    def get_num_partitions(self, **expr_map_funcs: Any) -> int:
        """Evaluate the ``num`` expression; keyword funcs provide values for
        ``ROWCOUNT`` / ``CONCURRENCY`` lazily."""
        expr = self.num_partitions
        for k, v in expr_map_funcs.items():
            if k in expr:
                expr = expr.replace(k, str(v()))
        return int(eval(expr, {"__builtins__": {}}, {}))

    @property
    def algo(self) -> str:
        return self._algo if self._algo != "" else "default"

    @property
    def partition_by(self) -> List[str]:
        return self._partition_by

    @property
    def presort(self) -> Dict[str, bool]:
        return self._presort

    @property
    def presort_expr(self) -> str:
        return ",".join(
            k + " " + ("ASC" if v else "DESC") for k, v in self._presort.items()
        )

    @property
    def jsondict(self) -> ParamDict:
        return ParamDict(
            dict(
                num_partitions=self._num_partitions,
                algo=self._algo,
                partition_by=self._partition_by,
                presort=self.presort_expr,
                size_limit=0,
                row_limit=0,
            )
        )

    def __uuid__(self) -> str:
        return to_uuid(self.jsondict)

    def get_sorts(
        self, schema: Schema, with_partition_keys: bool = True
    ) -> Dict[str, bool]:
        d: Dict[str, bool] = {}
        if with_partition_keys:
            for p in self.partition_by:
                if p not in schema:
                    raise KeyError(f"{p} not in {schema}")
                d[p] = True
        for p, v in self.presort.items():
            if p not in schema:
                raise KeyError(f"{p} not in {schema}")
            d[p] = v
        return d

    def get_key_schema(self, schema: Schema) -> Schema:
        return schema.extract(self.partition_by)

    def get_cursor(
        self, schema: Schema, physical_partition_no: int
    ) -> "PartitionCursor":
        return PartitionCursor(schema, self, physical_partition_no)


class DatasetPartitionCursor:
    """Cursor pointing at the first item of each logical partition inside a
    physical partition."""

    def __init__(self, physical_partition_no: int):
        self._physical_partition_no = physical_partition_no
        self._partition_no = 0
        self._slice_no = 0
        self._item: Any = None

    def set(self, item: Any, partition_no: int, slice_no: int) -> None:
        self._item = item
        self._partition_no = partition_no
        self._slice_no = slice_no

    @property
    def item(self) -> Any:
        if callable(self._item):
            self._item = self._item()
        return self._item

    @property
    def partition_no(self) -> int:
        return self._partition_no

    @property
    def physical_partition_no(self) -> int:
        return self._physical_partition_no

    @property
    def slice_no(self) -> int:
        return self._slice_no


class BagPartitionCursor(DatasetPartitionCursor):
    pass


class PartitionCursor(DatasetPartitionCursor):
    def __init__(self, schema: Schema, spec: PartitionSpec, physical_partition_no: int):
        super().__init__(physical_partition_no)
        self._orig_schema = schema
        self._key_index = [schema.index_of_key(key) for key in spec.partition_by]
        self._schema = schema.extract(spec.partition_by)

    def set(self, row: Any, partition_no: int, slice_no: int) -> None:
        super().set(
            list(row) if not callable(row) else (lambda: list(row())),
            partition_no=partition_no,
            slice_no=slice_no,
        )

    @property
    def row(self) -> List[Any]:
        return self.item

    @property
    def row_schema(self) -> Schema:
        return self._orig_schema

    @property
    def key_schema(self) -> Schema:
        return self._schema

    @property
    def key_value_dict(self) -> Dict[str, Any]:
        return {self.row_schema.names[i]: self.row[i] for i in self._key_index}

    @property
    def key_value_array(self) -> List[Any]:
        return [self.row[i] for i in self._key_index]

    def __getitem__(self, key: str) -> Any:
        return self.row[self.row_schema.index_of_key(key)]
