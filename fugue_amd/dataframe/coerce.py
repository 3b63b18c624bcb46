"""Cell-level type coercion so raw python rows conform to a schema
(the reference gets this from triad's type-safe conversions — e.g.
``"2020-01-01"`` in a ``datetime`` column becomes a ``datetime``)."""
import datetime as _dt
from typing import Any, Callable, List, Optional

import pyarrow as pa


def _to_datetime(v: Any) -> Any:
    if v is None:
        return v
    if v != v:  # NaT
        return None
    if isinstance(v, str):
        import pandas as pd

        return pd.to_datetime(v).to_pydatetime()
    if hasattr(v, "to_pydatetime"):  # pandas Timestamp (before datetime:
        # Timestamp subclasses datetime but carries pandas-only traits)
        return v.to_pydatetime()
    if isinstance(v, _dt.datetime):
        return v
    return v


def _to_float(v: Any) -> Any:
    if v is None:
        return None
    try:
        if v != v:  # NaN is null in type-safe row form
            return None
    except Exception:  # pragma: no cover
        return v
    return float(v)


def _to_date(v: Any) -> Any:
    if v is None:
        return v
    if isinstance(v, _dt.datetime):
        return v.date()
    if isinstance(v, _dt.date):
        return v
    if isinstance(v, str):
        import pandas as pd

        return pd.to_datetime(v).date()
    return v


def converter_for(t: "pa.DataType") -> Optional[Callable[[Any], Any]]:
    """Recursive cell converter for a type (None = passthrough)."""
    if pa.types.is_timestamp(t):
        return _to_datetime
    if pa.types.is_date(t):
        return _to_date
    if pa.types.is_floating(t):
        return _to_float
    if pa.types.is_list(t) or pa.types.is_large_list(t):
        inner = converter_for(t.value_type)
        if inner is None:
            return None
        return lambda v: None if v is None else [inner(x) for x in v]
    if pa.types.is_struct(t):
        convs = {f.name: converter_for(f.type) for f in t}
        if all(c is None for c in convs.values()):
            return None

        def conv_struct(v: Any) -> Any:
            if v is None:
                return None
            return {
                k: (c(v.get(k)) if c is not None else v.get(k))
                for k, c in convs.items()
            }

        return conv_struct
    return None


def column_coercers(schema: Any) -> List[Optional[Callable[[Any], Any]]]:
    """Per-column converter (or None when values pass through)."""
    return [converter_for(f.type) for f in schema.fields]


def coerce_rows(rows: List[List[Any]], schema: Any) -> List[List[Any]]:
    """Return rows with cells coerced to the schema's types; the input
    is returned unchanged (same object) when no column needs work."""
    convs = column_coercers(schema)
    if all(c is None for c in convs):
        return rows
    idx = [(i, c) for i, c in enumerate(convs) if c is not None]
    out = []
    for row in rows:
        row = list(row)
        for i, c in idx:
            row[i] = c(row[i])
        out.append(row)
    return out
