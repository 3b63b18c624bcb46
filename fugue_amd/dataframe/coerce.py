"""Cell-level type coercion so raw python rows conform to a schema
(the reference gets this from triad's type-safe conversions — e.g.
``"2020-01-01"`` in a ``datetime`` column becomes a ``datetime``)."""
import datetime as _dt
from typing import Any, Callable, List, Optional

import pyarrow as pa


def _to_datetime(v: Any) -> Any:
    if v is None or isinstance(v, _dt.datetime):
        return v
    if isinstance(v, str):
        import pandas as pd

        return pd.to_datetime(v).to_pydatetime()
    if hasattr(v, "to_pydatetime"):
        return v.to_pydatetime()
    return v


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


def column_coercers(schema: Any) -> List[Optional[Callable[[Any], Any]]]:
    """Per-column converter (or None when values pass through)."""
    out: List[Optional[Callable[[Any], Any]]] = []
    for f in schema.fields:
        t = f.type
        if pa.types.is_timestamp(t):
            out.append(_to_datetime)
        elif pa.types.is_date(t):
            out.append(_to_date)
        else:
            out.append(None)
    return out


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
