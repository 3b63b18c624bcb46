"""DataFrame utilities: equality assert, partition-blob serialization, and
join-schema inference.

Reference parity: ``fugue/dataframe/utils.py`` (``_df_eq`` :24,
``serialize_df``/``deserialize_df`` :97/:127, ``get_join_schemas`` :152).
The serialization format here is Arrow IPC (not pickle) so blobs can later
stay device-resident in the MI355X engine's zip/comap path.
"""
from typing import Any, Dict, List, Optional, Tuple

import pandas as pd
import pyarrow as pa

from fugue_amd.dataframe.array_dataframe import ArrayDataFrame
from fugue_amd.dataframe.arrow_dataframe import ArrowDataFrame
from fugue_amd.dataframe.dataframe import DataFrame, LocalBoundedDataFrame, as_fugue_df
from fugue_amd.exceptions import FugueDataFrameOperationError
from fugue_amd.schema import Schema

def parse_join_type(how: str) -> str:
    key = how.strip().lower().replace(" ", "").replace("_", "")
    key2 = {
        "inner": "inner",
        "join": "inner",
        "cross": "cross",
        "semi": "semi",
        "leftsemi": "semi",
        "anti": "anti",
        "leftanti": "anti",
        "left": "left_outer",
        "leftouter": "left_outer",
        "right": "right_outer",
        "rightouter": "right_outer",
        "outer": "full_outer",
        "full": "full_outer",
        "fullouter": "full_outer",
    }
    if key not in key2:
        raise NotImplementedError(f"unsupported join type {how!r}")
    return key2[key]


def get_join_schemas(
    df1: DataFrame, df2: DataFrame, how: str, on: Optional[Any]
) -> Tuple[Schema, Schema]:
    """Infer join key schema and output schema.  Keys default to the common
    columns of the two frames; validates explicit ``on`` against that set.
    Returns (key_schema, output_schema)."""
    how = parse_join_type(how)
    schema1, schema2 = df1.schema, df2.schema
    common = [n for n in schema1.names if n in schema2._index]
    if on is not None and len(list(on)) > 0:
        on = list(on)
        if set(on) != set(common):
            raise SyntaxError(
                f"join keys {on} differ from common columns {common}"
            )
        keys = on
    else:
        keys = common
    if how == "cross":
        if len(common) > 0:
            raise SyntaxError(f"cross join can't have common columns {common}")
        return Schema([]), schema1 + schema2
    if len(keys) == 0:
        raise SyntaxError("no join keys found")
    key_schema = schema1.extract(keys)
    if how in ("semi", "anti"):
        return key_schema, schema1.copy()
    output = schema1 + schema2.exclude(keys)
    return key_schema, output


def _df_eq(
    df: DataFrame,
    data: Any,
    schema: Any = None,
    check_order: bool = False,
    check_schema: bool = True,
    check_content: bool = True,
    no_pandas: bool = False,
    digits: int = 8,
    throw: bool = False,
) -> bool:
    """Compare a dataframe against expected data (another df or raw rows +
    schema)."""
    try:
        df1 = df.as_local_bounded()
        if isinstance(data, DataFrame):
            df2 = data.as_local_bounded()
        else:
            df2 = ArrayDataFrame(data, schema if schema is not None else df.schema)
        if check_schema and df1.schema != df2.schema:
            # column order is not identity (reference Schema.is_like):
            # same name->type mapping in any order is a match
            f1 = {f.name: f.type for f in df1.schema.fields}
            f2 = {f.name: f.type for f in df2.schema.fields}
            if f1 != f2:
                raise AssertionError(
                    f"schema mismatch {df1.schema} vs {df2.schema}"
                )
            df1 = df1[df2.schema.names].as_local_bounded()
        if not check_content:
            return True
        a1 = df1.as_array(type_safe=True)
        a2 = df2.as_array(type_safe=True)
        if len(a1) != len(a2):
            raise AssertionError(f"row count {len(a1)} vs {len(a2)}")
        if not check_order:
            a1 = sorted(a1, key=_row_key)
            a2 = sorted(a2, key=_row_key)
        for r1, r2 in zip(a1, a2):
            if not _rows_eq(r1, r2, digits):
                raise AssertionError(f"row mismatch {r1} vs {r2}")
        return True
    except AssertionError:
        if throw:
            raise
        return False


def _row_key(row: List[Any]) -> str:
    return repr([None if _is_na(x) else x for x in row])


def _is_na(x: Any) -> bool:
    if x is None:
        return True
    try:
        return bool(pd.isna(x))
    except (TypeError, ValueError):
        return False


def _rows_eq(r1: List[Any], r2: List[Any], digits: int) -> bool:
    if len(r1) != len(r2):
        return False
    for a, b in zip(r1, r2):
        na_a, na_b = _is_na(a), _is_na(b)
        if na_a or na_b:
            if na_a != na_b:
                return False
            continue
        if isinstance(a, float) or isinstance(b, float):
            if abs(float(a) - float(b)) >= 10 ** (-digits):
                return False
        elif a != b:
            if str(a) != str(b):
                return False
    return True


_FILE_BLOB_MARKER = b"\x00FUGUE-BLOB-FILE\x00"


def serialize_df(
    df: Optional[DataFrame],
    threshold: int = -1,
    file_path_root: Optional[str] = None,
) -> Optional[bytes]:
    """Serialize a local frame to Arrow IPC bytes.  When ``threshold`` is
    positive and the payload exceeds it, the bytes are written to a file
    under ``file_path_root`` and a path marker is returned instead
    (reference parity: ``to_file_threshold`` in
    ``fugue/execution/execution_engine.py:968-979``)."""
    if df is None:
        return None
    table = df.as_arrow()
    sink = pa.BufferOutputStream()
    with pa.ipc.new_stream(sink, table.schema) as writer:
        writer.write_table(table)
    data = sink.getvalue().to_pybytes()
    if threshold > 0 and len(data) > threshold:
        import os
        import tempfile
        import uuid as _uuid

        root = file_path_root or tempfile.gettempdir()
        os.makedirs(root, exist_ok=True)
        path = os.path.join(root, f"fugue-blob-{_uuid.uuid4().hex}.arrow")
        with open(path, "wb") as fh:
            fh.write(data)
        return _FILE_BLOB_MARKER + path.encode("utf-8")
    return data


def deserialize_df(data: Optional[bytes]) -> Optional[LocalBoundedDataFrame]:
    if data is None:
        return None
    if isinstance(data, (bytes, bytearray)) and bytes(data).startswith(
        _FILE_BLOB_MARKER
    ):
        path = bytes(data)[len(_FILE_BLOB_MARKER):].decode("utf-8")
        with open(path, "rb") as fh:
            data = fh.read()
    with pa.ipc.open_stream(pa.BufferReader(data)) as reader:
        table = reader.read_all()
    return ArrowDataFrame(table)


def normalize_dataframe_column_names(df: pd.DataFrame) -> Tuple[pd.DataFrame, Dict[str, Any]]:
    """Rename columns to safe identifiers; returns (renamed_df, inverse_map)."""
    names = list(df.columns)
    new_names: List[str] = []
    inverse: Dict[str, Any] = {}
    for i, n in enumerate(names):
        sn = str(n)
        if sn.isidentifier():
            new_names.append(sn)
        else:
            nn = f"_{i}"
            new_names.append(nn)
            inverse[nn] = n
    out = df.copy()
    out.columns = new_names
    return out, inverse
