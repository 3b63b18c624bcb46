"""ArrowDataFrame: pyarrow.Table-backed local frame.

Reference parity: ``fugue/dataframe/arrow_dataframe.py``.
"""
from typing import Any, Dict, Iterable, List, Optional

import pandas as pd
import pyarrow as pa

from fugue_amd.dataframe.dataframe import DataFrame, LocalBoundedDataFrame
from fugue_amd.exceptions import (
    FugueDataFrameEmptyError,
    FugueDataFrameInitError,
    FugueDataFrameOperationError,
)
from fugue_amd.schema import Schema


class ArrowDataFrame(LocalBoundedDataFrame):
    def __init__(self, df: Any = None, schema: Any = None):
        try:
            if df is None:
                schema = Schema(schema).assert_not_empty()
                table = schema.create_empty_arrow()
            elif isinstance(df, pa.Table):
                if schema is None:
                    table = df
                    schema = Schema(df.schema)
                else:
                    schema = Schema(schema).assert_not_empty()
                    if df.schema != schema.pa_schema:
                        table = df.select(schema.names).cast(schema.pa_schema)
                    else:
                        table = df
            elif isinstance(df, pa.RecordBatch):
                table = pa.Table.from_batches([df])
                schema = Schema(table.schema) if schema is None else Schema(schema)
            elif isinstance(df, pd.DataFrame):
                if schema is None:
                    table = pa.Table.from_pandas(
                        df.reset_index(drop=True), preserve_index=False
                    )
                    schema = Schema(table.schema)
                else:
                    schema = Schema(schema).assert_not_empty()
                    from fugue_amd.utils.pandas_like import cast_pandas

                    table = pa.Table.from_pandas(
                        cast_pandas(df.reset_index(drop=True), schema),
                        schema=schema.pa_schema,
                        preserve_index=False,
                    )
            elif isinstance(df, Iterable):
                schema = Schema(schema).assert_not_empty()
                rows = [
                    {c: row[i] for i, c in enumerate(schema.names)} for row in df
                ]
                table = pa.Table.from_pylist(rows, schema=schema.pa_schema)
            else:
                raise ValueError(f"{type(df)} is incompatible with ArrowDataFrame")
        except FugueDataFrameInitError:
            raise
        except Exception as e:
            raise FugueDataFrameInitError(str(e)) from e
        self._native: pa.Table = table
        super().__init__(schema)

    @property
    def native(self) -> pa.Table:
        return self._native

    def native_as_df(self) -> pa.Table:
        return self._native

    @property
    def empty(self) -> bool:
        return self._native.num_rows == 0

    def count(self) -> int:
        return self._native.num_rows

    def peek_array(self) -> List[Any]:
        if self.empty:
            raise FugueDataFrameEmptyError("dataframe is empty")
        return list(self._native.slice(0, 1).to_pylist()[0].values())

    def as_pandas(self) -> pd.DataFrame:
        return self._native.to_pandas()

    def as_arrow(self, type_safe: bool = False) -> pa.Table:
        return self._native

    def as_array(
        self, columns: Optional[List[str]] = None, type_safe: bool = False
    ) -> List[Any]:
        tbl = self._native if columns is None else self._native.select(columns)
        return [list(d.values()) for d in tbl.to_pylist()]

    def as_array_iterable(
        self, columns: Optional[List[str]] = None, type_safe: bool = False
    ) -> Iterable[Any]:
        tbl = self._native if columns is None else self._native.select(columns)
        for batch in tbl.to_batches():
            for d in batch.to_pylist():
                yield list(d.values())

    def _drop_cols(self, cols: List[str]) -> DataFrame:
        schema = self.schema.exclude(cols)
        return ArrowDataFrame(self._native.select(schema.names), schema)

    def _select_cols(self, cols: List[Any]) -> DataFrame:
        schema = self.schema.extract(cols)
        return ArrowDataFrame(self._native.select(schema.names), schema)

    def rename(self, columns: Dict[str, str]) -> DataFrame:
        try:
            schema = self.schema.rename(columns)
        except Exception as e:
            raise FugueDataFrameOperationError(str(e)) from e
        return ArrowDataFrame(self._native.rename_columns(schema.names), schema)

    def alter_columns(self, columns: Any) -> DataFrame:
        schema = self._get_altered_schema(columns)
        if schema == self.schema:
            return self
        # per-column cast via the pandas casting rules so temporal->str
        # uses python formatting, str->bool is case-insensitive, etc.
        from fugue_amd.utils.pandas_like import _cast_series

        arrays = []
        for f in schema.fields:
            col = self._native.column(f.name).combine_chunks()
            if col.type == f.type:
                arrays.append(col)
            else:
                casted = _cast_series(col.to_pandas(), f.type)
                arrays.append(pa.Array.from_pandas(casted, type=f.type))
        return ArrowDataFrame(
            pa.Table.from_arrays(arrays, schema=schema.pa_schema), schema
        )

    def head(
        self, n: int, columns: Optional[List[str]] = None
    ) -> LocalBoundedDataFrame:
        tbl = self._native if columns is None else self._native.select(columns)
        schema = self.schema if columns is None else self.schema.extract(columns)
        return ArrowDataFrame(tbl.slice(0, n), schema)
