"""LocalDataFrameIterableDataFrame: a stream of local frames.

Reference parity: ``fugue/dataframe/dataframe_iterable_dataframe.py:21`` —
lets a UDF consume/emit an iterable of chunk-frames without materializing a
whole partition (the copy/compute-overlap boundary type for the MI355X
engine's streaming map path).
"""
from typing import Any, Dict, Iterable, Iterator, List, Optional

import pandas as pd
import pyarrow as pa

from fugue_amd.dataframe.arrow_dataframe import ArrowDataFrame
from fugue_amd.dataframe.array_dataframe import ArrayDataFrame
from fugue_amd.dataframe.dataframe import (
    DataFrame,
    LocalBoundedDataFrame,
    LocalDataFrame,
    LocalUnboundedDataFrame,
)
from fugue_amd.dataframe.pandas_dataframe import PandasDataFrame
from fugue_amd.exceptions import (
    FugueDataFrameEmptyError,
    FugueDataFrameInitError,
    FugueDataFrameOperationError,
)
from fugue_amd.schema import Schema


class _FrameIter:
    def __init__(self, it: Iterator[LocalDataFrame]):
        self._it = it
        self._buffer: List[LocalDataFrame] = []

    def peek(self) -> LocalDataFrame:
        if not self._buffer:
            self._buffer.append(next(self._it))  # may raise StopIteration
        return self._buffer[0]

    def __iter__(self):
        while True:
            if self._buffer:
                yield self._buffer.pop(0)
            else:
                try:
                    yield next(self._it)
                except StopIteration:
                    return


class LocalDataFrameIterableDataFrame(LocalUnboundedDataFrame):
    def __init__(self, df: Any = None, schema: Any = None):
        try:
            if df is None:
                schema = Schema(schema).assert_not_empty()
                it: Iterator[LocalDataFrame] = iter([])
            elif isinstance(df, LocalDataFrameIterableDataFrame):
                it = iter(df.native)
                schema = df.schema if schema is None else Schema(schema)
            elif isinstance(df, Iterable):
                it = iter(df)
            else:
                raise ValueError(
                    f"{type(df)} is incompatible with LocalDataFrameIterableDataFrame"
                )
        except FugueDataFrameInitError:
            raise
        except Exception as e:
            raise FugueDataFrameInitError(str(e)) from e
        self._native = _FrameIter(it)
        if schema is None or (isinstance(schema, Schema) and len(schema) == 0):
            try:
                schema = self._native.peek().schema
            except StopIteration:
                raise FugueDataFrameInitError(
                    "schema can't be inferred from an empty iterable of dataframes"
                )
        super().__init__(schema)

    @property
    def native(self) -> _FrameIter:
        return self._native

    def native_as_df(self) -> Any:
        return self._native

    @property
    def empty(self) -> bool:
        try:
            return self._native.peek().empty
        except StopIteration:
            return True

    def peek_array(self) -> List[Any]:
        try:
            f = self._native.peek()
            return f.peek_array()
        except StopIteration:
            raise FugueDataFrameEmptyError("dataframe is empty")

    def as_local_bounded(self) -> LocalBoundedDataFrame:
        frames = [f.as_pandas() for f in self._native if f.count() > 0]
        if len(frames) == 0:
            res: LocalBoundedDataFrame = ArrayDataFrame([], self.schema)
        else:
            res = PandasDataFrame(
                pd.concat(frames, ignore_index=True), self.schema
            )
        if self.has_metadata:
            res.reset_metadata(self.metadata)
        return res

    def as_array(
        self, columns: Optional[List[str]] = None, type_safe: bool = False
    ) -> List[Any]:
        return self.as_local_bounded().as_array(columns, type_safe=type_safe)

    def as_array_iterable(
        self, columns: Optional[List[str]] = None, type_safe: bool = False
    ) -> Iterable[Any]:
        for f in self._native:
            yield from f.as_array_iterable(columns, type_safe=type_safe)

    def as_pandas(self) -> pd.DataFrame:
        return self.as_local_bounded().as_pandas()

    def as_arrow(self, type_safe: bool = False) -> pa.Table:
        return self.as_local_bounded().as_arrow(type_safe=type_safe)

    def _drop_cols(self, cols: List[str]) -> DataFrame:
        schema = self.schema.exclude(cols)
        return LocalDataFrameIterableDataFrame(
            (f.drop(cols) for f in self._native), schema
        )

    def _select_cols(self, cols: List[Any]) -> DataFrame:
        schema = self.schema.extract(cols)
        return LocalDataFrameIterableDataFrame(
            (f[schema.names] for f in self._native), schema
        )

    def rename(self, columns: Dict[str, str]) -> DataFrame:
        try:
            schema = self.schema.rename(columns)
        except Exception as e:
            raise FugueDataFrameOperationError(str(e)) from e
        return LocalDataFrameIterableDataFrame(
            (f.rename(columns) for f in self._native), schema
        )

    def alter_columns(self, columns: Any) -> DataFrame:
        schema = self._get_altered_schema(columns)
        if schema == self.schema:
            return self
        return LocalDataFrameIterableDataFrame(
            (f.alter_columns(columns) for f in self._native), schema
        )

    def head(
        self, n: int, columns: Optional[List[str]] = None
    ) -> LocalBoundedDataFrame:
        rows: List[Any] = []
        for f in self._native:
            for row in f.as_array_iterable(columns):
                if len(rows) >= n:
                    break
                rows.append(row)
            if len(rows) >= n:
                break
        schema = self.schema if columns is None else self.schema.extract(columns)
        return ArrayDataFrame(rows, schema)


class IterablePandasDataFrame(LocalDataFrameIterableDataFrame):
    """Iterable of PandasDataFrame chunks (format hint: pandas)."""


class IterableArrowDataFrame(LocalDataFrameIterableDataFrame):
    """Iterable of ArrowDataFrame chunks (format hint: pyarrow)."""
