"""IterableDataFrame: one-pass local unbounded frame.

Reference parity: ``fugue/dataframe/iterable_dataframe.py`` — consuming
operations (as_array etc.) exhaust the underlying iterator; ``peek_array``
is allowed before consumption.
"""
from typing import Any, Dict, Iterable, Iterator, List, Optional

from fugue_amd.dataframe.array_dataframe import ArrayDataFrame
from fugue_amd.dataframe.dataframe import (
    DataFrame,
    LocalBoundedDataFrame,
    LocalUnboundedDataFrame,
)
from fugue_amd.exceptions import (
    FugueDataFrameEmptyError,
    FugueDataFrameInitError,
    FugueDataFrameOperationError,
)
from fugue_amd.schema import Schema


class _PeekableIterator:
    def __init__(self, it: Iterator):
        self._it = it
        self._buffer: List[Any] = []

    def peek(self) -> Any:
        if len(self._buffer) == 0:
            self._buffer.append(next(self._it))
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

    @property
    def empty(self) -> bool:
        try:
            self.peek()
            return False
        except StopIteration:
            return True


class IterableDataFrame(LocalUnboundedDataFrame):
    def __init__(self, df: Any = None, schema: Any = None):
        try:
            if df is None:
                schema = Schema(schema).assert_not_empty()
                it: Iterator = iter([])
            elif isinstance(df, IterableDataFrame):
                it = iter(df._native)
                schema = df.schema if schema is None else Schema(schema)
            elif isinstance(df, DataFrame):
                schema = df.schema if schema is None else Schema(schema)
                it = iter(df.as_array_iterable())
            elif isinstance(df, Iterable):
                schema = Schema(schema).assert_not_empty()
                it = iter(df)
            else:
                raise ValueError(f"{type(df)} is incompatible with IterableDataFrame")
        except FugueDataFrameInitError:
            raise
        except Exception as e:
            raise FugueDataFrameInitError(str(e)) from e
        self._native = _PeekableIterator(it)
        super().__init__(schema)

    @property
    def native(self) -> _PeekableIterator:
        return self._native

    def native_as_df(self) -> Any:
        # the raw iterable carries no schema; present as pandas
        return self.as_pandas()

    @property
    def empty(self) -> bool:
        return self._native.empty

    @property
    def num_partitions(self) -> int:
        return 1

    def peek_array(self) -> List[Any]:
        try:
            return list(self._native.peek())
        except StopIteration:
            raise FugueDataFrameEmptyError("dataframe is empty")

    def as_local_bounded(self) -> LocalBoundedDataFrame:
        res = ArrayDataFrame(list(self._native), self.schema)
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
        if columns is None:
            yield from self._native
        else:
            pos = [self.schema.index_of_key(c) for c in columns]
            for row in self._native:
                yield [row[i] for i in pos]

    def _drop_cols(self, cols: List[str]) -> DataFrame:
        schema = self.schema.exclude(cols)
        pos = [self.schema.index_of_key(c) for c in schema.names]
        return IterableDataFrame(
            ([r[i] for i in pos] for r in self._native), schema
        )

    def _select_cols(self, cols: List[Any]) -> DataFrame:
        schema = self.schema.extract(cols)
        pos = [self.schema.index_of_key(c) for c in schema.names]
        return IterableDataFrame(
            ([r[i] for i in pos] for r in self._native), schema
        )

    def rename(self, columns: Dict[str, str]) -> DataFrame:
        try:
            schema = self.schema.rename(columns)
        except Exception as e:
            raise FugueDataFrameOperationError(str(e)) from e
        return IterableDataFrame(self._native, schema)

    def alter_columns(self, columns: Any) -> DataFrame:
        schema = self._get_altered_schema(columns)
        if schema == self.schema:
            return self
        arr = ArrayDataFrame(list(self._native), self.schema).alter_columns(columns)
        return IterableDataFrame(arr.as_array(), schema)

    def head(
        self, n: int, columns: Optional[List[str]] = None
    ) -> LocalBoundedDataFrame:
        it = self.as_array_iterable(columns)
        rows = []
        for row in it:
            if len(rows) >= n:
                break
            rows.append(row)
        schema = self.schema if columns is None else self.schema.extract(columns)
        return ArrayDataFrame(rows, schema)
