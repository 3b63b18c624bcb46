"""ArrayDataFrame: list-of-lists local frame (no type enforcement).

Reference parity: ``fugue/dataframe/array_dataframe.py``.
"""
from typing import Any, Dict, Iterable, List, Optional

from fugue_amd.dataframe.dataframe import DataFrame, LocalBoundedDataFrame
from fugue_amd.exceptions import (
    FugueDataFrameEmptyError,
    FugueDataFrameInitError,
    FugueDataFrameOperationError,
)
from fugue_amd.schema import Schema


class ArrayDataFrame(LocalBoundedDataFrame):
    def __init__(self, df: Any = None, schema: Any = None):
        try:
            if df is None:
                schema = Schema(schema).assert_not_empty()
                data: List[List[Any]] = []
            elif isinstance(df, DataFrame):
                if schema is None:
                    schema = df.schema
                    data = df.as_array(type_safe=False)
                else:
                    schema = Schema(schema).assert_not_empty()
                    data = df.as_array(columns=Schema(schema).names, type_safe=False)
            elif isinstance(df, Iterable):
                schema = Schema(schema).assert_not_empty()
                data = [list(row) for row in df]
            else:
                raise ValueError(f"{type(df)} is incompatible with ArrayDataFrame")
        except FugueDataFrameInitError:
            raise
        except Exception as e:
            raise FugueDataFrameInitError(str(e)) from e
        self._native = data
        super().__init__(schema)

    @property
    def native(self) -> List[List[Any]]:
        return self._native

    def native_as_df(self) -> Any:
        # the raw array carries no schema; the dataframe form of this
        # frame is pandas (reference ``dataframe.py:316``)
        return self.as_pandas()

    @property
    def empty(self) -> bool:
        return len(self._native) == 0

    def count(self) -> int:
        return len(self._native)

    def peek_array(self) -> List[Any]:
        if self.empty:
            raise FugueDataFrameEmptyError("dataframe is empty")
        return list(self._native[0])

    def _pos(self, columns: Optional[List[str]]) -> List[int]:
        if columns is None:
            return list(range(len(self.schema)))
        return [self.schema.index_of_key(c) for c in columns]

    def as_array(
        self, columns: Optional[List[str]] = None, type_safe: bool = False
    ) -> List[Any]:
        if columns is None and not type_safe:
            return self._native
        if type_safe:
            # round-trip through arrow: enforces the schema exactly
            # (NaT/NaN -> None, string datetimes parsed, struct fields
            # normalized) — reference triad semantics
            from fugue_amd.dataframe.arrow_dataframe import ArrowDataFrame

            return ArrowDataFrame(self.as_arrow()).as_array(
                columns, type_safe=True
            )
        pos = self._pos(columns)
        return [[row[i] for i in pos] for row in self._native]

    def as_array_iterable(
        self, columns: Optional[List[str]] = None, type_safe: bool = False
    ) -> Iterable[Any]:
        if columns is None and not type_safe:
            yield from self._native
        elif type_safe:
            yield from self.as_array(columns, type_safe=True)
        else:
            pos = self._pos(columns)
            for row in self._native:
                yield [row[i] for i in pos]

    def _drop_cols(self, cols: List[str]) -> DataFrame:
        schema = self.schema.exclude(cols)
        pos = self._pos(schema.names)
        return ArrayDataFrame([[r[i] for i in pos] for r in self._native], schema)

    def _select_cols(self, cols: List[Any]) -> DataFrame:
        schema = self.schema.extract(cols)
        pos = self._pos(schema.names)
        return ArrayDataFrame([[r[i] for i in pos] for r in self._native], schema)

    def rename(self, columns: Dict[str, str]) -> DataFrame:
        try:
            schema = self.schema.rename(columns)
        except Exception as e:
            raise FugueDataFrameOperationError(str(e)) from e
        return ArrayDataFrame(self._native, schema)

    def alter_columns(self, columns: Any) -> DataFrame:
        schema = self._get_altered_schema(columns)
        if schema == self.schema:
            return self
        from fugue_amd.dataframe.arrow_dataframe import ArrowDataFrame

        adf = ArrowDataFrame(self._native, self.schema)
        return ArrayDataFrame(
            adf.alter_columns(columns).as_array(), schema
        )

    def head(
        self, n: int, columns: Optional[List[str]] = None
    ) -> LocalBoundedDataFrame:
        pos = self._pos(columns)
        schema = (
            self.schema if columns is None else self.schema.extract(columns)
        )
        if columns is None:
            return ArrayDataFrame(self._native[:n], schema)
        return ArrayDataFrame(
            [[r[i] for i in pos] for r in self._native[:n]], schema
        )
