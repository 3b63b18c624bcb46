"""DataFrame abstraction hierarchy.

Reference parity: ``fugue/dataframe/dataframe.py`` — schema-carrying
immutable frame views with lazy schema discovery, conversion
(as_pandas/as_arrow/as_array/as_dicts), rename/alter/drop/head.
The implementation is new, written against pandas/pyarrow directly.
"""
from abc import abstractmethod
from typing import TypeVar, Any, Callable, Dict, Iterable, List, Optional, Tuple, Union

import pandas as pd
import pyarrow as pa

from fugue_amd.dataset.dataset import Dataset, DatasetDisplay
from fugue_amd.exceptions import FugueDataFrameOperationError
from fugue_amd.schema import Schema
from fugue_amd.utils.display import PrettyTable
from fugue_amd.utils.registry import register_plugin, try_run_plugin

# a TypeVar (not typing.Any) so annotation dispatch can target it
# (reference ``dataframe.py:26``)
AnyDataFrame = TypeVar("AnyDataFrame", "DataFrame", object)


class DataFrame(Dataset):
    """Abstract base: an immutable, schema'd collection of rows."""

    def __init__(self, schema: Any = None):
        super().__init__()
        if callable(schema):
            self._schema: Optional[Schema] = None
            self._schema_factory: Optional[Callable[[], Any]] = schema
        else:
            self._schema = _input_schema(schema).assert_not_empty() if schema is not None else None
            self._schema_factory = None

    @property
    def schema(self) -> Schema:
        if self._schema is None:
            if self._schema_factory is None:
                raise FugueDataFrameOperationError("schema is not known")
            self._schema = _input_schema(self._schema_factory()).assert_not_empty()
            self._schema_factory = None
        return self._schema

    @property
    def schema_discovered(self) -> bool:
        return self._schema is not None

    @property
    def columns(self) -> List[str]:
        return self.schema.names

    @abstractmethod
    def native_as_df(self) -> AnyDataFrame:
        """The native object this frame wraps (or itself if already native)."""

    @property
    def native(self) -> Any:
        return self.native_as_df()

    def as_local(self) -> "LocalDataFrame":
        return self.as_local_bounded()

    @abstractmethod
    def as_local_bounded(self) -> "LocalBoundedDataFrame":
        ...

    @abstractmethod
    def peek_array(self) -> List[Any]:
        """First row as a list; raise if empty."""

    def peek_dict(self) -> Dict[str, Any]:
        arr = self.peek_array()
        return {n: arr[i] for i, n in enumerate(self.columns)}

    def as_pandas(self) -> pd.DataFrame:
        return self.as_arrow().to_pandas()

    def as_arrow(self, type_safe: bool = False) -> pa.Table:
        from fugue_amd.dataframe.coerce import coerce_rows

        rows = coerce_rows(self.as_array(), self.schema)
        cols = self.columns
        pylist = [{c: row[i] for i, c in enumerate(cols)} for row in rows]
        return pa.Table.from_pylist(pylist, schema=self.schema.pa_schema)

    @abstractmethod
    def as_array(
        self, columns: Optional[List[str]] = None, type_safe: bool = False
    ) -> List[Any]:
        """Rows as list of lists."""

    @abstractmethod
    def as_array_iterable(
        self, columns: Optional[List[str]] = None, type_safe: bool = False
    ) -> Iterable[Any]:
        ...

    @abstractmethod
    def _drop_cols(self, cols: List[str]) -> "DataFrame":
        ...

    @abstractmethod
    def rename(self, columns: Dict[str, str]) -> "DataFrame":
        ...

    @abstractmethod
    def alter_columns(self, columns: Any) -> "DataFrame":
        """Cast columns to new types; ``columns`` is a schema-like of a
        subset of this frame's columns."""

    @abstractmethod
    def _select_cols(self, cols: List[Any]) -> "DataFrame":
        ...

    def drop(self, columns: List[str]) -> "DataFrame":
        try:
            schema = self.schema.exclude(columns)
        except Exception as e:
            raise FugueDataFrameOperationError(str(e)) from e
        if len(schema) == 0:
            raise FugueDataFrameOperationError("can't drop all columns")
        if len(schema) + len(columns) != len(self.schema):
            raise FugueDataFrameOperationError(
                f"can't drop {columns} from {self.schema}"
            )
        return self._drop_cols(columns)

    def __getitem__(self, columns: List[Any]) -> "DataFrame":
        if not isinstance(columns, list):
            columns = [columns]
        try:
            schema = self.schema.extract(columns)
            if len(schema) == 0:
                raise FugueDataFrameOperationError("can't select no columns")
        except FugueDataFrameOperationError:
            raise
        except Exception as e:
            raise FugueDataFrameOperationError(str(e)) from e
        return self._select_cols(schema.names)

    @abstractmethod
    def head(
        self, n: int, columns: Optional[List[str]] = None
    ) -> "LocalBoundedDataFrame":
        ...

    def as_dicts(self, columns: Optional[List[str]] = None) -> List[Dict[str, Any]]:
        return list(self.as_dict_iterable(columns))

    def as_dict_iterable(
        self, columns: Optional[List[str]] = None
    ) -> Iterable[Dict[str, Any]]:
        cols = columns if columns is not None else self.columns
        for row in self.as_array_iterable(columns, type_safe=True):
            yield {n: row[i] for i, n in enumerate(cols)}

    def get_info_str(self) -> str:
        return f"{type(self).__name__}({self.schema})"

    def __copy__(self) -> "DataFrame":
        return self

    def __deepcopy__(self, memo: Any) -> "DataFrame":
        return self

    def _get_altered_schema(self, subschema: Any) -> Schema:
        sub = Schema(subschema) if subschema is not None else None
        if sub is None or len(sub) == 0:
            return self.schema
        for f in sub.fields:
            if f.name not in self.schema:
                raise FugueDataFrameOperationError(
                    f"{f.name} not in {self.schema}"
                )
        return self.schema.alter(sub)


class LocalDataFrame(DataFrame):
    @property
    def is_local(self) -> bool:
        return True

    def native_as_df(self) -> AnyDataFrame:
        return self.native

    @property
    def num_partitions(self) -> int:
        return 1

    def as_local(self) -> "LocalDataFrame":
        return self


class LocalBoundedDataFrame(LocalDataFrame):
    @property
    def is_bounded(self) -> bool:
        return True

    def as_local_bounded(self) -> "LocalBoundedDataFrame":
        return self


class LocalUnboundedDataFrame(LocalDataFrame):
    @property
    def is_bounded(self) -> bool:
        return False

    def count(self) -> int:
        raise FugueDataFrameOperationError("can't count an unbounded dataframe")

    def as_local(self) -> "LocalDataFrame":
        return self


class YieldedDataFrame:
    """Handle for a dataframe yielded from a finished workflow
    (reference: ``fugue/dataframe/dataframe.py:384``)."""

    def __init__(self, yid: str):
        self._yid = yid
        self._df: Optional[DataFrame] = None

    @property
    def is_set(self) -> bool:
        return self._df is not None

    def set_value(self, df: DataFrame) -> None:
        self._df = df

    @property
    def result(self) -> DataFrame:
        if self._df is None:
            raise RuntimeError("yielded dataframe is not set")
        return self._df

    def __uuid__(self) -> str:
        from fugue_amd.utils.hash import to_uuid

        return to_uuid(self._yid)


class DataFrameDisplay(DatasetDisplay):
    @property
    def df(self) -> DataFrame:
        return self._ds  # type: ignore

    def show(
        self, n: int = 10, with_count: bool = False, title: Optional[str] = None
    ) -> None:
        head = self.df.head(n)
        if title is not None and title != "":
            print(title)
        print(self.df.get_info_str())
        pt = PrettyTable(self.df.columns, head.as_array())
        print(pt.to_string())
        if with_count:
            print(f"Total count: {self.df.count()}")
        print("")

    def repr(self) -> str:
        return self.df.get_info_str()


def _default_display_matcher(ds: Any) -> bool:
    return isinstance(ds, DataFrame)


register_plugin(
    "get_dataset_display",
    _default_display_matcher,
    lambda ds: DataFrameDisplay(ds),
    priority=0.0,
)


def as_fugue_df(df: AnyDataFrame, **kwargs: Any) -> DataFrame:
    """Convert any supported object to a fugue DataFrame."""
    if isinstance(df, DataFrame) and len(kwargs) == 0:
        return df
    ok, res = try_run_plugin("as_fugue_df", df, **kwargs)
    if ok:
        return res
    from fugue_amd.dataframe.arrow_dataframe import ArrowDataFrame
    from fugue_amd.dataframe.array_dataframe import ArrayDataFrame
    from fugue_amd.dataframe.iterable_dataframe import IterableDataFrame
    from fugue_amd.dataframe.pandas_dataframe import PandasDataFrame

    if isinstance(df, pd.DataFrame):
        return PandasDataFrame(df, **kwargs)
    if isinstance(df, pa.Table):
        return ArrowDataFrame(df, **kwargs)
    if isinstance(df, (list, tuple)):
        return ArrayDataFrame(list(df), **kwargs)
    if isinstance(df, Iterable):
        return IterableDataFrame(iter(df), **kwargs)
    raise ValueError(f"can't convert {type(df)} to a fugue DataFrame")


def _input_schema(schema: Any) -> Schema:
    return schema if isinstance(schema, Schema) else Schema(schema)


