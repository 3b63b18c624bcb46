"""NativeExecutionEngine: the local, single-process pandas engine.

Reference parity: ``fugue/execution/native_execution_engine.py`` — the
"plumbing, no GPU" path (BASELINE config #1) kept so
``fa.engine_context(None)`` works without a GPU.  SQL statements run on the
built-in SQL executor (``fugue_amd/sql/executor.py``) instead of qpd.
"""
import logging
from typing import Any, Callable, List, Optional, Union

import numpy as np
import pandas as pd

from fugue_amd.collections.partition import (
    PartitionCursor,
    PartitionSpec,
)
from fugue_amd.collections.sql import StructuredRawSQL
from fugue_amd.constants import KEYWORD_CORECOUNT, KEYWORD_ROWCOUNT
from fugue_amd.dataframe.array_dataframe import ArrayDataFrame
from fugue_amd.dataframe.dataframe import AnyDataFrame, DataFrame, LocalDataFrame
from fugue_amd.dataframe.dataframe_iterable_dataframe import (
    LocalDataFrameIterableDataFrame,
)
from fugue_amd.dataframe.dataframes import DataFrames
from fugue_amd.dataframe.pandas_dataframe import PandasDataFrame
from fugue_amd.dataframe.utils import get_join_schemas, parse_join_type
from fugue_amd.exceptions import FugueDataFrameInitError
from fugue_amd.execution.execution_engine import (
    ExecutionEngine,
    MapEngine,
    SQLEngine,
)
from fugue_amd.schema import Schema
from fugue_amd.utils import io as _io
from fugue_amd.utils.pandas_like import (
    cast_pandas,
    drop_duplicates,
    pandas_except,
    pandas_intersect,
    pandas_join,
    pandas_union,
    safe_groupby_apply,
)
from fugue_amd.utils.params import ParamDict


class PandasSQLEngine(SQLEngine):
    """SQL over pandas frames via the built-in SQL executor
    (replaces the reference's qpd dependency,
    ``fugue/execution/native_execution_engine.py:42``)."""

    @property
    def dialect(self) -> Optional[str]:
        return "spark"

    @property
    def is_distributed(self) -> bool:
        return False

    def select(self, dfs: DataFrames, statement: StructuredRawSQL) -> DataFrame:
        from fugue_amd.sql.executor import run_sql_on_pandas

        sql = statement.construct(log=self.log)
        pdfs = {k: v.as_pandas() for k, v in dfs.items()}
        schemas = {k: v.schema for k, v in dfs.items()}
        res, schema = run_sql_on_pandas(sql, pdfs, schemas)
        if schema is not None:
            return PandasDataFrame(res, schema)
        return PandasDataFrame(res)


class PandasMapEngine(MapEngine):
    @property
    def is_distributed(self) -> bool:
        return False

    @property
    def execution_engine_constraint(self) -> type:
        return NativeExecutionEngine

    def map_dataframe(
        self,
        df: DataFrame,
        map_func: Callable[[PartitionCursor, LocalDataFrame], LocalDataFrame],
        output_schema: Any,
        partition_spec: PartitionSpec,
        on_init: Optional[Callable[[int, DataFrame], Any]] = None,
        map_func_format_hint: Optional[str] = None,
    ) -> DataFrame:
        is_coarse = partition_spec.algo == "coarse"
        presort = partition_spec.get_sorts(df.schema, with_partition_keys=is_coarse)
        presort_keys = list(presort.keys())
        presort_asc = list(presort.values())
        output_schema = Schema(output_schema)
        cursor = partition_spec.get_cursor(df.schema, 0)
        if on_init is not None:
            on_init(0, df)
        if len(partition_spec.partition_by) == 0 or is_coarse:
            if len(presort_keys) > 0:
                pdf = (
                    df.as_pandas()
                    .sort_values(presort_keys, ascending=presort_asc)
                    .reset_index(drop=True)
                )
                input_df: LocalDataFrame = PandasDataFrame(
                    pdf, df.schema, pandas_df_wrapper=True
                )
            else:
                input_df = df.as_local()
                if isinstance(input_df, PandasDataFrame):
                    # shallow copy: a UDF assigning columns in place must
                    # not mutate the (possibly persisted) source frame
                    input_df = PandasDataFrame(
                        input_df.as_pandas().copy(deep=False),
                        df.schema,
                        pandas_df_wrapper=True,
                    )
            if (
                len(partition_spec.partition_by) == 0
                and partition_spec.num_partitions != "0"
            ):
                partitions = max(
                    1,
                    partition_spec.get_num_partitions(
                        **{
                            KEYWORD_ROWCOUNT: lambda: df.count(),
                            KEYWORD_CORECOUNT: lambda: 1,
                        }
                    ),
                )
                results: List[pd.DataFrame] = []
                _pdf_all = input_df.as_pandas()
                _bounds = [
                    (len(_pdf_all) * i) // partitions
                    for i in range(partitions + 1)
                ]
                for p, subdf in enumerate(
                    _pdf_all.iloc[_bounds[i] : _bounds[i + 1]]
                    for i in range(partitions)
                ):
                    if len(subdf) > 0:
                        sub = subdf.reset_index(drop=True)
                        tdf = PandasDataFrame(sub, df.schema, pandas_df_wrapper=True)
                        cursor.set(lambda: tdf.peek_array(), p, 0)
                        results.append(map_func(cursor, tdf).as_pandas())
                if len(results) == 0:
                    return PandasDataFrame(None, output_schema)
                out = pd.concat(results, ignore_index=True)
                return PandasDataFrame(out, output_schema)
            cursor.set(lambda: input_df.peek_array(), 0, 0)
            output_df = map_func(cursor, input_df)
            if output_df.schema != output_schema:
                raise ValueError(
                    f"map output {output_df.schema} mismatches {output_schema}"
                )
            return self.to_df(output_df)

        def _map(pdf: pd.DataFrame) -> pd.DataFrame:
            if len(presort_keys) > 0:
                pdf = pdf.sort_values(presort_keys, ascending=presort_asc)
            pdf = pdf.reset_index(drop=True)
            input_df = PandasDataFrame(pdf, df.schema, pandas_df_wrapper=True)
            if input_df.empty:
                return output_schema.create_empty_pandas()
            cursor.set(lambda: input_df.peek_array(), cursor.partition_no + 1, 0)
            return map_func(cursor, input_df).as_pandas()

        result = safe_groupby_apply(
            df.as_pandas(), partition_spec.partition_by, _map
        )
        return PandasDataFrame(result, output_schema)


class NativeExecutionEngine(ExecutionEngine):
    """Local single-threaded pandas engine.

    Reference parity: ``fugue/execution/native_execution_engine.py:172``
    (``repartition`` is a no-op, joins via pandas merge).
    """

    def __init__(self, conf: Any = None):
        super().__init__(conf)
        self._log = logging.getLogger("fugue_amd.native")

    @property
    def log(self) -> logging.Logger:
        return self._log

    @property
    def is_distributed(self) -> bool:
        return False

    def create_default_map_engine(self) -> MapEngine:
        return PandasMapEngine(self)

    def create_default_sql_engine(self) -> SQLEngine:
        return PandasSQLEngine(self)

    def get_current_parallelism(self) -> int:
        return 1

    def to_df(self, df: AnyDataFrame, schema: Any = None) -> LocalDataFrame:
        if isinstance(df, DataFrame):
            if schema is not None and df.schema != schema:
                raise FugueDataFrameInitError(
                    f"schema {schema} doesn't match {df.schema}"
                )
            return df.as_local()
        if isinstance(df, pd.DataFrame):
            return PandasDataFrame(df, schema)
        import pyarrow as pa

        if isinstance(df, pa.Table):
            from fugue_amd.dataframe.arrow_dataframe import ArrowDataFrame

            return ArrowDataFrame(df, schema)
        if isinstance(df, (list, tuple)) or hasattr(df, "__iter__"):
            if schema is None:
                raise FugueDataFrameInitError("schema is required for raw data")
            return ArrayDataFrame(list(df), schema)
        raise FugueDataFrameInitError(f"can't convert {type(df)} to DataFrame")

    def repartition(self, df: DataFrame, partition_spec: PartitionSpec) -> DataFrame:
        self.log.debug("%s doesn't respect repartition", self)
        return df

    def broadcast(self, df: DataFrame) -> DataFrame:
        return self.to_df(df)

    def persist(self, df: DataFrame, lazy: bool = False, **kwargs: Any) -> DataFrame:
        res = self.to_df(df).as_local_bounded()
        if df.has_metadata:
            res.reset_metadata(df.metadata)
        return res

    def join(
        self,
        df1: DataFrame,
        df2: DataFrame,
        how: str,
        on: Optional[List[str]] = None,
    ) -> DataFrame:
        how = parse_join_type(how)
        key_schema, output_schema = get_join_schemas(df1, df2, how=how, on=on)
        d = pandas_join(
            df1.as_pandas(), df2.as_pandas(), how=how, on=key_schema.names
        )
        return PandasDataFrame(d[output_schema.names], output_schema)

    def union(self, df1: DataFrame, df2: DataFrame, distinct: bool = True) -> DataFrame:
        self._assert_same_schema(df1, df2)
        d = pandas_union(df1.as_pandas(), df2.as_pandas(), unique=distinct)
        return PandasDataFrame(d, df1.schema)

    def subtract(
        self, df1: DataFrame, df2: DataFrame, distinct: bool = True
    ) -> DataFrame:
        self._assert_same_schema(df1, df2)
        d = pandas_except(df1.as_pandas(), df2.as_pandas(), unique=distinct)
        return PandasDataFrame(d, df1.schema)

    def intersect(
        self, df1: DataFrame, df2: DataFrame, distinct: bool = True
    ) -> DataFrame:
        self._assert_same_schema(df1, df2)
        d = pandas_intersect(df1.as_pandas(), df2.as_pandas(), unique=distinct)
        return PandasDataFrame(d, df1.schema)

    def distinct(self, df: DataFrame) -> DataFrame:
        d = drop_duplicates(df.as_pandas())
        return PandasDataFrame(d, df.schema)

    def dropna(
        self,
        df: DataFrame,
        how: str = "any",
        thresh: Optional[int] = None,
        subset: Optional[List[str]] = None,
    ) -> DataFrame:
        kw: dict = dict(axis=0, subset=subset)
        if thresh is not None:
            kw["thresh"] = thresh
        else:
            kw["how"] = how
        d = df.as_pandas().dropna(**kw).reset_index(drop=True)
        return PandasDataFrame(d, df.schema, pandas_df_wrapper=True)

    def fillna(
        self, df: DataFrame, value: Any, subset: Optional[List[str]] = None
    ) -> DataFrame:
        if isinstance(value, dict):
            if any(v is None for v in value.values()) or len(value) == 0:
                raise ValueError("fillna value can't be None or empty")
            mapping = value
        else:
            if value is None:
                raise ValueError("fillna value can't be None")
            subset = subset or df.columns
            mapping = {c: value for c in subset}
        d = df.as_pandas().fillna(mapping)
        return PandasDataFrame(d, df.schema)

    def sample(
        self,
        df: DataFrame,
        n: Optional[int] = None,
        frac: Optional[float] = None,
        replace: bool = False,
        seed: Optional[int] = None,
    ) -> DataFrame:
        if (n is None) == (frac is None):
            raise ValueError("one and only one of n and frac should be set")
        d = (
            df.as_pandas()
            .sample(n=n, frac=frac, replace=replace, random_state=seed)
            .reset_index(drop=True)
        )
        return PandasDataFrame(d, df.schema, pandas_df_wrapper=True)

    def take(
        self,
        df: DataFrame,
        n: int,
        presort: str,
        na_position: str = "last",
        partition_spec: Optional[PartitionSpec] = None,
    ) -> DataFrame:
        if not isinstance(n, int):
            raise ValueError("n needs to be an integer")
        partition_spec = partition_spec or PartitionSpec()
        from fugue_amd.collections.partition import parse_presort_exp

        d = df.as_pandas()
        _presort = (
            parse_presort_exp(presort)
            if presort is not None and presort != ""
            else partition_spec.presort
        )
        if len(_presort) > 0:
            d = d.sort_values(
                list(_presort.keys()),
                ascending=list(_presort.values()),
                na_position=na_position,
            )
        if len(partition_spec.partition_by) == 0:
            d = d.head(n)
        else:
            d = d.groupby(
                partition_spec.partition_by, dropna=False, sort=False
            ).head(n)
        return PandasDataFrame(
            d.reset_index(drop=True), df.schema, pandas_df_wrapper=True
        )

    def load_df(
        self,
        path: Union[str, List[str]],
        format_hint: Any = None,
        columns: Any = None,
        **kwargs: Any,
    ) -> DataFrame:
        pdf, schema = _io.load_df(
            path, format_hint=format_hint, columns=columns, **kwargs
        )
        if schema is not None:
            return PandasDataFrame(pdf, schema)
        return PandasDataFrame(pdf)

    def save_df(
        self,
        df: DataFrame,
        path: str,
        format_hint: Any = None,
        mode: str = "overwrite",
        partition_spec: Optional[PartitionSpec] = None,
        force_single: bool = False,
        **kwargs: Any,
    ) -> None:
        keys = list(partition_spec.partition_by) if partition_spec else []
        if len(keys) > 0 and not force_single:
            _io.save_df_partitioned(
                df.as_pandas(), df.schema, path, keys,
                format_hint=format_hint, mode=mode, **kwargs
            )
            return
        _io.save_df(
            df.as_pandas(), df.schema, path, format_hint=format_hint, mode=mode, **kwargs
        )

    def _assert_same_schema(self, df1: DataFrame, df2: DataFrame) -> None:
        if df1.schema != df2.schema:
            raise ValueError(
                f"schema mismatch: {df1.schema} vs {df2.schema}"
            )
