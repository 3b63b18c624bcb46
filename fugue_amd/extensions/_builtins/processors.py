"""Built-in processors: the workflow tasks' execution bodies.

Reference parity: ``fugue/extensions/_builtins/processors.py``.
"""
from typing import Any, List, Optional, Type

from fugue_amd.collections.partition import PartitionCursor, PartitionSpec
from fugue_amd.collections.sql import StructuredRawSQL
from fugue_amd.column.expressions import ColumnExpr
from fugue_amd.column.sql import SelectColumns as ColumnsSelect
from fugue_amd.dataframe.array_dataframe import ArrayDataFrame
from fugue_amd.dataframe.dataframe import DataFrame, LocalDataFrame
from fugue_amd.dataframe.dataframes import DataFrames
from fugue_amd.exceptions import FugueWorkflowError
from fugue_amd.execution.execution_engine import (
    _FUGUE_SERIALIZED_BLOB_SCHEMA,
)
from fugue_amd.execution.factory import make_sql_engine
from fugue_amd.extensions.processor.processor import Processor
from fugue_amd.extensions.transformer.convert import _to_transformer
from fugue_amd.extensions.transformer.transformer import CoTransformer, Transformer
from fugue_amd.rpc import EmptyRPCHandler, to_rpc_handler
from fugue_amd.schema import Schema
from fugue_amd.utils.convert import to_type
from fugue_amd.utils.params import ParamDict


class RunTransformer(Processor):
    def process(self, dfs: DataFrames) -> DataFrame:
        df = dfs[0]
        tf = _to_transformer(
            self.params.get_or_none("transformer", object),
            self.params.get_or_none("schema", object),
        )
        tf._workflow_conf = self.execution_engine.conf
        tf._params = ParamDict(self.params.get("params", ParamDict()))
        tf._partition_spec = self.partition_spec
        rpc_handler = to_rpc_handler(self.params.get_or_throw("rpc_handler", object))
        if not isinstance(rpc_handler, EmptyRPCHandler):
            tf._rpc_client = self.rpc_server.make_client(rpc_handler)
            tf._has_rpc_client = True
        ie = self.params.get("ignore_errors", [])
        self._ignore_errors = [to_type(x, Exception) for x in ie]
        tf.validate_on_compile()
        tf.validate_on_runtime(df)
        if isinstance(tf, Transformer):
            return self.transform(df, tf)
        return self.cotransform(df, tf)

    def transform(self, df: DataFrame, tf: Transformer) -> DataFrame:
        tf._key_schema = self.partition_spec.get_key_schema(df.schema)
        tf._output_schema = Schema(tf.get_output_schema(df))
        tr = _TransformerRunner(df, tf, self._ignore_errors)
        return self.execution_engine.map_engine.map_dataframe(
            df=df,
            map_func=tr.run,
            output_schema=tf.output_schema,
            partition_spec=tf.partition_spec,
            on_init=tr.on_init,
            map_func_format_hint=tf.get_format_hint(),
        )

    def cotransform(self, df: DataFrame, tf: CoTransformer) -> DataFrame:
        if not df.metadata.get("serialized", False):
            raise FugueWorkflowError("must use serialized (zipped) dataframe")
        tf._key_schema = df.schema - _FUGUE_SERIALIZED_BLOB_SCHEMA
        schemas = df.metadata["schemas"]
        named = df.metadata.get("serialized_has_name", False)
        empty_dfs = (
            DataFrames({k: ArrayDataFrame([], v) for k, v in schemas.items()})
            if named
            else DataFrames([ArrayDataFrame([], v) for v in schemas.values()])
        )
        tf._output_schema = Schema(tf.get_output_schema(empty_dfs))
        tr = _CoTransformerRunner(df, tf, self._ignore_errors)
        return self.execution_engine.comap(
            df=df,
            map_func=tr.run,
            output_schema=tf.output_schema,
            partition_spec=tf.partition_spec,
            on_init=tr.on_init,
        )


class RunJoin(Processor):
    def process(self, dfs: DataFrames) -> DataFrame:
        if len(dfs) == 1:
            return dfs[0]
        how = self.params.get_or_throw("how", str)
        on = self.params.get("on", [])
        df = dfs[0]
        for i in range(1, len(dfs)):
            df = self.execution_engine.join(df, dfs[i], how=how, on=on)
        return df


class RunSetOperation(Processor):
    def process(self, dfs: DataFrames) -> DataFrame:
        if len(dfs) == 1:
            return dfs[0]
        how = self.params.get_or_throw("how", str)
        func: Any = {
            "union": self.execution_engine.union,
            "subtract": self.execution_engine.subtract,
            "intersect": self.execution_engine.intersect,
        }[how]
        distinct = self.params.get("distinct", True)
        df = dfs[0]
        for i in range(1, len(dfs)):
            df = func(df, dfs[i], distinct=distinct)
        return df


class Distinct(Processor):
    def process(self, dfs: DataFrames) -> DataFrame:
        if len(dfs) != 1:
            raise FugueWorkflowError("not single input")
        return self.execution_engine.distinct(dfs[0])


class Dropna(Processor):
    def process(self, dfs: DataFrames) -> DataFrame:
        if len(dfs) != 1:
            raise FugueWorkflowError("not single input")
        how = self.params.get("how", "any")
        if how not in ("any", "all"):
            raise FugueWorkflowError("how' needs to be either 'any' or 'all'")
        thresh = self.params.get_or_none("thresh", int)
        subset = self.params.get_or_none("subset", list)
        return self.execution_engine.dropna(
            dfs[0], how=how, thresh=thresh, subset=subset
        )


class Fillna(Processor):
    def process(self, dfs: DataFrames) -> DataFrame:
        if len(dfs) != 1:
            raise FugueWorkflowError("not single input")
        value = self.params.get_or_none("value", object)
        if value is None:
            raise FugueWorkflowError("fillna value cannot be None")
        subset = self.params.get_or_none("subset", list)
        return self.execution_engine.fillna(dfs[0], value=value, subset=subset)


class RunSQLSelect(Processor):
    def process(self, dfs: DataFrames) -> DataFrame:
        statement = self.params.get_or_throw("statement", StructuredRawSQL)
        engine = self.params.get_or_none("sql_engine", object)
        engine_params = self.params.get("sql_engine_params", ParamDict())
        sql_engine = make_sql_engine(
            engine, self.execution_engine, **engine_params
        )
        return sql_engine.select(dfs, statement)


class Zip(Processor):
    def process(self, dfs: DataFrames) -> DataFrame:
        how = self.params.get("how", "inner")
        partition_spec = self.partition_spec
        return self.execution_engine.zip_all(
            dfs, how=how, partition_spec=partition_spec
        )


class Select(Processor):
    def process(self, dfs: DataFrames) -> DataFrame:
        if len(dfs) != 1:
            raise FugueWorkflowError("not single input")
        columns = self.params.get_or_throw("columns", ColumnsSelect)
        where = self.params.get_or_none("where", ColumnExpr)
        having = self.params.get_or_none("having", ColumnExpr)
        return self.execution_engine.select(
            dfs[0], cols=columns, where=where, having=having
        )


class Filter(Processor):
    def process(self, dfs: DataFrames) -> DataFrame:
        if len(dfs) != 1:
            raise FugueWorkflowError("not single input")
        condition = self.params.get_or_throw("condition", ColumnExpr)
        return self.execution_engine.filter(dfs[0], condition=condition)


class Assign(Processor):
    def process(self, dfs: DataFrames) -> DataFrame:
        if len(dfs) != 1:
            raise FugueWorkflowError("not single input")
        columns = self.params.get_or_throw("columns", list)
        return self.execution_engine.assign(dfs[0], columns=columns)


class Aggregate(Processor):
    def process(self, dfs: DataFrames) -> DataFrame:
        if len(dfs) != 1:
            raise FugueWorkflowError("not single input")
        columns = self.params.get_or_throw("columns", list)
        return self.execution_engine.aggregate(
            dfs[0], partition_spec=self.partition_spec, agg_cols=columns
        )


class Rename(Processor):
    def process(self, dfs: DataFrames) -> DataFrame:
        if len(dfs) != 1:
            raise FugueWorkflowError("not single input")
        columns = self.params.get_or_throw("columns", dict)
        return dfs[0].rename(columns)


class AlterColumns(Processor):
    def process(self, dfs: DataFrames) -> DataFrame:
        if len(dfs) != 1:
            raise FugueWorkflowError("not single input")
        columns = self.params.get_or_throw("columns", object)
        return dfs[0].alter_columns(columns)


class DropColumns(Processor):
    def process(self, dfs: DataFrames) -> DataFrame:
        if len(dfs) != 1:
            raise FugueWorkflowError("not single input")
        if_exists = self.params.get("if_exists", False)
        columns = self.params.get_or_throw("columns", list)
        if if_exists:
            columns = [c for c in columns if c in dfs[0].schema]
        if len(columns) == 0:
            return dfs[0]
        return dfs[0].drop(columns)


class SelectColumns(Processor):
    def process(self, dfs: DataFrames) -> DataFrame:
        if len(dfs) != 1:
            raise FugueWorkflowError("not single input")
        columns = self.params.get_or_throw("columns", list)
        return dfs[0][columns]


class Sample(Processor):
    def process(self, dfs: DataFrames) -> DataFrame:
        if len(dfs) != 1:
            raise FugueWorkflowError("not single input")
        n = self.params.get_or_none("n", int)
        frac = self.params.get_or_none("frac", float)
        replace = self.params.get("replace", False)
        seed = self.params.get_or_none("seed", int)
        return self.execution_engine.sample(
            dfs[0], n=n, frac=frac, replace=replace, seed=seed
        )


class Take(Processor):
    def process(self, dfs: DataFrames) -> DataFrame:
        if len(dfs) != 1:
            raise FugueWorkflowError("not single input")
        n = self.params.get_or_none("n", int)
        presort = self.params.get("presort", "")
        na_position = self.params.get("na_position", "last")
        partition_spec = self.partition_spec
        return self.execution_engine.take(
            dfs[0],
            n=n,
            presort=presort,
            na_position=na_position,
            partition_spec=partition_spec,
        )


class SaveAndUse(Processor):
    def process(self, dfs: DataFrames) -> DataFrame:
        if len(dfs) != 1:
            raise FugueWorkflowError("not single input")
        kwargs = self.params.get("params", dict())
        path = self.params.get_or_throw("path", str)
        format_hint = self.params.get("fmt", "")
        mode = self.params.get("mode", "overwrite")
        partition_spec = self.partition_spec
        force_single = self.params.get("single", False)
        self.execution_engine.save_df(
            df=dfs[0],
            path=path,
            format_hint=format_hint,
            mode=mode,
            partition_spec=partition_spec,
            force_single=force_single,
            **kwargs,
        )
        return self.execution_engine.load_df(path=path, format_hint=format_hint)


class _TransformerRunner:
    def __init__(
        self, df: DataFrame, transformer: Transformer, ignore_errors: List[type]
    ):
        self.schema = df.schema
        self.transformer = transformer
        self.ignore_errors = tuple(ignore_errors)

    def run(self, cursor: PartitionCursor, df: LocalDataFrame) -> LocalDataFrame:
        self.transformer._cursor = cursor
        if len(self.ignore_errors) == 0:
            return self.transformer.transform(df)
        try:
            return self.transformer.transform(df).as_local_bounded()
        except self.ignore_errors:
            return ArrayDataFrame([], self.transformer.output_schema)

    def on_init(self, partition_no: int, df: DataFrame) -> None:
        s = self.transformer.partition_spec
        self.transformer._cursor = s.get_cursor(self.schema, partition_no)
        self.transformer.on_init(df)


class _CoTransformerRunner:
    def __init__(
        self,
        df: DataFrame,
        transformer: CoTransformer,
        ignore_errors: List[Type[Exception]],
    ):
        self.schema = df.schema
        self.transformer = transformer
        self.ignore_errors = tuple(ignore_errors)

    def run(self, cursor: PartitionCursor, dfs: DataFrames) -> LocalDataFrame:
        self.transformer._cursor = cursor
        if len(self.ignore_errors) == 0:
            return self.transformer.transform(dfs)
        try:
            return self.transformer.transform(dfs).as_local_bounded()
        except self.ignore_errors:
            return ArrayDataFrame([], self.transformer.output_schema)

    def on_init(self, partition_no: int, dfs: DataFrames) -> None:
        s = self.transformer.partition_spec
        self.transformer._cursor = s.get_cursor(self.schema, partition_no)
        self.transformer.on_init(dfs)
