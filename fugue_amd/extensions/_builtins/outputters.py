"""Built-in outputters (reference: ``fugue/extensions/_builtins/outputters.py``)."""
from typing import Any, List, Type

from fugue_amd.collections.partition import PartitionCursor
from fugue_amd.dataframe.dataframe import DataFrame, LocalDataFrame
from fugue_amd.dataframe.dataframes import DataFrames
from fugue_amd.dataframe.utils import _df_eq
from fugue_amd.exceptions import FugueWorkflowError
from fugue_amd.extensions.outputter.outputter import Outputter
from fugue_amd.extensions.transformer.convert import _to_output_transformer
from fugue_amd.extensions.transformer.transformer import CoTransformer, Transformer
from fugue_amd.rpc import EmptyRPCHandler, to_rpc_handler
from fugue_amd.utils.convert import to_type
from fugue_amd.utils.params import ParamDict


class Show(Outputter):
    def process(self, dfs: DataFrames) -> None:
        n = self.params.get("n", 10)
        with_count = self.params.get("with_count", False)
        title = self.params.get_or_none("title", str)
        for df in dfs.values():
            df.show(n=n, with_count=with_count, title=title)


class AssertEqual(Outputter):
    def process(self, dfs: DataFrames) -> None:
        if len(dfs) < 2:
            raise FugueWorkflowError("at least two dataframes needed")
        expected = dfs[0]
        for i in range(1, len(dfs)):
            _df_eq(expected, dfs[i], throw=True, **self.params)


class AssertNotEqual(Outputter):
    def process(self, dfs: DataFrames) -> None:
        if len(dfs) < 2:
            raise FugueWorkflowError("at least two dataframes needed")
        expected = dfs[0]
        for i in range(1, len(dfs)):
            if _df_eq(expected, dfs[i], throw=False, **self.params):
                raise AssertionError(f"dataframe {i} equals to the first one")


class Save(Outputter):
    def process(self, dfs: DataFrames) -> None:
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


class RunOutputTransformer(Outputter):
    def process(self, dfs: DataFrames) -> None:
        df = dfs[0]
        tf = _to_output_transformer(
            self.params.get_or_none("transformer", object),
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
            self.transform(df, tf)
        else:
            self.cotransform(df, tf)

    def transform(self, df: DataFrame, tf: Transformer) -> None:
        from fugue_amd.extensions._builtins.processors import _TransformerRunner
        from fugue_amd.schema import Schema

        tf._key_schema = self.partition_spec.get_key_schema(df.schema)
        tf._output_schema = Schema(tf.get_output_schema(df))
        tr = _TransformerRunner(df, tf, self._ignore_errors)
        self.execution_engine.persist(
            self.execution_engine.map_engine.map_dataframe(
                df=df,
                map_func=tr.run,
                output_schema=tf.output_schema,
                partition_spec=tf.partition_spec,
                on_init=tr.on_init,
            ),
            lazy=False,
        )

    def cotransform(self, df: DataFrame, tf: CoTransformer) -> None:
        from fugue_amd.dataframe.array_dataframe import ArrayDataFrame
        from fugue_amd.execution.execution_engine import (
            _FUGUE_SERIALIZED_BLOB_SCHEMA,
        )
        from fugue_amd.extensions._builtins.processors import _CoTransformerRunner
        from fugue_amd.schema import Schema

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
        self.execution_engine.persist(
            self.execution_engine.comap(
                df=df,
                map_func=tr.run,
                output_schema=tf.output_schema,
                partition_spec=tf.partition_spec,
                on_init=tr.on_init,
            ),
            lazy=False,
        )
