"""Built-in outputters (reference: ``fugue/extensions/_builtins/outputters.py``)."""
from fugue_amd.dataframe.dataframes import DataFrames
from fugue_amd.dataframe.utils import _df_eq
from fugue_amd.exceptions import FugueWorkflowError
from fugue_amd.extensions.outputter.outputter import Outputter
from fugue_amd.extensions.transformer.convert import _to_output_transformer
from fugue_amd.extensions.transformer.transformer import Transformer


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
    """Runs an output (co)transformer for side effects: same execution
    shape as ``RunTransformer`` but the mapped result is forced to
    materialize via an eager persist and then discarded."""

    def process(self, dfs: DataFrames) -> None:
        from fugue_amd.extensions._builtins.processors import (
            _comap_empty_inputs,
            _CoTransformerRunner,
            _prepare_transformer,
            _TransformerRunner,
        )
        from fugue_amd.execution.execution_engine import (
            _FUGUE_SERIALIZED_BLOB_SCHEMA,
        )
        from fugue_amd.schema import Schema

        df = dfs[0]
        tf, ignorable = _prepare_transformer(
            self, convert=_to_output_transformer, with_schema=False
        )
        tf.validate_on_runtime(df)
        eng = self.execution_engine
        if isinstance(tf, Transformer):
            tf._key_schema = self.partition_spec.get_key_schema(df.schema)
            tf._output_schema = Schema(tf.get_output_schema(df))
            runner = _TransformerRunner(df, tf, list(ignorable))
            mapped = eng.map_engine.map_dataframe(
                df=df,
                map_func=runner.run,
                output_schema=tf.output_schema,
                partition_spec=tf.partition_spec,
                on_init=runner.on_init,
            )
        else:
            if not df.metadata.get("serialized", False):
                raise FugueWorkflowError(
                    "must use serialized (zipped) dataframe"
                )
            tf._key_schema = df.schema - _FUGUE_SERIALIZED_BLOB_SCHEMA
            tf._output_schema = Schema(
                tf.get_output_schema(_comap_empty_inputs(df))
            )
            co_runner = _CoTransformerRunner(df, tf, list(ignorable))
            mapped = eng.comap(
                df=df,
                map_func=co_runner.run,
                output_schema=tf.output_schema,
                partition_spec=tf.partition_spec,
                on_init=co_runner.on_init,
            )
        eng.persist(mapped, lazy=False)
