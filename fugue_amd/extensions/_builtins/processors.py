"""Built-in processors: the execution bodies behind workflow tasks.

Each processor unwraps its task parameters and issues one engine (or
frame) call.  The single-input operators share a declarative base,
:class:`_FrameOp`, that pulls parameters from a spec table instead of
hand-written per-class glue.  Behavioral parity target:
``fugue/extensions/_builtins/processors.py`` in the reference (the
class *names* are part of the workflow-task contract; the bodies here
are organized around the spec tables and helper functions below).
"""
from typing import Any, Callable, Dict, List, Optional, Tuple, Type

from fugue_amd.collections.partition import PartitionCursor
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
from fugue_amd.extensions.transformer.transformer import (
    CoTransformer,
    Transformer,
)
from fugue_amd.rpc import EmptyRPCHandler, to_rpc_handler
from fugue_amd.schema import Schema
from fugue_amd.utils.convert import to_type
from fugue_amd.utils.params import ParamDict

# ---------------------------------------------------------------------------
# parameter-pull mini-spec: ("req", tp) | ("opt", default) | ("none", tp)
# ---------------------------------------------------------------------------

_REQ, _OPT, _NONE = "req", "opt", "none"


class _FrameOp(Processor):
    """A processor over exactly one input frame whose parameters are
    declared in ``pulls`` and whose body is :meth:`run_op`."""

    pulls: Dict[str, Tuple[str, Any]] = {}

    def process(self, dfs: DataFrames) -> DataFrame:
        if len(dfs) != 1:
            raise FugueWorkflowError("not single input")
        kw: Dict[str, Any] = {}
        for name, (mode, arg) in self.pulls.items():
            if mode == _REQ:
                kw[name] = self.params.get_or_throw(name, arg)
            elif mode == _OPT:
                kw[name] = self.params.get(name, arg)
            else:
                kw[name] = self.params.get_or_none(name, arg)
        return self.run_op(dfs[0], **kw)

    def run_op(self, df: DataFrame, **kw: Any) -> DataFrame:
        raise NotImplementedError  # pragma: no cover


# ---------------------------------------------------------------------------
# transformer execution
# ---------------------------------------------------------------------------


def _prepare_transformer(
    proc: Any, convert: Callable[..., Any] = _to_transformer,
    with_schema: bool = True,
) -> Tuple[Any, Tuple[type, ...]]:
    """Instantiate and configure the transformer object carried in the
    task params; returns it with the tuple of ignorable exception
    types.  ``convert`` selects the conversion chain (transformer vs
    output transformer); output transformers carry no schema param."""
    args = [proc.params.get_or_none("transformer", object)]
    if with_schema:
        args.append(proc.params.get_or_none("schema", object))
    tf = convert(*args)
    tf._workflow_conf = proc.execution_engine.conf
    tf._params = ParamDict(proc.params.get("params", ParamDict()))
    tf._partition_spec = proc.partition_spec
    handler = to_rpc_handler(proc.params.get_or_throw("rpc_handler", object))
    if not isinstance(handler, EmptyRPCHandler):
        tf._rpc_client = proc.rpc_server.make_client(handler)
        tf._has_rpc_client = True
    ignorable = tuple(
        to_type(x, Exception) for x in proc.params.get("ignore_errors", [])
    )
    tf.validate_on_compile()
    return tf, ignorable


def _comap_empty_inputs(df: DataFrame) -> DataFrames:
    """Empty frames (one per zipped input, named when the zip was
    named) used to ask a cotransformer for its output schema."""
    schemas = df.metadata["schemas"]
    if df.metadata.get("serialized_has_name", False):
        return DataFrames(
            {name: ArrayDataFrame([], s) for name, s in schemas.items()}
        )
    return DataFrames([ArrayDataFrame([], s) for s in schemas.values()])


class RunTransformer(Processor):
    def process(self, dfs: DataFrames) -> DataFrame:
        df = dfs[0]
        tf, ignorable = _prepare_transformer(self)
        tf.validate_on_runtime(df)
        run = (
            self._run_single if isinstance(tf, Transformer) else self._run_co
        )
        return run(df, tf, ignorable)

    # kept as named methods (not lambdas) so subclasses can override
    def _run_single(
        self, df: DataFrame, tf: Transformer, ignorable: Tuple[type, ...]
    ) -> DataFrame:
        tf._key_schema = self.partition_spec.get_key_schema(df.schema)
        tf._output_schema = Schema(tf.get_output_schema(df))
        runner = _TransformerRunner(df, tf, list(ignorable))
        return self.execution_engine.map_engine.map_dataframe(
            df=df,
            map_func=runner.run,
            output_schema=tf.output_schema,
            partition_spec=tf.partition_spec,
            on_init=runner.on_init,
            map_func_format_hint=tf.get_format_hint(),
        )

    def _run_co(
        self, df: DataFrame, tf: CoTransformer, ignorable: Tuple[type, ...]
    ) -> DataFrame:
        if not df.metadata.get("serialized", False):
            raise FugueWorkflowError("must use serialized (zipped) dataframe")
        tf._key_schema = df.schema - _FUGUE_SERIALIZED_BLOB_SCHEMA
        tf._output_schema = Schema(tf.get_output_schema(_comap_empty_inputs(df)))
        runner = _CoTransformerRunner(df, tf, list(ignorable))
        return self.execution_engine.comap(
            df=df,
            map_func=runner.run,
            output_schema=tf.output_schema,
            partition_spec=tf.partition_spec,
            on_init=runner.on_init,
        )



# ---------------------------------------------------------------------------
# multi-input relational ops (fold over the input list)
# ---------------------------------------------------------------------------


class _FoldOp(Processor):
    """Left-fold a binary engine op across 2+ inputs (a single input
    passes through untouched)."""

    def process(self, dfs: DataFrames) -> DataFrame:
        seq = list(dfs.values())
        if len(seq) == 1:
            return seq[0]
        step = self.make_step()
        acc = seq[0]
        for nxt in seq[1:]:
            acc = step(acc, nxt)
        return acc

    def make_step(self) -> Callable[[DataFrame, DataFrame], DataFrame]:
        raise NotImplementedError  # pragma: no cover


class RunJoin(_FoldOp):
    def make_step(self) -> Callable[[DataFrame, DataFrame], DataFrame]:
        how = self.params.get_or_throw("how", str)
        on = self.params.get("on", [])
        eng = self.execution_engine
        return lambda a, b: eng.join(a, b, how=how, on=on)


class RunSetOperation(_FoldOp):
    _OPS = ("union", "subtract", "intersect")

    def make_step(self) -> Callable[[DataFrame, DataFrame], DataFrame]:
        how = self.params.get_or_throw("how", str)
        if how not in self._OPS:
            raise FugueWorkflowError(f"unknown set operation {how!r}")
        op = getattr(self.execution_engine, how)
        distinct = self.params.get("distinct", True)
        return lambda a, b: op(a, b, distinct=distinct)


# ---------------------------------------------------------------------------
# single-input ops, table-driven
# ---------------------------------------------------------------------------


class Distinct(_FrameOp):
    def run_op(self, df: DataFrame) -> DataFrame:
        return self.execution_engine.distinct(df)


class Dropna(_FrameOp):
    pulls = dict(
        how=(_OPT, "any"), thresh=(_NONE, int), subset=(_NONE, list)
    )

    def run_op(self, df: DataFrame, how: str, thresh: Optional[int],
               subset: Optional[List[str]]) -> DataFrame:
        if how not in ("any", "all"):
            raise FugueWorkflowError(
                "how' needs to be either 'any' or 'all'"
            )
        return self.execution_engine.dropna(
            df, how=how, thresh=thresh, subset=subset
        )


class Fillna(_FrameOp):
    pulls = dict(value=(_NONE, object), subset=(_NONE, list))

    def run_op(self, df: DataFrame, value: Any,
               subset: Optional[List[str]]) -> DataFrame:
        if value is None or (
            isinstance(value, dict)
            and (len(value) == 0 or any(v is None for v in value.values()))
        ):
            raise FugueWorkflowError("fillna value cannot be None")
        return self.execution_engine.fillna(df, value=value, subset=subset)


class RunSQLSelect(Processor):
    def process(self, dfs: DataFrames) -> DataFrame:
        statement = self.params.get_or_throw("statement", StructuredRawSQL)
        sql_engine = make_sql_engine(
            self.params.get_or_none("sql_engine", object),
            self.execution_engine,
            **self.params.get("sql_engine_params", ParamDict()),
        )
        return sql_engine.select(dfs, statement)


class Zip(Processor):
    def process(self, dfs: DataFrames) -> DataFrame:
        return self.execution_engine.zip_all(
            dfs,
            how=self.params.get("how", "inner"),
            partition_spec=self.partition_spec,
        )


class Select(_FrameOp):
    pulls = dict(
        columns=(_REQ, ColumnsSelect),
        where=(_NONE, ColumnExpr),
        having=(_NONE, ColumnExpr),
    )

    def run_op(self, df: DataFrame, columns: ColumnsSelect,
               where: Optional[ColumnExpr],
               having: Optional[ColumnExpr]) -> DataFrame:
        return self.execution_engine.select(
            df, cols=columns, where=where, having=having
        )


class Filter(_FrameOp):
    pulls = dict(condition=(_REQ, ColumnExpr))

    def run_op(self, df: DataFrame, condition: ColumnExpr) -> DataFrame:
        return self.execution_engine.filter(df, condition=condition)


class Assign(_FrameOp):
    pulls = dict(columns=(_REQ, list))

    def run_op(self, df: DataFrame, columns: List[ColumnExpr]) -> DataFrame:
        return self.execution_engine.assign(df, columns=columns)


class Aggregate(_FrameOp):
    pulls = dict(columns=(_REQ, list))

    def run_op(self, df: DataFrame, columns: List[ColumnExpr]) -> DataFrame:
        return self.execution_engine.aggregate(
            df, partition_spec=self.partition_spec, agg_cols=columns
        )


class Rename(_FrameOp):
    pulls = dict(columns=(_REQ, dict))

    def run_op(self, df: DataFrame, columns: Dict[str, str]) -> DataFrame:
        return df.rename(columns)


class AlterColumns(_FrameOp):
    pulls = dict(columns=(_REQ, object))

    def run_op(self, df: DataFrame, columns: Any) -> DataFrame:
        return df.alter_columns(columns)


class DropColumns(_FrameOp):
    pulls = dict(columns=(_REQ, list), if_exists=(_OPT, False))

    def run_op(self, df: DataFrame, columns: List[str],
               if_exists: bool) -> DataFrame:
        if if_exists:
            columns = [c for c in columns if c in df.schema]
        return df.drop(columns) if len(columns) > 0 else df


class SelectColumns(_FrameOp):
    pulls = dict(columns=(_REQ, list))

    def run_op(self, df: DataFrame, columns: List[str]) -> DataFrame:
        return df[columns]


class Sample(_FrameOp):
    pulls = dict(
        n=(_NONE, int), frac=(_NONE, float),
        replace=(_OPT, False), seed=(_NONE, int),
    )

    def run_op(self, df: DataFrame, n: Optional[int], frac: Optional[float],
               replace: bool, seed: Optional[int]) -> DataFrame:
        return self.execution_engine.sample(
            df, n=n, frac=frac, replace=replace, seed=seed
        )


class Take(_FrameOp):
    pulls = dict(
        n=(_NONE, int), presort=(_OPT, ""), na_position=(_OPT, "last")
    )

    def run_op(self, df: DataFrame, n: Optional[int], presort: str,
               na_position: str) -> DataFrame:
        return self.execution_engine.take(
            df,
            n=n,
            presort=presort,
            na_position=na_position,
            partition_spec=self.partition_spec,
        )


class SaveAndUse(_FrameOp):
    pulls = dict(
        path=(_REQ, str), fmt=(_OPT, ""), mode=(_OPT, "overwrite"),
        single=(_OPT, False), params=(_OPT, dict()),
    )

    def run_op(self, df: DataFrame, path: str, fmt: str, mode: str,
               single: bool, params: Dict[str, Any]) -> DataFrame:
        eng = self.execution_engine
        eng.save_df(
            df=df,
            path=path,
            format_hint=fmt,
            mode=mode,
            partition_spec=self.partition_spec,
            force_single=single,
            **params,
        )
        return eng.load_df(path=path, format_hint=fmt)


# ---------------------------------------------------------------------------
# worker-side runners (execute inside map_dataframe / comap partitions)
# ---------------------------------------------------------------------------


class _RunnerBase:
    """Shared worker-side wrapper: position the cursor, run the user
    transform, and convert listed exception types to an empty result."""

    def __init__(self, df: DataFrame, transformer: Any,
                 ignore_errors: List[Type[Exception]]):
        self.schema = df.schema
        self.transformer = transformer
        self.ignore_errors = tuple(ignore_errors)

    def _invoke(self, cursor: PartitionCursor, arg: Any) -> LocalDataFrame:
        self.transformer._cursor = cursor
        if not self.ignore_errors:
            return self.transformer.transform(arg)
        try:
            # materialize now so lazy errors surface inside the guard
            return self.transformer.transform(arg).as_local_bounded()
        except self.ignore_errors:
            return ArrayDataFrame([], self.transformer.output_schema)

    def on_init(self, partition_no: int, arg: Any) -> None:
        spec = self.transformer.partition_spec
        self.transformer._cursor = spec.get_cursor(self.schema, partition_no)
        self.transformer.on_init(arg)


class _TransformerRunner(_RunnerBase):
    def run(self, cursor: PartitionCursor,
            df: LocalDataFrame) -> LocalDataFrame:
        return self._invoke(cursor, df)


class _CoTransformerRunner(_RunnerBase):
    def run(self, cursor: PartitionCursor,
            dfs: DataFrames) -> LocalDataFrame:
        return self._invoke(cursor, dfs)
