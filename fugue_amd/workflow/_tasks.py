"""Workflow task nodes: Create / Process / Output.

Reference parity: ``fugue/workflow/_tasks.py`` — spec-UUID determinism,
checkpoint → broadcast → yield handling on ``set_result``.  The task
runner itself is ``fugue_amd/workflow/_runner.py`` (replaces adagio).
"""
import sys
from typing import Any, Callable, Dict, List, Optional

from fugue_amd.collections.partition import PartitionSpec
from fugue_amd.collections.yielded import PhysicalYielded, Yielded
from fugue_amd.dataframe.dataframe import DataFrame, YieldedDataFrame
from fugue_amd.dataframe.dataframes import DataFrames
from fugue_amd.exceptions import (
    FugueWorkflowCompileError,
    FugueWorkflowError,
)
from fugue_amd.extensions.creator.creator import Creator
from fugue_amd.extensions.outputter.outputter import Outputter
from fugue_amd.extensions.processor.processor import Processor
from fugue_amd.schema import Schema
from fugue_amd.utils.exception import frames_to_traceback, make_prune_predicate, modify_traceback
from fugue_amd.utils.hash import to_uuid
from fugue_amd.utils.params import ParamDict
from fugue_amd.workflow._checkpoint import Checkpoint


class FugueTask:
    def __init__(
        self,
        extension: Any,
        inputs: List["FugueTask"],
        params: Any = None,
        partition_spec: Optional[PartitionSpec] = None,
        input_names: Optional[List[Optional[str]]] = None,
    ):
        self._extension = extension
        self._inputs = inputs
        self._input_names = input_names
        self._params = ParamDict(params)
        self._partition_spec = partition_spec or PartitionSpec()
        self._checkpoint = Checkpoint()
        self._broadcast = False
        self._yields: List[Any] = []
        self._result: Optional[DataFrame] = None
        self._executed = False
        self._traceback = None
        self.name = ""
        self._uuid: Optional[str] = None

    def reset_traceback(self, limit: int, should_prune: Optional[Callable] = None) -> None:
        """Capture the user's compile-time call site for error splicing
        (reference ``_tasks.py:77-83``)."""
        if limit <= 0:
            self._traceback = None
            return
        frame = sys._getframe(2)
        self._traceback = frames_to_traceback(frame, limit, should_prune)

    @property
    def params(self) -> ParamDict:
        return self._params

    @property
    def partition_spec(self) -> PartitionSpec:
        return self._partition_spec

    @property
    def inputs(self) -> List["FugueTask"]:
        return self._inputs

    def __uuid__(self) -> str:
        # identity fields are fixed at construction, so the spec UUID is
        # computed once (the DAG recomputes uuids heavily for dedup)
        if self._uuid is None:
            self._uuid = to_uuid(
                str(type(self).__name__),
                self._extension.__uuid__()
                if hasattr(self._extension, "__uuid__")
                else str(self._extension),
                dict(self._params),
                self._partition_spec,
                [t.__uuid__() for t in self._inputs],
                self._input_names,
            )
        return self._uuid

    def set_checkpoint(self, checkpoint: Checkpoint) -> "FugueTask":
        self._checkpoint = checkpoint
        return self

    @property
    def has_checkpoint(self) -> bool:
        return not self._checkpoint.is_null

    def broadcast(self) -> "FugueTask":
        self._broadcast = True
        return self

    def set_yield(self, name: str, obj: Any) -> None:
        self._yields.append(obj)

    @property
    def executed(self) -> bool:
        return self._executed

    @property
    def result(self) -> DataFrame:
        if self._result is None:
            raise FugueWorkflowError("task has no result (not executed)")
        return self._result

    def _bind(self, ctx: Any) -> None:
        e = self._extension
        e._execution_engine = ctx.execution_engine
        e._params = self._params.get("params", ParamDict())
        e._partition_spec = self._partition_spec
        e._rpc_server = ctx.rpc_server
        e._workflow_conf = ctx.execution_engine.conf

    def execute(self, ctx: Any) -> None:
        try:
            self._execute(ctx)
        except Exception as e:
            raise modify_traceback(
                e, self._traceback, make_prune_predicate(("fugue_amd",))
            ) from None

    def _execute(self, ctx: Any) -> None:
        raise NotImplementedError

    def set_result(self, ctx: Any, df: DataFrame) -> None:
        df = self._checkpoint.run(df, ctx.checkpoint_path, self.__uuid__())
        if self._broadcast:
            df = ctx.execution_engine.broadcast(df)
        self._result = df
        self._executed = True
        for obj in self._yields:
            if isinstance(obj, PhysicalYielded):
                if obj.storage_type == "file":
                    path = ctx.checkpoint_path.get_temp_file(self.__uuid__(), True)
                    ctx.execution_engine.save_df(df, path, format_hint="parquet")
                    obj.set_value(path)
                else:
                    table = "_fugue_yield_" + self.__uuid__().replace("-", "")
                    ctx.execution_engine.sql_engine.save_table(df, table)
                    obj.set_value(table)
            elif isinstance(obj, YieldedDataFrame):
                obj.set_value(
                    ctx.execution_engine.convert_yield_dataframe(
                        df, as_local=ctx.yields_as_local
                    )
                )


class Create(FugueTask):
    """Reference parity: ``fugue/workflow/_tasks.py:214``."""

    def __init__(self, creator: Creator, params: Any = None):
        super().__init__(creator, [], params=params)

    def _execute(self, ctx: Any) -> None:
        self._bind(ctx)
        df = self._extension.create()
        self.set_result(ctx, ctx.execution_engine.to_df(df))


class Process(FugueTask):
    """Reference parity: ``fugue/workflow/_tasks.py:243``."""

    def __init__(
        self,
        processor: Processor,
        inputs: List[FugueTask],
        params: Any = None,
        partition_spec: Optional[PartitionSpec] = None,
        input_names: Optional[List[Optional[str]]] = None,
    ):
        super().__init__(
            processor,
            inputs,
            params=params,
            partition_spec=partition_spec,
            input_names=input_names,
        )

    def _execute(self, ctx: Any) -> None:
        self._bind(ctx)
        self._extension.validate_on_compile()
        dfs = self._collect_inputs(ctx)
        self._extension.validate_on_runtime(dfs)
        df = self._extension.process(dfs)
        self.set_result(ctx, ctx.execution_engine.to_df(df))

    def _collect_inputs(self, ctx: Any) -> DataFrames:
        if self._input_names is not None and any(
            n is not None for n in self._input_names
        ):
            return DataFrames(
                {n: t.result for n, t in zip(self._input_names, self._inputs)}
            )
        return DataFrames([t.result for t in self._inputs])


class Output(FugueTask):
    """Reference parity: ``fugue/workflow/_tasks.py:297``."""

    def __init__(
        self,
        outputter: Outputter,
        inputs: List[FugueTask],
        params: Any = None,
        partition_spec: Optional[PartitionSpec] = None,
    ):
        super().__init__(outputter, inputs, params=params, partition_spec=partition_spec)

    def _execute(self, ctx: Any) -> None:
        self._bind(ctx)
        self._extension.validate_on_compile()
        dfs = DataFrames([t.result for t in self._inputs])
        self._extension.validate_on_runtime(dfs)
        self._extension.process(dfs)
        self._executed = True
        # outputs pass through their first input so downstream `show` chains work
        if len(self._inputs) > 0:
            self._result = self._inputs[0].result
