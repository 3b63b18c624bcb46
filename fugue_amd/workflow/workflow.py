"""Lazy workflow DAG: every API call appends a spec-UUID'd task; ``run()``
executes via the built-in thread-pool runner.

Reference parity: ``fugue/workflow/workflow.py`` (``FugueWorkflow`` :1499,
``WorkflowDataFrame`` :88).  New implementation on
``fugue_amd/workflow/_tasks.py`` + ``_workflow_context.py``.
"""
from typing import Any, Callable, Dict, Iterable, List, Optional, Tuple, Union

from fugue_amd.collections.partition import PartitionSpec
from fugue_amd.collections.sql import StructuredRawSQL, TempTableName
from fugue_amd.collections.yielded import PhysicalYielded, Yielded
from fugue_amd.column.expressions import ColumnExpr, col, lit
from fugue_amd.column.sql import SelectColumns as ColumnsSelect
from fugue_amd.constants import (
    FUGUE_COMPILE_TIME_CONFIGS,
    FUGUE_CONF_WORKFLOW_EXCEPTION_INJECT,
    get_global_conf,
)
from fugue_amd.dataframe.dataframe import DataFrame, YieldedDataFrame
from fugue_amd.dataframe.dataframes import DataFrames
from fugue_amd.exceptions import (
    FugueWorkflowCompileError,
    FugueWorkflowError,
)
from fugue_amd.execution.factory import make_execution_engine
from fugue_amd.extensions._builtins import (
    Aggregate,
    AlterColumns,
    Assign,
    AssertEqual,
    AssertNotEqual,
    CreateData,
    Distinct,
    DropColumns,
    Dropna,
    Fillna,
    Filter,
    Load,
    LoadYielded,
    Rename,
    RunJoin,
    RunOutputTransformer,
    RunSetOperation,
    RunSQLSelect,
    RunTransformer,
    Sample,
    Save,
    SaveAndUse,
    Select,
    SelectColumns,
    Show,
    Take,
    Zip,
)
from fugue_amd.extensions.creator.convert import _to_creator
from fugue_amd.extensions.outputter.convert import _to_outputter
from fugue_amd.extensions.processor.convert import _to_processor
from fugue_amd.extensions.transformer.convert import (
    _to_output_transformer,
    _to_transformer,
)
from fugue_amd.extensions._utils import to_validation_rules
from fugue_amd.rpc import to_rpc_handler
from fugue_amd.utils.exception import make_prune_predicate
from fugue_amd.utils.hash import to_uuid
from fugue_amd.utils.params import ParamDict
from fugue_amd.workflow._checkpoint import (
    Checkpoint,
    StrongCheckpoint,
    WeakCheckpoint,
)
from fugue_amd.workflow._tasks import Create, FugueTask, Output, Process
from fugue_amd.workflow._workflow_context import FugueWorkflowContext

_DEFAULT_IGNORE_ERRORS: List[Any] = []


def _compile_validate(ext: Any, spec: PartitionSpec, callback: Any) -> None:
    """DAG-build-time validation: partition-spec rules
    (``partitionby_has``/``partitionby_is``) and required-callback checks
    fail at compile, before any engine runs (reference
    ``builtin_suite`` validation tests)."""
    from fugue_amd.exceptions import FugueInterfacelessError

    ext._partition_spec = spec
    ext.validate_on_compile()
    if getattr(ext, "_requires_callback", False) and callback is None:
        raise FugueInterfacelessError(
            f"{ext} requires a callback but none was provided"
        )


class WorkflowDataFrame:
    """A lazy node handle in the workflow DAG."""

    def __init__(
        self,
        workflow: "FugueWorkflow",
        task: FugueTask,
        metadata: Any = None,
    ):
        self._workflow = workflow
        self._task = task
        self._metadata = ParamDict(metadata)

    @property
    def workflow(self) -> "FugueWorkflow":
        return self._workflow

    @property
    def task(self) -> FugueTask:
        return self._task

    @property
    def partition_spec(self) -> PartitionSpec:
        return PartitionSpec(self._metadata.get("pre_partition", PartitionSpec()))

    def __uuid__(self) -> str:
        return to_uuid(self._task.__uuid__(), self._metadata.get("pre_partition", ""))

    def spec_uuid(self) -> str:
        """Deterministic spec id of this node (reference
        ``workflow.py`` WorkflowDataFrame.spec_uuid)."""
        return self._task.__uuid__()

    @property
    def result(self) -> DataFrame:
        return self._task.result

    def compute(self, *args: Any, **kwargs: Any) -> DataFrame:
        self.workflow.run(*args, **kwargs)
        return self.result

    # --- partitioning -------------------------------------------------- #
    def partition(self, *args: Any, **kwargs: Any) -> "WorkflowDataFrame":
        return WorkflowDataFrame(
            self._workflow,
            self._task,
            {"pre_partition": PartitionSpec(*args, **kwargs)},
        )

    def partition_by(self, *keys: str, **kwargs: Any) -> "WorkflowDataFrame":
        return self.partition(by=list(keys), **kwargs)

    def per_partition_by(self, *keys: str) -> "WorkflowDataFrame":
        return self.partition(by=list(keys), algo="coarse")

    def per_row(self) -> "WorkflowDataFrame":
        return self.partition("per_row")

    @property
    def _pre_partition(self) -> PartitionSpec:
        return PartitionSpec(self._metadata.get("pre_partition", PartitionSpec()))

    # --- transforms ---------------------------------------------------- #
    def transform(
        self,
        using: Any,
        schema: Any = None,
        params: Any = None,
        pre_partition: Any = None,
        ignore_errors: Optional[List[Any]] = None,
        callback: Any = None,
    ) -> "WorkflowDataFrame":
        if pre_partition is None:
            pre_partition = self._pre_partition
        return self.workflow.transform(
            self,
            using=using,
            schema=schema,
            params=params,
            pre_partition=pre_partition,
            ignore_errors=ignore_errors or [],
            callback=callback,
        )

    def out_transform(
        self,
        using: Any,
        params: Any = None,
        pre_partition: Any = None,
        ignore_errors: Optional[List[Any]] = None,
        callback: Any = None,
    ) -> None:
        if pre_partition is None:
            pre_partition = self._pre_partition
        self.workflow.out_transform(
            self,
            using=using,
            params=params,
            pre_partition=pre_partition,
            ignore_errors=ignore_errors or [],
            callback=callback,
        )

    def process(
        self,
        using: Any,
        schema: Any = None,
        params: Any = None,
        pre_partition: Any = None,
    ) -> "WorkflowDataFrame":
        if pre_partition is None:
            pre_partition = self._pre_partition
        return self.workflow.process(
            self, using=using, schema=schema, params=params, pre_partition=pre_partition
        )

    def output(self, using: Any, params: Any = None, pre_partition: Any = None) -> None:
        if pre_partition is None:
            pre_partition = self._pre_partition
        self.workflow.output(
            self, using=using, params=params, pre_partition=pre_partition
        )

    # --- relational ops ------------------------------------------------ #
    def _op(self, processor_cls: type, params: Any, *others: Any, partition_spec: Any = None) -> "WorkflowDataFrame":
        dfs = [self] + [self.workflow._to_wdf(o) for o in others]
        task = Process(
            processor_cls(),
            [d.task for d in dfs],
            params=dict(params=ParamDict(params)),
            partition_spec=PartitionSpec(partition_spec)
            if partition_spec is not None
            else None,
        )
        return self.workflow.add(task)

    def join(self, *dfs: Any, how: str, on: Optional[List[str]] = None) -> "WorkflowDataFrame":
        return self._op(RunJoin, dict(how=how, on=list(on or [])), *dfs)

    def inner_join(self, *dfs: Any, on: Optional[List[str]] = None) -> "WorkflowDataFrame":
        return self.join(*dfs, how="inner", on=on)

    def semi_join(self, *dfs: Any, on: Optional[List[str]] = None) -> "WorkflowDataFrame":
        return self.join(*dfs, how="semi", on=on)

    def left_semi_join(self, *dfs: Any, on: Optional[List[str]] = None) -> "WorkflowDataFrame":
        return self.join(*dfs, how="semi", on=on)

    def anti_join(self, *dfs: Any, on: Optional[List[str]] = None) -> "WorkflowDataFrame":
        return self.join(*dfs, how="anti", on=on)

    def left_anti_join(self, *dfs: Any, on: Optional[List[str]] = None) -> "WorkflowDataFrame":
        return self.join(*dfs, how="anti", on=on)

    def left_outer_join(self, *dfs: Any, on: Optional[List[str]] = None) -> "WorkflowDataFrame":
        return self.join(*dfs, how="left_outer", on=on)

    def right_outer_join(self, *dfs: Any, on: Optional[List[str]] = None) -> "WorkflowDataFrame":
        return self.join(*dfs, how="right_outer", on=on)

    def full_outer_join(self, *dfs: Any, on: Optional[List[str]] = None) -> "WorkflowDataFrame":
        return self.join(*dfs, how="full_outer", on=on)

    def cross_join(self, *dfs: Any) -> "WorkflowDataFrame":
        return self.join(*dfs, how="cross")

    def union(self, *dfs: Any, distinct: bool = True) -> "WorkflowDataFrame":
        return self._op(RunSetOperation, dict(how="union", distinct=distinct), *dfs)

    def subtract(self, *dfs: Any, distinct: bool = True) -> "WorkflowDataFrame":
        return self._op(RunSetOperation, dict(how="subtract", distinct=distinct), *dfs)

    def intersect(self, *dfs: Any, distinct: bool = True) -> "WorkflowDataFrame":
        return self._op(RunSetOperation, dict(how="intersect", distinct=distinct), *dfs)

    def distinct(self) -> "WorkflowDataFrame":
        return self._op(Distinct, {})

    def dropna(
        self,
        how: str = "any",
        thresh: Optional[int] = None,
        subset: Optional[List[str]] = None,
    ) -> "WorkflowDataFrame":
        params: Dict[str, Any] = dict(how=how)
        if thresh is not None:
            params["thresh"] = thresh
        if subset is not None:
            params["subset"] = subset
        return self._op(Dropna, params)

    def fillna(self, value: Any, subset: Optional[List[str]] = None) -> "WorkflowDataFrame":
        params: Dict[str, Any] = dict(value=value)
        if subset is not None:
            params["subset"] = subset
        return self._op(Fillna, params)

    def sample(
        self,
        n: Optional[int] = None,
        frac: Optional[float] = None,
        replace: bool = False,
        seed: Optional[int] = None,
    ) -> "WorkflowDataFrame":
        if (n is None) == (frac is None):
            raise ValueError("one and only one of n and frac must be set")
        params: Dict[str, Any] = dict(replace=replace)
        if n is not None:
            params["n"] = n
        if frac is not None:
            params["frac"] = frac
        if seed is not None:
            params["seed"] = seed
        return self._op(Sample, params)

    def take(self, n: int, presort: str = "", na_position: str = "last") -> "WorkflowDataFrame":
        if not isinstance(n, int):
            raise ValueError("n must be an integer")
        if na_position not in ("first", "last"):
            raise ValueError("na_position must be 'first' or 'last'")
        task = Process(
            Take(),
            [self.task],
            params=dict(
                params=ParamDict(dict(n=n, presort=presort, na_position=na_position))
            ),
            partition_spec=self._pre_partition,
        )
        return self.workflow.add(task)

    def select(
        self,
        *columns: Union[str, ColumnExpr],
        where: Optional[ColumnExpr] = None,
        having: Optional[ColumnExpr] = None,
        distinct: bool = False,
    ) -> "WorkflowDataFrame":
        cols = ColumnsSelect(
            *[col(c) if isinstance(c, str) else c for c in columns],
            arg_distinct=distinct,
        )
        params: Dict[str, Any] = dict(columns=cols)
        if where is not None:
            params["where"] = where
        if having is not None:
            params["having"] = having
        return self._op(Select, params)

    def filter(self, condition: ColumnExpr) -> "WorkflowDataFrame":
        return self._op(Filter, dict(condition=condition))

    def assign(self, *args: ColumnExpr, **kwargs: Any) -> "WorkflowDataFrame":
        cols = list(args) + [
            v.alias(k) if isinstance(v, ColumnExpr) else lit(v).alias(k)
            for k, v in kwargs.items()
        ]
        return self._op(Assign, dict(columns=cols))

    def aggregate(self, *agg_cols: ColumnExpr, **kwagg_cols: ColumnExpr) -> "WorkflowDataFrame":
        cols = list(agg_cols) + [v.alias(k) for k, v in kwagg_cols.items()]
        task = Process(
            Aggregate(),
            [self.task],
            params=dict(params=ParamDict(dict(columns=cols))),
            partition_spec=self._pre_partition,
        )
        return self.workflow.add(task)

    def rename(self, *args: Any, **kwargs: str) -> "WorkflowDataFrame":
        m: Dict[str, str] = {}
        for a in args:
            m.update(a)
        m.update(kwargs)
        return self._op(Rename, dict(columns=m))

    def alter_columns(self, columns: Any) -> "WorkflowDataFrame":
        return self._op(AlterColumns, dict(columns=str(columns)))

    def drop(self, columns: List[str], if_exists: bool = False) -> "WorkflowDataFrame":
        return self._op(DropColumns, dict(columns=columns, if_exists=if_exists))

    def __getitem__(self, columns: List[Any]) -> "WorkflowDataFrame":
        return self._op(SelectColumns, dict(columns=columns))

    def zip(
        self,
        *dfs: Any,
        how: str = "inner",
        partition: Any = None,
    ) -> "WorkflowDataFrame":
        return self.workflow.zip(
            self, *dfs, how=how, partition=partition or self._pre_partition
        )

    # --- checkpoints / persist / broadcast / yields --------------------- #
    def persist(self) -> "WorkflowDataFrame":
        return self.weak_checkpoint(lazy=False)

    def weak_checkpoint(self, lazy: bool = False, **kwargs: Any) -> "WorkflowDataFrame":
        self._task.set_checkpoint(WeakCheckpoint(lazy=lazy, **kwargs))
        return self

    def checkpoint(self, storage_type: str = "file") -> "WorkflowDataFrame":
        return self.strong_checkpoint(storage_type=storage_type)

    def strong_checkpoint(
        self,
        storage_type: str = "file",
        lazy: bool = False,
        partition: Any = None,
        single: bool = False,
        **kwargs: Any,
    ) -> "WorkflowDataFrame":
        self._task.set_checkpoint(
            StrongCheckpoint(
                storage_type=storage_type,
                lazy=lazy,
                partition=partition,
                single=single,
                **kwargs,
            )
        )
        return self

    def deterministic_checkpoint(
        self,
        storage_type: str = "file",
        lazy: bool = False,
        partition: Any = None,
        single: bool = False,
        namespace: Any = None,
        **kwargs: Any,
    ) -> "WorkflowDataFrame":
        self._task.set_checkpoint(
            StrongCheckpoint(
                storage_type=storage_type,
                lazy=lazy,
                partition=partition,
                single=single,
                deterministic=True,
                namespace=namespace,
                **kwargs,
            )
        )
        return self

    def broadcast(self) -> "WorkflowDataFrame":
        self._task.broadcast()
        return self

    def _assert_can_yield_physical(self) -> None:
        # physical yields are allowed on unpersisted nodes (the yield
        # itself becomes a strong checkpoint) or on deterministic
        # checkpoints; any other checkpoint state is a compile error
        # (persist: ValueError — reference workflow.py:987 note)
        ck = self._task._checkpoint
        if ck.is_null or ck.deterministic:
            return
        if not ck.to_file:
            raise ValueError(
                "can't yield file/table after persist; use "
                "deterministic_checkpoint"
            )
        raise FugueWorkflowCompileError(
            "can't yield file/table after a non-deterministic checkpoint"
        )

    def _physical_yield_key(self) -> str:
        # a direct yield is not reusable across executions, so its
        # identity is random per DAG build; yielding a deterministic
        # checkpoint is stable (consumers of dag.yields[...] then hash
        # identically across builds — reference test_yield_file)
        ck = self._task._checkpoint
        if not ck.is_null and ck.deterministic:
            return self.__uuid__()
        import uuid as _uuid_mod

        return str(_uuid_mod.uuid4())

    def yield_file_as(self, name: str) -> None:
        self._assert_can_yield_physical()
        y = PhysicalYielded(self._physical_yield_key(), "file")
        self.workflow._register_yield(name, y, self._task)

    def yield_table_as(self, name: str) -> None:
        self._assert_can_yield_physical()
        y = PhysicalYielded(self._physical_yield_key(), "table")
        self.workflow._register_yield(name, y, self._task)

    def yield_dataframe_as(self, name: str, as_local: bool = False) -> None:
        y = YieldedDataFrame(self.__uuid__())
        self.workflow._register_yield(name, y, self._task, as_local=as_local)

    # --- IO / display --------------------------------------------------- #
    def save(
        self,
        path: str,
        fmt: str = "",
        mode: str = "overwrite",
        partition: Any = None,
        single: bool = False,
        **kwargs: Any,
    ) -> None:
        if partition is None:
            partition = self._pre_partition
        task = Output(
            Save(),
            [self.task],
            params=dict(
                params=ParamDict(
                    dict(path=path, fmt=fmt, mode=mode, single=single, params=kwargs)
                )
            ),
            partition_spec=PartitionSpec(partition),
        )
        self.workflow.add(task)

    def save_and_use(
        self,
        path: str,
        fmt: str = "",
        mode: str = "overwrite",
        partition: Any = None,
        single: bool = False,
        **kwargs: Any,
    ) -> "WorkflowDataFrame":
        if partition is None:
            partition = self._pre_partition
        task = Process(
            SaveAndUse(),
            [self.task],
            params=dict(
                params=ParamDict(
                    dict(path=path, fmt=fmt, mode=mode, single=single, params=kwargs)
                )
            ),
            partition_spec=PartitionSpec(partition),
        )
        return self.workflow.add(task)

    def show(
        self,
        n: int = 10,
        with_count: bool = False,
        title: Optional[str] = None,
    ) -> None:
        params: Dict[str, Any] = dict(n=n, with_count=with_count)
        if title is not None:
            params["title"] = title
        task = Output(
            Show(), [self.task], params=dict(params=ParamDict(params))
        )
        self.workflow.add(task)

    def assert_eq(self, *dfs: Any, **params: Any) -> None:
        self.workflow.assert_eq(self, *dfs, **params)

    def assert_not_eq(self, *dfs: Any, **params: Any) -> None:
        self.workflow.assert_not_eq(self, *dfs, **params)


class WorkflowDataFrames(DataFrames):
    """Dict-like holder of WorkflowDataFrames (reference ``workflow.py:1413``)."""

    def __setitem__(self, key: str, value: Any) -> None:  # type: ignore
        if not isinstance(value, WorkflowDataFrame):
            raise ValueError(f"{value} is not a WorkflowDataFrame")
        self._has_key = True
        dict.__setitem__(self, key, value)

    def _append(self, value: Any) -> None:
        if not isinstance(value, WorkflowDataFrame):
            raise ValueError(f"{value} is not a WorkflowDataFrame")
        dict.__setitem__(self, f"_{len(self)}", value)


class FugueWorkflowResult:
    """Result of ``FugueWorkflow.run``: yields by name."""

    def __init__(self, yields: Dict[str, Yielded]):
        self._yields = yields

    @property
    def yields(self) -> Dict[str, Any]:
        return self._yields

    def __getitem__(self, name: str) -> Any:
        y = self._yields[name]
        # dataframe yields resolve to their result (reference
        # FugueWorkflowResult is a DataFrames of results)
        return y.result if isinstance(y, YieldedDataFrame) else y


class FugueWorkflow:
    """The lazy DAG builder + runner."""

    def __init__(self, compile_conf: Any = None):
        self._conf = ParamDict(
            {k: v for k, v in get_global_conf().items() if k in FUGUE_COMPILE_TIME_CONFIGS}
        )
        self._conf.update_params(compile_conf)
        self._tasks: Dict[str, FugueTask] = {}
        self._task_order: List[FugueTask] = []
        self._yields: Dict[str, Yielded] = {}
        self._computed = False
        self._last_df: Optional[WorkflowDataFrame] = None
        self._consumers: Dict[int, int] = {}

    @property
    def conf(self) -> ParamDict:
        return self._conf

    @property
    def yields(self) -> Dict[str, Yielded]:
        return self._yields

    def spec_uuid(self) -> str:
        """Deterministic id of the whole DAG spec: stable across runs
        for identical workflows (reference FugueWorkflow.spec_uuid)."""
        return to_uuid([t.__uuid__() for t in self._tasks.values()])

    def reset_execution(self) -> None:
        """Clear task results so a built DAG can run again (used by the
        fugue_sql plan cache: construction is reused, execution is not)."""
        for t in self._tasks.values():
            t._executed = False
            t._result = None

    @property
    def last_df(self) -> Optional[WorkflowDataFrame]:
        return self._last_df

    def __enter__(self) -> "FugueWorkflow":
        return self

    def __exit__(self, exc_type: Any, exc_val: Any, exc_tb: Any) -> None:
        # the `with` form only scopes the DAG build; execution is the
        # explicit dag.run(...) (reference ``workflow.py:1619``)
        return

    def add(self, task: FugueTask) -> WorkflowDataFrame:
        """Add a task, dedup by spec uuid (determinism)."""
        from fugue_amd.constants import (
            FUGUE_CONF_WORKFLOW_EXCEPTION_HIDE,
            FUGUE_CONF_WORKFLOW_EXCEPTION_OPTIMIZE,
        )

        inject = int(self._conf.get(FUGUE_CONF_WORKFLOW_EXCEPTION_INJECT, 3))
        optimize = bool(self._conf.get(FUGUE_CONF_WORKFLOW_EXCEPTION_OPTIMIZE, True))
        hide = str(self._conf.get(FUGUE_CONF_WORKFLOW_EXCEPTION_HIDE, "fugue_amd."))
        prefixes = tuple(x for x in hide.split(",") if x != "")
        task.reset_traceback(
            inject if optimize else 0,
            make_prune_predicate(prefixes) if inject > 0 and optimize else None,
        )
        uid = task.__uuid__()
        if uid in self._tasks:
            task = self._tasks[uid]
        else:
            self._tasks[uid] = task
            self._task_order.append(task)
            # auto-persist: an upstream consumed by >1 downstream tasks is
            # persisted so it isn't recomputed per consumer (reference
            # ``workflow/workflow.py:2228-2241``)
            from fugue_amd.constants import (
                FUGUE_CONF_WORKFLOW_AUTO_PERSIST,
                FUGUE_CONF_WORKFLOW_AUTO_PERSIST_VALUE,
            )

            for inp in task._inputs:
                k = id(inp)
                self._consumers[k] = self._consumers.get(k, 0) + 1
                if (
                    self._consumers[k] > 1
                    and self._conf.get(FUGUE_CONF_WORKFLOW_AUTO_PERSIST, False)
                    and inp._checkpoint.is_null
                ):
                    kwargs: Dict[str, Any] = {}
                    level = self._conf.get_or_none(
                        FUGUE_CONF_WORKFLOW_AUTO_PERSIST_VALUE, object
                    )
                    if level is not None:
                        kwargs["level"] = level
                    inp.set_checkpoint(WeakCheckpoint(lazy=False, **kwargs))
        res = WorkflowDataFrame(self, task)
        self._last_df = res
        return res

    def _to_wdf(self, data: Any) -> WorkflowDataFrame:
        if isinstance(data, WorkflowDataFrame):
            if data.workflow is not self:
                raise FugueWorkflowCompileError(
                    "dataframe belongs to another workflow"
                )
            return data
        return self.df(data)

    def _register_yield(
        self, name: str, obj: Any, task: FugueTask, as_local: bool = False
    ) -> None:
        if name in self._yields:
            raise FugueWorkflowCompileError(f"yield name {name} already used")
        task.set_yield(name, obj)
        self._yields[name] = obj
        self._yields_as_local = as_local

    # --- creation ------------------------------------------------------- #
    def df(self, data: Any, schema: Any = None) -> WorkflowDataFrame:
        return self.create_data(data, schema)

    def create_data(self, data: Any, schema: Any = None) -> WorkflowDataFrame:
        if isinstance(data, WorkflowDataFrame):
            if schema is not None:
                raise FugueWorkflowCompileError(
                    "can't reset schema for WorkflowDataFrame"
                )
            return self._to_wdf(data)
        if isinstance(data, Yielded):
            task: FugueTask = Create(
                LoadYielded(), params=dict(params=ParamDict(dict(yielded=data)))
            )
            return self.add(task)
        task = Create(
            CreateData(), params=dict(params=ParamDict(dict(df=data, schema=schema)))
        )
        return self.add(task)

    def create(
        self, using: Any, schema: Any = None, params: Any = None
    ) -> WorkflowDataFrame:
        # a dataframe object used as a "creator" is just data: emit the
        # same CreateData task as dag.df so the spec uuids coincide
        import pandas as _pd
        import pyarrow as _pa

        if isinstance(using, (DataFrame, Yielded, _pd.DataFrame, _pa.Table)):
            return self.create_data(using, schema)
        creator = _to_creator(using, schema)
        task = Create(creator, params=dict(params=ParamDict(params)))
        return self.add(task)

    def load(
        self, path: str, fmt: str = "", columns: Any = None, **kwargs: Any
    ) -> WorkflowDataFrame:
        task = Create(
            Load(),
            params=dict(
                params=ParamDict(
                    dict(path=path, fmt=fmt, columns=columns, params=kwargs)
                )
            ),
        )
        return self.add(task)

    # --- generic extension entry points ---------------------------------- #
    def process(
        self,
        *dfs: Any,
        using: Any,
        schema: Any = None,
        params: Any = None,
        pre_partition: Any = None,
    ) -> WorkflowDataFrame:
        _dfs = self._build_dataframes(dfs)
        proc = _to_processor(using, schema)
        _compile_validate(proc, PartitionSpec(pre_partition), None)
        task = Process(
            proc,
            [d.task for d in _dfs.values()],
            params=dict(params=ParamDict(params)),
            partition_spec=PartitionSpec(pre_partition),
            input_names=list(_dfs.keys()) if _dfs.has_key else None,
        )
        return self.add(task)

    def output(
        self, *dfs: Any, using: Any, params: Any = None, pre_partition: Any = None
    ) -> None:
        _dfs = self._build_dataframes(dfs)
        out = _to_outputter(using)
        _compile_validate(out, PartitionSpec(pre_partition), None)
        task = Output(
            out,
            [d.task for d in _dfs.values()],
            params=dict(params=ParamDict(params)),
            partition_spec=PartitionSpec(pre_partition),
        )
        self.add(task)

    def transform(
        self,
        *dfs: Any,
        using: Any,
        schema: Any = None,
        params: Any = None,
        pre_partition: Any = None,
        ignore_errors: Optional[List[Any]] = None,
        callback: Any = None,
    ) -> WorkflowDataFrame:
        if len(dfs) != 1:
            raise NotImplementedError("transform can only take one input dataframe")
        tf = _to_transformer(using, schema)
        spec = PartitionSpec(pre_partition)
        _compile_validate(tf, spec, callback)
        handler = to_rpc_handler(callback)
        _dfs = self._build_dataframes(dfs)
        wdf = list(_dfs.values())[0]
        task = Process(
            RunTransformer(),
            [wdf.task],
            params=dict(
                params=ParamDict(
                    dict(
                        transformer=tf,
                        schema=None,
                        ignore_errors=ignore_errors or [],
                        params=ParamDict(params),
                        rpc_handler=handler,
                    )
                )
            ),
            partition_spec=PartitionSpec(pre_partition),
        )
        return self.add(task)

    def out_transform(
        self,
        *dfs: Any,
        using: Any,
        params: Any = None,
        pre_partition: Any = None,
        ignore_errors: Optional[List[Any]] = None,
        callback: Any = None,
    ) -> None:
        if len(dfs) != 1:
            raise NotImplementedError("transform can only take one input dataframe")
        tf = _to_output_transformer(using)
        _compile_validate(tf, PartitionSpec(pre_partition), callback)
        handler = to_rpc_handler(callback)
        _dfs = self._build_dataframes(dfs)
        wdf = list(_dfs.values())[0]
        task = Output(
            RunOutputTransformer(),
            [wdf.task],
            params=dict(
                params=ParamDict(
                    dict(
                        transformer=tf,
                        schema=None,
                        ignore_errors=ignore_errors or [],
                        params=ParamDict(params),
                        rpc_handler=handler,
                    )
                )
            ),
            partition_spec=PartitionSpec(pre_partition),
        )
        self.add(task)

    # --- joins / set ops / zip ------------------------------------------ #
    def join(
        self, *dfs: Any, how: str, on: Optional[List[str]] = None
    ) -> WorkflowDataFrame:
        _dfs = [self._to_wdf(d) for d in dfs]
        if len(_dfs) == 0:
            raise FugueWorkflowCompileError("no dataframes to join")
        res = _dfs[0]
        return res.join(*_dfs[1:], how=how, on=on)

    def union(self, *dfs: Any, distinct: bool = True) -> WorkflowDataFrame:
        _dfs = [self._to_wdf(d) for d in dfs]
        return _dfs[0].union(*_dfs[1:], distinct=distinct)

    def subtract(self, *dfs: Any, distinct: bool = True) -> WorkflowDataFrame:
        _dfs = [self._to_wdf(d) for d in dfs]
        return _dfs[0].subtract(*_dfs[1:], distinct=distinct)

    def intersect(self, *dfs: Any, distinct: bool = True) -> WorkflowDataFrame:
        _dfs = [self._to_wdf(d) for d in dfs]
        return _dfs[0].intersect(*_dfs[1:], distinct=distinct)

    def zip(
        self,
        *dfs: Any,
        how: str = "inner",
        partition: Any = None,
    ) -> WorkflowDataFrame:
        _dfs = self._build_dataframes(dfs)
        task = Process(
            Zip(),
            [d.task for d in _dfs.values()],
            params=dict(params=ParamDict(dict(how=how))),
            partition_spec=PartitionSpec(partition),
            input_names=list(_dfs.keys()) if _dfs.has_key else None,
        )
        return self.add(task)

    # --- SQL select ------------------------------------------------------ #
    def select(
        self, *statements: Any, sql_engine: Any = None, sql_engine_params: Any = None, dialect: Optional[str] = "spark"
    ) -> WorkflowDataFrame:
        """Raw SQL select; statements are a mix of strings and
        WorkflowDataFrames (referenced as tables)."""
        parts: List[Tuple[bool, str]] = []
        deps: List[FugueTask] = []
        names: Dict[int, str] = {}
        for s in statements:
            if isinstance(s, str):
                # pad string fragments so refs don't glue to keywords
                parts.append((False, " " + s + " "))
            else:
                wdf = self._to_wdf(s)
                tid = id(wdf.task)
                if tid not in names:
                    names[tid] = f"_fugue_tmp_{len(names)}"
                    deps.append(wdf.task)
                parts.append((True, names[tid]))
        # "SELECT" may be omitted (reference: dag.select("* FROM", df))
        first_str = next((p1 for has, p1 in parts if not has), "")
        first_tok = first_str.strip().split(" ")[0].upper() if first_str.strip() else ""
        lead_is_df = len(parts) > 0 and parts[0][0]
        if first_tok != "SELECT" or (lead_is_df and parts[0][0]):
            if not any(
                not has and p1.strip().upper().startswith("SELECT")
                for has, p1 in parts[:1]
            ):
                parts = [(False, "SELECT ")] + parts
        statement = StructuredRawSQL(parts, dialect=dialect)
        task = Process(
            RunSQLSelect(),
            deps,
            params=dict(
                params=ParamDict(
                    dict(
                        statement=statement,
                        sql_engine=sql_engine,
                        sql_engine_params=ParamDict(sql_engine_params),
                    )
                )
            ),
            input_names=[names[id(t)] for t in deps] if deps else None,
        )
        return self.add(task)

    def assert_eq(self, *dfs: Any, **params: Any) -> None:
        _dfs = [self._to_wdf(d) for d in dfs]
        task = Output(
            AssertEqual(),
            [d.task for d in _dfs],
            params=dict(params=ParamDict(params)),
        )
        self.add(task)

    def assert_not_eq(self, *dfs: Any, **params: Any) -> None:
        _dfs = [self._to_wdf(d) for d in dfs]
        task = Output(
            AssertNotEqual(),
            [d.task for d in _dfs],
            params=dict(params=ParamDict(params)),
        )
        self.add(task)

    # --- run -------------------------------------------------------------- #
    def run(self, engine: Any = None, conf: Any = None, **kwargs: Any) -> FugueWorkflowResult:
        import sys as _sys

        from fugue_amd.constants import (
            FUGUE_CONF_WORKFLOW_EXCEPTION_HIDE,
            FUGUE_CONF_WORKFLOW_EXCEPTION_OPTIMIZE,
        )
        from fugue_amd.utils.exception import modify_traceback

        e = make_execution_engine(engine, conf, **kwargs)
        ctx = FugueWorkflowContext(
            e, yields_as_local=getattr(self, "_yields_as_local", False)
        )
        try:
            ctx.run(self._task_order)
        except Exception as ex:
            if not bool(
                self._conf.get(FUGUE_CONF_WORKFLOW_EXCEPTION_OPTIMIZE, True)
            ):
                raise
            hide = str(self._conf.get(FUGUE_CONF_WORKFLOW_EXCEPTION_HIDE, ""))
            prefixes = tuple(
                x.rstrip(".") for x in hide.split(",") if x != ""
            )
            if len(prefixes) == 0:
                raise
            raise modify_traceback(
                ex, None, make_prune_predicate(prefixes)
            ) from None
        self._computed = True
        return FugueWorkflowResult(self._yields)

    def _build_dataframes(self, dfs: Tuple[Any, ...]) -> WorkflowDataFrames:
        if (
            len(dfs) == 1
            and isinstance(dfs[0], (list, tuple))
            and not (
                isinstance(dfs[0], tuple)
                and len(dfs[0]) == 2
                and isinstance(dfs[0][0], str)
            )
            and all(
                isinstance(x, (WorkflowDataFrame, DataFrame, Yielded))
                for x in dfs[0]
            )
        ):
            # a single list argument is the unpacked input set
            dfs = tuple(dfs[0])
        if len(dfs) == 1 and isinstance(dfs[0], dict):
            res = WorkflowDataFrames()
            for k, v in dfs[0].items():
                res[k] = self._to_wdf(v)
            return res
        res = WorkflowDataFrames()
        for d in dfs:
            if (
                isinstance(d, tuple)
                and len(d) == 2
                and isinstance(d[0], str)
            ):
                res[d[0]] = self._to_wdf(d[1])
            else:
                res._append(self._to_wdf(d))
        return res
