"""Eager functional API (the fa.* surface).

Reference parity: ``fugue/execution/api.py`` — each function makes/infers
an engine, runs one op, converts the output back to the input's native
type family.
"""
from contextlib import contextmanager
from typing import Any, Callable, Iterator, List, Optional, Tuple, Union

from fugue_amd.collections.partition import PartitionSpec
from fugue_amd.column.expressions import ColumnExpr
from fugue_amd.column.sql import SelectColumns
from fugue_amd.dataframe.dataframe import AnyDataFrame, DataFrame, as_fugue_df
from fugue_amd.execution.execution_engine import (
    ExecutionEngine,
    _GLOBAL_ENGINE,
)
from fugue_amd.execution.factory import (
    make_execution_engine,
    try_get_context_execution_engine,
)
from fugue_amd.utils.params import ParamDict


@contextmanager
def engine_context(
    engine: Any = None, engine_conf: Any = None, infer_by: Optional[List[Any]] = None
) -> Iterator[ExecutionEngine]:
    e = make_execution_engine(engine, engine_conf, infer_by=infer_by)
    with e.as_context() as ctx:
        yield ctx


def set_global_engine(engine: Any = None, engine_conf: Any = None) -> ExecutionEngine:
    e = make_execution_engine(engine, engine_conf)
    return e.set_global()


def clear_global_engine() -> None:
    current = _GLOBAL_ENGINE[0]
    if current is not None:
        current._is_global = False
        current._exit_context()
        _GLOBAL_ENGINE[0] = None


def get_context_engine() -> ExecutionEngine:
    e = try_get_context_execution_engine()
    if e is None:
        raise RuntimeError("no context/global execution engine is set")
    return e


def get_current_conf() -> ParamDict:
    e = try_get_context_execution_engine()
    if e is not None:
        return e.conf
    from fugue_amd.constants import get_global_conf

    return ParamDict(get_global_conf())


def get_current_parallelism() -> int:
    return make_execution_engine().get_current_parallelism()


def as_fugue_engine_df(engine: ExecutionEngine, df: AnyDataFrame, schema: Any = None) -> DataFrame:
    return engine.to_df(df, schema)


def run_engine_function(
    func: Callable[[ExecutionEngine], Any],
    engine: Any = None,
    engine_conf: Any = None,
    as_fugue: bool = False,
    as_local: bool = False,
    infer_by: Optional[List[Any]] = None,
) -> Any:
    e = make_execution_engine(engine, engine_conf, infer_by=infer_by)
    with e.as_context():
        res = func(e)
        if isinstance(res, DataFrame):
            res = e.convert_yield_dataframe(res, as_local)
            if as_fugue:
                return res
            return res.native_as_df()
        return res


def _one_df_func(
    df: AnyDataFrame,
    func: Callable[[ExecutionEngine, DataFrame], Any],
    engine: Any,
    engine_conf: Any,
    as_fugue: bool,
    as_local: bool = False,
    extra_infer: Optional[List[Any]] = None,
) -> Any:
    infer = [df] + (extra_infer or [])
    return run_engine_function(
        lambda e: func(e, e.to_df(df)),
        engine=engine,
        engine_conf=engine_conf,
        as_fugue=as_fugue or isinstance(df, DataFrame),
        as_local=as_local,
        infer_by=infer,
    )


def repartition(
    df: AnyDataFrame,
    partition: Any,
    engine: Any = None,
    engine_conf: Any = None,
    as_fugue: bool = False,
) -> AnyDataFrame:
    return _one_df_func(
        df,
        lambda e, d: e.repartition(d, PartitionSpec(partition)),
        engine,
        engine_conf,
        as_fugue,
    )


def broadcast(
    df: AnyDataFrame, engine: Any = None, engine_conf: Any = None, as_fugue: bool = False
) -> AnyDataFrame:
    return _one_df_func(df, lambda e, d: e.broadcast(d), engine, engine_conf, as_fugue)


def persist(
    df: AnyDataFrame,
    lazy: bool = False,
    engine: Any = None,
    engine_conf: Any = None,
    as_fugue: bool = False,
    **kwargs: Any,
) -> AnyDataFrame:
    return _one_df_func(
        df, lambda e, d: e.persist(d, lazy=lazy, **kwargs), engine, engine_conf, as_fugue
    )


def distinct(
    df: AnyDataFrame, engine: Any = None, engine_conf: Any = None, as_fugue: bool = False
) -> AnyDataFrame:
    return _one_df_func(df, lambda e, d: e.distinct(d), engine, engine_conf, as_fugue)


def dropna(
    df: AnyDataFrame,
    how: str = "any",
    thresh: Optional[int] = None,
    subset: Optional[List[str]] = None,
    engine: Any = None,
    engine_conf: Any = None,
    as_fugue: bool = False,
) -> AnyDataFrame:
    return _one_df_func(
        df,
        lambda e, d: e.dropna(d, how=how, thresh=thresh, subset=subset),
        engine,
        engine_conf,
        as_fugue,
    )


def fillna(
    df: AnyDataFrame,
    value: Any,
    subset: Optional[List[str]] = None,
    engine: Any = None,
    engine_conf: Any = None,
    as_fugue: bool = False,
) -> AnyDataFrame:
    return _one_df_func(
        df, lambda e, d: e.fillna(d, value, subset=subset), engine, engine_conf, as_fugue
    )


def sample(
    df: AnyDataFrame,
    n: Optional[int] = None,
    frac: Optional[float] = None,
    replace: bool = False,
    seed: Optional[int] = None,
    engine: Any = None,
    engine_conf: Any = None,
    as_fugue: bool = False,
) -> AnyDataFrame:
    return _one_df_func(
        df,
        lambda e, d: e.sample(d, n=n, frac=frac, replace=replace, seed=seed),
        engine,
        engine_conf,
        as_fugue,
    )


def take(
    df: AnyDataFrame,
    n: int,
    presort: str,
    na_position: str = "last",
    partition: Any = None,
    engine: Any = None,
    engine_conf: Any = None,
    as_fugue: bool = False,
) -> AnyDataFrame:
    spec = PartitionSpec(partition) if partition is not None else None
    return _one_df_func(
        df,
        lambda e, d: e.take(
            d, n, presort=presort, na_position=na_position, partition_spec=spec
        ),
        engine,
        engine_conf,
        as_fugue,
    )


def load(
    path: Union[str, List[str]],
    format_hint: Any = None,
    columns: Any = None,
    engine: Any = None,
    engine_conf: Any = None,
    as_fugue: bool = False,
    **kwargs: Any,
) -> AnyDataFrame:
    return run_engine_function(
        lambda e: e.load_df(path, format_hint=format_hint, columns=columns, **kwargs),
        engine=engine,
        engine_conf=engine_conf,
        as_fugue=as_fugue,
    )


def save(
    df: AnyDataFrame,
    path: str,
    format_hint: Any = None,
    mode: str = "overwrite",
    partition: Any = None,
    force_single: bool = False,
    engine: Any = None,
    engine_conf: Any = None,
    **kwargs: Any,
) -> None:
    spec = PartitionSpec(partition) if partition is not None else None
    run_engine_function(
        lambda e: e.save_df(
            e.to_df(df),
            path,
            format_hint=format_hint,
            mode=mode,
            partition_spec=spec,
            force_single=force_single,
            **kwargs,
        ),
        engine=engine,
        engine_conf=engine_conf,
        infer_by=[df],
    )


def _two_df_func(
    df1: AnyDataFrame,
    df2: AnyDataFrame,
    func: Callable[[ExecutionEngine, DataFrame, DataFrame], Any],
    engine: Any,
    engine_conf: Any,
    as_fugue: bool,
) -> Any:
    return run_engine_function(
        lambda e: func(e, e.to_df(df1), e.to_df(df2)),
        engine=engine,
        engine_conf=engine_conf,
        as_fugue=as_fugue or isinstance(df1, DataFrame),
        infer_by=[df1, df2],
    )


def join(
    df1: AnyDataFrame,
    df2: AnyDataFrame,
    *dfs: AnyDataFrame,
    how: str,
    on: Optional[List[str]] = None,
    engine: Any = None,
    engine_conf: Any = None,
    as_fugue: bool = False,
) -> AnyDataFrame:
    def _join(e: ExecutionEngine) -> DataFrame:
        res = e.join(e.to_df(df1), e.to_df(df2), how=how, on=on)
        for df in dfs:
            res = e.join(res, e.to_df(df), how=how, on=on)
        return res

    return run_engine_function(
        _join,
        engine=engine,
        engine_conf=engine_conf,
        as_fugue=as_fugue or isinstance(df1, DataFrame),
        infer_by=[df1, df2, *dfs],
    )


def _make_join(join_type: str) -> Callable:
    def _join(
        df1: AnyDataFrame,
        df2: AnyDataFrame,
        *dfs: AnyDataFrame,
        on: Optional[List[str]] = None,
        engine: Any = None,
        engine_conf: Any = None,
        as_fugue: bool = False,
    ) -> AnyDataFrame:
        return join(
            df1,
            df2,
            *dfs,
            how=join_type,
            on=on,
            engine=engine,
            engine_conf=engine_conf,
            as_fugue=as_fugue,
        )

    _join.__name__ = join_type + "_join"
    return _join


inner_join = _make_join("inner")
semi_join = _make_join("semi")
anti_join = _make_join("anti")
left_outer_join = _make_join("left_outer")
right_outer_join = _make_join("right_outer")
full_outer_join = _make_join("full_outer")
cross_join = _make_join("cross")


def union(
    df1: AnyDataFrame,
    df2: AnyDataFrame,
    *dfs: AnyDataFrame,
    distinct: bool = True,
    engine: Any = None,
    engine_conf: Any = None,
    as_fugue: bool = False,
) -> AnyDataFrame:
    def _union(e: ExecutionEngine) -> DataFrame:
        res = e.union(e.to_df(df1), e.to_df(df2), distinct=distinct)
        for df in dfs:
            res = e.union(res, e.to_df(df), distinct=distinct)
        return res

    return run_engine_function(
        _union,
        engine=engine,
        engine_conf=engine_conf,
        as_fugue=as_fugue or isinstance(df1, DataFrame),
        infer_by=[df1, df2, *dfs],
    )


def subtract(
    df1: AnyDataFrame,
    df2: AnyDataFrame,
    *dfs: AnyDataFrame,
    distinct: bool = True,
    engine: Any = None,
    engine_conf: Any = None,
    as_fugue: bool = False,
) -> AnyDataFrame:
    def _sub(e: ExecutionEngine) -> DataFrame:
        res = e.subtract(e.to_df(df1), e.to_df(df2), distinct=distinct)
        for df in dfs:
            res = e.subtract(res, e.to_df(df), distinct=distinct)
        return res

    return run_engine_function(
        _sub,
        engine=engine,
        engine_conf=engine_conf,
        as_fugue=as_fugue or isinstance(df1, DataFrame),
        infer_by=[df1, df2, *dfs],
    )


def intersect(
    df1: AnyDataFrame,
    df2: AnyDataFrame,
    *dfs: AnyDataFrame,
    distinct: bool = True,
    engine: Any = None,
    engine_conf: Any = None,
    as_fugue: bool = False,
) -> AnyDataFrame:
    def _int(e: ExecutionEngine) -> DataFrame:
        res = e.intersect(e.to_df(df1), e.to_df(df2), distinct=distinct)
        for df in dfs:
            res = e.intersect(res, e.to_df(df), distinct=distinct)
        return res

    return run_engine_function(
        _int,
        engine=engine,
        engine_conf=engine_conf,
        as_fugue=as_fugue or isinstance(df1, DataFrame),
        infer_by=[df1, df2, *dfs],
    )


def select(
    df: AnyDataFrame,
    *columns: Union[str, ColumnExpr],
    where: Optional[ColumnExpr] = None,
    having: Optional[ColumnExpr] = None,
    distinct: bool = False,
    engine: Any = None,
    engine_conf: Any = None,
    as_fugue: bool = False,
) -> AnyDataFrame:
    from fugue_amd.column.expressions import col

    cols = SelectColumns(
        *[col(c) if isinstance(c, str) else c for c in columns],
        arg_distinct=distinct,
    )
    return _one_df_func(
        df,
        lambda e, d: e.select(d, cols, where=where, having=having),
        engine,
        engine_conf,
        as_fugue,
    )


def filter(
    df: AnyDataFrame,
    condition: ColumnExpr,
    engine: Any = None,
    engine_conf: Any = None,
    as_fugue: bool = False,
) -> AnyDataFrame:
    return _one_df_func(
        df, lambda e, d: e.filter(d, condition), engine, engine_conf, as_fugue
    )


def assign(
    df: AnyDataFrame,
    engine: Any = None,
    engine_conf: Any = None,
    as_fugue: bool = False,
    **columns: Any,
) -> AnyDataFrame:
    from fugue_amd.column.expressions import lit

    cols = [
        (v.alias(k) if isinstance(v, ColumnExpr) else lit(v).alias(k))
        for k, v in columns.items()
    ]
    return _one_df_func(
        df, lambda e, d: e.assign(d, cols), engine, engine_conf, as_fugue
    )


def aggregate(
    df: AnyDataFrame,
    partition_by: Union[None, str, List[str]] = None,
    engine: Any = None,
    engine_conf: Any = None,
    as_fugue: bool = False,
    **agg_kwcols: ColumnExpr,
) -> AnyDataFrame:
    from fugue_amd.column.expressions import ColumnExpr as _CE

    bad = [k for k, v in agg_kwcols.items() if not isinstance(v, _CE)]
    if bad:
        raise ValueError(
            f"aggregate args must be column expressions: {bad}"
        )
    cols = [v.alias(k) for k, v in agg_kwcols.items()]
    spec = (
        PartitionSpec(by=partition_by)
        if partition_by is not None
        else None
    )
    return _one_df_func(
        df,
        lambda e, d: e.aggregate(d, spec, cols),
        engine,
        engine_conf,
        as_fugue,
    )
