"""transform / out_transform / raw_sql: the flagship single-shot API.

Reference parity: ``fugue/workflow/api.py:34,187,253`` — build a throwaway
DAG around one transformer, run it, return the native type.
"""
import os
from typing import Any, Callable, Dict, List, Optional

from fugue_amd.collections.yielded import Yielded
from fugue_amd.constants import FUGUE_CONF_WORKFLOW_EXCEPTION_INJECT
from fugue_amd.dataframe.dataframe import DataFrame
from fugue_amd.execution.factory import make_execution_engine
from fugue_amd.workflow.workflow import FugueWorkflow


def transform(
    df: Any,
    using: Any,
    schema: Any = None,
    params: Any = None,
    partition: Any = None,
    callback: Any = None,
    ignore_errors: Optional[List[Any]] = None,
    persist: bool = False,
    as_local: bool = False,
    save_path: Optional[str] = None,
    checkpoint: bool = False,
    engine: Any = None,
    engine_conf: Any = None,
    as_fugue: bool = False,
) -> Any:
    """Transform a dataframe using one transformer; returns the result in
    a native format (or saves it when ``save_path`` is given)."""
    dag = FugueWorkflow(
        compile_conf={FUGUE_CONF_WORKFLOW_EXCEPTION_INJECT: 0}
    )
    if isinstance(df, str):
        src = dag.load(df)
    else:
        src = dag.df(df)
    if partition is not None:
        src = src.partition(partition)
    tdf = src.transform(
        using=using,
        schema=schema,
        params=params,
        ignore_errors=ignore_errors or [],
        callback=callback,
    )
    if persist:
        tdf = tdf.persist()
    if checkpoint:
        tdf = tdf.strong_checkpoint()
    if save_path is None:
        tdf.yield_dataframe_as("result", as_local=as_local)
    else:
        tdf.save(save_path, fmt="parquet")
    e = make_execution_engine(engine, engine_conf, infer_by=[df])
    dag.run(e)
    if save_path is not None:
        return save_path
    result = dag.yields["result"].result  # type: ignore
    if as_fugue:
        return result
    if isinstance(df, DataFrame):
        return result
    import pandas as pd
    import pyarrow as pa

    if isinstance(df, pd.DataFrame):
        return result.as_pandas()
    if isinstance(df, pa.Table):
        return result.as_arrow()
    return result.native_as_df()


def out_transform(
    df: Any,
    using: Any,
    params: Any = None,
    partition: Any = None,
    callback: Any = None,
    ignore_errors: Optional[List[Any]] = None,
    engine: Any = None,
    engine_conf: Any = None,
) -> None:
    """Transform with no output (side-effect only)."""
    dag = FugueWorkflow(
        compile_conf={FUGUE_CONF_WORKFLOW_EXCEPTION_INJECT: 0}
    )
    if isinstance(df, str):
        src = dag.load(df)
    else:
        src = dag.df(df)
    if partition is not None:
        src = src.partition(partition)
    src.out_transform(
        using=using,
        params=params,
        ignore_errors=ignore_errors or [],
        callback=callback,
    )
    e = make_execution_engine(engine, engine_conf, infer_by=[df])
    dag.run(e)


def raw_sql(
    *statements: Any,
    engine: Any = None,
    engine_conf: Any = None,
    as_fugue: bool = False,
    as_local: bool = False,
) -> Any:
    """Run a raw SQL statement mixing string fragments and dataframes."""
    dag = FugueWorkflow(
        compile_conf={FUGUE_CONF_WORKFLOW_EXCEPTION_INJECT: 0}
    )
    converted: List[Any] = []
    infer_by: List[Any] = []
    for s in statements:
        if isinstance(s, str):
            converted.append(s)
        else:
            infer_by.append(s)
            converted.append(dag.df(s))
    res = dag.select(*converted)
    res.yield_dataframe_as("result", as_local=as_local)
    e = make_execution_engine(engine, engine_conf, infer_by=infer_by)
    dag.run(e)
    result = dag.yields["result"].result  # type: ignore
    if as_fugue:
        return result
    if any(isinstance(x, DataFrame) for x in infer_by):
        return result
    return result.native_as_df()
