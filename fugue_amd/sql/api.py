"""fugue_sql / fugue_sql_flow: the FugueSQL entry points.

Reference parity: ``fugue/sql/api.py:18,111`` (caller local-variable
capture, single-result yield convention).
"""
import inspect
from typing import Any, Dict, Optional

from fugue_amd.dataframe.dataframe import DataFrame
from fugue_amd.exceptions import FugueSQLError
from fugue_amd.sql.workflow import FugueSQLWorkflow


def _capture_caller_vars(depth: int = 2) -> Dict[str, Any]:
    # sys._getframe, not inspect.stack(): the latter resolves source info
    # for every frame (~20ms per call)
    import sys

    frame = sys._getframe(depth)
    res: Dict[str, Any] = {}
    res.update(frame.f_globals)
    res.update(frame.f_locals)
    return {k: v for k, v in res.items() if not k.startswith("__")}


def fugue_sql_flow(query: str, *args: Any, **kwargs: Any) -> FugueSQLWorkflow:
    """Parse a FugueSQL script into a workflow (lazy; call ``.run()``)."""
    dag = FugueSQLWorkflow()
    variables = _capture_caller_vars()
    dag._sql(query, variables, *args, **kwargs)
    return dag


def fugue_sql(
    query: str,
    *args: Any,
    engine: Any = None,
    engine_conf: Any = None,
    as_fugue: bool = False,
    as_local: bool = False,
    **kwargs: Any,
) -> Any:
    """Run a FugueSQL script eagerly and return the single result (the
    script's last dataframe, or its single YIELD)."""
    dag = FugueSQLWorkflow()
    variables = _capture_caller_vars()
    dag._sql(query, variables, *args, **kwargs)
    auto_yield = False
    if len(dag.yields) == 0:
        if dag.last_df is None:
            raise FugueSQLError("no dataframe to return from the SQL")
        dag.last_df.yield_dataframe_as("result", as_local=as_local)
        auto_yield = True
    elif len(dag.yields) != 1:
        raise FugueSQLError(
            "fugue_sql can only have one yield; use fugue_sql_flow instead"
        )
    dag.run(engine, engine_conf)
    key = "result" if auto_yield else list(dag.yields.keys())[0]
    y = dag.yields[key]
    if hasattr(y, "result"):
        result = y.result
        if as_fugue:
            return result
        return result.native_as_df()
    return y
