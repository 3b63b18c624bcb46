"""fugue_sql / fugue_sql_flow: the FugueSQL entry points.

Reference parity: ``fugue/sql/api.py:18,111`` (caller local-variable
capture, single-result yield convention).

``fugue_sql`` additionally keeps a **plan cache**: the built DAG for a
script is replayed when the same query text runs again with the same
variable bindings (validated per consulted name — same object identity,
same immutable value, or still-absent).  Execution always reruns; only
construction (parse, DAG build, spec-uuid hashing) is reused — the same
idea as a database's prepared-statement/plan cache.  Disable with
``FUGUE_SQL_PLAN_CACHE=0`` or ``plan_cache=False``.
"""
import os
import threading
from collections import OrderedDict
from typing import Any, Dict, List, Optional, Tuple

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


class _PlanEntry:
    __slots__ = ("dag", "guards", "yield_key", "lock")

    def __init__(self, dag: Any, guards: Dict[str, Any], yield_key: str):
        self.dag = dag
        self.guards = guards
        self.yield_key = yield_key
        self.lock = threading.Lock()


# NOTE: a cached entry holds the built DAG, which keeps strong
# references to its input dataframes — the cap bounds that retention
# and clear_plan_cache() releases it eagerly.
_PLAN_CACHE: "OrderedDict[Tuple[str, bool], List[_PlanEntry]]" = OrderedDict()
_PLAN_CACHE_CAP = 8
_PLAN_CACHE_LOCK = threading.Lock()


def clear_plan_cache() -> None:
    """Drop all cached FugueSQL plans (releases their input frames)."""
    with _PLAN_CACHE_LOCK:
        _PLAN_CACHE.clear()


def _guards_ok(guards: Dict[str, Any], variables: Dict[str, Any]) -> bool:
    for name, g in guards.items():
        kind = g[0]
        if kind == "absent":
            if name in variables:
                return False
        elif kind == "val":
            v = variables.get(name, guards)  # sentinel
            if type(v) is not g[1] or v != g[2]:
                return False
        elif kind == "ref":
            tgt = g[1]()
            if tgt is None or variables.get(name, None) is not tgt:
                return False
        else:  # nocache
            return False
    return True


def _plan_cacheable(guards: Dict[str, Any]) -> bool:
    return all(g[0] != "nocache" for g in guards.values())


def _cache_lookup(
    key: Tuple[str, bool], variables: Dict[str, Any]
) -> Optional[_PlanEntry]:
    with _PLAN_CACHE_LOCK:
        entries = _PLAN_CACHE.get(key)
        if not entries:
            return None
        for e in entries:
            if _guards_ok(e.guards, variables) and e.lock.acquire(
                blocking=False
            ):
                _PLAN_CACHE.move_to_end(key)  # LRU refresh
                return e  # caller releases e.lock
    return None


def _cache_store(key: Tuple[str, bool], entry: _PlanEntry) -> None:
    with _PLAN_CACHE_LOCK:
        entries = _PLAN_CACHE.setdefault(key, [])
        entries.append(entry)
        del entries[:-4]  # at most 4 bindings per query text
        while len(_PLAN_CACHE) > _PLAN_CACHE_CAP:
            _PLAN_CACHE.popitem(last=False)


def _run_and_extract(
    dag: FugueSQLWorkflow,
    yield_key: str,
    engine: Any,
    engine_conf: Any,
    as_fugue: bool,
) -> Any:
    dag.run(engine, engine_conf)
    y = dag.yields[yield_key]
    if hasattr(y, "result"):
        result = y.result
        if as_fugue:
            return result
        return result.native_as_df()
    return y


def fugue_sql(
    query: str,
    *args: Any,
    engine: Any = None,
    engine_conf: Any = None,
    as_fugue: bool = False,
    as_local: bool = False,
    plan_cache: Optional[bool] = None,
    **kwargs: Any,
) -> Any:
    """Run a FugueSQL script eagerly and return the single result (the
    script's last dataframe, or its single YIELD)."""
    variables = _capture_caller_vars()
    use_cache = (
        plan_cache
        if plan_cache is not None
        else os.environ.get("FUGUE_SQL_PLAN_CACHE", "1") != "0"
    )
    key = (query, as_local)
    if use_cache:
        merged = dict(variables)
        for a in args:
            if isinstance(a, dict):
                merged.update(a)
            else:
                use_cache = False  # non-dict positional: don't cache
        merged.update(kwargs)
    if use_cache:
        entry = _cache_lookup(key, merged)
        if entry is not None:
            try:
                entry.dag.reset_execution()
                return _run_and_extract(
                    entry.dag, entry.yield_key, engine, engine_conf,
                    as_fugue,
                )
            finally:
                entry.lock.release()
    dag = FugueSQLWorkflow()
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
    yield_key = "result" if auto_yield else list(dag.yields.keys())[0]
    res = _run_and_extract(dag, yield_key, engine, engine_conf, as_fugue)
    if use_cache and _plan_cacheable(dag._var_guards):
        _cache_store(key, _PlanEntry(dag, dict(dag._var_guards), yield_key))
    return res
