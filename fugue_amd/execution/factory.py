"""Engine factory/registry: name/type/instance → engine resolution.

Reference parity: ``fugue/execution/factory.py`` — resolution priority is
context > global > infer > default.
"""
import threading
from typing import Any, Callable, Dict, List, Optional, Tuple, Type, Union

from fugue_amd.dataframe.dataframe import DataFrame
from fugue_amd.exceptions import FuguePluginsRegistrationError
from fugue_amd.execution.execution_engine import (
    ExecutionEngine,
    SQLEngine,
    _FUGUE_EXECUTION_ENGINE_CONTEXT,
    _GLOBAL_ENGINE,
)
from fugue_amd.utils.params import ParamDict
from fugue_amd.utils.registry import ConditionalDispatcher

# plugin points (reference ``parse_execution_engine`` factory.py:342 /
# ``parse_sql_engine`` :508): candidates convert non-standard engine
# descriptors (e.g. namespace tuples) before the built-in resolution
parse_execution_engine = ConditionalDispatcher("parse_execution_engine")
parse_sql_engine = ConditionalDispatcher("parse_sql_engine")

_LOCK = threading.RLock()
_ENGINE_REGISTRY: Dict[str, Callable[[Any], ExecutionEngine]] = {}
_ENGINE_TYPE_REGISTRY: Dict[type, Callable[[Any, Any], ExecutionEngine]] = {}
_SQL_ENGINE_REGISTRY: Dict[str, Callable[[ExecutionEngine], SQLEngine]] = {}
_DEFAULT_FACTORY: List[Optional[Callable[[Any], ExecutionEngine]]] = [None]
_INFER_FUNCS: List[Callable[[List[Any]], Optional[Any]]] = []


def register_execution_engine(
    name_or_type: Union[str, type],
    func: Callable,
    on_dup: str = "overwrite",
) -> None:
    with _LOCK:
        if isinstance(name_or_type, str):
            if name_or_type in _ENGINE_REGISTRY and on_dup == "throw":
                raise FuguePluginsRegistrationError(
                    f"engine {name_or_type} already registered"
                )
            if name_or_type in _ENGINE_REGISTRY and on_dup == "ignore":
                return
            _ENGINE_REGISTRY[name_or_type] = func
        else:
            _ENGINE_TYPE_REGISTRY[name_or_type] = func


def register_default_execution_engine(func: Callable, on_dup: str = "overwrite") -> None:
    with _LOCK:
        if _DEFAULT_FACTORY[0] is not None and on_dup == "throw":
            raise FuguePluginsRegistrationError("default engine already registered")
        if _DEFAULT_FACTORY[0] is not None and on_dup == "ignore":
            return
        _DEFAULT_FACTORY[0] = func


def register_sql_engine(name: str, func: Callable, on_dup: str = "overwrite") -> None:
    with _LOCK:
        if name in _SQL_ENGINE_REGISTRY and on_dup == "throw":
            raise FuguePluginsRegistrationError(f"sql engine {name} already registered")
        if name in _SQL_ENGINE_REGISTRY and on_dup == "ignore":
            return
        _SQL_ENGINE_REGISTRY[name] = func


def register_engine_inference(func: Callable[[List[Any]], Optional[Any]]) -> None:
    """Register a function that looks at input objects and may return an
    engine-identifying object (reference: ``infer_execution_engine``
    plugin, ``fugue/execution/factory.py:420``)."""
    with _LOCK:
        _INFER_FUNCS.insert(0, func)


def infer_execution_engine(objs: List[Any]) -> Optional[Any]:
    with _LOCK:
        funcs = list(_INFER_FUNCS)
    for f in funcs:
        try:
            res = f(objs)
        except Exception:
            continue
        if res is not None:
            return res
    return None


def try_get_context_execution_engine() -> Optional[ExecutionEngine]:
    """Context engine (``with engine.as_context()``), else global engine."""
    engine = _FUGUE_EXECUTION_ENGINE_CONTEXT.get()
    if engine is not None:
        return engine
    return _GLOBAL_ENGINE[0]


def make_execution_engine(
    engine: Any = None,
    conf: Any = None,
    infer_by: Optional[List[Any]] = None,
    **kwargs: Any,
) -> ExecutionEngine:
    if isinstance(engine, ExecutionEngine):
        if conf is not None:
            engine.conf.update_params(conf)
        engine.conf.update_params(kwargs)
        return engine
    if isinstance(engine, tuple):
        e = make_execution_engine(engine[0], conf, infer_by=infer_by, **kwargs)
        e.sql_engine = make_sql_engine(engine[1], e)
        return e
    if engine is None:
        ctx = try_get_context_execution_engine()
        if ctx is not None:
            if conf is not None:
                ctx.conf.update_params(conf)
            return ctx
        if infer_by is not None:
            inferred = infer_execution_engine(infer_by)
            if inferred is not None:
                return make_execution_engine(inferred, conf, **kwargs)
        return _make_default(conf, **kwargs)
    merged = ParamDict(conf)
    merged.update_params(kwargs)
    if isinstance(engine, str):
        with _LOCK:
            func = _ENGINE_REGISTRY.get(engine)
        if func is None and engine in ("hip", "mi355x", "gpu"):
            import fugue_amd.hip  # noqa: F401  (registers the engine)

            with _LOCK:
                func = _ENGINE_REGISTRY.get(engine)
        if func is None:
            raise ValueError(f"execution engine {engine!r} is not registered")
        return func(merged)
    with _LOCK:
        items = list(_ENGINE_TYPE_REGISTRY.items())
    for tp, func in items:
        if isinstance(engine, tp):
            return func(engine, merged)
    ok, parsed = parse_execution_engine.run(engine, merged)
    if ok:
        return parsed
    raise ValueError(f"can't make execution engine from {engine!r}")


def _make_default(conf: Any, **kwargs: Any) -> ExecutionEngine:
    merged = ParamDict(conf)
    merged.update_params(kwargs)
    with _LOCK:
        factory = _DEFAULT_FACTORY[0]
    if factory is not None:
        return factory(merged)
    from fugue_amd.execution.native_execution_engine import NativeExecutionEngine

    return NativeExecutionEngine(merged)


def register_default_sql_engine(func: Callable, on_dup: str = "overwrite") -> None:
    register_sql_engine("__default__", func, on_dup=on_dup)


def make_sql_engine(
    engine: Any = None,
    execution_engine: Optional[ExecutionEngine] = None,
    **kwargs: Any,
) -> SQLEngine:
    if isinstance(engine, SQLEngine):
        return engine
    if engine is None:
        with _LOCK:
            func = _SQL_ENGINE_REGISTRY.get("__default__")
        if func is not None:
            return func(execution_engine)
        assert execution_engine is not None
        return execution_engine.create_default_sql_engine()
    if isinstance(engine, str):
        with _LOCK:
            func = _SQL_ENGINE_REGISTRY.get(engine)
        if func is None:
            raise ValueError(f"sql engine {engine!r} is not registered")
        return func(execution_engine, **kwargs)
    if isinstance(engine, type) and issubclass(engine, SQLEngine):
        return engine(execution_engine, **kwargs)
    ok, parsed = parse_sql_engine.run(engine, execution_engine, **kwargs)
    if ok:
        return parsed
    raise ValueError(f"can't make sql engine from {engine!r}")


def is_pandas_or(objs: List[Any], obj_type: Any) -> bool:
    """Whether all objs are pandas-like or of obj_type (used by engine
    inference)."""
    import pandas as pd

    return all(
        isinstance(o, (pd.DataFrame, obj_type))
        or (isinstance(o, DataFrame) and o.is_local)
        for o in objs
    )
