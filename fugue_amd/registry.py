"""Builtin engine registration (reference parity: ``fugue/registry.py:8`` —
aliases "native"/"pandas" and the default engine)."""
from typing import Any

from fugue_amd.execution.factory import register_execution_engine


def _make_native(conf: Any):
    from fugue_amd.execution.native_execution_engine import NativeExecutionEngine

    return NativeExecutionEngine(conf)


_done = [False]
_plugins_done = [False]

FUGUE_ENTRYPOINT_GROUPS = ("fugue.plugins", "fugue_amd.plugins")


def _make_pandas_sql(engine: Any, conf: Any = None, **kwargs: Any):
    from fugue_amd.execution.native_execution_engine import PandasSQLEngine

    return PandasSQLEngine(engine, **kwargs)


def register_builtins() -> None:
    if _done[0]:
        return
    _done[0] = True
    from fugue_amd.execution.factory import register_sql_engine

    for alias in ("native", "pandas"):
        register_execution_engine(alias, _make_native, on_dup="ignore")
    # the local pandas SQL facet (reference alias "qpdpandas")
    for alias in ("native", "pandas", "qpdpandas"):
        register_sql_engine(alias, _make_pandas_sql, on_dup="ignore")


def load_entry_point_plugins() -> int:
    """Load third-party plugins advertised via setuptools entry points in
    the ``fugue.plugins`` / ``fugue_amd.plugins`` groups (reference
    parity: ``fugue/constants.py:7`` ``FUGUE_ENTRYPOINT`` +
    triad's ``load_entry_point``).  Each entry point is imported once;
    a plugin that raises is skipped with a warning so one broken package
    can't take down the framework.  Returns the number of plugins
    loaded on this call."""
    if _plugins_done[0]:
        return 0
    _plugins_done[0] = True
    import logging
    from importlib.metadata import entry_points

    n = 0
    for group in FUGUE_ENTRYPOINT_GROUPS:
        try:
            eps = entry_points(group=group)  # type: ignore[call-arg]
        except TypeError:  # pragma: no cover - py<3.10 dict API
            eps = entry_points().get(group, [])  # type: ignore[attr-defined]
        except Exception:  # pragma: no cover
            continue
        for ep in eps:
            try:
                obj = ep.load()
                if callable(obj):
                    obj()
                n += 1
            except Exception as e:  # pragma: no cover
                logging.getLogger("fugue_amd").warning(
                    "failed to load plugin %s: %s", ep, e
                )
    return n
