"""Builtin engine registration (reference parity: ``fugue/registry.py:8`` —
aliases "native"/"pandas" and the default engine)."""
from typing import Any

from fugue_amd.execution.factory import register_execution_engine


def _make_native(conf: Any):
    from fugue_amd.execution.native_execution_engine import NativeExecutionEngine

    return NativeExecutionEngine(conf)


_done = [False]


def register_builtins() -> None:
    if _done[0]:
        return
    _done[0] = True
    for alias in ("native", "pandas"):
        register_execution_engine(alias, _make_native, on_dup="ignore")
