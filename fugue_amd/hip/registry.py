"""Engine registration: aliases "hip" / "mi355x" / "gpu", frame inference.

Reference parity: the per-backend registries
(``fugue_spark/registry.py`` pattern).
"""
from typing import Any, List, Optional

from fugue_amd.execution.factory import (
    register_engine_inference,
    register_execution_engine,
)
from fugue_amd.utils.registry import register_plugin


def _make_engine(conf: Any):
    from fugue_amd.hip.execution_engine import HipExecutionEngine

    return HipExecutionEngine(conf)


def _infer_hip(objs: List[Any]) -> Optional[str]:
    from fugue_amd.hip.frame import HipDataFrame

    for o in objs:
        if isinstance(o, HipDataFrame):
            return "hip"
    return None


_registered = [False]


def register_hip_engine() -> None:
    if _registered[0]:
        return
    _registered[0] = True
    for alias in ("hip", "mi355x", "gpu"):
        register_execution_engine(alias, _make_engine, on_dup="ignore")
    register_engine_inference(_infer_hip)

    def _is_hip_df(df: Any, **kwargs: Any) -> bool:
        from fugue_amd.hip.frame import HipDataFrame

        return isinstance(df, HipDataFrame)

    register_plugin("as_fugue_df", _is_hip_df, lambda df, **k: df)
