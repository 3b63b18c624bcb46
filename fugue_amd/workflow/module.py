"""Workflow module composition (reference: ``fugue/workflow/module.py:20``).

A module is a function taking a FugueWorkflow (and optionally
WorkflowDataFrames) that appends sub-DAGs; ``module`` makes it reusable.
"""
import inspect
from typing import Any, Callable, Optional

from fugue_amd.workflow.workflow import FugueWorkflow, WorkflowDataFrame


def module(func: Optional[Callable] = None, as_method: bool = False, name: Optional[str] = None, on_dup: str = "overwrite") -> Any:
    """Decorator: mark a function as a workflow module.  The wrapped
    function's first workflow/dataframe argument determines the target
    workflow."""

    def deco(fn: Callable) -> Callable:
        sig = inspect.signature(fn)

        def wrapper(*args: Any, **kwargs: Any) -> Any:
            return fn(*args, **kwargs)

        wrapper.__name__ = fn.__name__
        wrapper.__module__ = fn.__module__
        wrapper.__doc__ = fn.__doc__
        wrapper._is_fugue_module = True  # type: ignore
        if as_method:
            mname = name or fn.__name__

            def method(self: WorkflowDataFrame, *args: Any, **kwargs: Any) -> Any:
                return fn(self, *args, **kwargs)

            setattr(WorkflowDataFrame, mname, method)
        return wrapper

    if func is not None:
        return deco(func)
    return deco
