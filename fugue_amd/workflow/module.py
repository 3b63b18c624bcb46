"""Workflow module composition (reference: ``fugue/workflow/module.py``).

A *module* is a plain function whose annotated parameters are
``FugueWorkflow`` / ``WorkflowDataFrame`` / ``WorkflowDataFrames`` (plus
ordinary keyword params) and whose return annotation is one of the
dataframe types (or nothing).  ``module`` wraps it so that

* the target workflow is inferred from the input dataframes when the
  function wants a ``FugueWorkflow`` first argument but the caller did
  not pass one (reference ``_ModuleFunctionWrapper.__call__``), and
* FugueSQL's ``SUB ... USING fn`` statement can introspect the inputs /
  outputs to dispatch correctly (reference ``_visitors.py:697``
  ``visitFugueModuleTask``).

Unlike the reference this does not route through a generic
function-wrapper registry; the signature classification is done directly
with ``inspect`` since modules only distinguish four parameter kinds.
"""
import inspect
from typing import Any, Callable, Dict, List, Optional, Tuple

from fugue_amd.exceptions import FugueInterfacelessError
from fugue_amd.workflow.workflow import (
    FugueWorkflow,
    WorkflowDataFrame,
    WorkflowDataFrames,
)


class ModuleFunction:
    """A callable wrapping a module function, exposing its input/output
    shape for the FugueSQL ``SUB`` statement."""

    def __init__(self, fn: Callable):
        if isinstance(fn, ModuleFunction):  # idempotent
            fn = fn.fn
        self.fn = fn
        self.__name__ = getattr(fn, "__name__", repr(fn))
        self.__doc__ = fn.__doc__
        sig = inspect.signature(fn)
        self._param_kinds: List[Tuple[str, str]] = []  # (name, kind)
        for name, p in sig.parameters.items():
            ann = p.annotation
            if inspect.isclass(ann) and issubclass(ann, FugueWorkflow):
                kind = "workflow"
            elif ann is WorkflowDataFrame:
                kind = "df"
            elif ann is WorkflowDataFrames:
                kind = "dfs"
            else:
                kind = "param"
            self._param_kinds.append((name, kind))
        kinds = [k for _, k in self._param_kinds]
        if "workflow" in kinds and kinds.index("workflow") != 0:
            raise FugueInterfacelessError(
                f"{self.__name__}: FugueWorkflow must be the first parameter"
            )
        ret = sig.return_annotation
        if ret is WorkflowDataFrame:
            self._output = "df"
        elif ret is WorkflowDataFrames:
            self._output = "dfs"
        else:
            self._output = "none"

    # --- shape ---------------------------------------------------------- #
    @property
    def has_input(self) -> bool:
        return any(k in ("df", "dfs") for _, k in self._param_kinds)

    @property
    def has_dfs_input(self) -> bool:
        return any(k == "dfs" for _, k in self._param_kinds)

    @property
    def has_single_output(self) -> bool:
        return self._output == "df"

    @property
    def has_multiple_output(self) -> bool:
        return self._output == "dfs"

    @property
    def has_no_output(self) -> bool:
        return self._output == "none"

    # --- invocation ------------------------------------------------------ #
    def __call__(self, *args: Any, **kwargs: Any) -> Any:
        if self._needs_workflow_prepended(*args, **kwargs):
            wf = self._infer_workflow(*args, **kwargs)
            if wf is None:
                raise ValueError(
                    f"can't infer workflow for module {self.__name__}"
                )
            return self.fn(wf, *args, **kwargs)
        return self.fn(*args, **kwargs)

    def _needs_workflow_prepended(self, *args: Any, **kwargs: Any) -> bool:
        if len(self._param_kinds) == 0 or self._param_kinds[0][1] != "workflow":
            return False
        if self._param_kinds[0][0] in kwargs:
            return False
        if len(args) > 0 and isinstance(args[0], FugueWorkflow):
            return False
        return True

    @staticmethod
    def _infer_workflow(*args: Any, **kwargs: Any) -> Optional[FugueWorkflow]:
        wf: Optional[FugueWorkflow] = None

        def visit(v: Any) -> None:
            nonlocal wf
            if isinstance(v, WorkflowDataFrame):
                if wf is not None and v.workflow is not wf:
                    raise ValueError(
                        "different parent workflows found on input dataframes"
                    )
                wf = v.workflow
            elif isinstance(v, WorkflowDataFrames):
                for item in v.values():
                    visit(item)

        for a in args:
            visit(a)
        for v in kwargs.values():
            visit(v)
        return wf


def module(
    func: Optional[Callable] = None,
    as_method: bool = False,
    name: Optional[str] = None,
    on_dup: str = "overwrite",
) -> Any:
    """Decorator: mark a function as a workflow module.  With
    ``as_method=True`` the module is also attached to
    ``WorkflowDataFrame`` under ``name`` (reference ``module.py:20``)."""

    if func is None:
        return lambda fn: module(fn, as_method=as_method, name=name, on_dup=on_dup)
    res = ModuleFunction(func)
    if as_method:
        mname = name or func.__name__
        if on_dup == "error" and hasattr(WorkflowDataFrame, mname):
            raise ValueError(f"WorkflowDataFrame.{mname} already exists")

        def method(self: WorkflowDataFrame, *args: Any, **kwargs: Any) -> Any:
            return func(self, *args, **kwargs)

        setattr(WorkflowDataFrame, mname, method)
    return res


def to_module(
    obj: Any, resolvers: Optional[Dict[str, Any]] = None
) -> ModuleFunction:
    """Resolve ``obj`` (a ModuleFunction, a callable, a variable name, or
    a dotted import path) into a :class:`ModuleFunction`.  ``resolvers``
    is the name→object map to consult for string names (FugueSQL passes
    the captured caller variables)."""
    if isinstance(obj, ModuleFunction):
        return obj
    if callable(obj):
        return ModuleFunction(obj)
    if isinstance(obj, str):
        if resolvers and obj in resolvers:
            return to_module(resolvers[obj])
        if "." in obj:  # dotted import path
            import importlib

            mod_name, _, attr = obj.rpartition(".")
            try:
                m = importlib.import_module(mod_name)
                return to_module(getattr(m, attr))
            except (ImportError, AttributeError) as e:
                raise FugueInterfacelessError(
                    f"{obj} is not a valid module"
                ) from e
    raise FugueInterfacelessError(f"{obj} is not a valid module")
