"""Object → Outputter conversion (reference: ``fugue/extensions/outputter/convert.py``)."""
import copy
from typing import Any, Callable, Dict, Optional

from fugue_amd.dataframe.dataframes import DataFrames
from fugue_amd.dataframe.function_wrapper import DataFrameFunctionWrapper
from fugue_amd.exceptions import FugueInterfacelessError
from fugue_amd.extensions._utils import (
    ExtensionRegistry,
    parse_validation_rules_from_comment,
    to_validation_rules,
)
from fugue_amd.extensions.outputter.outputter import Outputter
from fugue_amd.utils.convert import to_function, to_instance
from fugue_amd.utils.hash import to_uuid
from fugue_amd.utils.registry import ConditionalDispatcher

_OUTPUTTER_REGISTRY = ExtensionRegistry()


def register_outputter(alias: str, obj: Any, on_dup: str = "overwrite") -> None:
    _OUTPUTTER_REGISTRY.register(alias, obj, on_dup=on_dup)


def outputter(**validation_rules: Any) -> Callable[[Callable], "_FuncAsOutputter"]:
    def deco(func: Callable) -> _FuncAsOutputter:
        return _FuncAsOutputter.from_func(
            func, validation_rules=to_validation_rules(validation_rules)
        )

    return deco


# plugin point (reference ``parse_outputter`` conditional dispatcher)
parse_outputter = ConditionalDispatcher("parse_outputter")


def _to_outputter(
    obj: Any,
    global_vars: Optional[Dict[str, Any]] = None,
    local_vars: Optional[Dict[str, Any]] = None,
    validation_rules: Optional[Dict[str, Any]] = None,
) -> Outputter:
    ok, parsed = parse_outputter.run(obj)
    if ok:
        obj = parsed
    if isinstance(obj, str):
        reg = _OUTPUTTER_REGISTRY.get(obj)
        if reg is not None:
            obj = reg
    exp: Optional[Exception] = None
    try:
        if isinstance(obj, Outputter):
            return copy.copy(obj)
        if isinstance(obj, type) and issubclass(obj, Outputter):
            return to_instance(obj)
    except Exception as e:
        exp = e
    try:
        f = to_function(obj, global_vars={**(global_vars or {}), **(local_vars or {})})
        return _FuncAsOutputter.from_func(f, validation_rules=validation_rules or {})
    except Exception as e:
        exp = e
    raise FugueInterfacelessError(f"{obj} can't be converted to an outputter: {exp}")


class _FuncAsOutputter(Outputter):
    """Plain function as Outputter; signature ``^e?(c|[dlspq]+)x*z?$ → ^n$``."""

    @property
    def validation_rules(self) -> Dict[str, Any]:
        return self._validation_rules

    def process(self, dfs: DataFrames) -> None:
        args: list = []
        if self._engine_param:
            args.append(self.execution_engine)
        if self._use_dfs:
            args.append(dfs)
            self._wrapper.run(args, dict(self.params), ignore_unknown=False, output=False)
            return
        if not dfs.has_key:
            args.extend(dfs.values())
            self._wrapper.run(args, dict(self.params), ignore_unknown=False, output=False)
            return
        p = dict(dfs)
        p.update(self.params)
        self._wrapper.run(args, p, ignore_unknown=False, output=False)

    def __call__(self, *args: Any, **kwargs: Any) -> Any:
        return self._wrapper(*args, **kwargs)

    def __uuid__(self) -> str:
        return to_uuid(self._wrapper.__uuid__())

    @staticmethod
    def from_func(func: Callable, validation_rules: Dict[str, Any]) -> "_FuncAsOutputter":
        validation_rules.update(parse_validation_rules_from_comment(func))
        tr = _FuncAsOutputter()
        tr._wrapper = DataFrameFunctionWrapper(func, "^e?(c|[dlspq]+)x*z?$", "^n$")
        tr._engine_param = tr._wrapper.input_code.startswith("e")
        tr._use_dfs = "c" in tr._wrapper.input_code
        tr._validation_rules = validation_rules
        return tr
