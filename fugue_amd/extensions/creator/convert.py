"""Object → Creator conversion (reference: ``fugue/extensions/creator/convert.py``)."""
import copy
from typing import Any, Callable, Dict, Optional

from fugue_amd.dataframe.dataframe import DataFrame
from fugue_amd.dataframe.function_wrapper import DataFrameFunctionWrapper
from fugue_amd.exceptions import FugueInterfacelessError
from fugue_amd.extensions._utils import ExtensionRegistry
from fugue_amd.extensions.creator.creator import Creator
from fugue_amd.schema import Schema
from fugue_amd.utils.convert import to_function, to_instance
from fugue_amd.utils.hash import to_uuid
from fugue_amd.utils.registry import ConditionalDispatcher
from fugue_amd.utils.interfaceless import parse_output_schema_from_comment

_CREATOR_REGISTRY = ExtensionRegistry()


def register_creator(alias: str, obj: Any, on_dup: str = "overwrite") -> None:
    _CREATOR_REGISTRY.register(alias, obj, on_dup=on_dup)


def creator(schema: Any = None) -> Callable[[Callable], "_FuncAsCreator"]:
    def deco(func: Callable) -> _FuncAsCreator:
        return _FuncAsCreator.from_func(func, schema)

    return deco


# plugin point (reference ``parse_creator`` conditional dispatcher)
parse_creator = ConditionalDispatcher("parse_creator")


def _to_creator(
    obj: Any,
    schema: Any = None,
    global_vars: Optional[Dict[str, Any]] = None,
    local_vars: Optional[Dict[str, Any]] = None,
) -> Creator:
    ok, parsed = parse_creator.run(obj)
    if ok:
        obj = parsed
    if isinstance(obj, str):
        reg = _CREATOR_REGISTRY.get(obj)
        if reg is not None:
            obj = reg
    exp: Optional[Exception] = None
    try:
        if isinstance(obj, Creator):
            return copy.copy(obj)
        if isinstance(obj, type) and issubclass(obj, Creator):
            return to_instance(obj)
    except Exception as e:
        exp = e
    try:
        f = to_function(obj, global_vars={**(global_vars or {}), **(local_vars or {})})
        return _FuncAsCreator.from_func(f, schema)
    except Exception as e:
        exp = e
    raise FugueInterfacelessError(f"{obj} can't be converted to a creator: {exp}")


class _FuncAsCreator(Creator):
    """Plain function as Creator; signature ``^e?x*z?$ → ^[dlspq]$``."""

    def create(self) -> DataFrame:
        args = []
        if self._engine_param:
            args.append(self.execution_engine)
        schema = (
            None if self._output_schema_arg is None else Schema(self._output_schema_arg)
        )
        return self._wrapper.run(
            args,
            dict(self.params),
            ignore_unknown=False,
            output_schema=schema,
        )

    def __call__(self, *args: Any, **kwargs: Any) -> Any:
        return self._wrapper(*args, **kwargs)

    def __uuid__(self) -> str:
        return to_uuid(self._wrapper.__uuid__(), self._output_schema_arg)

    @staticmethod
    def from_func(func: Callable, schema: Any) -> "_FuncAsCreator":
        if schema is None:
            schema = parse_output_schema_from_comment(func)
        if isinstance(schema, Schema):
            schema = str(schema)
        tr = _FuncAsCreator()
        tr._wrapper = DataFrameFunctionWrapper(func, "^e?x*z?$", "^[dlspq]$")
        tr._engine_param = tr._wrapper.input_code.startswith("e")
        tr._output_schema_arg = schema
        tr._need_output_schema = tr._wrapper.need_output_schema
        if tr._need_output_schema and schema is None:
            raise FugueInterfacelessError(
                f"schema is required for creator {func} (output type needs schema)"
            )
        return tr
