"""Object → Processor conversion (reference: ``fugue/extensions/processor/convert.py``)."""
import copy
from typing import Any, Callable, Dict, Optional

from fugue_amd.dataframe.dataframe import DataFrame
from fugue_amd.dataframe.dataframes import DataFrames
from fugue_amd.dataframe.function_wrapper import DataFrameFunctionWrapper
from fugue_amd.exceptions import FugueInterfacelessError
from fugue_amd.extensions._utils import (
    ExtensionRegistry,
    parse_validation_rules_from_comment,
    to_validation_rules,
)
from fugue_amd.extensions.processor.processor import Processor
from fugue_amd.schema import Schema
from fugue_amd.utils.convert import to_function, to_instance
from fugue_amd.utils.hash import to_uuid
from fugue_amd.utils.registry import ConditionalDispatcher
from fugue_amd.utils.interfaceless import parse_output_schema_from_comment

_PROCESSOR_REGISTRY = ExtensionRegistry()


def register_processor(alias: str, obj: Any, on_dup: str = "overwrite") -> None:
    _PROCESSOR_REGISTRY.register(alias, obj, on_dup=on_dup)


def processor(schema: Any = None, **validation_rules: Any) -> Callable[[Callable], "_FuncAsProcessor"]:
    def deco(func: Callable) -> _FuncAsProcessor:
        return _FuncAsProcessor.from_func(
            func, schema, validation_rules=to_validation_rules(validation_rules)
        )

    return deco


# plugin point (reference ``parse_processor`` conditional dispatcher)
parse_processor = ConditionalDispatcher("parse_processor")


def _to_processor(
    obj: Any,
    schema: Any = None,
    global_vars: Optional[Dict[str, Any]] = None,
    local_vars: Optional[Dict[str, Any]] = None,
    validation_rules: Optional[Dict[str, Any]] = None,
) -> Processor:
    ok, parsed = parse_processor.run(obj)
    if ok:
        obj = parsed
    if isinstance(obj, str):
        reg = _PROCESSOR_REGISTRY.get(obj)
        if reg is not None:
            obj = reg
    exp: Optional[Exception] = None
    try:
        if isinstance(obj, Processor):
            return copy.copy(obj)
        if isinstance(obj, type) and issubclass(obj, Processor):
            return to_instance(obj)
    except Exception as e:
        exp = e
    try:
        f = to_function(obj, global_vars={**(global_vars or {}), **(local_vars or {})})
        return _FuncAsProcessor.from_func(
            f, schema, validation_rules=validation_rules or {}
        )
    except Exception as e:
        exp = e
    raise FugueInterfacelessError(f"{obj} can't be converted to a processor: {exp}")


class _FuncAsProcessor(Processor):
    """Plain function as Processor; signature ``^e?(c|[dlspq]+)x*z?$ → ^[dlspq]$``."""

    @property
    def validation_rules(self) -> Dict[str, Any]:
        return self._validation_rules

    def process(self, dfs: DataFrames) -> DataFrame:
        args: list = []
        if self._engine_param:
            args.append(self.execution_engine)
        schema = (
            None if self._output_schema_arg is None else Schema(self._output_schema_arg)
        )
        if self._use_dfs:
            args.append(dfs)
            return self._wrapper.run(
                args, dict(self.params), ignore_unknown=False, output_schema=schema
            )
        if not dfs.has_key:
            args.extend(dfs.values())
            return self._wrapper.run(
                args, dict(self.params), ignore_unknown=False, output_schema=schema
            )
        p = dict(dfs)
        p.update(self.params)
        return self._wrapper.run(
            args, p, ignore_unknown=False, output_schema=schema
        )

    def __call__(self, *args: Any, **kwargs: Any) -> Any:
        return self._wrapper(*args, **kwargs)

    def __uuid__(self) -> str:
        return to_uuid(self._wrapper.__uuid__(), self._output_schema_arg)

    @staticmethod
    def from_func(
        func: Callable, schema: Any, validation_rules: Dict[str, Any]
    ) -> "_FuncAsProcessor":
        if schema is None:
            schema = parse_output_schema_from_comment(func)
        if isinstance(schema, Schema):
            schema = str(schema)
        validation_rules.update(parse_validation_rules_from_comment(func))
        tr = _FuncAsProcessor()
        tr._wrapper = DataFrameFunctionWrapper(
            func, "^e?(c|[dlspq]+)x*z?$", "^[dlspq]$"
        )
        tr._engine_param = tr._wrapper.input_code.startswith("e")
        tr._use_dfs = "c" in tr._wrapper.input_code
        tr._output_schema_arg = schema
        tr._validation_rules = validation_rules
        if tr._wrapper.need_output_schema and schema is None:
            raise FugueInterfacelessError(
                f"schema is required for processor {func} (output type needs schema)"
            )
        return tr
