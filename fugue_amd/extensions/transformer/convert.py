"""Object → Transformer conversion chain.

Reference parity: ``fugue/extensions/transformer/convert.py`` — class
instance / decorated function / plain function with ``# schema:`` comment /
registered alias all become Transformer/CoTransformer objects.
"""
import copy
import weakref
from typing import Any, Callable, Dict, List, Optional, Tuple, Union

from fugue_amd.dataframe.array_dataframe import ArrayDataFrame
from fugue_amd.dataframe.dataframe import DataFrame, LocalDataFrame
from fugue_amd.dataframe.dataframes import DataFrames
from fugue_amd.dataframe.function_wrapper import DataFrameFunctionWrapper
from fugue_amd.exceptions import FugueInterfacelessError
from fugue_amd.extensions._utils import (
    ExtensionRegistry,
    parse_validation_rules_from_comment,
    to_validation_rules,
)
from fugue_amd.extensions.transformer.constants import (
    OUTPUT_TRANSFORMER_DUMMY_SCHEMA,
)
from fugue_amd.extensions.transformer.transformer import (
    CoTransformer,
    OutputCoTransformer,
    OutputTransformer,
    Transformer,
)
from fugue_amd.schema import Schema
from fugue_amd.utils.convert import get_full_type_path, to_function, to_instance
from fugue_amd.utils.registry import ConditionalDispatcher
from fugue_amd.utils.hash import to_uuid
from fugue_amd.utils.interfaceless import parse_output_schema_from_comment
from fugue_amd.utils.params import ParamDict

_TRANSFORMER_REGISTRY = ExtensionRegistry()
_OUTPUT_TRANSFORMER_REGISTRY = ExtensionRegistry()


def register_transformer(alias: str, obj: Any, on_dup: str = "overwrite") -> None:
    _TRANSFORMER_REGISTRY.register(alias, obj, on_dup=on_dup)


def register_output_transformer(alias: str, obj: Any, on_dup: str = "overwrite") -> None:
    _OUTPUT_TRANSFORMER_REGISTRY.register(alias, obj, on_dup=on_dup)


def transformer(schema: Any, **validation_rules: Any) -> Callable[[Callable], "_FuncAsTransformer"]:
    """Decorator: plain function → Transformer."""

    def deco(func: Callable) -> _FuncAsTransformer:
        return _FuncAsTransformer.from_func(
            func, schema, validation_rules=to_validation_rules(validation_rules)
        )

    return deco


def output_transformer(**validation_rules: Any) -> Callable[[Callable], "_FuncAsOutputTransformer"]:
    def deco(func: Callable) -> _FuncAsOutputTransformer:
        return _FuncAsOutputTransformer.from_func(
            func, None, validation_rules=to_validation_rules(validation_rules)
        )

    return deco


def cotransformer(schema: Any, **validation_rules: Any) -> Callable[[Callable], "_FuncAsCoTransformer"]:
    def deco(func: Callable) -> _FuncAsCoTransformer:
        return _FuncAsCoTransformer.from_func(
            func, schema, validation_rules=to_validation_rules(validation_rules)
        )

    return deco


def output_cotransformer(**validation_rules: Any) -> Callable[[Callable], "_FuncAsOutputCoTransformer"]:
    def deco(func: Callable) -> _FuncAsOutputCoTransformer:
        return _FuncAsOutputCoTransformer.from_func(
            func, None, validation_rules=to_validation_rules(validation_rules)
        )

    return deco


# conversion prototypes per callable (weak keys: dropping the function
# drops its cache entry); values: {(schema, rules, is_output): prototype}
_CONVERT_CACHE: "weakref.WeakKeyDictionary" = weakref.WeakKeyDictionary()


def _to_general_transformer(
    obj: Any,
    schema: Any,
    global_vars: Optional[Dict[str, Any]],
    local_vars: Optional[Dict[str, Any]],
    validation_rules: Dict[str, Any],
    is_output: bool,
    registry: ExtensionRegistry,
    func_single: Callable,
    func_co: Callable,
) -> Any:
    if isinstance(obj, str):
        reg = registry.get(obj)
        if reg is not None:
            obj = reg
    exp: Optional[Exception] = None
    try:
        if isinstance(obj, (Transformer, CoTransformer)):
            return copy.copy(obj)
        if isinstance(obj, type) and issubclass(obj, (Transformer, CoTransformer)):
            return to_instance(obj)
    except Exception as e:
        exp = e
    # plain-callable conversions are memoized: signature inspection and
    # `# schema:` comment parsing are pure in (func, schema, rules), and
    # the call sites receive a shallow copy exactly like the
    # Transformer-instance path above
    cache_key: Optional[Tuple[Any, ...]] = None
    if callable(obj) and not isinstance(obj, type):
        try:
            cache_key = (repr(schema), repr(validation_rules), is_output)
            proto_map = _CONVERT_CACHE.get(obj)
            if proto_map is not None and cache_key in proto_map:
                return copy.copy(proto_map[cache_key])
        except TypeError:  # not weak-referenceable
            cache_key = None
    try:
        f = to_function(obj, global_vars={**(global_vars or {}), **(local_vars or {})})
        # single vs co: co if first param is DataFrames or multiple df params
        wrapper = DataFrameFunctionWrapper(f)
        code = wrapper.input_code
        n_df = len([c for c in code if c in "dlspq"])
        use_dfs = "c" in code
        if use_dfs or n_df > 1:
            res = func_co(f, schema, validation_rules)
        else:
            res = func_single(f, schema, validation_rules)
        if cache_key is not None:
            try:
                _CONVERT_CACHE.setdefault(obj, {})[cache_key] = res
            except TypeError:
                pass
        return copy.copy(res) if cache_key is not None else res
    except Exception as e:
        exp = e
    raise FugueInterfacelessError(
        f"{obj} can't be converted to a transformer: {exp}"
    )


# plugin points (reference ``parse_transformer``/``parse_output_transformer``
# conditional dispatchers, ``fugue/extensions/transformer/convert.py:27``):
# backends add candidates (often with ``namespace_candidate``) that turn
# non-standard objects into transformers before the default chain runs
parse_transformer = ConditionalDispatcher("parse_transformer")
parse_output_transformer = ConditionalDispatcher("parse_output_transformer")


def _to_transformer(
    obj: Any,
    schema: Any = None,
    global_vars: Optional[Dict[str, Any]] = None,
    local_vars: Optional[Dict[str, Any]] = None,
    validation_rules: Optional[Dict[str, Any]] = None,
) -> Union[Transformer, CoTransformer]:
    ok, parsed = parse_transformer.run(obj)
    if ok:
        obj = parsed
    return _to_general_transformer(
        obj,
        schema,
        global_vars,
        local_vars,
        validation_rules or {},
        is_output=False,
        registry=_TRANSFORMER_REGISTRY,
        func_single=lambda f, s, v: _FuncAsTransformer.from_func(
            f, s, validation_rules=v
        ),
        func_co=lambda f, s, v: _FuncAsCoTransformer.from_func(
            f, s, validation_rules=v
        ),
    )


def _to_output_transformer(
    obj: Any,
    global_vars: Optional[Dict[str, Any]] = None,
    local_vars: Optional[Dict[str, Any]] = None,
    validation_rules: Optional[Dict[str, Any]] = None,
) -> Union[Transformer, CoTransformer]:
    ok, parsed = parse_output_transformer.run(obj)
    if ok:
        obj = parsed
    return _to_general_transformer(
        obj,
        None,
        global_vars,
        local_vars,
        validation_rules or {},
        is_output=True,
        registry=_OUTPUT_TRANSFORMER_REGISTRY,
        func_single=lambda f, s, v: _FuncAsOutputTransformer.from_func(
            f, s, validation_rules=v
        ),
        func_co=lambda f, s, v: _FuncAsOutputCoTransformer.from_func(
            f, s, validation_rules=v
        ),
    )


def _get_callback(ctx: Any) -> List[Any]:
    uses = getattr(ctx, "_uses_callback", False)
    if not uses:
        return []
    requires = getattr(ctx, "_requires_callback", False)
    has = ctx.has_callback
    if requires and not has:
        raise FugueInterfacelessError("callback is required but not provided")
    return [ctx.callback if has else None]


class _FuncAsTransformer(Transformer):
    """Plain function as Transformer; signature
    ``^[lspq][fF]?x*z?$ → ^[lspq]$``."""

    def validate_on_compile(self) -> None:
        super().validate_on_compile()

    def get_output_schema(self, df: DataFrame) -> Any:
        return self._parse_schema(self._output_schema_arg, df)

    def get_format_hint(self) -> Optional[str]:
        return self._format_hint

    @property
    def validation_rules(self) -> Dict[str, Any]:
        return self._validation_rules

    def transform(self, df: LocalDataFrame) -> LocalDataFrame:
        args = [df] + _get_callback(self)
        return self._wrapper.run(
            args,
            dict(self.params),
            ignore_unknown=False,
            output_schema=self.output_schema,
        )

    def __call__(self, *args: Any, **kwargs: Any) -> Any:
        return self._wrapper(*args, **kwargs)

    def __uuid__(self) -> str:
        return to_uuid(self._wrapper.__uuid__(), self._output_schema_arg)

    def _parse_schema(self, obj: Any, df: DataFrame) -> Schema:
        if callable(obj):
            return obj(df, **self.params)
        if isinstance(obj, str):
            return df.schema.transform(obj)
        if isinstance(obj, list):
            return df.schema.transform(*obj)
        raise NotImplementedError(f"can't parse schema from {obj}")

    @staticmethod
    def from_func(
        func: Callable, schema: Any, validation_rules: Dict[str, Any]
    ) -> "_FuncAsTransformer":
        if schema is None:
            schema = parse_output_schema_from_comment(func)
        if isinstance(schema, Schema):
            schema = str(schema)
        validation_rules.update(parse_validation_rules_from_comment(func))
        if schema is None:
            raise FugueInterfacelessError(
                f"schema is required for transformer {func}"
            )
        tr = _FuncAsTransformer()
        tr._wrapper = DataFrameFunctionWrapper(func, "^[dlspq][fF]?x*z?$", "^[dlspqr]$")
        tr._output_schema_arg = schema
        tr._validation_rules = validation_rules
        tr._uses_callback = "f" in tr._wrapper.input_code.lower()
        tr._requires_callback = "F" in tr._wrapper.input_code
        tr._format_hint = tr._wrapper.get_format_hint()
        return tr


class _FuncAsOutputTransformer(_FuncAsTransformer):
    def get_output_schema(self, df: DataFrame) -> Any:
        return OUTPUT_TRANSFORMER_DUMMY_SCHEMA

    def transform(self, df: LocalDataFrame) -> LocalDataFrame:
        args = [df] + _get_callback(self)
        self._wrapper.run(args, dict(self.params), ignore_unknown=False, output=False)
        return ArrayDataFrame([], OUTPUT_TRANSFORMER_DUMMY_SCHEMA)

    @staticmethod
    def from_func(
        func: Callable, schema: Any, validation_rules: Dict[str, Any]
    ) -> "_FuncAsOutputTransformer":
        if schema is not None:
            raise FugueInterfacelessError(
                "schema must be None for output transformers"
            )
        validation_rules.update(parse_validation_rules_from_comment(func))
        tr = _FuncAsOutputTransformer()
        tr._wrapper = DataFrameFunctionWrapper(
            func, "^[dlspq][fF]?x*z?$", "^[dlspqrn]$"
        )
        tr._output_schema_arg = None
        tr._validation_rules = validation_rules
        tr._uses_callback = "f" in tr._wrapper.input_code.lower()
        tr._requires_callback = "F" in tr._wrapper.input_code
        tr._format_hint = tr._wrapper.get_format_hint()
        return tr


class _FuncAsCoTransformer(CoTransformer):
    def validate_on_compile(self) -> None:
        super().validate_on_compile()

    def get_output_schema(self, dfs: DataFrames) -> Any:
        return self._parse_schema(self._output_schema_arg, dfs)

    def get_format_hint(self) -> Optional[str]:
        return self._format_hint

    @property
    def validation_rules(self) -> Dict[str, Any]:
        return self._validation_rules

    def transform(self, dfs: DataFrames) -> LocalDataFrame:
        cb = _get_callback(self)
        if self._dfs_input:
            return self._wrapper.run(
                [dfs] + cb,
                dict(self.params),
                ignore_unknown=False,
                output_schema=self.output_schema,
            )
        if not dfs.has_key:
            return self._wrapper.run(
                list(dfs.values()) + cb,
                dict(self.params),
                ignore_unknown=False,
                output_schema=self.output_schema,
            )
        p = dict(dfs)
        p.update(self.params)
        return self._wrapper.run(
            [] + cb, p, ignore_unknown=False, output_schema=self.output_schema
        )

    def __call__(self, *args: Any, **kwargs: Any) -> Any:
        return self._wrapper(*args, **kwargs)

    def __uuid__(self) -> str:
        return to_uuid(
            self._wrapper.__uuid__(), self._output_schema_arg, self._dfs_input
        )

    def _parse_schema(self, obj: Any, dfs: DataFrames) -> Schema:
        if callable(obj):
            return obj(dfs, **self.params)
        if isinstance(obj, str):
            return Schema(obj)
        if isinstance(obj, list):
            return Schema(*obj)
        raise NotImplementedError(f"can't parse schema from {obj}")

    @staticmethod
    def from_func(
        func: Callable, schema: Any, validation_rules: Dict[str, Any]
    ) -> "_FuncAsCoTransformer":
        if len(validation_rules) > 0:
            for k in validation_rules:
                if k.startswith("input"):
                    raise NotImplementedError(
                        "input validation is not supported for cotransformers"
                    )
        validation_rules.update(parse_validation_rules_from_comment(func))
        if schema is None:
            schema = parse_output_schema_from_comment(func)
        if isinstance(schema, Schema):
            schema = str(schema)
        if schema is None:
            raise FugueInterfacelessError(
                f"schema is required for cotransformer {func}"
            )
        tr = _FuncAsCoTransformer()
        tr._wrapper = DataFrameFunctionWrapper(
            func, "^(c|[dlspq]+)[fF]?x*z?$", "^[dlspqr]$"
        )
        tr._dfs_input = tr._wrapper.input_code.startswith("c")
        tr._output_schema_arg = schema
        tr._validation_rules = validation_rules
        tr._uses_callback = "f" in tr._wrapper.input_code.lower()
        tr._requires_callback = "F" in tr._wrapper.input_code
        tr._format_hint = tr._wrapper.get_format_hint()
        return tr


class _FuncAsOutputCoTransformer(_FuncAsCoTransformer):
    def get_output_schema(self, dfs: DataFrames) -> Any:
        return OUTPUT_TRANSFORMER_DUMMY_SCHEMA

    def transform(self, dfs: DataFrames) -> LocalDataFrame:
        cb = _get_callback(self)
        if self._dfs_input:
            self._wrapper.run(
                [dfs] + cb, dict(self.params), ignore_unknown=False, output=False
            )
        elif not dfs.has_key:
            self._wrapper.run(
                list(dfs.values()) + cb,
                dict(self.params),
                ignore_unknown=False,
                output=False,
            )
        else:
            p = dict(dfs)
            p.update(self.params)
            self._wrapper.run([] + cb, p, ignore_unknown=False, output=False)
        return ArrayDataFrame([], OUTPUT_TRANSFORMER_DUMMY_SCHEMA)

    @staticmethod
    def from_func(
        func: Callable, schema: Any, validation_rules: Dict[str, Any]
    ) -> "_FuncAsOutputCoTransformer":
        if schema is not None:
            raise FugueInterfacelessError(
                "schema must be None for output cotransformers"
            )
        validation_rules.update(parse_validation_rules_from_comment(func))
        tr = _FuncAsOutputCoTransformer()
        tr._wrapper = DataFrameFunctionWrapper(
            func, "^(c|[dlspq]+)[fF]?x*z?$", "^[dlspqrn]$"
        )
        tr._dfs_input = tr._wrapper.input_code.startswith("c")
        tr._output_schema_arg = None
        tr._validation_rules = validation_rules
        tr._uses_callback = "f" in tr._wrapper.input_code.lower()
        tr._requires_callback = "F" in tr._wrapper.input_code
        tr._format_hint = tr._wrapper.get_format_hint()
        return tr
