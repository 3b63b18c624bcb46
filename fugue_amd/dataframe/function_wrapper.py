"""Annotation-dispatched UDF wrapping (the "interfaceless" machinery).

Reference parity: ``fugue/dataframe/function_wrapper.py`` +
``triad.collections.function_wrapper`` — maps Python type annotations ⇄
frame representations so plain functions become transformers / processors /
creators / outputters.  New implementation; the code-char scheme kept for
regex-based signature validation:

=====  =======================================================
code   annotation
=====  =======================================================
``e``  ExecutionEngine (and subclasses)
``c``  DataFrames (multi-input)
``d``  DataFrame (any fugue frame)
``l``  LocalDataFrame
``s``  schema-less local collections: List[List[Any]],
       Iterable[List[Any]], EmptyAwareIterable[List[Any]],
       List[Dict[str, Any]], Iterable[Dict[str, Any]],
       EmptyAwareIterable[Dict[str, Any]]
``p``  pd.DataFrame, Iterable[pd.DataFrame]
``q``  pa.Table, Iterable[pa.Table]
``f``  Optional[Callable] (callback, optional)
``F``  Callable (callback, required)
``x``  any other named parameter
``z``  ``**kwargs``
``n``  None return
=====  =======================================================
"""
import inspect
import re
from typing import (
    Any,
    Callable,
    Dict,
    Iterable,
    Iterator,
    List,
    Optional,
    Union,
    get_args,
    get_origin,
)

import pandas as pd
import pyarrow as pa

from fugue_amd.dataframe.array_dataframe import ArrayDataFrame
from fugue_amd.dataframe.arrow_dataframe import ArrowDataFrame
from fugue_amd.dataframe.dataframe import (
    DataFrame,
    LocalDataFrame,
    as_fugue_df,
)
from fugue_amd.dataframe.dataframe_iterable_dataframe import (
    IterableArrowDataFrame,
    IterablePandasDataFrame,
    LocalDataFrameIterableDataFrame,
)
from fugue_amd.dataframe.dataframes import DataFrames
from fugue_amd.dataframe.iterable_dataframe import IterableDataFrame
from fugue_amd.dataframe.pandas_dataframe import PandasDataFrame
from fugue_amd.schema import Schema
from fugue_amd.utils.hash import to_uuid


from typing import Generic, TypeVar

_T = TypeVar("_T")


class EmptyAwareIterable(Generic[_T]):
    """An iterable that can report emptiness without consuming
    (triad's ``EmptyAwareIterable`` analog)."""

    def __init__(self, it: Iterable):
        self._it = iter(it)
        self._buffer: List[Any] = []
        self._checked = False

    @property
    def empty(self) -> bool:
        if not self._buffer:
            try:
                self._buffer.append(next(self._it))
            except StopIteration:
                return True
        return False

    def peek(self) -> Any:
        if self.empty:
            raise StopIteration("iterable is empty")
        return self._buffer[0]

    def __iter__(self) -> Iterator:
        while True:
            if self._buffer:
                yield self._buffer.pop(0)
            else:
                try:
                    yield next(self._it)
                except StopIteration:
                    return


def make_empty_aware(it: Iterable) -> EmptyAwareIterable:
    return it if isinstance(it, EmptyAwareIterable) else EmptyAwareIterable(it)


def _anno_eq(anno: Any, target: Any) -> bool:
    if anno is target:
        return True
    try:
        if anno == target:
            return True
    except Exception:
        pass
    return False


def _is_iterable_of(anno: Any, elem: Any) -> bool:
    import collections.abc as cabc

    origin = get_origin(anno)
    if origin not in (cabc.Iterable, cabc.Iterator):
        return False
    args = get_args(anno)
    if len(args) != 1:
        return False
    return _anno_eq(args[0], elem)


def _is_empty_aware_of(anno: Any, elem: Any) -> bool:
    if get_origin(anno) is not EmptyAwareIterable:
        return False
    args = get_args(anno)
    return len(args) == 1 and _anno_eq(args[0], elem)


def _is_list_of(anno: Any, elem: Any) -> bool:
    if get_origin(anno) is not list:
        return False
    args = get_args(anno)
    return len(args) == 1 and _anno_eq(args[0], elem)


class AnnotatedParam:
    code = "x"
    need_schema_: Optional[bool] = False
    format_hint_: Optional[str] = None

    def __init__(self, param: Optional[inspect.Parameter]):
        self.param = param

    @staticmethod
    def matches(anno: Any) -> bool:
        return False

    def to_input_data(self, df: DataFrame, ctx: Any) -> Any:
        raise NotImplementedError

    def to_output_df(self, output: Any, schema: Any, ctx: Any) -> DataFrame:
        raise NotImplementedError

    def count(self, obj: Any) -> int:
        raise NotImplementedError

    def need_schema(self) -> Optional[bool]:
        return self.need_schema_

    def format_hint(self) -> Optional[str]:
        return self.format_hint_


class DataFrameParam(AnnotatedParam):
    code = "d"
    need_schema_ = None

    @staticmethod
    def matches(anno: Any) -> bool:
        return anno is DataFrame

    def to_input_data(self, df: DataFrame, ctx: Any) -> Any:
        return df

    def to_output_df(self, output: Any, schema: Any, ctx: Any) -> DataFrame:
        df = as_fugue_df(output)
        if schema is not None:
            sc = schema if isinstance(schema, Schema) else Schema(schema)
            if df.schema != sc:
                raise ValueError(f"schema mismatch: {df.schema} vs {sc}")
        return df

    def count(self, obj: Any) -> int:
        return obj.count()


class LocalDataFrameParam(DataFrameParam):
    code = "l"
    need_schema_ = False

    @staticmethod
    def matches(anno: Any) -> bool:
        return anno is LocalDataFrame

    def to_input_data(self, df: DataFrame, ctx: Any) -> Any:
        return df.as_local()

    def to_output_df(self, output: Any, schema: Any, ctx: Any) -> DataFrame:
        if not isinstance(output, DataFrame):
            raise ValueError(f"{output} is not a DataFrame")
        return output

    def count(self, obj: Any) -> int:
        return obj.count()


class _ListListParam(AnnotatedParam):
    code = "s"
    need_schema_ = True

    @staticmethod
    def matches(anno: Any) -> bool:
        return _is_list_of(anno, List[Any])

    def to_input_data(self, df: DataFrame, ctx: Any) -> Any:
        return df.as_array(type_safe=True)

    def to_output_df(self, output: Any, schema: Any, ctx: Any) -> DataFrame:
        return ArrayDataFrame(output, schema)

    def count(self, obj: Any) -> int:
        return len(obj)


class _IterableListParam(AnnotatedParam):
    code = "s"
    need_schema_ = True

    @staticmethod
    def matches(anno: Any) -> bool:
        return _is_iterable_of(anno, List[Any])

    def to_input_data(self, df: DataFrame, ctx: Any) -> Any:
        return iter(df.as_array_iterable(type_safe=True))

    def to_output_df(self, output: Any, schema: Any, ctx: Any) -> DataFrame:
        return IterableDataFrame(output, schema)

    def count(self, obj: Any) -> int:
        raise NotImplementedError("can't count an iterable")


class _EmptyAwareIterableListParam(_IterableListParam):
    @staticmethod
    def matches(anno: Any) -> bool:
        return _is_empty_aware_of(anno, List[Any])

    def to_input_data(self, df: DataFrame, ctx: Any) -> Any:
        return make_empty_aware(df.as_array_iterable(type_safe=True))


class _ListDictParam(AnnotatedParam):
    code = "s"
    need_schema_ = True

    @staticmethod
    def matches(anno: Any) -> bool:
        return _is_list_of(anno, Dict[str, Any])

    def to_input_data(self, df: DataFrame, ctx: Any) -> Any:
        return df.as_local().as_dicts()

    def to_output_df(self, output: Any, schema: Any, ctx: Any) -> DataFrame:
        sc = schema if isinstance(schema, Schema) else Schema(schema)
        rows = [[d.get(n, None) for n in sc.names] for d in output]
        return ArrayDataFrame(rows, sc)

    def count(self, obj: Any) -> int:
        return len(obj)


class _IterableDictParam(AnnotatedParam):
    code = "s"
    need_schema_ = True

    @staticmethod
    def matches(anno: Any) -> bool:
        return _is_iterable_of(anno, Dict[str, Any])

    def to_input_data(self, df: DataFrame, ctx: Any) -> Any:
        return iter(df.as_dict_iterable())

    def to_output_df(self, output: Any, schema: Any, ctx: Any) -> DataFrame:
        sc = schema if isinstance(schema, Schema) else Schema(schema)

        def gen():
            for d in output:
                yield [d.get(n, None) for n in sc.names]

        return IterableDataFrame(gen(), sc)

    def count(self, obj: Any) -> int:
        raise NotImplementedError("can't count an iterable")


class _EmptyAwareIterableDictParam(_IterableDictParam):
    @staticmethod
    def matches(anno: Any) -> bool:
        return _is_empty_aware_of(anno, Dict[str, Any])

    def to_input_data(self, df: DataFrame, ctx: Any) -> Any:
        return make_empty_aware(df.as_dict_iterable())


class _AnyDataFrameParam(AnnotatedParam):
    """``AnyDataFrame`` annotation: any supported frame object
    (reference ``_AnyDataFrameParam``, code "d")."""

    code = "d"
    need_schema_ = None

    @staticmethod
    def matches(anno: Any) -> bool:
        from fugue_amd.dataframe.dataframe import AnyDataFrame

        return anno is AnyDataFrame

    def to_input_data(self, df: DataFrame, ctx: Any) -> Any:
        return df

    def to_output_df(self, output: Any, schema: Any, ctx: Any) -> DataFrame:
        from fugue_amd.dataframe.dataframe import as_fugue_df

        res = as_fugue_df(output) if schema is None else as_fugue_df(
            output, schema=schema
        )
        return res

    def count(self, obj: Any) -> int:
        from fugue_amd.dataframe.dataframe import as_fugue_df

        return as_fugue_df(obj).count()


class _DictRowParam(AnnotatedParam):
    """``Dict[str, Any]`` output: one row (reference ``DictParam``,
    code "r")."""

    code = "r"
    need_schema_ = True

    @staticmethod
    def matches(anno: Any) -> bool:
        import typing

        return anno == Dict[str, Any] or anno == typing.Dict[str, Any]

    def to_output_df(self, output: Any, schema: Any, ctx: Any) -> DataFrame:
        sc = schema if isinstance(schema, Schema) else Schema(schema)
        return ArrayDataFrame([[output.get(n, None) for n in sc.names]], sc)

    def count(self, obj: Any) -> int:
        return 1


class _PandasParam(AnnotatedParam):
    code = "p"
    need_schema_ = False
    format_hint_ = "pandas"

    @staticmethod
    def matches(anno: Any) -> bool:
        return anno is pd.DataFrame

    def to_input_data(self, df: DataFrame, ctx: Any) -> Any:
        return df.as_pandas()

    def to_output_df(self, output: Any, schema: Any, ctx: Any) -> DataFrame:
        return PandasDataFrame(output, schema)

    def count(self, obj: Any) -> int:
        return len(obj)


class _IterablePandasParam(AnnotatedParam):
    code = "p"
    need_schema_ = False
    format_hint_ = "pandas"

    @staticmethod
    def matches(anno: Any) -> bool:
        return _is_iterable_of(anno, pd.DataFrame)

    def to_input_data(self, df: DataFrame, ctx: Any) -> Any:
        if isinstance(df, LocalDataFrameIterableDataFrame):
            return (f.as_pandas() for f in df.native)

        def gen():
            yield df.as_pandas()

        return gen()

    def to_output_df(self, output: Any, schema: Any, ctx: Any) -> DataFrame:
        sc = Schema(schema) if schema is not None else None

        def gen():
            for pdf in output:
                yield PandasDataFrame(pdf, sc)

        return IterablePandasDataFrame(gen(), sc)

    def count(self, obj: Any) -> int:
        raise NotImplementedError("can't count an iterable")


class _PyArrowTableParam(AnnotatedParam):
    code = "q"
    need_schema_ = False
    format_hint_ = "pyarrow"

    @staticmethod
    def matches(anno: Any) -> bool:
        return anno is pa.Table

    def to_input_data(self, df: DataFrame, ctx: Any) -> Any:
        return df.as_arrow()

    def to_output_df(self, output: Any, schema: Any, ctx: Any) -> DataFrame:
        return ArrowDataFrame(output, schema)

    def count(self, obj: Any) -> int:
        return obj.num_rows


class _IterableArrowParam(AnnotatedParam):
    code = "q"
    need_schema_ = False
    format_hint_ = "pyarrow"

    @staticmethod
    def matches(anno: Any) -> bool:
        return _is_iterable_of(anno, pa.Table)

    def to_input_data(self, df: DataFrame, ctx: Any) -> Any:
        if isinstance(df, LocalDataFrameIterableDataFrame):
            return (f.as_arrow() for f in df.native)

        def gen():
            yield df.as_arrow()

        return gen()

    def to_output_df(self, output: Any, schema: Any, ctx: Any) -> DataFrame:
        sc = Schema(schema) if schema is not None else None

        def gen():
            for tbl in output:
                yield ArrowDataFrame(tbl, sc)

        return IterableArrowDataFrame(gen(), sc)

    def count(self, obj: Any) -> int:
        raise NotImplementedError("can't count an iterable")


class _DataFramesParam(AnnotatedParam):
    code = "c"

    @staticmethod
    def matches(anno: Any) -> bool:
        return anno is DataFrames

    def to_input_data(self, df: Any, ctx: Any) -> Any:
        return df


class ExecutionEngineParam(AnnotatedParam):
    """An ExecutionEngine (or subclass) annotation — the engine instance is
    injected by the caller (code ``e``)."""

    code = "e"

    @staticmethod
    def matches(anno: Any) -> bool:
        from fugue_amd.execution.execution_engine import ExecutionEngine

        return isinstance(anno, type) and issubclass(anno, ExecutionEngine)


class _CallableParam(AnnotatedParam):
    code = "F"

    @staticmethod
    def matches(anno: Any) -> bool:
        return anno is Callable or anno is callable or get_origin(anno) is _cabc_callable()


class _OptionalCallableParam(AnnotatedParam):
    code = "f"

    @staticmethod
    def matches(anno: Any) -> bool:
        if get_origin(anno) is Union:
            args = get_args(anno)
            if len(args) == 2 and type(None) in args:
                other = args[0] if args[1] is type(None) else args[1]
                return _CallableParam.matches(other)
        return False


def _cabc_callable():
    import collections.abc as cabc

    return cabc.Callable


class _NoneParam(AnnotatedParam):
    code = "n"

    @staticmethod
    def matches(anno: Any) -> bool:
        return anno is None or anno is type(None)


class _OtherParam(AnnotatedParam):
    code = "x"


class _KeywordParam(AnnotatedParam):
    code = "z"


_PARAM_CLASSES: List[type] = []


def register_annotated_param(cls: type, prepend: bool = False) -> type:
    """Register a custom AnnotatedParam subclass (plugin point; the HIP
    engine registers device-frame param types through this).  Also
    exposed as ``fugue_annotated_param`` for reference-name parity
    (``fugue/dataframe/function_wrapper.py:151``)."""
    if prepend:
        _PARAM_CLASSES.insert(0, cls)
    else:
        _PARAM_CLASSES.append(cls)
    return cls


for _c in [
    DataFrameParam,
    LocalDataFrameParam,
    _ListListParam,
    _IterableListParam,
    _EmptyAwareIterableListParam,
    _ListDictParam,
    _IterableDictParam,
    _EmptyAwareIterableDictParam,
    _DictRowParam,
    _AnyDataFrameParam,
    _PandasParam,
    _IterablePandasParam,
    _PyArrowTableParam,
    _IterableArrowParam,
    _DataFramesParam,
    ExecutionEngineParam,
    _OptionalCallableParam,
    _CallableParam,
]:
    register_annotated_param(_c)


def _resolve_param(param: inspect.Parameter, extra: Optional[List[type]] = None) -> AnnotatedParam:
    anno = param.annotation
    if param.kind == param.VAR_KEYWORD:
        return _KeywordParam(param)
    if param.kind == param.VAR_POSITIONAL:
        raise TypeError("*args is not supported in wrapped functions")
    classes = (extra or []) + _PARAM_CLASSES
    for cls in classes:
        try:
            if cls.matches(anno):
                return cls(param)
        except Exception:
            continue
    return _OtherParam(param)


def _resolve_return(anno: Any, extra: Optional[List[type]] = None) -> AnnotatedParam:
    if anno is None or anno is type(None):
        return _NoneParam(None)
    if anno is inspect.Signature.empty:
        return _NoneParam(None)
    classes = (extra or []) + _PARAM_CLASSES
    for cls in classes:
        try:
            if cls.matches(anno):
                return cls(None)
        except Exception:
            continue
    return _OtherParam(None)


class DataFrameFunctionWrapper:
    """Wrap a plain function; validate its signature shape against regexes
    over the param-code string; convert frames at call boundaries."""

    def __init__(self, func: Callable, params_re: str = ".*", return_re: str = ".*"):
        self._func = func
        sig = inspect.signature(func)
        self._params: Dict[str, AnnotatedParam] = {}
        for name, param in sig.parameters.items():
            self._params[name] = _resolve_param(param)
        self._rt = _resolve_return(sig.return_annotation)
        self.input_code = "".join(p.code for p in self._params.values())
        self.output_code = self._rt.code
        if not re.match(params_re, self.input_code):
            raise TypeError(
                f"input signature {self.input_code!r} of {func} "
                f"doesn't match {params_re!r}"
            )
        if not re.match(return_re, self.output_code):
            raise TypeError(
                f"return annotation {self.output_code!r} of {func} "
                f"doesn't match {return_re!r}"
            )

    @property
    def need_output_schema(self) -> Optional[bool]:
        return self._rt.need_schema() if self._rt.code in "dlspq" else False

    def get_format_hint(self) -> Optional[str]:
        for v in self._params.values():
            if v.format_hint() is not None:
                return v.format_hint()
        return self._rt.format_hint()

    def get_param_by_index(self, i: int) -> AnnotatedParam:
        return list(self._params.values())[i]

    def __call__(self, *args: Any, **kwargs: Any) -> Any:
        return self._func(*args, **kwargs)

    def __uuid__(self) -> str:
        return to_uuid(self._func, self.input_code, self.output_code)

    def run(
        self,
        args: List[Any],
        kwargs: Dict[str, Any],
        ignore_unknown: bool = False,
        output_schema: Any = None,
        output: bool = True,
        ctx: Any = None,
    ) -> Any:
        p: Dict[str, Any] = {}
        names = list(self._params.keys())
        for i, a in enumerate(args):
            p[names[i]] = a
        p.update(kwargs)
        has_kw = False
        rargs: Dict[str, Any] = {}
        for k, v in self._params.items():
            if isinstance(v, _KeywordParam):
                has_kw = True
            elif k in p:
                if v.code in "dlspqc" and isinstance(p[k], (DataFrame, DataFrames)):
                    rargs[k] = v.to_input_data(p[k], ctx=ctx)
                else:
                    rargs[k] = p[k]
                del p[k]
            elif v.param is not None and v.param.default is not inspect.Parameter.empty:
                continue
            else:
                raise ValueError(f"parameter {k} is required but not given")
        if has_kw:
            rargs.update(p)
        elif not ignore_unknown and len(p) > 0:
            raise ValueError(f"{list(p.keys())} are not acceptable parameters")
        rt = self._func(**rargs)
        if not output:
            if isinstance(rt, Iterable) and not isinstance(
                rt, (list, str, bytes, dict, pd.DataFrame, pa.Table, DataFrame)
            ):
                for _ in rt:
                    pass
            return None
        if self._rt.code in "dlspqr":
            return self._rt.to_output_df(rt, output_schema, ctx=ctx)
        return rt


fugue_annotated_param = register_annotated_param
