"""Object/string → type/instance/function resolution helpers.

Replaces the ``triad.utils.convert`` helpers the reference relies on for its
"using" parameters (extensions referenced by name, class, or instance).
"""
import importlib
import inspect
from typing import Any, Callable, Optional, Type, TypeVar

T = TypeVar("T")


def str_to_object(expr: str, global_vars: Optional[dict] = None) -> Any:
    """Resolve ``"module.sub:attr"`` / ``"module.sub.attr"`` / plain names
    (searched in ``global_vars``) to a python object."""
    if global_vars is not None and expr in global_vars:
        return global_vars[expr]
    if ":" in expr:
        mod_name, attr = expr.split(":", 1)
        mod = importlib.import_module(mod_name)
        obj: Any = mod
        for part in attr.split("."):
            obj = getattr(obj, part)
        return obj
    parts = expr.split(".")
    for i in range(len(parts) - 1, 0, -1):
        try:
            mod = importlib.import_module(".".join(parts[:i]))
        except ImportError:
            continue
        obj = mod
        try:
            for part in parts[i:]:
                obj = getattr(obj, part)
            return obj
        except AttributeError:
            continue
    raise ValueError(f"can't resolve object from {expr!r}")


def to_type(obj: Any, expected_base: Type[T] = object) -> Type[T]:  # type: ignore
    if isinstance(obj, str):
        obj = str_to_object(obj)
    if not isinstance(obj, type):
        raise TypeError(f"{obj!r} is not a type")
    if expected_base is not object and not issubclass(obj, expected_base):
        raise TypeError(f"{obj} is not a subclass of {expected_base}")
    return obj


def to_instance(obj: Any, expected_base: Type[T] = object, args: tuple = (), kwargs: Optional[dict] = None) -> T:  # type: ignore
    kwargs = kwargs or {}
    if isinstance(obj, str):
        obj = str_to_object(obj)
    if isinstance(obj, type):
        obj = obj(*args, **kwargs)
    if expected_base is not object and not isinstance(obj, expected_base):
        raise TypeError(f"{obj} is not an instance of {expected_base}")
    return obj


def to_function(obj: Any, global_vars: Optional[dict] = None) -> Callable:
    if isinstance(obj, str):
        obj = str_to_object(obj, global_vars)
    if not callable(obj):
        raise TypeError(f"{obj!r} is not callable")
    return obj


def get_full_type_path(obj: Any) -> str:
    if inspect.isclass(obj) or inspect.isfunction(obj):
        return obj.__module__ + ":" + obj.__qualname__
    return type(obj).__module__ + ":" + type(obj).__qualname__
