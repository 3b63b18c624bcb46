"""ParamDict: a dict with typed getters; replaces ``triad.ParamDict``."""
from typing import Any, Dict, Iterable, Optional, Tuple, Type, TypeVar, Union, no_type_check

T = TypeVar("T")

_BOOL_TRUE = {"true", "yes", "1", "on"}
_BOOL_FALSE = {"false", "no", "0", "off"}


class ParamDict(Dict[str, Any]):
    def __init__(self, data: Any = None, deep: bool = True):
        super().__init__()
        self.update_params(data)

    def update_params(self, data: Any) -> "ParamDict":
        if data is None:
            return self
        if isinstance(data, dict):
            for k, v in data.items():
                self[str(k)] = v
            return self
        if isinstance(data, Iterable):
            for item in data:
                k, v = item
                self[str(k)] = v
            return self
        raise ValueError(f"can't construct ParamDict from {data!r}")

    def get(self, key: Union[str, int], default: Any = None) -> Any:  # type: ignore
        if isinstance(key, int):
            key = list(self.keys())[key]
        if key in self:
            v = self[key]
            if default is not None:
                return _convert_value(v, type(default))
            return v
        if default is None:
            raise KeyError(f"{key} not found and no default")
        return default

    def get_or_none(self, key: str, expected_type: Any = object) -> Any:
        if key not in self:
            return None
        return _convert_value(self[key], expected_type)

    def get_or_throw(self, key: str, expected_type: Any = object) -> Any:
        if key not in self:
            raise KeyError(f"{key} not found in ParamDict")
        return _convert_value(self[key], expected_type)

    def __uuid__(self) -> str:
        from fugue_amd.utils.hash import to_uuid

        return to_uuid(dict(self))


@no_type_check
def _convert_value(v: Any, tp: Any) -> Any:
    if tp is object or tp is None:
        return v
    if isinstance(v, tp):
        return v
    if tp is bool:
        if isinstance(v, str):
            lv = v.lower()
            if lv in _BOOL_TRUE:
                return True
            if lv in _BOOL_FALSE:
                return False
            raise ValueError(f"can't convert {v!r} to bool")
        return bool(v)
    if tp in (int, float, str):
        return tp(v)
    raise TypeError(f"can't convert {v!r} to {tp}")
