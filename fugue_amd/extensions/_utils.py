"""Validation rules + named-extension registries.

Reference parity: ``fugue/extensions/_utils.py``.
"""
import threading
from typing import Any, Callable, Dict, List, Optional

from fugue_amd.collections.partition import PartitionSpec, parse_presort_exp
from fugue_amd.exceptions import (
    FugueWorkflowCompileValidationError,
    FugueWorkflowRuntimeValidationError,
)
from fugue_amd.schema import Schema
from fugue_amd.utils.interfaceless import parse_comment_annotation

def is_namespace_extension(obj: Any) -> bool:
    """Whether ``obj`` is a namespace extension reference: a 2-tuple of
    (namespace str, payload) (reference ``fugue/extensions/_utils.py:14``)."""
    return (
        isinstance(obj, tuple)
        and len(obj) == 2
        and isinstance(obj[0], str)
        and obj[0] != ""
    )


def namespace_candidate(
    namespace: str, matcher: Callable[..., bool]
) -> Callable[..., bool]:
    """Build a plugin matcher that fires only for ``(namespace, payload)``
    tuples whose payload passes ``matcher`` (reference
    ``fugue/extensions/_utils.py:25``); used with the ``parse_*``
    dispatchers to register e.g. ``("sparksql", "...")``-style creators."""

    def _matcher(obj: Any, *args: Any, **kwargs: Any) -> bool:
        return (
            is_namespace_extension(obj)
            and obj[0] == namespace
            and matcher(obj[1], *args, **kwargs)
        )

    return _matcher


_VALIDATION_KEYS = [
    "partitionby_has",
    "partitionby_is",
    "presort_has",
    "presort_is",
    "input_has",
    "input_is",
]


def parse_validation_rules_from_comment(func: Callable) -> Dict[str, Any]:
    res: Dict[str, Any] = {}
    for key in _VALIDATION_KEYS:
        v = parse_comment_annotation(func, key)
        if v is None:
            continue
        if v == "":
            raise SyntaxError(f"{key} can't be empty")
        res[key] = v
    return to_validation_rules(res)


def to_validation_rules(data: Dict[str, Any]) -> Dict[str, Any]:
    res: Dict[str, Any] = {}
    for k, v in data.items():
        if k in ("partitionby_has", "partitionby_is"):
            if isinstance(v, str):
                v = [x.strip() for x in v.split(",")]
            res[k] = PartitionSpec(by=v).partition_by
        elif k in ("presort_has", "presort_is"):
            res[k] = list(parse_presort_exp(v).items())
        elif k == "input_has":
            if isinstance(v, str):
                res[k] = v.replace(" ", "").split(",")
            elif isinstance(v, list):
                res[k] = [x.replace(" ", "") for x in v]
            else:
                raise SyntaxError(f"{v} is neither a string nor a list")
        elif k == "input_is":
            try:
                res[k] = str(Schema(v))
            except Exception:
                raise SyntaxError(
                    f"for input_is, the input must be a schema expression: {v}"
                )
        else:
            raise NotImplementedError(k)
    return res


def validate_partition_spec(spec: PartitionSpec, rules: Dict[str, Any]) -> None:
    for k, v in rules.items():
        if k in ("partitionby_has", "partitionby_is"):
            for x in v:
                if x not in spec.partition_by:
                    raise FugueWorkflowCompileValidationError(
                        f"required partition key {x} is not in {spec}"
                    )
            if k == "partitionby_is" and len(v) != len(spec.partition_by):
                raise FugueWorkflowCompileValidationError(
                    f"{v} does not match {spec}"
                )
        if k in ("presort_has", "presort_is"):
            expected = spec.presort
            for pk, pv in v:
                if pk not in expected:
                    raise FugueWorkflowCompileValidationError(
                        f"required presort key {pk} is not in presort of {spec}"
                    )
                if pv != expected[pk]:
                    raise FugueWorkflowCompileValidationError(
                        f"({pk}) order doesn't match presort of {spec}"
                    )
            if k == "presort_is":
                if v != list(expected.items()):
                    raise FugueWorkflowCompileValidationError(
                        f"{v} does not match {spec}"
                    )


def validate_input_schema(schema: Schema, rules: Dict[str, Any]) -> None:
    for k, v in rules.items():
        if k == "input_has":
            for x in v:
                if x not in schema:
                    raise FugueWorkflowRuntimeValidationError(
                        f"required column {x} is not in {schema}"
                    )
        if k == "input_is":
            if schema != v:
                raise FugueWorkflowRuntimeValidationError(
                    f"{v} does not match {schema}"
                )


class ExtensionRegistry:
    """Registry of named extensions (per extension type)."""

    def __init__(self):
        self._lock = threading.RLock()
        self._items: Dict[str, Any] = {}

    def register(self, name: str, extension: Any, on_dup: str = "overwrite") -> None:
        with self._lock:
            if name in self._items:
                if on_dup == "throw":
                    raise KeyError(f"extension {name} already registered")
                if on_dup == "ignore":
                    return
            self._items[name] = extension

    def get(self, name: str) -> Optional[Any]:
        with self._lock:
            return self._items.get(name)
