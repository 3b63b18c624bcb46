"""Yields: handles for data escaping a finished workflow.

Reference parity: ``fugue/collections/yielded.py``.
"""
from typing import Any

from fugue_amd.utils.hash import to_uuid


class Yielded:
    def __init__(self, yid: str):
        self._yid = to_uuid(yid)

    def __uuid__(self) -> str:
        return self._yid

    @property
    def is_set(self) -> bool:
        raise NotImplementedError

    def __copy__(self) -> "Yielded":
        return self

    def __deepcopy__(self, memo: Any) -> "Yielded":
        return self


class PhysicalYielded(Yielded):
    """Yield by file or table name. ``storage_type`` ∈ {"file", "table"}."""

    def __init__(self, yid: str, storage_type: str):
        super().__init__(yid)
        if storage_type not in ("file", "table"):
            raise ValueError(f"invalid storage type {storage_type}")
        self._name = ""
        self._storage_type = storage_type

    @property
    def is_set(self) -> bool:
        return self._name != ""

    def set_value(self, name: str) -> None:
        self._name = name

    @property
    def name(self) -> str:
        if not self.is_set:
            raise RuntimeError("value is not set")
        return self._name

    @property
    def storage_type(self) -> str:
        return self._storage_type
