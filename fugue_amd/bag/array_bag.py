"""ArrayBag (reference parity: ``fugue/bag/array_bag.py``)."""
from typing import Any, Iterable, List

from fugue_amd.bag.bag import Bag, LocalBag


class ArrayBag(LocalBag):
    def __init__(self, data: Any):
        super().__init__()
        if isinstance(data, list):
            self._native = list(data)
        elif isinstance(data, Iterable):
            self._native = list(data)
        else:
            raise ValueError(f"can't create ArrayBag from {type(data)}")

    @property
    def native(self) -> List[Any]:
        return self._native

    @property
    def empty(self) -> bool:
        return len(self._native) == 0

    def count(self) -> int:
        return len(self._native)

    def peek(self) -> Any:
        if self.empty:
            from fugue_amd.exceptions import FugueDatasetEmptyError

            raise FugueDatasetEmptyError("bag is empty")
        return self._native[0]

    def head(self, n: int) -> "ArrayBag":
        return ArrayBag(self._native[:n])

    def as_array(self) -> List[Any]:
        return list(self._native)

    def as_array_iterable(self) -> Iterable[Any]:
        yield from self._native
