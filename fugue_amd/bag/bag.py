"""Bag: unordered object collection (reference parity: ``fugue/bag/bag.py``)."""
from abc import abstractmethod
from typing import Any, Iterable, List

from fugue_amd.dataset.dataset import Dataset, DatasetDisplay
from fugue_amd.utils.registry import register_plugin


class Bag(Dataset):
    def __copy__(self) -> "Bag":
        return self

    def __deepcopy__(self, memo: Any) -> "Bag":
        return self

    def head(self, n: int) -> "LocalBag":
        raise NotImplementedError  # pragma: no cover

    @abstractmethod
    def as_local(self) -> "LocalBag":
        ...

    @abstractmethod
    def peek(self) -> Any:
        ...

    @abstractmethod
    def as_array(self) -> List[Any]:
        ...

    @abstractmethod
    def as_array_iterable(self) -> Iterable[Any]:
        ...

    @property
    def is_bounded(self) -> bool:
        return True


class LocalBag(Bag):
    @property
    def is_local(self) -> bool:
        return True

    @property
    def num_partitions(self) -> int:
        return 1

    def as_local(self) -> "LocalBag":
        return self


class BagDisplay(DatasetDisplay):
    def show(self, n: int = 10, with_count: bool = False, title: Any = None) -> None:
        if title:
            print(title)
        arr = self._ds.as_local().as_array()[:n]  # type: ignore
        print(arr)
        if with_count:
            print(f"Total count: {self._ds.count()}")


register_plugin(
    "get_dataset_display",
    lambda ds: isinstance(ds, Bag),
    lambda ds: BagDisplay(ds),
    priority=0.5,
)
