"""Dataset: the abstract base of DataFrame and Bag.

Reference parity: ``fugue/dataset/dataset.py:14`` (metadata, native,
is_local/is_bounded/num_partitions/empty/count/show).
"""
from abc import ABC, abstractmethod
from typing import Any, Optional

from fugue_amd.utils.params import ParamDict
from fugue_amd.utils.registry import run_plugin, try_run_plugin


AnyDataset = Any  # anything convertible to a Dataset (reference fugue/dataset/dataset.py)


class Dataset(ABC):
    def __init__(self):
        self._metadata: Optional[ParamDict] = None

    @property
    def metadata(self) -> ParamDict:
        if self._metadata is None:
            self._metadata = ParamDict()
        return self._metadata

    @property
    def has_metadata(self) -> bool:
        return self._metadata is not None and len(self._metadata) > 0

    def reset_metadata(self, metadata: Any) -> None:
        self._metadata = ParamDict(metadata) if metadata is not None else None

    @property
    @abstractmethod
    def native(self) -> Any:
        """The underlying object of this dataset"""

    @property
    @abstractmethod
    def is_local(self) -> bool:
        """Whether this dataset is a local object"""

    @property
    @abstractmethod
    def is_bounded(self) -> bool:
        """Whether this dataset is bounded (finite)"""

    @property
    @abstractmethod
    def num_partitions(self) -> int:
        """Number of physical partitions"""

    @property
    @abstractmethod
    def empty(self) -> bool:
        """Whether this dataset is empty"""

    @abstractmethod
    def count(self) -> int:
        """Number of elements in this dataset"""

    def assert_not_empty(self) -> None:
        if self.empty:
            raise AssertionError("dataset is empty")

    def show(
        self, n: int = 10, with_count: bool = False, title: Optional[str] = None
    ) -> None:
        get_dataset_display(self).show(n=n, with_count=with_count, title=title)

    def __repr__(self):
        return get_dataset_display(self).repr()

    def _repr_html_(self):
        return get_dataset_display(self).repr_html()


class DatasetDisplay(ABC):
    def __init__(self, ds: Dataset):
        self._ds = ds

    @abstractmethod
    def show(
        self, n: int = 10, with_count: bool = False, title: Optional[str] = None
    ) -> None:
        ...

    def repr(self) -> str:
        return type(self._ds).__name__

    def repr_html(self) -> str:
        return "<pre>" + self.repr() + "</pre>"


def get_dataset_display(ds: Dataset) -> DatasetDisplay:
    ok, res = try_run_plugin("get_dataset_display", ds)
    if ok:
        return res
    raise NotImplementedError(f"no display registered for {type(ds)}")
