"""Functional API over Datasets (reference: ``fugue/dataset/api.py``)."""
from typing import Any, Optional

from fugue_amd.dataset.dataset import Dataset
from fugue_amd.utils.registry import run_plugin, try_run_plugin


def as_fugue_dataset(data: Any, **kwargs: Any) -> Dataset:
    if isinstance(data, Dataset) and len(kwargs) == 0:
        return data
    ok, res = try_run_plugin("as_fugue_dataset", data, **kwargs)
    if ok:
        return res
    from fugue_amd.dataframe.dataframe import as_fugue_df

    return as_fugue_df(data, **kwargs)


def show(data: Any, n: int = 10, with_count: bool = False, title: Optional[str] = None) -> None:
    as_fugue_dataset(data).show(n=n, with_count=with_count, title=title)


def count(data: Any) -> int:
    return as_fugue_dataset(data).count()


def is_local(data: Any) -> bool:
    return as_fugue_dataset(data).is_local


def is_bounded(data: Any) -> bool:
    return as_fugue_dataset(data).is_bounded


def is_empty(data: Any) -> bool:
    return as_fugue_dataset(data).empty


def as_local(data: Any) -> Any:
    """Convert the dataset to a local dataset (reference
    ``fugue/dataset/api.py:39``); plugin-overridable via
    ``as_local_bounded``."""
    return as_local_bounded(data)


def as_local_bounded(data: Any) -> Any:
    """Convert the dataset to a local bounded dataset (reference
    ``fugue/dataset/api.py:48``)."""
    ok, res = try_run_plugin("as_local_bounded", data)
    if ok:
        return res
    try:
        ds = as_fugue_dataset(data)
    except Exception:
        ds = None
    if ds is not None and hasattr(ds, "as_local_bounded"):
        return ds.as_local_bounded()
    raise NotImplementedError(
        f"no registered function to convert {type(data)} to a local bounded dataset"
    )


def get_num_partitions(data: Any) -> int:
    """Number of partitions of the dataset (reference
    ``fugue/dataset/api.py:95``)."""
    return as_fugue_dataset(data).num_partitions
