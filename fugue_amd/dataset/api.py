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
