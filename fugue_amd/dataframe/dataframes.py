"""DataFrames: ordered dict of (optionally named) DataFrames.

Reference parity: ``fugue/dataframe/dataframes.py:9``.
"""
from typing import Any, Callable, Dict, List, Union

from fugue_amd.dataframe.dataframe import DataFrame, as_fugue_df


class DataFrames(Dict[str, DataFrame]):
    def __init__(self, *args: Any, **kwargs: Any):
        super().__init__()
        self._has_key = False
        for a in args:
            self._append_arg(a)
        for k, v in kwargs.items():
            self[k] = v

    def _append_arg(self, value: Any) -> None:
        if value is None:
            return
        if isinstance(value, DataFrames):
            for k, v in value.items():
                if value.has_key:
                    self[k] = v
                else:
                    self._append(v)
            return
        if isinstance(value, dict):
            for k, v in value.items():
                self[k] = v
            return
        if isinstance(value, DataFrame):
            self._append(value)
            return
        if isinstance(value, (list, tuple)):
            for v in value:
                self._append_arg(v)
            return
        self._append(value)

    @property
    def has_key(self) -> bool:
        return self._has_key

    def __setitem__(self, key: str, value: Any) -> None:
        if not isinstance(key, str) or key == "":
            raise ValueError(f"invalid dataframe name {key!r}")
        if len(self) > 0 and not self._has_key:
            raise ValueError("can't mix named and unnamed dataframes")
        df = value if isinstance(value, DataFrame) else as_fugue_df(value)
        self._has_key = True
        super().__setitem__(key, df)

    def __getitem__(self, key: Union[str, int]) -> DataFrame:
        if isinstance(key, int):
            return list(self.values())[key]
        return super().__getitem__(key)

    def _append(self, value: Any) -> None:
        if len(self) > 0 and self._has_key:
            raise ValueError("can't mix named and unnamed dataframes")
        df = value if isinstance(value, DataFrame) else as_fugue_df(value)
        super().__setitem__(f"_{len(self)}", df)

    def convert(self, func: Callable[[DataFrame], DataFrame]) -> "DataFrames":
        if self._has_key:
            return DataFrames({k: func(v) for k, v in self.items()})
        return DataFrames([func(v) for v in self.values()])
