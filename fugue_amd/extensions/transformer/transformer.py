"""Transformer / CoTransformer interfaces.

Reference parity: ``fugue/extensions/transformer/transformer.py``.
"""
from typing import Any, Optional

from fugue_amd.dataframe.array_dataframe import ArrayDataFrame
from fugue_amd.dataframe.dataframe import DataFrame, LocalDataFrame
from fugue_amd.dataframe.dataframes import DataFrames
from fugue_amd.extensions.context import ExtensionContext
from fugue_amd.extensions.transformer.constants import (
    OUTPUT_TRANSFORMER_DUMMY_SCHEMA,
)


class Transformer(ExtensionContext):
    """Process logical partitions of a dataframe, one local frame at a
    time.  Not an ABC (to stay picklable across process boundaries)."""

    def get_output_schema(self, df: DataFrame) -> Any:  # pragma: no cover
        raise NotImplementedError

    def get_format_hint(self) -> Optional[str]:
        return None

    def on_init(self, df: DataFrame) -> None:  # pragma: no cover
        pass

    def transform(self, df: LocalDataFrame) -> LocalDataFrame:  # pragma: no cover
        raise NotImplementedError


class OutputTransformer(Transformer):
    def process(self, df: LocalDataFrame) -> None:  # pragma: no cover
        raise NotImplementedError

    def get_output_schema(self, df: DataFrame) -> Any:
        return OUTPUT_TRANSFORMER_DUMMY_SCHEMA

    def transform(self, df: LocalDataFrame) -> LocalDataFrame:
        self.process(df)
        return ArrayDataFrame([], OUTPUT_TRANSFORMER_DUMMY_SCHEMA)


class CoTransformer(ExtensionContext):
    """Process logical partitions of a zipped dataframe."""

    def get_output_schema(self, dfs: DataFrames) -> Any:  # pragma: no cover
        raise NotImplementedError

    def get_format_hint(self) -> Optional[str]:
        return None

    def on_init(self, dfs: DataFrames) -> None:  # pragma: no cover
        pass

    def transform(self, dfs: DataFrames) -> LocalDataFrame:  # pragma: no cover
        raise NotImplementedError


class OutputCoTransformer(CoTransformer):
    def process(self, dfs: DataFrames) -> None:  # pragma: no cover
        raise NotImplementedError

    def get_output_schema(self, dfs: DataFrames) -> Any:
        return OUTPUT_TRANSFORMER_DUMMY_SCHEMA

    def transform(self, dfs: DataFrames) -> LocalDataFrame:
        self.process(dfs)
        return ArrayDataFrame([], OUTPUT_TRANSFORMER_DUMMY_SCHEMA)
