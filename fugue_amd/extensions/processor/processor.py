"""Processor interface (reference: ``fugue/extensions/processor/processor.py:7``)."""
from fugue_amd.dataframe.dataframe import DataFrame
from fugue_amd.dataframe.dataframes import DataFrames
from fugue_amd.extensions.context import ExtensionContext


class Processor(ExtensionContext):
    """Driver-side transformation: DataFrames → DataFrame."""

    def process(self, dfs: DataFrames) -> DataFrame:  # pragma: no cover
        raise NotImplementedError
