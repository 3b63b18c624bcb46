"""Outputter interface (reference: ``fugue/extensions/outputter/outputter.py:7``)."""
from fugue_amd.dataframe.dataframes import DataFrames
from fugue_amd.extensions.context import ExtensionContext


class Outputter(ExtensionContext):
    """Driver-side terminal action: DataFrames → None."""

    def process(self, dfs: DataFrames) -> None:  # pragma: no cover
        raise NotImplementedError
