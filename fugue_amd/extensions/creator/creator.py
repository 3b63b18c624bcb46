"""Creator interface (reference: ``fugue/extensions/creator/creator.py:7``)."""
from fugue_amd.dataframe.dataframe import DataFrame
from fugue_amd.extensions.context import ExtensionContext


class Creator(ExtensionContext):
    """Create a DataFrame from nothing (driver side)."""

    def create(self) -> DataFrame:  # pragma: no cover
        raise NotImplementedError
