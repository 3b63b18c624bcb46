"""Built-in creators (reference: ``fugue/extensions/_builtins/creators.py``)."""
from typing import Any

from fugue_amd.collections.yielded import Yielded
from fugue_amd.dataframe.dataframe import DataFrame
from fugue_amd.extensions.creator.creator import Creator


class Load(Creator):
    def process(self) -> DataFrame:  # pragma: no cover
        return self.create()

    def create(self) -> DataFrame:
        kwargs = self.params.get("params", dict())
        path = self.params.get_or_throw("path", str)
        format_hint = self.params.get("fmt", "")
        columns = self.params.get_or_none("columns", object)
        return self.execution_engine.load_df(
            path=path, format_hint=format_hint, columns=columns, **kwargs
        )


class CreateData(Creator):
    def create(self) -> DataFrame:
        df = self.params.get_or_throw("df", object)
        schema = self.params.get_or_none("schema", object)
        return self.execution_engine.to_df(df, schema=schema)


class LoadYielded(Creator):
    def create(self) -> DataFrame:
        yielded = self.params.get_or_throw("yielded", Yielded)
        return self.execution_engine.load_yielded(yielded)
