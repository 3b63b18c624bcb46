"""Visualization outputters (reference parity:
``fugue_contrib/viz/_ext.py``): per-partition plotting through pandas'
plot backend, registered as the ``("viz", func)`` extension namespace so
``df.partition_by(...).output(("viz", "scatter"), params=...)`` and
FugueSQL ``OUTPUT ... USING viz:scatter`` work on any engine.
"""
import json
from abc import ABC, abstractmethod
from typing import Any, Tuple

import pandas as pd

from fugue_amd.dataframe.dataframes import DataFrames
from fugue_amd.exceptions import FugueWorkflowError
from fugue_amd.extensions.outputter.convert import parse_outputter
from fugue_amd.extensions.outputter.outputter import Outputter
from fugue_amd.plugins import namespace_candidate


class Visualize(Outputter, ABC):
    """Base: sort by presort, split by the partition keys, and plot each
    logical partition (title carries the key values)."""

    def __init__(self, func: str) -> None:
        super().__init__()
        self._func = func

    def process(self, dfs: DataFrames) -> None:
        if len(dfs) != 1:
            raise FugueWorkflowError("not single input")
        df = dfs[0].as_pandas()
        presort = self.partition_spec.presort
        if len(presort) > 0:
            df = df.sort_values(
                list(presort.keys()), ascending=list(presort.values())
            ).reset_index(drop=True)
        by = self.partition_spec.partition_by
        if len(by) == 0:
            self._plot(df)
        else:
            keys: Any = by if len(by) > 1 else by[0]
            for _, gp in df.groupby(keys, dropna=False):
                self._plot(gp.reset_index(drop=True))

    def _title_params(self, df: pd.DataFrame) -> Tuple[pd.DataFrame, dict]:
        """Drop the partition keys from the frame and fold their values
        into the plot title."""
        params = dict(self.params)
        by = self.partition_spec.partition_by
        if len(by) > 0:
            keys = df[by].head(1).to_dict("records")[0]
            kt = json.dumps(keys)[1:-1]
            params["title"] = (
                params["title"] + " -- " + kt if "title" in params else kt
            )
            df = df.drop(by, axis=1)
        return df, params

    @abstractmethod
    def _plot(self, df: pd.DataFrame) -> None:  # pragma: no cover
        raise NotImplementedError


class _PandasVisualize(Visualize):
    def __init__(self, func: str) -> None:
        super().__init__(func)
        if func != "plot":
            getattr(pd.DataFrame.plot, func)  # validate early

    def _plot(self, df: pd.DataFrame) -> None:
        df, params = self._title_params(df)
        fn = df.plot if self._func == "plot" else getattr(df.plot, self._func)
        fn(**params)


class _SeabornVisualize(Visualize):
    def __init__(self, func: str) -> None:
        super().__init__(func)
        import seaborn  # validate availability + func early

        getattr(seaborn, func)

    def _plot(self, df: pd.DataFrame) -> None:
        import matplotlib.pyplot as plt
        import seaborn

        df, params = self._title_params(df)
        title = params.pop("title", None)
        getattr(seaborn, self._func)(data=df, **params)
        if title is not None:
            plt.title(title)
        plt.show()


@parse_outputter.candidate(
    namespace_candidate("viz", lambda x: isinstance(x, str))
)
def _parse_pandas_plot(obj: Tuple[str, str]) -> Outputter:
    return _PandasVisualize(obj[1])


@parse_outputter.candidate(
    namespace_candidate("sns", lambda x: isinstance(x, str))
)
def _parse_seaborn(obj: Tuple[str, str]) -> Outputter:
    return _SeabornVisualize(obj[1])


# compatibility helpers kept from the earlier minimal module
def plot(df: pd.DataFrame, kind: str = "line", **kwargs: Any) -> None:
    """Plot one (logical partition of a) dataframe with pandas' plotting
    backend (requires matplotlib)."""
    df.plot(kind=kind, **kwargs)


def seaborn_plot(df: pd.DataFrame, func: str = "lineplot", **kwargs: Any) -> None:
    """Plot with seaborn (requires seaborn)."""
    import seaborn as sns

    getattr(sns, func)(data=df, **kwargs)
