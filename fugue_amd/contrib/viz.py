"""Visualization output-transformers (reference parity:
``fugue_contrib/viz/_ext.py`` — pandas ``.plot``-based plotting per
partition).  Usable as ``out_transform(df, plot, params=...)`` or in
FugueSQL ``OUTTRANSFORM ... USING plot``."""
from typing import Any

import pandas as pd


def plot(df: pd.DataFrame, kind: str = "line", **kwargs: Any) -> None:
    """Plot one (logical partition of a) dataframe with pandas' plotting
    backend (requires matplotlib)."""
    df.plot(kind=kind, **kwargs)


def seaborn_plot(df: pd.DataFrame, func: str = "lineplot", **kwargs: Any) -> None:
    """Plot with seaborn (requires seaborn)."""
    import seaborn as sns

    getattr(sns, func)(data=df, **kwargs)
