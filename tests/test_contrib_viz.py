"""Contrib visualization outputters (reference
``fugue_contrib/viz/_ext.py`` parity)."""
import matplotlib

matplotlib.use("Agg")

import pandas as pd
import pytest

import fugue_amd.contrib.viz as viz  # noqa: F401  (registers namespaces)
from fugue_amd.execution import NativeExecutionEngine
from fugue_amd.workflow import FugueWorkflow


def test_viz_namespace_outputter(monkeypatch):
    seen = []
    from pandas.plotting import PlotAccessor

    def spy_line(self, **kw):
        seen.append((self._parent.copy(), dict(kw)))

    monkeypatch.setattr(PlotAccessor, "line", spy_line)
    with FugueWorkflow() as dag:
        df = dag.df([[1, 2.0], [1, 3.0], [2, 4.0]], "g:int,v:double")
        df.partition(by=["g"], presort="v desc").output(
            ("viz", "line"), params=dict(title="t")
        )
    dag.run(NativeExecutionEngine())
    assert len(seen) == 2  # one plot per partition
    for frame, kw in seen:
        assert "g" not in frame.columns  # keys folded into the title
        assert kw["title"].startswith("t -- ")
    # presort applied within each partition
    assert seen[0][0]["v"].tolist() == sorted(
        seen[0][0]["v"].tolist(), reverse=True
    )


def test_viz_invalid_func():
    with pytest.raises(Exception):
        viz._PandasVisualize("definitely_not_a_plot_kind")
