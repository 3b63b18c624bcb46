"""Conformance suites on the real MI355X engine (device tensors + HIP
kernels)."""
import pytest

pytestmark = pytest.mark.gpu

torch = pytest.importorskip("torch")
if not torch.cuda.is_available():
    pytest.skip("no GPU available", allow_module_level=True)

from typing import Any

from fugue_amd import ArrayDataFrame
from fugue_amd.testing import (
    BuiltInWorkflowTestSuite,
    DataFrameTestSuite,
    ExecutionEngineTestSuite,
)


class TestHipGpuExecutionEngine(ExecutionEngineTestSuite):
    @classmethod
    def make_engine(cls):
        from fugue_amd.hip.execution_engine import HipExecutionEngine

        return HipExecutionEngine()


class TestHipGpuBuiltIn(BuiltInWorkflowTestSuite):
    @classmethod
    def make_engine(cls):
        from fugue_amd.hip.execution_engine import HipExecutionEngine

        return HipExecutionEngine()


class TestHipDataFrameGpu(DataFrameTestSuite):
    supports_nested = False  # device columns are flat (validity + data)

    @classmethod
    def make_df(cls, data: Any, schema: Any):
        from fugue_amd.hip.frame import HipDataFrame

        return HipDataFrame(ArrayDataFrame(data, schema).as_arrow(), schema)


def test_sql_planner_on_gpu():
    import numpy as np
    import pandas as pd

    from fugue_amd.hip.execution_engine import HipExecutionEngine
    from fugue_amd.sql.executor import parse_select
    from fugue_amd.sql.planner import execute_plan

    e = HipExecutionEngine()
    rng = np.random.default_rng(0)
    a = pd.DataFrame(dict(k=rng.integers(0, 50, 50000), v=rng.random(50000)))
    b = pd.DataFrame(dict(k=np.arange(50), w=np.arange(50) * 2.0))
    stmt = parse_select(
        "SELECT a.k, w, SUM(v) AS s FROM a INNER JOIN b ON a.k = b.k "
        "WHERE v > 0.25 GROUP BY k, w ORDER BY s DESC LIMIT 10"
    )
    res = execute_plan(stmt, dict(a=e.to_df(a), b=e.to_df(b)), e)
    from fugue_amd.hip.frame import HipDataFrame

    assert isinstance(res, HipDataFrame) or res.count() == 10
    exp = (
        a[a.v > 0.25]
        .merge(b, on="k")
        .groupby(["k", "w"], as_index=False)
        .agg(s=("v", "sum"))
        .nlargest(10, "s")
    )
    got = res.as_pandas().sort_values("s", ascending=False).reset_index(drop=True)
    np.testing.assert_allclose(got["s"].values, exp["s"].values, rtol=1e-9)


from fugue_amd.testing.suites import ExecutionEngineEdgeCaseTests


class TestHipGpuEdgeCases(ExecutionEngineEdgeCaseTests):
    @classmethod
    def make_engine(cls):
        from fugue_amd.hip.execution_engine import HipExecutionEngine

        return HipExecutionEngine()


from fugue_amd.testing.execution_conformance import ExecutionEngineConformance


class TestHipGpuExecutionConformance(ExecutionEngineConformance):
    """All 42 reference ExecutionEngineTests cases on device tensors."""

    native_is_fugue = True  # HipDataFrame is both native and fugue

    @classmethod
    def make_engine(cls):
        from fugue_amd.hip.execution_engine import HipExecutionEngine

        return HipExecutionEngine()


from fugue_amd.testing.builtin_conformance import BuiltInConformance


class TestHipGpuBuiltInConformance(BuiltInConformance):
    """All 45 reference BuiltInTests cases on device tensors."""

    @classmethod
    def make_engine(cls):
        from fugue_amd.hip.execution_engine import HipExecutionEngine

        return HipExecutionEngine()


from fugue_amd.testing.dataframe_conformance import DataFrameConformance


class TestHipGpuDataFrameConformance(DataFrameConformance):
    """All 24 reference DataFrameTests cases on device tensors."""

    supports_nested = False  # flat device columns (documented deviation)
    supports_map = False
    native_is_fugue = True

    def df(self, data: Any = None, schema: Any = None):
        from fugue_amd.hip.frame import HipDataFrame

        return HipDataFrame(
            ArrayDataFrame(data, schema).as_arrow(), schema, device="cuda:0"
        )
