"""Conformance suites instantiated for every engine/frame type
(the reference's 2-3-line backend test files)."""
from typing import Any

import pytest

from fugue_amd import ArrayDataFrame, ArrowDataFrame, PandasDataFrame
from fugue_amd.execution import NativeExecutionEngine
from fugue_amd.testing import (
    BuiltInWorkflowTestSuite,
    DataFrameTestSuite,
    ExecutionEngineTestSuite,
)


class TestNativeExecutionEngine(ExecutionEngineTestSuite):
    @classmethod
    def make_engine(cls):
        return NativeExecutionEngine()


class TestNativeBuiltIn(BuiltInWorkflowTestSuite):
    @classmethod
    def make_engine(cls):
        return NativeExecutionEngine()


class TestHipCpuExecutionEngine(ExecutionEngineTestSuite):
    """The MI355X engine with CPU-resident tensors (no GPU in CI); the
    same code paths run the HIP kernels when tensors are device-resident."""

    @classmethod
    def make_engine(cls):
        from fugue_amd.hip.execution_engine import HipExecutionEngine

        return HipExecutionEngine()


class TestHipCpuBuiltIn(BuiltInWorkflowTestSuite):
    @classmethod
    def make_engine(cls):
        from fugue_amd.hip.execution_engine import HipExecutionEngine

        return HipExecutionEngine()


class TestArrayDataFrame(DataFrameTestSuite):
    @classmethod
    def make_df(cls, data: Any, schema: Any):
        return ArrayDataFrame(data, schema)


class TestPandasDataFrame(DataFrameTestSuite):
    @classmethod
    def make_df(cls, data: Any, schema: Any):
        return PandasDataFrame(data, schema)


class TestArrowDataFrame(DataFrameTestSuite):
    @classmethod
    def make_df(cls, data: Any, schema: Any):
        return ArrowDataFrame(data, schema)


class TestHipDataFrameCpu(DataFrameTestSuite):
    supports_nested = False  # device columns are flat (validity + data)
    @classmethod
    def make_df(cls, data: Any, schema: Any):
        from fugue_amd.hip.frame import HipDataFrame

        return HipDataFrame(
            ArrayDataFrame(data, schema).as_arrow(), schema, device="cpu"
        )


from fugue_amd.testing.suites import ExecutionEngineEdgeCaseTests


class TestNativeEdgeCases(ExecutionEngineEdgeCaseTests):
    @classmethod
    def make_engine(cls):
        return NativeExecutionEngine()


class TestHipCpuEdgeCases(ExecutionEngineEdgeCaseTests):
    @classmethod
    def make_engine(cls):
        from fugue_amd.hip.execution_engine import HipExecutionEngine

        return HipExecutionEngine()


from fugue_amd.testing.suites import BagTestSuite


class TestArrayBag(BagTestSuite):
    pass
