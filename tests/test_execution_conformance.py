"""Reference ExecutionEngineTests (42 cases) instantiated for native and
hip(CPU) — the GPU instantiation is in tests/test_suites_gpu.py."""
from fugue_amd.execution import NativeExecutionEngine
from fugue_amd.testing.execution_conformance import ExecutionEngineConformance


class TestNativeExecutionConformance(ExecutionEngineConformance):
    @classmethod
    def make_engine(cls):
        return NativeExecutionEngine()


class TestHipCpuExecutionConformance(ExecutionEngineConformance):
    native_is_fugue = True  # HipDataFrame is both native and fugue

    @classmethod
    def make_engine(cls):
        from fugue_amd.hip.execution_engine import HipExecutionEngine

        return HipExecutionEngine()
