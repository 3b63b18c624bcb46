"""Reference BuiltInTests (45 cases) instantiated for native and
hip(CPU); the GPU instantiation is in tests/test_suites_gpu.py."""
from fugue_amd.execution import NativeExecutionEngine
from fugue_amd.testing.builtin_conformance import BuiltInConformance


class TestNativeBuiltInConformance(BuiltInConformance):
    @classmethod
    def make_engine(cls):
        return NativeExecutionEngine()


class TestHipCpuBuiltInConformance(BuiltInConformance):
    @classmethod
    def make_engine(cls):
        from fugue_amd.hip.execution_engine import HipExecutionEngine

        return HipExecutionEngine()
