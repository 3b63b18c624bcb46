from fugue_amd.testing.suites import (
    BuiltInWorkflowTestSuite,
    DataFrameTestSuite,
    ExecutionEngineTestSuite,
)
