from fugue_amd.test.plugins import (
    FugueTestBackend,
    FugueTestContext,
    FugueTestSuite,
    extract_conf,
    fugue_test_backend,
    fugue_test_suite,
    with_backend,
    _NativeTestBackend as NativeTestBackend,
    _PandasTestBackend as PandasTestBackend,
)
