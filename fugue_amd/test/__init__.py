from fugue_amd.test.plugins import (
    FugueTestBackend,
    FugueTestContext,
    fugue_test_backend,
    fugue_test_suite,
    with_backend,
)
