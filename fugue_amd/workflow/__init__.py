from fugue_amd.workflow.workflow import (
    FugueWorkflow,
    FugueWorkflowResult,
    WorkflowDataFrame,
    WorkflowDataFrames,
)
from fugue_amd.workflow.api import out_transform, raw_sql, transform
from fugue_amd.workflow.module import module
from fugue_amd.workflow._workflow_context import (  # noqa: E402
    FugueWorkflowContext,
)


def register_raw_df_type(df_type: type) -> None:  # pragma: no cover
    """Deprecated in the reference (``fugue/workflow/input.py:4``):
    register via ``fugue_amd.api.is_df`` plugins instead."""
    raise DeprecationWarning("use fugue_amd.api.is_df to register the dataframe")
