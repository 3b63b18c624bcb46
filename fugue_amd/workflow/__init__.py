from fugue_amd.workflow.workflow import (
    FugueWorkflow,
    FugueWorkflowResult,
    WorkflowDataFrame,
    WorkflowDataFrames,
)
from fugue_amd.workflow.api import out_transform, raw_sql, transform
from fugue_amd.workflow.module import module
