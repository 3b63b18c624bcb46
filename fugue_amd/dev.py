"""Developer-facing re-exports for extending fugue_amd (reference
parity: ``fugue/dev.py`` — everything needed to build a new engine,
frame type, or annotated UDF param in one import)."""
# flake8: noqa
from fugue_amd.bag.bag import BagDisplay
from fugue_amd.collections.partition import (
    BagPartitionCursor,
    PartitionCursor,
    PartitionSpec,
)
from fugue_amd.collections.sql import StructuredRawSQL, TempTableName
from fugue_amd.collections.yielded import PhysicalYielded, Yielded
from fugue_amd.dataframe.function_wrapper import (
    AnnotatedParam,
    DataFrameFunctionWrapper,
    DataFrameParam,
    ExecutionEngineParam,
    LocalDataFrameParam,
    register_annotated_param,
)
from fugue_amd.dataset import DatasetDisplay
from fugue_amd.execution.execution_engine import (
    EngineFacet,
    ExecutionEngine,
    MapEngine,
    SQLEngine,
)
from fugue_amd.execution.factory import (
    make_execution_engine,
    make_sql_engine,
    register_default_execution_engine,
    register_execution_engine,
    register_sql_engine,
)
from fugue_amd.utils.params import ParamDict
from fugue_amd.utils.hash import to_uuid
