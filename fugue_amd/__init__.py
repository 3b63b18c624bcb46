"""fugue_amd: MI355X-native distributed DataFrame execution engine with
Fugue-compatible contracts (see SURVEY.md for the reference analysis)."""

__version__ = "0.1.0"

from fugue_amd.schema import Schema
from fugue_amd.exceptions import *  # noqa
from fugue_amd.dataframe import (
    AnyDataFrame,
    ArrayDataFrame,
    ArrowDataFrame,
    DataFrame,
    DataFrames,
    IterableArrowDataFrame,
    IterableDataFrame,
    IterablePandasDataFrame,
    LocalBoundedDataFrame,
    LocalDataFrame,
    LocalDataFrameIterableDataFrame,
    LocalUnboundedDataFrame,
    PandasDataFrame,
    YieldedDataFrame,
    as_fugue_df,
)
from fugue_amd.dataset import Dataset
from fugue_amd.registry import load_entry_point_plugins
from fugue_amd.registry import register_builtins as _register_builtins

_register_builtins()
load_entry_point_plugins()

# top-level convenience exports (reference parity: ``fugue/__init__.py``)
from fugue_amd.bag import ArrayBag, Bag, LocalBag  # noqa: E402
from fugue_amd.collections.partition import (  # noqa: E402
    BagPartitionCursor,
    PartitionCursor,
    PartitionSpec,
)
from fugue_amd.collections.sql import StructuredRawSQL, TempTableName  # noqa: E402
from fugue_amd.collections.yielded import PhysicalYielded, Yielded  # noqa: E402
from fugue_amd.constants import register_global_conf  # noqa: E402
from fugue_amd.execution import (  # noqa: E402
    ExecutionEngine,
    MapEngine,
    NativeExecutionEngine,
    SQLEngine,
)
from fugue_amd.execution.factory import (  # noqa: E402
    make_execution_engine,
    register_execution_engine,
)
from fugue_amd.extensions import (  # noqa: E402
    CoTransformer,
    Creator,
    OutputCoTransformer,
    Outputter,
    OutputTransformer,
    Processor,
    Transformer,
)
from fugue_amd.workflow import (  # noqa: E402
    FugueWorkflow,
    WorkflowDataFrame,
    out_transform,
    transform,
)
from fugue_amd.workflow.module import module  # noqa: E402
from fugue_amd.workflow._workflow_context import (  # noqa: E402
    FugueWorkflowContext,
)
from fugue_amd.sql.api import fugue_sql, fugue_sql_flow  # noqa: E402
from fugue_amd.sql.workflow import FugueSQLWorkflow  # noqa: E402
