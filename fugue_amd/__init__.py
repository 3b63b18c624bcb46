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

# --- full reference top-level surface (reference fugue/__init__.py) --- #
from fugue_amd.bag.bag import BagDisplay  # noqa: E402
from fugue_amd.dataframe.dataframe import (  # noqa: E402
    AnyDataFrame,
    DataFrameDisplay,
)
from fugue_amd.dataset.api import as_fugue_dataset  # noqa: E402
from fugue_amd.dataset.dataset import (  # noqa: E402
    AnyDataset,
    DatasetDisplay,
    get_dataset_display,
)
from fugue_amd.execution.execution_engine import (  # noqa: E402
    AnyExecutionEngine,
    EngineFacet,
)
from fugue_amd.execution.native_execution_engine import (  # noqa: E402
    PandasMapEngine,
    PandasSQLEngine,
)

# the reference exports its qpd-backed pandas SQL engine under this name
# (fugue/execution/native_execution_engine.py:42); here the same role is
# played by the built-in pandas SQL executor
QPDPandasEngine = PandasSQLEngine
from fugue_amd.execution.factory import (  # noqa: E402
    is_pandas_or,
    make_sql_engine,
    register_default_execution_engine,
    register_default_sql_engine,
    register_sql_engine,
)
from fugue_amd.extensions import (  # noqa: E402
    cotransformer,
    creator,
    output_cotransformer,
    output_transformer,
    outputter,
    processor,
    register_creator,
    register_output_transformer,
    register_outputter,
    register_processor,
    register_transformer,
    transformer,
)
from fugue_amd.rpc import (  # noqa: E402
    EmptyRPCHandler,
    RPCClient,
    RPCFunc,
    RPCHandler,
    RPCServer,
    make_rpc_server,
    to_rpc_handler,
)
from fugue_amd.workflow.workflow import WorkflowDataFrames  # noqa: E402
from fugue_amd.sql.api import fugue_sql_flow as fsql  # noqa: E402
