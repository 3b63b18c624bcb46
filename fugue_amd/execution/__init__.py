from fugue_amd.execution.execution_engine import (
    AnyExecutionEngine,
    EngineFacet,
    ExecutionEngine,
    FugueEngineBase,
    MapEngine,
    SQLEngine,
)
from fugue_amd.execution.native_execution_engine import (
    NativeExecutionEngine,
    PandasMapEngine,
    PandasSQLEngine,
)
from fugue_amd.execution.factory import (
    infer_execution_engine,
    is_pandas_or,
    make_execution_engine,
    make_sql_engine,
    register_default_execution_engine,
    register_default_sql_engine,
    register_execution_engine,
    register_sql_engine,
    try_get_context_execution_engine,
)
from fugue_amd.execution.native_execution_engine import (  # noqa: E402
    PandasMapEngine,
    PandasSQLEngine,
)

# reference name for the pandas SQL facet (see fugue_amd/__init__.py)
QPDPandasEngine = PandasSQLEngine
from fugue_amd.execution.factory import (  # noqa: E402
    infer_execution_engine,
    parse_execution_engine,
    parse_sql_engine,
)
