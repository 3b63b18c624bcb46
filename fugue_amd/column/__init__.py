from fugue_amd.column.expressions import (
    ColumnExpr,
    all_cols,
    col,
    function,
    lit,
    null,
)
from fugue_amd.column import functions
from fugue_amd.column.sql import SelectColumns, SQLExpressionGenerator
from fugue_amd.column.functions import is_agg  # noqa: E402
