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
