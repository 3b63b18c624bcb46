"""Public plugin points (reference parity: ``fugue/plugins.py``).

Third-party packages extend fugue_amd through these:

- ``register_execution_engine`` / ``register_sql_engine`` /
  ``register_default_execution_engine`` — engine aliases and types
- ``register_annotated_param`` — new UDF parameter frame types
  (the MI355X engine registers :class:`~fugue_amd.hip.udf.HipDataFrameParam`
  through this)
- ``register_creator`` / ``register_processor`` / ``register_outputter``
  / ``register_transformer`` — named extensions
- ``register_global_conf`` — configuration defaults
- setuptools entry points in the ``fugue.plugins`` / ``fugue_amd.plugins``
  groups are loaded at import (``fugue_amd.registry.load_entry_point_plugins``)
"""
# flake8: noqa
from fugue_amd.constants import register_global_conf
from fugue_amd.dataframe.function_wrapper import register_annotated_param
from fugue_amd.execution.factory import (
    make_execution_engine,
    make_sql_engine,
    register_default_execution_engine,
    register_execution_engine,
    register_sql_engine,
)
from fugue_amd.extensions import (
    register_creator,
    register_outputter,
    register_processor,
    register_transformer,
)
from fugue_amd.registry import load_entry_point_plugins

# full reference plugin surface (fugue/plugins.py)
from fugue_amd.collections.sql import transpile_sql
from fugue_amd.dataframe.api import (
    alter_columns,
    as_array,
    as_array_iterable,
    as_arrow,
    as_dict_iterable,
    as_dicts,
    as_pandas,
    drop_columns,
    get_column_names,
    get_schema,
    head,
    is_df,
    peek_array,
    peek_dict,
    rename,
    select_columns,
)
from fugue_amd.dataframe.function_wrapper import fugue_annotated_param
from fugue_amd.dataset.api import (
    as_fugue_dataset,
    as_local,
    as_local_bounded,
    count,
    get_num_partitions,
    is_bounded,
    is_empty,
    is_local,
)
from fugue_amd.dataset.dataset import get_dataset_display
from fugue_amd.execution.api import as_fugue_engine_df
from fugue_amd.execution.factory import (
    infer_execution_engine,
    parse_execution_engine,
    parse_sql_engine,
)
from fugue_amd.extensions import (
    namespace_candidate,
    parse_creator,
    parse_output_transformer,
    parse_outputter,
    parse_processor,
    parse_transformer,
)
