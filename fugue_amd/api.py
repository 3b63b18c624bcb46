"""The ``fa.*`` functional namespace (reference parity: ``fugue/api.py``).

Usage::

    import fugue_amd.api as fa
    fa.transform(df, fn, schema="*")
"""
# flake8: noqa
from fugue_amd.dataset.api import (
    as_fugue_dataset,
    as_local,
    as_local_bounded,
    count,
    get_num_partitions,
    is_bounded,
    is_empty,
    is_local,
    show,
)
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
    get_native_as_df,
    get_schema,
    head,
    is_df,
    normalize_column_names,
    peek_array,
    peek_dict,
    rename,
    select_columns,
)
from fugue_amd.dataframe.dataframe import as_fugue_df
from fugue_amd.execution.api import (
    aggregate,
    anti_join,
    as_fugue_engine_df,
    assign,
    broadcast,
    clear_global_engine,
    cross_join,
    distinct,
    dropna,
    engine_context,
    fillna,
    filter,
    full_outer_join,
    get_context_engine,
    get_current_conf,
    get_current_parallelism,
    inner_join,
    intersect,
    join,
    left_outer_join,
    load,
    persist,
    repartition,
    right_outer_join,
    run_engine_function,
    sample,
    save,
    select,
    semi_join,
    set_global_engine,
    subtract,
    take,
    union,
)
from fugue_amd.workflow.api import out_transform, raw_sql, transform
from fugue_amd.sql.api import fugue_sql, fugue_sql_flow
