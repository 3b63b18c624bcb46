from fugue_amd.dataframe.array_dataframe import ArrayDataFrame
from fugue_amd.dataframe.arrow_dataframe import ArrowDataFrame
from fugue_amd.dataframe.dataframe import (
    AnyDataFrame,
    DataFrame,
    DataFrameDisplay,
    LocalBoundedDataFrame,
    LocalDataFrame,
    LocalUnboundedDataFrame,
    YieldedDataFrame,
    as_fugue_df,
)
from fugue_amd.dataframe.dataframe_iterable_dataframe import (
    IterableArrowDataFrame,
    IterablePandasDataFrame,
    LocalDataFrameIterableDataFrame,
)
from fugue_amd.dataframe.dataframes import DataFrames
from fugue_amd.dataframe.function_wrapper import (
    AnnotatedParam,
    DataFrameFunctionWrapper,
    DataFrameParam,
    EmptyAwareIterable,
    LocalDataFrameParam,
    make_empty_aware,
    register_annotated_param,
)
from fugue_amd.dataframe.iterable_dataframe import IterableDataFrame
from fugue_amd.dataframe.pandas_dataframe import PandasDataFrame
from fugue_amd.dataframe.utils import (
    deserialize_df,
    get_join_schemas,
    parse_join_type,
    serialize_df,
)
from fugue_amd.dataframe.api import (  # noqa: E402
    get_column_names,
    rename,
)
from fugue_amd.dataframe.function_wrapper import (  # noqa: E402
    fugue_annotated_param,
    register_annotated_param,
)
from fugue_amd.dataframe.utils import (  # noqa: E402
    normalize_dataframe_column_names,
)
