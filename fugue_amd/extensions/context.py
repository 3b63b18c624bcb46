"""ExtensionContext: the context variables every extension can access.

Reference parity: ``fugue/extensions/context.py:13``.
"""
from typing import Any, Dict, Union

from fugue_amd.collections.partition import PartitionCursor, PartitionSpec
from fugue_amd.dataframe.dataframe import DataFrame
from fugue_amd.dataframe.dataframes import DataFrames
from fugue_amd.extensions._utils import validate_input_schema, validate_partition_spec
from fugue_amd.schema import Schema
from fugue_amd.utils.convert import get_full_type_path
from fugue_amd.utils.hash import to_uuid
from fugue_amd.utils.params import ParamDict


class ExtensionContext:
    @property
    def params(self) -> ParamDict:
        return self._params  # type: ignore

    @property
    def workflow_conf(self) -> ParamDict:
        if "_workflow_conf" in self.__dict__:
            return self._workflow_conf  # type: ignore
        return self.execution_engine.conf

    @property
    def execution_engine(self) -> Any:
        return self._execution_engine  # type: ignore

    @property
    def output_schema(self) -> Schema:
        return self._output_schema  # type: ignore

    @property
    def key_schema(self) -> Schema:
        return self._key_schema  # type: ignore

    @property
    def partition_spec(self) -> PartitionSpec:
        return self._partition_spec  # type: ignore

    @property
    def cursor(self) -> PartitionCursor:
        return self._cursor  # type: ignore

    @property
    def has_callback(self) -> bool:
        return "_has_rpc_client" in self.__dict__ and self._has_rpc_client  # type: ignore

    @property
    def callback(self) -> Any:
        return self._rpc_client  # type: ignore

    @property
    def rpc_server(self) -> Any:
        return self._rpc_server  # type: ignore

    @property
    def validation_rules(self) -> Dict[str, Any]:
        return {}

    def validate_on_compile(self) -> None:
        validate_partition_spec(self.partition_spec, self.validation_rules)

    def validate_on_runtime(self, data: Union[DataFrame, DataFrames]) -> None:
        if isinstance(data, DataFrame):
            validate_input_schema(data.schema, self.validation_rules)
        else:
            for df in data.values():
                validate_input_schema(df.schema, self.validation_rules)

    def __uuid__(self) -> str:
        return to_uuid(get_full_type_path(self))
