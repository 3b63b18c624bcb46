"""Device-resident UDF boundary: transformers annotated with
:class:`~fugue_amd.hip.frame.HipDataFrame` receive the HBM-resident shard
directly — no D2H/H2D staging, no pandas conversion.

This is the MI355X analog of the reference's ``fugue_polars`` pattern
(``fugue_polars/registry.py:24-41`` registers Polars frames as annotated
transformer params on any engine): the param class plugs into
``register_annotated_param`` and sets a ``"device"`` format hint that
:class:`~fugue_amd.hip.execution_engine.HipMapEngine` honors with a
zero-copy per-partition slice path.
"""
from typing import Any

from fugue_amd.dataframe.dataframe import DataFrame
from fugue_amd.dataframe.function_wrapper import (
    DataFrameParam,
    register_annotated_param,
)
from fugue_amd.hip.frame import HipDataFrame


class HipDataFrameParam(DataFrameParam):
    """``def f(df: HipDataFrame) -> HipDataFrame`` transformer param."""

    code = "d"
    need_schema_ = None
    format_hint_ = "device"

    @staticmethod
    def matches(anno: Any) -> bool:
        return anno is HipDataFrame

    def to_input_data(self, df: DataFrame, ctx: Any) -> Any:
        if isinstance(df, HipDataFrame):
            return df
        import torch

        device = "cuda" if torch.cuda.is_available() else "cpu"
        return HipDataFrame(df.as_arrow(), df.schema, device=device)

    def to_output_df(self, output: Any, schema: Any, ctx: Any) -> DataFrame:
        if not isinstance(output, HipDataFrame):
            raise ValueError(
                f"device transformer must return HipDataFrame, got {type(output)}"
            )
        if schema is not None:
            from fugue_amd.schema import Schema

            sc = schema if isinstance(schema, Schema) else Schema(schema)
            if output.schema != sc:
                raise ValueError(f"schema mismatch: {output.schema} vs {sc}")
        return output

    def count(self, obj: Any) -> int:
        return obj.count()


def register_device_params() -> None:
    register_annotated_param(HipDataFrameParam, prepend=True)
