"""Reference DataFrameTests (24 cases) + BagTests (6) instantiated for
every frame type; the device-frame GPU instantiation is in
tests/test_suites_gpu.py."""
from typing import Any

import pandas as pd

from fugue_amd import ArrayDataFrame, ArrowDataFrame, PandasDataFrame
from fugue_amd.bag.array_bag import ArrayBag
from fugue_amd.testing.dataframe_conformance import (
    BagConformance,
    DataFrameConformance,
    NativeDataFrameConformance,
)


class TestArrayDataFrameConformance(DataFrameConformance):
    native_is_fugue = True  # the raw array carries no schema

    def df(self, data: Any = None, schema: Any = None):
        return ArrayDataFrame(data, schema)


class TestPandasDataFrameConformance(NativeDataFrameConformance):
    def df(self, data: Any = None, schema: Any = None):
        return PandasDataFrame(data, schema)

    def to_native_df(self, pdf: pd.DataFrame):
        return pdf


class TestArrowDataFrameConformance(DataFrameConformance):
    def df(self, data: Any = None, schema: Any = None):
        return ArrowDataFrame(ArrayDataFrame(data, schema).as_arrow())


class TestHipCpuDataFrameConformance(DataFrameConformance):
    # device columns are flat buffers (validity + data); nested and map
    # data stays on host frames (documented deviation)
    supports_nested = False
    supports_map = False
    native_is_fugue = True

    def df(self, data: Any = None, schema: Any = None):
        from fugue_amd.hip.frame import HipDataFrame

        return HipDataFrame(
            ArrayDataFrame(data, schema).as_arrow(), schema, device="cpu"
        )


class TestArrayBagConformance(BagConformance):
    def bg(self, data: Any = None):
        return ArrayBag(data)
