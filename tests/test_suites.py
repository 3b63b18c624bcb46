"""Conformance suites instantiated for every engine/frame type
(the reference's 2-3-line backend test files)."""
from typing import Any

import pytest

from fugue_amd import ArrayDataFrame, ArrowDataFrame, PandasDataFrame
from fugue_amd.execution import NativeExecutionEngine
from fugue_amd.testing import (
    BuiltInWorkflowTestSuite,
    DataFrameTestSuite,
    ExecutionEngineTestSuite,
)


class TestNativeExecutionEngine(ExecutionEngineTestSuite):
    @classmethod
    def make_engine(cls):
        return NativeExecutionEngine()


class TestNativeBuiltIn(BuiltInWorkflowTestSuite):
    @classmethod
    def make_engine(cls):
        return NativeExecutionEngine()


class TestHipCpuExecutionEngine(ExecutionEngineTestSuite):
    """The MI355X engine with CPU-resident tensors (no GPU in CI); the
    same code paths run the HIP kernels when tensors are device-resident."""

    @classmethod
    def make_engine(cls):
        from fugue_amd.hip.execution_engine import HipExecutionEngine

        return HipExecutionEngine()


class TestHipCpuBuiltIn(BuiltInWorkflowTestSuite):
    @classmethod
    def make_engine(cls):
        from fugue_amd.hip.execution_engine import HipExecutionEngine

        return HipExecutionEngine()


class TestArrayDataFrame(DataFrameTestSuite):
    @classmethod
    def make_df(cls, data: Any, schema: Any):
        return ArrayDataFrame(data, schema)


class TestPandasDataFrame(DataFrameTestSuite):
    @classmethod
    def make_df(cls, data: Any, schema: Any):
        return PandasDataFrame(data, schema)


class TestArrowDataFrame(DataFrameTestSuite):
    @classmethod
    def make_df(cls, data: Any, schema: Any):
        return ArrowDataFrame(data, schema)


class TestHipDataFrameCpu(DataFrameTestSuite):
    supports_nested = False  # device columns are flat (validity + data)
    @classmethod
    def make_df(cls, data: Any, schema: Any):
        from fugue_amd.hip.frame import HipDataFrame

        return HipDataFrame(
            ArrayDataFrame(data, schema).as_arrow(), schema, device="cpu"
        )


from fugue_amd.testing.suites import ExecutionEngineEdgeCaseTests


class TestNativeEdgeCases(ExecutionEngineEdgeCaseTests):
    @classmethod
    def make_engine(cls):
        return NativeExecutionEngine()


class TestHipCpuEdgeCases(ExecutionEngineEdgeCaseTests):
    @classmethod
    def make_engine(cls):
        from fugue_amd.hip.execution_engine import HipExecutionEngine

        return HipExecutionEngine()


from fugue_amd.testing.suites import BagTestSuite


class TestArrayBag(BagTestSuite):
    pass


def test_hip_cpu_distinct_aggregates():
    """SUM/AVG/MIN DISTINCT on the device engine (CPU tensors), checked
    against the pandas groupby comparator."""
    import numpy as np
    import pandas as pd

    import fugue_amd.api as fa
    from fugue_amd.column import functions as f
    from fugue_amd.column.expressions import _UnaryAggFuncExpr, col
    from fugue_amd.hip.execution_engine import HipExecutionEngine

    e = HipExecutionEngine()
    rng = np.random.default_rng(11)
    pdf = pd.DataFrame(
        dict(
            k=rng.integers(0, 15, 2000),
            v=rng.integers(0, 6, 2000).astype("f8"),
        )
    )
    res = fa.aggregate(
        pdf,
        partition_by="k",
        engine=e,
        sd=_UnaryAggFuncExpr("SUM", col("v"), arg_distinct=True),
        ad=_UnaryAggFuncExpr("AVG", col("v"), arg_distinct=True),
        md=_UnaryAggFuncExpr("MIN", col("v"), arg_distinct=True),
        tot=f.sum(col("v")),
    )
    got = fa.as_pandas(res).sort_values("k").reset_index(drop=True)
    exp = (
        pdf.groupby("k", as_index=False)
        .agg(
            sd=("v", lambda s: s.drop_duplicates().sum()),
            ad=("v", lambda s: s.drop_duplicates().mean()),
            md=("v", "min"),
            tot=("v", "sum"),
        )
        .sort_values("k")
        .reset_index(drop=True)
    )
    for c in ("sd", "ad", "md", "tot"):
        assert np.allclose(got[c].to_numpy(float), exp[c].to_numpy(float)), c


def test_hip_cpu_like_general_patterns():
    """Full LIKE wildcard support on the device engine (CPU tensors):
    `_`, interior `%`, and mixes, vs a regex comparator."""
    import re as _re

    import numpy as np
    import pandas as pd

    import fugue_amd.api as fa
    from fugue_amd.hip.execution_engine import HipExecutionEngine

    e = HipExecutionEngine()
    rng = np.random.default_rng(7)
    alphabet = list("abcdez")
    vals = [
        "".join(rng.choice(alphabet, rng.integers(0, 8)).tolist())
        for _ in range(500)
    ] + ["", "abc", "abcde", "azc", "aXc"]
    pdf = pd.DataFrame(dict(s=vals, i=range(len(vals))))

    def like_to_re(p):
        out = "^"
        for ch in p:
            if ch == "%":
                out += ".*"
            elif ch == "_":
                out += "."
            else:
                out += _re.escape(ch)
        return out + "$"

    patterns = [
        "a_c", "%b_d%", "a%c%e", "_bc", "a__%", "%_z", "abc",
        "a%c", "%de", "%%", "a_%_e", "__", "%a%b%",
    ]
    from fugue_amd.hip import expr as hexpr

    for p in patterns:
        got = fa.as_pandas(
            fa.fugue_sql(
                f"SELECT i FROM pdf WHERE s LIKE '{p}'", engine=e
            )
        )["i"].sort_values().tolist()
        rx = _re.compile(like_to_re(p))
        exp = pdf[pdf["s"].map(lambda s: rx.match(s) is not None)][
            "i"
        ].sort_values().tolist()
        assert got == exp, (p, got[:10], exp[:10])


def test_hip_cpu_global_distinct_aggregates():
    """Keyless SUM/AVG/COUNT DISTINCT on the device engine (CPU
    tensors) vs pandas comparator."""
    import numpy as np
    import pandas as pd

    import fugue_amd.api as fa
    from fugue_amd.hip.execution_engine import HipExecutionEngine

    e = HipExecutionEngine()
    rng = np.random.default_rng(21)
    pdf = pd.DataFrame(dict(v=rng.integers(0, 9, 5000).astype("f8"),
                            w=rng.random(5000)))
    r = fa.as_pandas(
        fa.fugue_sql(
            "SELECT SUM(DISTINCT v) AS s, AVG(DISTINCT v) AS a, "
            "COUNT(DISTINCT v) AS c, SUM(w) AS tw FROM pdf",
            engine=e,
        )
    )
    dd = pdf["v"].drop_duplicates()
    assert float(r["s"][0]) == dd.sum()
    assert abs(float(r["a"][0]) - dd.mean()) < 1e-9
    assert int(r["c"][0]) == dd.nunique()
    assert abs(float(r["tw"][0]) - pdf["w"].sum()) < 1e-6


def test_hip_cpu_like_all_empty_strings():
    """LIKE over a column where every row is the empty string: the byte
    buffer is empty and the simple/general matchers must not index into
    it (r01 advisor finding — IndexError on index_select)."""
    import pandas as pd

    import fugue_amd.api as fa
    from fugue_amd.hip.execution_engine import HipExecutionEngine

    e = HipExecutionEngine()
    pdf = pd.DataFrame(dict(s=["", "", ""], i=[0, 1, 2]))
    for p, expect in [
        ("a", []), ("%a%", []), ("a%", []), ("%a", []),
        ("%", [0, 1, 2]), ("%%", [0, 1, 2]), ("", [0, 1, 2]),
        ("_", []), ("a_c", []), ("%_%", []), ("a%c%e", []),
    ]:
        got = fa.as_pandas(
            fa.fugue_sql(
                f"SELECT i FROM pdf WHERE s LIKE '{p}'", engine=e
            )
        )["i"].sort_values().tolist()
        assert got == expect, (p, got)


def test_sql_first_last_distinct_not_dropped():
    """FIRST(DISTINCT x)/LAST(DISTINCT x) must not silently plan as
    plain FIRST/LAST on the device path (r01 advisor finding): the
    planner refuses the distinct qualifier and the host executor
    computes it over the deduplicated sequence (LAST over [1,2,1]
    distinct-ordered [1,2] is 2, not 1)."""
    import pandas as pd

    import fugue_amd.api as fa
    from fugue_amd.hip.execution_engine import HipExecutionEngine

    e = HipExecutionEngine()
    pdf = pd.DataFrame(dict(k=[1, 1, 1], v=[1, 2, 1]))
    r = fa.as_pandas(
        fa.fugue_sql(
            "SELECT k, LAST(DISTINCT v) AS lv, FIRST(DISTINCT v) AS fv "
            "FROM pdf GROUP BY k",
            engine=e,
        )
    )
    assert int(r["lv"][0]) == 2
    assert int(r["fv"][0]) == 1


def test_sql_sub_unknown_named_input_errors():
    """SUB with an unknown named input variable raises a syntax error
    naming the variable, not a bare KeyError (r01 advisor finding)."""
    import pytest as _pytest

    import fugue_amd.api as fa
    from fugue_amd.exceptions import FugueSQLSyntaxError

    with _pytest.raises(FugueSQLSyntaxError) as ei:
        fa.fugue_sql_flow("SUB a:unknown_df USING mymod")
    assert "unknown_df" in str(ei.value)


def test_keyless_device_transform():
    """A HipDataFrame-annotated transformer with no partition keys must
    stay on the device path (no pandas round trip) — the bench's
    TRANSFORM stage shape.  CPU tensors here; same code path on GPU."""
    import pandas as pd
    import pyarrow as pa
    import torch

    import fugue_amd.api as fa
    from fugue_amd.hip.execution_engine import HipExecutionEngine
    from fugue_amd.hip.frame import DeviceColumn, HipDataFrame
    from fugue_amd.schema import Schema

    e = HipExecutionEngine()
    n = 1000
    fact = HipDataFrame.from_columns(
        {
            "k": DeviceColumn(
                torch.arange(n, dtype=torch.int64), None, pa.int64()
            ),
            "v": DeviceColumn(
                torch.ones(n, dtype=torch.float64), None, pa.float64()
            ),
        },
        Schema("k:long,v:double"),
        e.device,
    )

    calls = []

    def scale(df: HipDataFrame) -> HipDataFrame:
        calls.append(type(df).__name__)
        v = df.col("v")
        return HipDataFrame.from_columns(
            {"k": df.col("k"),
             "v": DeviceColumn(v.data * 2.0, v.valid, pa.float64())},
            Schema("k:long,v:double"),
            df.device,
        )

    out = fa.transform(fact, scale, schema="k:long,v:double", engine=e,
                       as_fugue=True)
    assert calls == ["HipDataFrame"]
    assert isinstance(out, HipDataFrame) or isinstance(
        e.to_df(out), HipDataFrame
    )
    pdf = out.as_pandas()
    assert len(pdf) == n and (pdf["v"] == 2.0).all()

    # same through FugueSQL TRANSFORM + SELECT (the bench pipeline)
    res = fa.as_pandas(
        fa.fugue_sql(
            "t = TRANSFORM fact USING scale SCHEMA k:long,v:double\n"
            "SELECT SUM(v) AS s FROM t",
            fact=fact, scale=scale, engine=e,
        )
    )
    assert float(res["s"][0]) == 2.0 * n


def test_hip_engine_nested_decimal_fallback():
    """The device engine accepts frames with nested / decimal columns:
    they stay host-resident and every op still works (documented
    deviation — device columns are flat)."""
    import decimal

    import pandas as pd
    import pyarrow as pa

    import fugue_amd.api as fa
    from fugue_amd import ArrayDataFrame
    from fugue_amd.hip.execution_engine import HipExecutionEngine

    e = HipExecutionEngine()
    nested = ArrayDataFrame(
        [[1, {"x": 1}, [1, 2]], [2, {"x": 2}, [3]], [1, None, []]],
        "k:long,s:{x:long},l:[long]",
    )
    # relational ops via the engine: distinct/filter/take/join
    d = e.to_df(nested)
    assert e.take(d, 2, presort="k").count() == 2
    dim = ArrayDataFrame([[1, "a"], [2, "b"]], "k:long,name:str")
    joined = e.join(d, e.to_df(dim), how="inner")
    got = joined.as_array(type_safe=True)
    assert len(got) == 3
    # map/transform over nested input
    # schema: k:long,n:long
    def count_list(df: pd.DataFrame) -> pd.DataFrame:
        return pd.DataFrame(
            dict(k=[df["k"].iloc[0]], n=[sum(len(x) for x in df["l"])])
        )

    res = fa.transform(
        d, count_list, partition=dict(by=["k"]), engine=e, as_fugue=True
    )
    out = {r[0]: r[1] for r in res.as_array()}
    assert out == {1: 2, 2: 1}

    # decimal columns round-trip through engine ops
    dec = ArrayDataFrame(
        [[1, decimal.Decimal("1.50")], [2, decimal.Decimal("2.25")]],
        "k:long,v:decimal(10,2)",
    )
    dd = e.to_df(dec)
    assert e.distinct(dd).count() == 2
    arr = e.take(dd, 1, presort="k desc").as_array(type_safe=True)
    assert arr[0][0] == 2
