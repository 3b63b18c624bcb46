"""Correctness of the TPC-H-Q3-like FugueSQL pipeline on the MI355X
engine (CPU tensors; the GPU run is in tests/test_hip_gpu.py)."""
import sys
import os

import numpy as np
import pandas as pd
import pytest

sys.path.insert(0, os.path.join(os.path.dirname(__file__), ".."))

from benchmarks.q3_bench import Q3, gen_tables  # noqa: E402
import fugue_amd.api as fa  # noqa: E402
from fugue_amd.hip.execution_engine import HipExecutionEngine  # noqa: E402
from fugue_amd.sql.executor import parse_select  # noqa: E402
from fugue_amd.sql.planner import execute_plan  # noqa: E402


def _expected(customer, orders, lineitem):
    c = customer.as_pandas()
    o = orders.as_pandas()
    l = lineitem.as_pandas()
    m = (
        c[c.mktsegment == "BUILDING"]
        .merge(o, on="custkey")
        .merge(l, on="orderkey")
    )
    m = m[(m.orderdate < 9204) & (m.shipdate > 9204)]
    m["rev"] = m.extendedprice * (1 - m.discount)
    g = (
        m.groupby(["orderkey", "orderdate", "shippriority"], as_index=False)
        .agg(revenue=("rev", "sum"))
        .nlargest(10, "revenue")
    )
    return g


def test_q3_correctness_and_plan_lowering():
    engine = HipExecutionEngine()
    customer, orders, lineitem, _ = gen_tables(0.01, engine.device, 0)
    # 1. the plan must lower (no pandas fallback): execute_plan directly
    stmt = parse_select(
        Q3.replace("customer", "c_tbl")
        .replace("orders", "o_tbl")
        .replace("lineitem", "l_tbl")
    )
    res = execute_plan(
        stmt,
        dict(c_tbl=customer, o_tbl=orders, l_tbl=lineitem),
        engine,
    )
    exp = _expected(customer, orders, lineitem)
    got = res.as_pandas()
    assert len(got) == len(exp)
    np.testing.assert_allclose(
        got["revenue"].values, exp["revenue"].values, rtol=1e-9
    )
    # 2. through the public fugue_sql API
    res2 = fa.fugue_sql(
        Q3, customer=customer, orders=orders, lineitem=lineitem,
        engine=engine, as_fugue=True,
    )
    got2 = res2.as_pandas()
    np.testing.assert_allclose(
        got2["revenue"].values, exp["revenue"].values, rtol=1e-9
    )
