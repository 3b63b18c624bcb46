import os
import tempfile
from typing import Any, Dict, Iterable, List

import numpy as np
import pandas as pd
import pytest

import fugue_amd.api as fa
from fugue_amd.exceptions import FugueSQLError, FugueSQLSyntaxError
from fugue_amd.sql import FugueSQLWorkflow, fugue_sql, fugue_sql_flow


def test_select_basic():
    a = pd.DataFrame(dict(x=[1, 2, 3]))
    res = fa.fugue_sql("SELECT x, x*2 AS x2 FROM a WHERE x > 1", a=a)
    assert res.values.tolist() == [[2, 4], [3, 6]]


def test_assignment_chain():
    a = pd.DataFrame(dict(x=[1, 2, 3]))
    res = fa.fugue_sql(
        """
        b = SELECT x FROM a WHERE x > 1
        c = SELECT x+10 AS y FROM b
        SELECT * FROM c
        """,
        a=a,
    )
    assert sorted(res["y"].tolist()) == [12, 13]


def test_implicit_from():
    res = fa.fugue_sql(
        """
        CREATE [[1],[2],[3]] SCHEMA x:int
        SELECT * WHERE x >= 2
        SELECT x+1 AS y
        """
    )
    assert sorted(res["y"].tolist()) == [3, 4]


def test_transform_prepartition():
    a = pd.DataFrame(dict(g=["a", "a", "b"], v=[3, 1, 2]))

    # schema: g:str,top:long
    def top1(df: pd.DataFrame) -> pd.DataFrame:
        return pd.DataFrame(dict(g=[df["g"].iloc[0]], top=[df["v"].iloc[0]]))

    res = fa.fugue_sql(
        "TRANSFORM a PREPARTITION BY g PRESORT v DESC USING top1",
        a=a,
        top1=top1,
    )
    assert sorted(res.values.tolist()) == [["a", 3], ["b", 2]]


def test_take_sample_drop_rename_alter():
    a = pd.DataFrame(dict(x=[3.0, 1.0, None], y=["a", "b", "c"]))
    r1 = fa.fugue_sql("TAKE 2 ROWS FROM a PRESORT x", a=a)
    assert len(r1) == 2
    r2 = fa.fugue_sql("SAMPLE 2 ROWS FROM a", a=a)
    assert len(r2) == 2
    r3 = fa.fugue_sql("DROP ROWS IF ANY NULL FROM a", a=a)
    assert len(r3) == 2
    r4 = fa.fugue_sql("DROP COLUMNS y FROM a", a=a)
    assert list(r4.columns) == ["x"]
    r5 = fa.fugue_sql("RENAME COLUMNS x:xx FROM a", a=a)
    assert list(r5.columns) == ["xx", "y"]
    r6 = fa.fugue_sql("ALTER COLUMNS x:str FROM a", a=a)
    assert r6["x"].iloc[0] == "3"
    r7 = fa.fugue_sql("FILL NULLS (x=0) FROM a", a=a)
    assert r7["x"].tolist() == [3.0, 1.0, 0.0]


def test_save_and_load():
    a = pd.DataFrame(dict(x=[1, 2]))
    with tempfile.TemporaryDirectory() as tmp:
        pq = os.path.join(tmp, "t.parquet")
        fa.fugue_sql_flow(f"SAVE a OVERWRITE PARQUET '{pq}'", a=a).run()
        res = fa.fugue_sql(f"LOAD '{pq}'")
        assert len(res) == 2
        csv = os.path.join(tmp, "t.csv")
        fa.fugue_sql_flow(
            f"SAVE a OVERWRITE CSV '{csv}' (header=true)", a=a
        ).run()
        res2 = fa.fugue_sql(
            f"LOAD CSV '{csv}' (header=true, infer_schema=true)"
        )
        assert len(res2) == 2


def test_yield_and_checkpoint():
    dag = fugue_sql_flow(
        """
        a = CREATE [[1],[2]] SCHEMA x:int
        b = SELECT x+1 AS y FROM a PERSIST
        YIELD DATAFRAME AS out
        """
    )
    res = dag.run()
    assert sorted(r[0] for r in res["out"].as_array()) == [2, 3]


def test_print_and_outtransform(capsys):
    side: List[int] = []

    def sink(rows: List[List[Any]]) -> None:
        side.append(len(rows))

    fa.fugue_sql_flow(
        """
        a = CREATE [[1],[2]] SCHEMA x:int
        PRINT a TITLE 'hello'
        OUTTRANSFORM a USING sink
        """,
        sink=sink,
    ).run()
    out = capsys.readouterr().out
    assert "hello" in out
    assert sum(side) == 2


def test_zip_cotransform_sql():
    a = pd.DataFrame(dict(k=[1, 2], v=[10, 20]))
    b = pd.DataFrame(dict(k=[1, 1], w=[5.0, 6.0]))

    # schema: k:long,n:long
    def count_pair(df1: pd.DataFrame, df2: pd.DataFrame) -> Iterable[Dict[str, Any]]:
        yield dict(k=int(df1["k"].iloc[0]), n=len(df2))

    res = fa.fugue_sql(
        """
        z = ZIP a, b LEFT OUTER BY k
        TRANSFORM z USING count_pair
        """,
        a=a,
        b=b,
        count_pair=count_pair,
    )
    assert sorted(res.values.tolist()) == [[1, 2], [2, 0]]


def test_process_output_sql():
    a = pd.DataFrame(dict(x=[1, 2]))
    collected: List[int] = []

    def doubler(df: pd.DataFrame) -> pd.DataFrame:
        return df * 2

    def collect(rows: List[List[Any]]) -> None:
        collected.extend(r[0] for r in rows)

    res = fa.fugue_sql(
        """
        b = PROCESS a USING doubler SCHEMA x:long
        OUTPUT b USING collect
        SELECT * FROM b
        """,
        a=a,
        doubler=doubler,
        collect=collect,
    )
    assert sorted(collected) == [2, 4]
    assert sorted(res["x"].tolist()) == [2, 4]


def test_jinja_template():
    a = pd.DataFrame(dict(x=[1, 2, 3]))
    res = fa.fugue_sql(
        "SELECT * FROM a WHERE x > {{threshold}}", a=a, threshold=1
    )
    assert len(res) == 2


def test_union_sql():
    a = pd.DataFrame(dict(x=[1]))
    b = pd.DataFrame(dict(x=[2]))
    res = fa.fugue_sql("SELECT * FROM a UNION ALL SELECT * FROM b", a=a, b=b)
    assert sorted(res["x"].tolist()) == [1, 2]


def test_sql_caller_var_capture():
    captured_df = pd.DataFrame(dict(q=[42]))
    res = fugue_sql("SELECT * FROM captured_df")
    assert res["q"].tolist() == [42]


def test_sql_errors():
    with pytest.raises((FugueSQLSyntaxError, FugueSQLError)):
        fa.fugue_sql("NONSENSE STATEMENT HERE")


def test_subquery_in_from():
    import fugue_amd.api as fa

    df = pd.DataFrame(dict(k=[1, 1, 2, 2, 3], v=[1.0, 2.0, 3.0, 4.0, 5.0]))
    res = fa.fugue_sql(
        """
        SELECT k, s FROM (
          SELECT k, SUM(v) AS s FROM df GROUP BY k HAVING SUM(v) > 3
        ) WHERE s < 8
        UNION ALL
        SELECT k, v AS s FROM df WHERE k = 3
        """,
        df=df,
    )
    got = sorted((int(a), float(b)) for a, b in res.values.tolist())
    assert got == [(2, 7.0), (3, 5.0), (3, 5.0)]


def test_case_when_in_fsql():
    import fugue_amd.api as fa

    df = pd.DataFrame(dict(k=[1, 2], v=[1.0, 3.0]))
    res = fa.fugue_sql(
        "SELECT k, CASE WHEN v > 2.5 THEN 'big' ELSE 'small' END AS c FROM df",
        df=df,
    )
    assert res["c"].tolist() == ["small", "big"]


def test_window_functions():
    from fugue_amd.sql.executor import run_sql_on_pandas

    df = pd.DataFrame(dict(k=["a", "a", "a", "b", "b"], v=[3.0, 1.0, 2.0, 5.0, 4.0]))
    r = run_sql_on_pandas(
        "SELECT k, v, ROW_NUMBER() OVER (PARTITION BY k ORDER BY v) AS rn, "
        "SUM(v) OVER (PARTITION BY k ORDER BY v) AS cs, "
        "SUM(v) OVER (PARTITION BY k) AS ts, "
        "LAG(v) OVER (PARTITION BY k ORDER BY v) AS pv, "
        "RANK() OVER (ORDER BY v DESC) AS r FROM a",
        dict(a=df),
    )[0]
    assert r["rn"].tolist() == [3, 1, 2, 2, 1]
    assert r["cs"].tolist() == [6.0, 1.0, 3.0, 9.0, 4.0]
    assert r["ts"].tolist() == [6.0, 6.0, 6.0, 9.0, 9.0]
    assert r["r"].tolist() == [3, 5, 4, 1, 2]
    assert pd.isna(r["pv"].iloc[1]) and r["pv"].iloc[2] == 1.0


def test_window_dedup_pattern():
    import fugue_amd.api as fa

    df = pd.DataFrame(dict(k=["a", "a", "b"], v=[1.0, 3.0, 5.0]))
    r = fa.fugue_sql(
        "SELECT k, v FROM ("
        "  SELECT k, v, ROW_NUMBER() OVER (PARTITION BY k ORDER BY v DESC) AS rn"
        "  FROM df) WHERE rn = 1",
        df=df,
    )
    assert sorted(map(tuple, r.values.tolist())) == [("a", 3.0), ("b", 5.0)]


def test_window_rank_dense_and_values():
    from fugue_amd.sql.executor import run_sql_on_pandas

    df = pd.DataFrame(dict(k=["a", "a", "a", "a"], v=[1.0, 2.0, 2.0, 3.0]))
    r = run_sql_on_pandas(
        "SELECT v, RANK() OVER (ORDER BY v) AS r, "
        "DENSE_RANK() OVER (ORDER BY v) AS d, "
        "FIRST_VALUE(v) OVER (PARTITION BY k ORDER BY v) AS fv, "
        "LAST_VALUE(v) OVER (PARTITION BY k) AS lv FROM a",
        dict(a=df),
    )[0]
    assert r["r"].tolist() == [1, 2, 2, 4]
    assert r["d"].tolist() == [1, 2, 2, 3]
    assert r["fv"].tolist() == [1.0] * 4
    assert r["lv"].tolist() == [3.0] * 4


def test_in_and_scalar_subqueries():
    from fugue_amd.sql.executor import run_sql_on_pandas

    df = pd.DataFrame(dict(k=[1, 2, 3], v=[1.0, 2.0, 3.0]))
    d2 = pd.DataFrame(dict(k=[2, 3]))
    r = run_sql_on_pandas(
        "SELECT k FROM a WHERE k IN (SELECT k FROM b)", dict(a=df, b=d2)
    )[0]
    assert r["k"].tolist() == [2, 3]
    r = run_sql_on_pandas(
        "SELECT k FROM a WHERE k NOT IN (SELECT k FROM b)", dict(a=df, b=d2)
    )[0]
    assert r["k"].tolist() == [1]
    r = run_sql_on_pandas(
        "SELECT k FROM a WHERE v > (SELECT AVG(v) FROM a)", dict(a=df)
    )[0]
    assert r["k"].tolist() == [3]


def test_sql_module_sub():
    """SUB statement invoking @module functions (reference
    tests/fugue/sql/test_workflow_parse.py:711 test_module scenario)."""
    from fugue_amd.workflow.module import module
    from fugue_amd.workflow.workflow import (
        FugueWorkflow,
        WorkflowDataFrame,
        WorkflowDataFrames,
    )

    def create(wf: FugueWorkflow, n: int = 1) -> WorkflowDataFrame:
        return wf.df([[n]], "a:int")

    def merge(
        df1: WorkflowDataFrame, df2: WorkflowDataFrame, k: str = "aa"
    ) -> WorkflowDataFrames:
        return WorkflowDataFrames({k: df1, "bb": df2})

    def merge2(
        wf: FugueWorkflow, dfs: WorkflowDataFrames, k: int = 0
    ) -> WorkflowDataFrame:
        return dfs[k]

    def merge3(df1: WorkflowDataFrame, df2: WorkflowDataFrame) -> WorkflowDataFrames:
        return WorkflowDataFrames(df1, df2)

    collected = []

    @module()
    def out1(wf: FugueWorkflow, df: WorkflowDataFrame) -> None:
        def grab(pdf: pd.DataFrame) -> None:
            collected.append(pdf["a"].tolist())

        df.output(grab)

    dag = FugueSQLWorkflow()
    dag(
        """
        a=sub using create
        b=sub using create(n=2)
        dfs=sub a,b using merge(k="a1")
        r1=select * from dfs[a1]
        r1 yield dataframe as r1
        r2=select * from dfs[bb]
        r2 yield dataframe as r2
        m2=sub a,b using merge2(k=1)
        sub using out1
        dfs2=sub df2:a,df1:b using merge3
        r3=select * from dfs2[0]
        r3 yield dataframe as r3
        """,
        create=create,
        merge=merge,
        merge2=merge2,
        merge3=merge3,
        out1=out1,
    )
    res = dag.run()
    assert res["r1"].as_array() == [[1]]
    assert res["r2"].as_array() == [[2]]
    # merge2 selects dfs[1] == b == [[2]]; out1 prints the last frame (m2)
    assert collected == [[2]]
    # merge3 named refs: df1=b, df2=a → positional order df1,df2 = b,a
    assert res["r3"].as_array() == [[2]]


def test_distinct_aggregates_sql_executor():
    from fugue_amd.sql.executor import run_sql_on_pandas

    df = pd.DataFrame(dict(k=[1, 1, 1, 2, 2], v=[1.0, 1.0, 3.0, 2.0, 2.0]))
    r = run_sql_on_pandas(
        "SELECT k, SUM(DISTINCT v) AS s, AVG(DISTINCT v) AS a, "
        "MIN(DISTINCT v) AS mn, COUNT(DISTINCT v) AS c "
        "FROM t GROUP BY k ORDER BY k",
        dict(t=df),
    )[0]
    assert r["s"].tolist() == [4.0, 2.0]
    assert r["a"].tolist() == [2.0, 2.0]
    assert r["mn"].tolist() == [1.0, 2.0]
    assert r["c"].tolist() == [2, 1]


def test_connect_sql_engine():
    """CONNECT <engine> SELECT ...: per-query SQL engine (reference
    test_workflow_parse.py test_select_plus_engine)."""
    from fugue_amd.execution.execution_engine import SQLEngine
    from fugue_amd.execution.factory import register_sql_engine

    seen = []

    class MockEngine(SQLEngine):
        def __init__(self, execution_engine, p: int = 0):
            super().__init__(execution_engine)
            self.p = p

        @property
        def is_distributed(self):
            return False

        def select(self, dfs, statement):
            seen.append(self.p)
            from fugue_amd.sql.executor import run_sql_on_pandas

            sql = statement.construct(dialect=None)
            pdfs = {k: v.as_pandas() for k, v in dfs.items()}
            from fugue_amd.dataframe.pandas_dataframe import PandasDataFrame

            return PandasDataFrame(run_sql_on_pandas(sql, pdfs)[0])

    register_sql_engine("_mock_sql", lambda e, **kw: MockEngine(e, **kw))
    df = pd.DataFrame(dict(a=[1, 2, 3]))
    r = fugue_sql("connect _mock_sql(p=2) select a from df where a>1")
    assert sorted(r["a"].tolist()) == [2, 3]
    r2 = fugue_sql("connect MockEngine select a from df where a>2")
    assert r2["a"].tolist() == [3]
    assert seen == [2, 0]
