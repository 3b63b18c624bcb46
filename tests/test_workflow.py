import os
import tempfile
from typing import Any, Callable, Dict, Iterable, List

import pandas as pd
import pytest

import fugue_amd.api as fa
from fugue_amd import ArrayDataFrame, DataFrame, LocalDataFrame, PandasDataFrame
from fugue_amd.column.expressions import col
from fugue_amd.column import functions as f
from fugue_amd.constants import FUGUE_CONF_WORKFLOW_CHECKPOINT_PATH
from fugue_amd.exceptions import (
    FugueInterfacelessError,
    FugueWorkflowCompileValidationError,
)
from fugue_amd.workflow import FugueWorkflow, transform, out_transform


def test_basic_dag_run():
    dag = FugueWorkflow()
    a = dag.df(pd.DataFrame(dict(x=[1, 2, 3])))
    b = a.filter(col("x") > 1)
    b.yield_dataframe_as("r")
    res = dag.run()
    assert res["r"].as_array() == [[2], [3]]


def test_dag_determinism():
    def build():
        dag = FugueWorkflow()
        a = dag.df([[1, "a"]], "x:long,y:str")
        b = a.transform(_double, schema="*")
        b.yield_dataframe_as("r")
        return [t.__uuid__() for t in dag._task_order]

    assert build() == build()


# schema: *
def _double(df: pd.DataFrame) -> pd.DataFrame:
    df["x"] = df["x"] * 2
    return df


def test_transform_styles():
    pdf = pd.DataFrame(dict(x=[1, 2], g=["a", "b"]))

    # plain function with schema comment
    res = transform(pdf, _double)
    assert res["x"].tolist() == [2, 4]

    # plain function with explicit schema, list output
    def to_arr(df: List[List[Any]]) -> List[List[Any]]:
        return [[r[0] * 10] for r in df]

    res2 = transform(pdf[["x"]], to_arr, schema="x:long")
    assert res2["x"].tolist() == [10, 20]

    # iterable of dicts
    def gen(df: Iterable[Dict[str, Any]]) -> Iterable[Dict[str, Any]]:
        for row in df:
            row["x"] += 1
            yield row

    res3 = transform(pdf, gen, schema="*")
    assert res3["x"].tolist() == [2, 3]

    # transformer class instance via decorator
    from fugue_amd.extensions import transformer

    @transformer("*,z:long")
    def with_z(df: pd.DataFrame) -> pd.DataFrame:
        df["z"] = 1
        return df

    res4 = transform(pdf, with_z)
    assert "z" in res4.columns


def test_transform_partition():
    pdf = pd.DataFrame(dict(g=["a", "a", "b"], v=[3, 1, 2]))

    # schema: g:str,first_v:long
    def first_v(df: pd.DataFrame) -> pd.DataFrame:
        return pd.DataFrame(dict(g=[df["g"].iloc[0]], first_v=[df["v"].iloc[0]]))

    res = transform(
        pdf, first_v, partition=dict(by=["g"], presort="v")
    )
    assert sorted(res.values.tolist()) == [["a", 1], ["b", 2]]


def test_out_transform_and_callback():
    collected = []

    def cb(x: int) -> None:
        collected.append(x)

    def sink(df: pd.DataFrame, callback: Callable) -> None:
        callback(len(df))

    out_transform(pd.DataFrame(dict(a=[1, 2])), sink, callback=cb)
    assert collected == [2]


def test_ignore_errors():
    def bad(df: pd.DataFrame) -> pd.DataFrame:
        if df["g"].iloc[0] == "a":
            raise ValueError("boom")
        return df

    pdf = pd.DataFrame(dict(g=["a", "b"], v=[1, 2]))
    res = transform(
        pdf,
        bad,
        schema="*",
        partition=dict(by=["g"]),
        ignore_errors=[ValueError],
    )
    assert res.values.tolist() == [["b", 2]]


def test_workflow_join_setops():
    dag = FugueWorkflow()
    a = dag.df([[1, "x"], [2, "y"]], "k:long,a:str")
    b = dag.df([[2, 5.0]], "k:long,b:double")
    j = a.inner_join(b)
    j.yield_dataframe_as("j")
    u = a.union(a, distinct=False)
    u.yield_dataframe_as("u")
    res = dag.run()
    assert res["j"].as_array() == [[2, "y", 5.0]]
    assert res["u"].count() == 4


def test_checkpoints_and_persist():
    with tempfile.TemporaryDirectory() as tmp:
        dag = FugueWorkflow()
        a = dag.df([[1]], "x:long").persist()
        b = a.strong_checkpoint()
        b.yield_dataframe_as("r")
        res = dag.run(None, {FUGUE_CONF_WORKFLOW_CHECKPOINT_PATH: tmp})
        assert res["r"].as_array() == [[1]]


def test_deterministic_checkpoint_reuse():
    with tempfile.TemporaryDirectory() as tmp:
        conf = {FUGUE_CONF_WORKFLOW_CHECKPOINT_PATH: tmp}

        def run_once():
            dag = FugueWorkflow()
            a = dag.df([[1]], "x:long")
            b = a.transform(_double_x, schema="*").deterministic_checkpoint()
            b.yield_dataframe_as("r")
            return dag.run(None, conf)["r"].as_array()

        assert run_once() == [[2]]
        files = os.listdir(tmp)
        assert run_once() == [[2]]
        assert os.listdir(tmp) == files  # second run reused the checkpoint


def _double_x(df: pd.DataFrame) -> pd.DataFrame:
    df["x"] = df["x"] * 2
    return df


def test_validation_rules():
    # partitionby_has: g
    def needs_g(df: pd.DataFrame) -> pd.DataFrame:
        return df

    with pytest.raises(FugueWorkflowCompileValidationError):
        transform(
            pd.DataFrame(dict(g=[1], v=[2])),
            needs_g,
            schema="*",
        )


def test_zip_comap_workflow():
    dag = FugueWorkflow()
    a = dag.df([[1, "a"], [2, "b"]], "k:long,x:str")
    b = dag.df([[1, 1.0], [1, 2.0]], "k:long,y:double")
    z = dag.zip(a, b)

    def merge_counts(df1: pd.DataFrame, df2: pd.DataFrame) -> List[List[Any]]:
        return [[len(df1), len(df2)]]

    r = z.transform(merge_counts, schema="n1:long,n2:long")
    r.yield_dataframe_as("r")
    res = dag.run()
    assert res["r"].as_array() == [[1, 2]]


def test_workflow_parallelism():
    dag = FugueWorkflow()
    a = dag.df([[1]], "x:long")
    for i in range(5):
        a.transform(_double_x, schema="*").yield_dataframe_as(f"r{i}")
    res = dag.run(None, {"fugue.workflow.concurrency": 4})
    for i in range(5):
        assert res[f"r{i}"].as_array() == [[2]]


def test_runtime_exception_traceback():
    def bad(df: pd.DataFrame) -> pd.DataFrame:
        raise RuntimeError("inner failure")

    with pytest.raises(RuntimeError, match="inner failure"):
        transform(pd.DataFrame(dict(a=[1])), bad, schema="*")


def test_yield_file_and_table_cross_workflow():
    import tempfile

    from fugue_amd.execution import NativeExecutionEngine

    with tempfile.TemporaryDirectory() as tmp:
        conf = {FUGUE_CONF_WORKFLOW_CHECKPOINT_PATH: tmp}
        engine = NativeExecutionEngine(conf)
        dag1 = FugueWorkflow()
        dag1.df([[1], [2]], "x:long").yield_file_as("f1")
        dag1.df([[3]], "x:long").yield_table_as("t1")
        res1 = dag1.run(engine)
        # second workflow consumes the yields
        dag2 = FugueWorkflow()
        a = dag2.df(res1["f1"])
        b = dag2.df(res1["t1"])
        a.union(b, distinct=False).yield_dataframe_as("out")
        res2 = dag2.run(engine)
        assert sorted(r[0] for r in res2["out"].as_array()) == [1, 2, 3]


def test_auto_persist():
    from fugue_amd.constants import FUGUE_CONF_WORKFLOW_AUTO_PERSIST

    calls = []

    def count_calls(df: pd.DataFrame) -> pd.DataFrame:
        calls.append(1)
        return df

    dag = FugueWorkflow({FUGUE_CONF_WORKFLOW_AUTO_PERSIST: True})
    a = dag.df([[1]], "x:long").transform(count_calls, schema="*")
    a.transform(_double_x, schema="*").yield_dataframe_as("r1")
    a.transform(lambda df: df, schema="*") if False else None
    b = a.filter(col("x") >= 0)
    b.yield_dataframe_as("r2")
    res = dag.run()
    assert res["r1"].as_array() == [[2]]
    assert res["r2"].as_array() == [[1]]
    # `a` is consumed twice; the auto-persist marks it with a weak
    # checkpoint so its transform runs once
    assert len(calls) == 1


def test_module_composition():
    from fugue_amd.workflow.module import module
    from fugue_amd.workflow.workflow import WorkflowDataFrame

    @module
    def doubled(df: WorkflowDataFrame) -> WorkflowDataFrame:
        return df.transform(_double_x, schema="*")

    @module(as_method=True, name="tripled")
    def _tripled(df: WorkflowDataFrame) -> WorkflowDataFrame:
        # schema: *
        def t3(pdf: pd.DataFrame) -> pd.DataFrame:
            pdf["x"] = pdf["x"] * 3
            return pdf

        return df.transform(t3, schema="*")

    dag = FugueWorkflow()
    a = dag.df([[1]], "x:long")
    doubled(a).yield_dataframe_as("d")
    a.tripled().yield_dataframe_as("t")
    res = dag.run()
    assert res["d"].as_array() == [[2]]
    assert res["t"].as_array() == [[3]]


def test_save_and_use():
    import tempfile

    with tempfile.TemporaryDirectory() as tmp:
        path = os.path.join(tmp, "out.parquet")
        dag = FugueWorkflow()
        a = dag.df([[1], [2]], "x:long")
        b = a.save_and_use(path)
        b.yield_dataframe_as("r")
        dag.run()
        assert os.path.exists(path)
        # the saved file loads back with the same data
        dag2 = FugueWorkflow()
        dag2.load(path).yield_dataframe_as("r")
        res2 = dag2.run()
        assert sorted(r[0] for r in res2["r"].as_array()) == [1, 2]


def test_transform_iterable_dfs_output():
    from typing import Iterable

    pdf = pd.DataFrame(dict(x=[1, 2, 3]))

    def splitter(df: pd.DataFrame) -> Iterable[pd.DataFrame]:
        for _, row in df.iterrows():
            yield pd.DataFrame([row])

    res = transform(pdf, splitter, schema="*")
    assert sorted(res["x"].tolist()) == [1, 2, 3]


def test_extension_registry_aliases():
    from fugue_amd.extensions import (
        register_creator,
        register_outputter,
        register_processor,
        register_transformer,
    )

    # schema: x:long
    def make_one(ctx=None) -> pd.DataFrame:
        return pd.DataFrame(dict(x=[7]))

    def double_proc(df: pd.DataFrame) -> pd.DataFrame:
        df = df.copy()
        df["x"] = df["x"] * 2
        return df

    seen = []

    def sink(df: pd.DataFrame) -> None:
        seen.append(df["x"].tolist())

    register_creator("mk1", make_one)
    register_transformer("dbl", double_proc)
    register_outputter("snk", sink)

    dag = FugueWorkflow()
    a = dag.create("mk1")
    b = a.transform("dbl", schema="*")
    b.output("snk")
    b.yield_dataframe_as("r")
    dag.run()
    assert seen == [[14]]


def test_cotransformer_decorator():
    from fugue_amd.extensions import cotransformer

    @cotransformer("k:long,n1:long,n2:long")
    def merge(df1: pd.DataFrame, df2: pd.DataFrame) -> pd.DataFrame:
        return pd.DataFrame(
            dict(k=[df1["k"].iloc[0]], n1=[len(df1)], n2=[len(df2)])
        )

    dag = FugueWorkflow()
    a = dag.df([[1, "a"], [1, "b"], [2, "c"]], "k:long,x:str")
    b = dag.df([[1, 1.0], [3, 2.0]], "k:long,y:double")
    z = dag.zip(a, b)  # inner: only k=1 survives
    z.transform(merge).yield_dataframe_as("r")
    res = dag.run()
    assert res["r"].as_array() == [[1, 2, 1]]


def test_zip_to_file_threshold(tmp_path):
    from fugue_amd.execution import NativeExecutionEngine
    from fugue_amd.dataframe.dataframes import DataFrames

    e = NativeExecutionEngine()
    a = e.to_df(pd.DataFrame(dict(k=[1] * 50, x=list(range(50)))))
    b = e.to_df(pd.DataFrame(dict(k=[1] * 30, y=list(range(30)))))
    z = e.zip(
        DataFrames(a, b),
        partition_spec=None,
        temp_path=str(tmp_path),
        to_file_threshold=64,  # tiny: force file-backed blobs
    )
    import os

    blobs = [f for f in os.listdir(tmp_path) if f.startswith("fugue-blob-")]
    assert len(blobs) > 0  # payloads spilled to files

    def cm(cursor, dfs):
        from fugue_amd.dataframe.array_dataframe import ArrayDataFrame

        return ArrayDataFrame(
            [[dfs[0].count(), dfs[1].count()]], "n1:long,n2:long"
        )

    from fugue_amd.collections.partition import PartitionSpec

    res = e.comap(z, cm, "n1:long,n2:long", PartitionSpec())
    assert res.as_array() == [[50, 30]]


def test_namespace_extension_plugins():
    """parse_creator + namespace_candidate plugin point (reference
    fugue/plugins.py dispatchers): a ("myns", payload) tuple resolves
    to a creator through a registered candidate."""
    from fugue_amd.extensions import namespace_candidate, parse_creator
    from fugue_amd.workflow.workflow import FugueWorkflow

    @parse_creator.candidate(
        namespace_candidate("myns", lambda x: isinstance(x, str))
    )
    def _parse(obj):
        n = int(obj[1])

        # schema: a:int
        def create() -> List[List[Any]]:
            return [[n]]

        return create

    dag = FugueWorkflow()
    dag.create(("myns", "7")).yield_dataframe_as("r")
    res = dag.run()
    assert res["r"].as_array() == [[7]]


def test_fugue_test_suite_base():
    """ft.FugueTestSuite + fugue_test_suite binding (reference
    fugue/test/plugins.py:139)."""
    import fugue_amd.test as ft

    @ft.fugue_test_suite("pandas")
    class MySuite(ft.FugueTestSuite):
        pass

    s = MySuite()
    assert s.backend == "pandas"
    assert s.engine is s.engine  # lazily built, cached
    from fugue_amd import PandasDataFrame

    assert s.df_eq(
        PandasDataFrame(pd.DataFrame(dict(a=[1]))), [[1]], "a:long"
    )
    assert ft.extract_conf({"x.a": 1, "y.b": 2}, "x.", True) == {"a": 1}
