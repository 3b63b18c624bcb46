"""DAG-branch concurrency (reference
``tests/fugue/workflow/test_workflow_parallel.py``): independent
branches run concurrently up to ``fugue.workflow.concurrency``; results
stay correct; failures in one branch propagate."""
import threading
import time

import pandas as pd
import pytest

from fugue_amd.exceptions import FugueWorkflowError
from fugue_amd.execution import NativeExecutionEngine
from fugue_amd.workflow import FugueWorkflow


class _Gauge:
    def __init__(self):
        self.lock = threading.Lock()
        self.cur = 0
        self.peak = 0

    def enter(self):
        with self.lock:
            self.cur += 1
            self.peak = max(self.peak, self.cur)

    def exit(self):
        with self.lock:
            self.cur -= 1


def test_branches_run_concurrently():
    gauge = _Gauge()

    def make_creator(i):
        def create() -> pd.DataFrame:
            gauge.enter()
            time.sleep(0.2)
            gauge.exit()
            return pd.DataFrame({"a": [i]})

        return create

    dag = FugueWorkflow()
    for i in range(4):
        dag.create(make_creator(i), schema="a:long").yield_dataframe_as(
            f"r{i}"
        )
    t0 = time.perf_counter()
    res = dag.run(NativeExecutionEngine({"fugue.workflow.concurrency": 4}))
    el = time.perf_counter() - t0
    assert gauge.peak >= 2, f"no concurrency observed (peak={gauge.peak})"
    assert el < 0.75, f"branches serialized ({el:.2f}s)"
    for i in range(4):
        assert res[f"r{i}"].as_array() == [[i]]


def test_concurrency_one_serializes():
    gauge = _Gauge()

    def make_creator(i):
        def create() -> pd.DataFrame:
            gauge.enter()
            time.sleep(0.05)
            gauge.exit()
            return pd.DataFrame({"a": [i]})

        return create

    dag = FugueWorkflow()
    for i in range(3):
        dag.create(make_creator(i), schema="a:long").yield_dataframe_as(
            f"r{i}"
        )
    dag.run(NativeExecutionEngine({"fugue.workflow.concurrency": 1}))
    assert gauge.peak == 1


def test_branch_failure_propagates():
    def boom() -> pd.DataFrame:
        raise RuntimeError("branch failed")

    def ok() -> pd.DataFrame:
        return pd.DataFrame({"a": [1]})

    dag = FugueWorkflow()
    dag.create(ok, schema="a:long").show()
    dag.create(boom, schema="a:long").show()
    with pytest.raises(Exception) as ei:
        dag.run(NativeExecutionEngine({"fugue.workflow.concurrency": 4}))
    assert "branch failed" in str(ei.value)


def test_dependent_tasks_ordered():
    order = []
    lock = threading.Lock()

    def create() -> pd.DataFrame:
        with lock:
            order.append("create")
        return pd.DataFrame({"a": [1, 2]})

    # schema: *
    def tr(df: pd.DataFrame) -> pd.DataFrame:
        with lock:
            order.append("transform")
        return df

    dag = FugueWorkflow()
    dag.create(create, schema="a:long").transform(tr).yield_dataframe_as("r")
    res = dag.run(NativeExecutionEngine({"fugue.workflow.concurrency": 8}))
    assert order == ["create", "transform"]
    assert sorted(res["r"].as_array()) == [[1], [2]]
