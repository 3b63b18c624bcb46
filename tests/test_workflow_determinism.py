"""Workflow spec determinism (reference
``tests/fugue/workflow/test_workflow_determinism.py``): spec uuids are
stable across builds and processes, sensitive to every identity field,
and insensitive to execution."""
import subprocess
import sys

import pandas as pd

from fugue_amd.collections.partition import PartitionSpec
from fugue_amd.workflow import FugueWorkflow


# schema: *
def _tr(df: pd.DataFrame, p: int = 1) -> pd.DataFrame:
    return df


def _build(p=1, by=("a",), data=((0, 1),)):
    dag = FugueWorkflow()
    a = dag.df([list(r) for r in data], "a:int,b:int")
    b = a.partition(by=list(by)).transform(_tr, params=dict(p=p))
    c = a.join(b, how="inner")
    c.select("a", "b")
    return dag


def test_same_build_same_uuid():
    assert _build().spec_uuid() == _build().spec_uuid()


def test_identity_fields_change_uuid():
    base = _build().spec_uuid()
    assert _build(p=2).spec_uuid() != base          # params
    assert _build(by=("b",)).spec_uuid() != base    # partition spec
    assert _build(data=((0, 2),)).spec_uuid() != base  # input data


def test_uuid_not_affected_by_run():
    dag = _build()
    before = dag.spec_uuid()
    dag.run()
    assert dag.spec_uuid() == before


def test_uuid_stable_across_processes():
    """No id()-based hashing: two fresh interpreters compute the same
    spec uuid for the same DAG.  (The uuid legitimately includes the
    extension functions' module paths, so the comparison runs both
    sides under identical import roots.)"""
    code = (
        "import sys; sys.path.insert(0, %r);"
        "from tests.test_workflow_determinism import _build;"
        "print(_build().spec_uuid())"
    ) % (__file__.rsplit("/tests/", 1)[0],)

    def run_once() -> str:
        out = subprocess.run(
            [sys.executable, "-c", code], capture_output=True, text=True,
            timeout=120,
        )
        assert out.returncode == 0, out.stderr[-2000:]
        return out.stdout.strip()

    assert run_once() == run_once()


def test_partition_spec_uuid_components():
    a = PartitionSpec(by=["a"], presort="b desc", num=4)
    b = PartitionSpec(by=["a"], presort="b desc", num=4)
    assert __import__("fugue_amd.utils.hash", fromlist=["to_uuid"]).to_uuid(
        a
    ) == __import__("fugue_amd.utils.hash", fromlist=["to_uuid"]).to_uuid(b)
    c = PartitionSpec(by=["a"], presort="b", num=4)
    assert __import__("fugue_amd.utils.hash", fromlist=["to_uuid"]).to_uuid(
        a
    ) != __import__("fugue_amd.utils.hash", fromlist=["to_uuid"]).to_uuid(c)
