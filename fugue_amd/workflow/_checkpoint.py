"""Checkpoints: weak (persist), strong (save+reload), deterministic
(permanent, keyed by task spec uuid).

Reference parity: ``fugue/workflow/_checkpoint.py``.
"""
import os
import uuid as _uuid
from typing import Any, Optional

from fugue_amd.collections.partition import PartitionSpec
from fugue_amd.collections.yielded import PhysicalYielded, Yielded
from fugue_amd.constants import FUGUE_CONF_WORKFLOW_CHECKPOINT_PATH
from fugue_amd.dataframe.dataframe import DataFrame
from fugue_amd.exceptions import FugueWorkflowCompileError, FugueWorkflowRuntimeError
from fugue_amd.execution.execution_engine import ExecutionEngine
from fugue_amd.utils.hash import to_uuid
from fugue_amd.utils.params import ParamDict


class Checkpoint:
    def __init__(
        self,
        to_file: bool = False,
        deterministic: bool = False,
        permanent: bool = False,
        lazy: bool = False,
        **kwargs: Any,
    ):
        if deterministic and not permanent:
            raise ValueError("deterministic checkpoints must be permanent")
        self.to_file = to_file
        self.deterministic = deterministic
        self.permanent = permanent
        self.lazy = lazy
        self.kwargs = dict(kwargs)

    @property
    def is_null(self) -> bool:
        return True

    def run(self, df: DataFrame, path: "CheckpointPath", object_id: str) -> DataFrame:
        return df

    def __uuid__(self) -> str:
        return to_uuid(
            self.to_file, self.deterministic, self.permanent, self.lazy, self.kwargs
        )


class WeakCheckpoint(Checkpoint):
    """Engine persist only (reference ``_checkpoint.py:111``)."""

    def __init__(self, lazy: bool = False, **kwargs: Any):
        super().__init__(to_file=False, deterministic=False, permanent=False, lazy=lazy, **kwargs)

    @property
    def is_null(self) -> bool:
        return False

    def run(self, df: DataFrame, path: "CheckpointPath", object_id: str) -> DataFrame:
        return path.execution_engine.persist(df, lazy=self.lazy, **self.kwargs)


class StrongCheckpoint(Checkpoint):
    """Save to file + reload (reference ``_checkpoint.py:38``)."""

    def __init__(
        self,
        storage_type: str = "file",
        lazy: bool = False,
        partition: Any = None,
        single: bool = False,
        deterministic: bool = False,
        permanent: bool = False,
        namespace: Any = None,
        **kwargs: Any,
    ):
        super().__init__(
            to_file=True,
            deterministic=deterministic,
            permanent=deterministic or permanent,
            lazy=lazy,
            **kwargs,
        )
        self._storage_type = storage_type
        self._partition = PartitionSpec(partition)
        self._single = single
        self._namespace = namespace

    @property
    def is_null(self) -> bool:
        return False

    def run(self, df: DataFrame, path: "CheckpointPath", object_id: str) -> DataFrame:
        fpath = path.get_temp_file(
            object_id if self.deterministic else str(_uuid.uuid4()), self.permanent
        )
        if not self.deterministic or not path.temp_file_exists(fpath):
            path.execution_engine.save_df(
                df,
                fpath,
                format_hint="parquet",
                mode="overwrite",
                partition_spec=self._partition,
                force_single=self._single,
                **self.kwargs,
            )
        return path.execution_engine.load_df(fpath, format_hint="parquet")

    def __uuid__(self) -> str:
        return to_uuid(
            super().__uuid__(), self._storage_type, self._partition, self._single, self._namespace
        )


class CheckpointPath:
    """Manages the per-run temp dir + the permanent checkpoint root
    (reference ``_checkpoint.py:131``)."""

    def __init__(self, engine: ExecutionEngine):
        self._engine = engine
        self._path = engine.conf.get(FUGUE_CONF_WORKFLOW_CHECKPOINT_PATH, "").strip()
        self._temp_path = ""

    @property
    def execution_engine(self) -> ExecutionEngine:
        return self._engine

    def init_temp_path(self, execution_id: str) -> str:
        if self._path == "":
            self._temp_path = ""
            return ""
        self._temp_path = os.path.join(self._path, execution_id)
        os.makedirs(self._temp_path, exist_ok=True)
        return self._temp_path

    def remove_temp_path(self) -> None:
        if self._temp_path != "":
            import shutil

            try:
                shutil.rmtree(self._temp_path)
            except Exception:  # pragma: no cover
                pass

    def get_temp_file(self, obj_id: str, permanent: bool) -> str:
        path = self._path if permanent else self._temp_path
        if path == "":
            raise FugueWorkflowRuntimeError(
                f"{FUGUE_CONF_WORKFLOW_CHECKPOINT_PATH} is not set for checkpoints"
            )
        return os.path.join(path, obj_id + ".parquet")

    def temp_file_exists(self, path: str) -> bool:
        return os.path.exists(path)
