"""Workflow runtime context + the DAG runner (replaces adagio).

Reference parity: ``fugue/workflow/_workflow_context.py`` (RPC server +
checkpoint path + parallel task execution with
``fugue.workflow.concurrency``).
"""
import threading
import uuid as _uuid
from concurrent.futures import Future, ThreadPoolExecutor
from typing import Any, Dict, List, Optional

from fugue_amd.constants import FUGUE_CONF_WORKFLOW_CONCURRENCY
from fugue_amd.execution.execution_engine import ExecutionEngine
from fugue_amd.rpc import make_rpc_server
from fugue_amd.workflow._checkpoint import CheckpointPath
from fugue_amd.workflow._tasks import FugueTask


class FugueWorkflowContext:
    def __init__(
        self,
        engine: ExecutionEngine,
        compile_conf: Any = None,
        yields_as_local: bool = False,
    ):
        self._engine = engine
        self._rpc_server = make_rpc_server(engine.conf)
        self._checkpoint_path = CheckpointPath(engine)
        self._concurrency = int(
            engine.conf.get(FUGUE_CONF_WORKFLOW_CONCURRENCY, 1)
        )
        self.yields_as_local = yields_as_local
        self._lock = threading.RLock()

    @property
    def execution_engine(self) -> ExecutionEngine:
        return self._engine

    @property
    def rpc_server(self) -> Any:
        return self._rpc_server

    @property
    def checkpoint_path(self) -> CheckpointPath:
        return self._checkpoint_path

    def run(self, tasks: List[FugueTask]) -> None:
        """Execute the DAG: topological order, independent branches run
        concurrently up to the configured concurrency."""
        execution_id = str(_uuid.uuid4())
        self._rpc_server.start()
        self._checkpoint_path.init_temp_path(execution_id)
        try:
            with self._engine.as_context():
                if self._concurrency <= 1:
                    self._run_sequential(tasks)
                else:
                    self._run_parallel(tasks)
        finally:
            self._checkpoint_path.remove_temp_path()
            self._rpc_server.stop()

    def _run_sequential(self, tasks: List[FugueTask]) -> None:
        done = set()

        def _run(task: FugueTask) -> None:
            if id(task) in done:
                return
            for dep in task.inputs:
                _run(dep)
            if not task.executed:
                task.execute(self)
            done.add(id(task))

        for t in tasks:
            _run(t)

    def _run_parallel(self, tasks: List[FugueTask]) -> None:
        # collect the full graph (unique by identity)
        all_tasks: Dict[int, FugueTask] = {}

        def _collect(t: FugueTask) -> None:
            if id(t) in all_tasks:
                return
            all_tasks[id(t)] = t
            for dep in t.inputs:
                _collect(dep)

        for t in tasks:
            _collect(t)
        dependents: Dict[int, List[FugueTask]] = {k: [] for k in all_tasks}
        remaining: Dict[int, int] = {}
        for t in all_tasks.values():
            deps = {id(d) for d in t.inputs}
            remaining[id(t)] = len(deps)
            for d in deps:
                dependents[d].append(t)
        lock = threading.Lock()
        done = threading.Event()
        errors: List[BaseException] = []
        pending = [len(all_tasks)]
        pool = ThreadPoolExecutor(self._concurrency)

        def _on_finish(task: FugueTask, err: Optional[BaseException]) -> None:
            with lock:
                if err is not None:
                    errors.append(err)
                pending[0] -= 1
                if pending[0] == 0 or err is not None:
                    done.set()
                ready = []
                if err is None:
                    for dep_t in dependents[id(task)]:
                        remaining[id(dep_t)] -= 1
                        if remaining[id(dep_t)] == 0:
                            ready.append(dep_t)
            for r in ready:
                pool.submit(_work, r)

        def _work(task: FugueTask) -> None:
            err: Optional[BaseException] = None
            try:
                if not task.executed:
                    task.execute(self)
            except BaseException as e:
                err = e
            _on_finish(task, err)

        try:
            initial = [t for t in all_tasks.values() if remaining[id(t)] == 0]
            if len(initial) == 0 and len(all_tasks) > 0:
                raise RuntimeError("workflow graph has a cycle")
            for t in initial:
                pool.submit(_work, t)
            if len(all_tasks) > 0:
                done.wait()
            if errors:
                raise errors[0]
        finally:
            pool.shutdown(wait=True)
