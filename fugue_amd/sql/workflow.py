"""FugueSQLWorkflow: FugueWorkflow + the FugueSQL language.

Reference parity: ``fugue/sql/workflow.py`` + ``fugue/sql/_visitors.py``
(``_Extensions`` DAG emission), on the hand-written parser in
``fugue_amd/sql/_parser.py``.  Parser hooks use the ``sql_`` prefix to
avoid clashing with the FugueWorkflow builder methods.
"""
from typing import Any, Dict, List, Optional, Tuple

from fugue_amd.collections.partition import PartitionSpec
from fugue_amd.dataframe.dataframe import DataFrame
from fugue_amd.exceptions import FugueSQLError
from fugue_amd.sql._parser import FugueSQLParser
from fugue_amd.sql._utils import fill_sql_template
from fugue_amd.utils.params import ParamDict
from fugue_amd.workflow.workflow import FugueWorkflow, WorkflowDataFrame


class FugueSQLWorkflow(FugueWorkflow):
    """A workflow whose DAG can be extended with FugueSQL code."""

    def __init__(self, compile_conf: Any = None):
        super().__init__(compile_conf)
        self._sql_vars: Dict[str, WorkflowDataFrame] = {}
        self._captured: Dict[str, Any] = {}
        # (name -> guard) for every caller variable the script build
        # consulted; the fugue_sql plan cache replays a built DAG only
        # while every guard still holds (same object / value / absence)
        self._var_guards: Dict[str, Any] = {}

    def __call__(self, code: str, *args: Any, **kwargs: Any) -> None:
        self._sql(code, *args, **kwargs)

    def _sql(self, code: str, *args: Any, **kwargs: Any) -> None:
        variables: Dict[str, Any] = {}
        for a in args:
            if isinstance(a, dict):
                variables.update(a)
        variables.update(kwargs)
        template_vars = {
            k: v
            for k, v in variables.items()
            if not self._is_dfable(v)
        }
        rendered = fill_sql_template(code, template_vars)
        if rendered != code:
            # jinja-templated scripts depend on arbitrary rendered text;
            # the plan cache must not replay them
            self._var_guards["__template__"] = ("nocache",)
        code = rendered
        self._captured.update(variables)
        parser = FugueSQLParser(code, self)
        parser.parse()

    @staticmethod
    def _is_dfable(v: Any) -> bool:
        import pandas as pd
        import pyarrow as pa

        return isinstance(v, (pd.DataFrame, pa.Table, DataFrame, WorkflowDataFrame))

    # --- variable management (parser hooks) ---------------------------- #
    def set_var(self, name: str, df: WorkflowDataFrame) -> None:
        self._sql_vars[name] = df

    def _touch(self, name: str) -> None:
        if name in self._sql_vars or name in self._var_guards:
            return
        if name in self._captured:
            v = self._captured[name]
            if isinstance(v, (str, int, float, bool, bytes)):
                self._var_guards[name] = ("val", type(v), v)
            else:
                import weakref as _wr

                try:
                    self._var_guards[name] = ("ref", _wr.ref(v))
                except TypeError:
                    self._var_guards[name] = ("nocache",)
        else:
            self._var_guards[name] = ("absent",)

    def has_var(self, name: str) -> bool:
        if name in self._sql_vars:
            return True
        self._touch(name)
        v = self._captured.get(name)
        return v is not None and self._is_dfable(v)

    def get_var(self, name: str) -> WorkflowDataFrame:
        if name in self._sql_vars:
            return self._sql_vars[name]
        self._touch(name)
        v = self._captured.get(name)
        if v is not None and self._is_dfable(v):
            wdf = self.df(v)
            self._sql_vars[name] = wdf
            return wdf
        raise FugueSQLError(f"dataframe {name} is not defined")

    def _dfs_or_last(self, dfs: List[Any]) -> List[WorkflowDataFrame]:
        if len(dfs) > 0:
            return dfs
        if self.last_df is None:
            raise FugueSQLError("no dataframe available in context")
        return [self.last_df]

    # --- statement hooks ------------------------------------------------ #
    def select_statement(
        self,
        parts: List[Tuple[bool, str]],
        sql_engine: Any = None,
        sql_engine_params: Any = None,
    ) -> WorkflowDataFrame:
        has_ref = any(r for r, _ in parts)
        text = "".join(v for r, v in parts if not r)
        norm = " " + text.upper().replace("\n", " ") + " "
        if not has_ref and " FROM " not in norm:
            if self.last_df is not None:
                import re as _re

                self._sql_vars["__last__"] = self.last_df
                m = _re.search(
                    r"\b(WHERE|GROUP|HAVING|ORDER|LIMIT|UNION|INTERSECT|EXCEPT)\b",
                    text,
                    _re.IGNORECASE,
                )
                if m is None:
                    parts = list(parts) + [(False, " FROM "), (True, "__last__")]
                else:
                    head, tail = text[: m.start()], text[m.start() :]
                    parts = [
                        (False, head + " FROM "),
                        (True, "__last__"),
                        (False, " " + tail),
                    ]
        statements: List[Any] = []
        for is_ref, v in parts:
            if is_ref:
                statements.append(self.get_var(v) if isinstance(v, str) else v)
            else:
                statements.append(v)
        return self.select(
            *statements,
            sql_engine=self._resolve_ext(sql_engine)
            if sql_engine is not None
            else None,
            sql_engine_params=sql_engine_params,
        )

    def sql_create(self, ext: str, schema: Any, params: ParamDict) -> WorkflowDataFrame:
        return self.create(self._resolve_ext(ext), schema=schema, params=params)

    def sql_create_data(self, data: Any, schema: Any) -> WorkflowDataFrame:
        return self.create_data(data, schema)

    def sql_load(
        self, path: str, fmt: str, columns: Any, params: ParamDict
    ) -> WorkflowDataFrame:
        return self.load(path, fmt=fmt, columns=columns, **params)

    def sql_transform(
        self,
        dfs: List[Any],
        ext: Any,
        schema: Any,
        params: Any,
        spec: Optional[PartitionSpec],
        callback: Any,
    ) -> WorkflowDataFrame:
        _dfs = self._dfs_or_last(dfs)
        return self.transform(
            *_dfs,
            using=self._resolve_ext(ext),
            schema=schema,
            params=params,
            pre_partition=spec,
            callback=self._resolve_callback(callback),
        )

    def _resolve_ext(self, ext: Any) -> Any:
        if isinstance(ext, str):
            self._touch(ext)
        if isinstance(ext, str) and ext in self._captured:
            return self._captured[ext]
        return ext

    def _resolve_callback(self, callback: Any) -> Any:
        if callback is None:
            return None
        if isinstance(callback, str):
            self._touch(callback)
        if isinstance(callback, str) and callback in self._captured:
            return self._captured[callback]
        return callback

    def sql_out_transform(
        self,
        dfs: List[Any],
        ext: Any,
        params: Any,
        spec: Optional[PartitionSpec],
        callback: Any,
    ) -> None:
        _dfs = self._dfs_or_last(dfs)
        self.out_transform(
            *_dfs,
            using=self._resolve_ext(ext),
            params=params,
            pre_partition=spec,
            callback=self._resolve_callback(callback),
        )

    def sql_process(
        self,
        dfs: List[Any],
        ext: Any,
        schema: Any,
        params: Any,
        spec: Optional[PartitionSpec],
    ) -> WorkflowDataFrame:
        _dfs = self._dfs_or_last(dfs)
        return self.process(
            *_dfs, using=self._resolve_ext(ext), schema=schema, params=params, pre_partition=spec
        )

    def sql_output(
        self, dfs: List[Any], ext: Any, params: Any, spec: Optional[PartitionSpec]
    ) -> None:
        _dfs = self._dfs_or_last(dfs)
        self.output(*_dfs, using=self._resolve_ext(ext), params=params, pre_partition=spec)

    def print_dfs(
        self, dfs: List[Any], n: int, with_count: bool, title: Optional[str]
    ) -> None:
        for df in self._dfs_or_last(dfs):
            df.show(n, with_count=with_count, title=title)

    def sql_save(
        self,
        dfs: List[Any],
        path: str,
        fmt: str,
        mode: str,
        single: bool,
        spec: Optional[PartitionSpec],
        params: ParamDict,
        and_use: bool,
    ) -> Optional[WorkflowDataFrame]:
        df = self._dfs_or_last(dfs)[0]
        if and_use:
            return df.save_and_use(
                path, fmt=fmt, mode=mode, partition=spec, single=single, **params
            )
        df.save(path, fmt=fmt, mode=mode, partition=spec, single=single, **params)
        return None

    def sql_take(
        self,
        dfs: List[Any],
        n: int,
        presort: Optional[str],
        na_position: str,
        spec: Optional[PartitionSpec],
    ) -> WorkflowDataFrame:
        df = self._dfs_or_last(dfs)[0]
        if spec is not None:
            df = df.partition(spec)
        return df.take(n, presort=presort or "", na_position=na_position)

    def sql_zip(
        self,
        dfs: List[Any],
        how: str,
        by: Optional[List[str]],
        presort: Optional[str],
    ) -> WorkflowDataFrame:
        _dfs = self._dfs_or_last(dfs)
        spec_args: Dict[str, Any] = {}
        if by is not None:
            spec_args["by"] = by
        if presort is not None:
            spec_args["presort"] = presort
        return self.zip(*_dfs, how=how, partition=PartitionSpec(**spec_args))

    def sql_drop_columns(
        self, dfs: List[Any], cols: List[str], if_exists: bool
    ) -> WorkflowDataFrame:
        return self._dfs_or_last(dfs)[0].drop(cols, if_exists=if_exists)

    def sql_dropna(
        self, dfs: List[Any], how: str, subset: Optional[List[str]]
    ) -> WorkflowDataFrame:
        return self._dfs_or_last(dfs)[0].dropna(how=how, subset=subset)

    def sql_rename(self, dfs: List[Any], columns: Dict[str, str]) -> WorkflowDataFrame:
        return self._dfs_or_last(dfs)[0].rename(columns)

    def sql_alter_columns(self, dfs: List[Any], schema: str) -> WorkflowDataFrame:
        return self._dfs_or_last(dfs)[0].alter_columns(schema)

    def sql_sample(
        self,
        dfs: List[Any],
        n: Optional[int],
        frac: Optional[float],
        replace: bool,
        seed: Optional[int],
    ) -> WorkflowDataFrame:
        return self._dfs_or_last(dfs)[0].sample(
            n=n, frac=frac, replace=replace, seed=seed
        )

    def sql_fillna(self, dfs: List[Any], value: Any) -> WorkflowDataFrame:
        return self._dfs_or_last(dfs)[0].fillna(value)

    def sql_module(
        self,
        ordered: List[Any],
        named: Dict[str, Any],
        ext: Any,
        params: ParamDict,
    ) -> Any:
        """``SUB`` statement: invoke a workflow module (reference
        ``_visitors.py:697``).  Input dispatch: ``WorkflowDataFrames``
        annotation gets the whole collection; otherwise positional
        frames (or the previous statement's frame when none given)."""
        from fugue_amd.workflow.module import to_module
        from fugue_amd.workflow.workflow import WorkflowDataFrames

        sub = to_module(self._resolve_ext(ext), self._captured)
        if sub.has_input:
            if len(ordered) == 0 and len(named) == 0:
                dfs = WorkflowDataFrames(self._dfs_or_last([])[0])
            elif len(named) > 0:
                dfs = WorkflowDataFrames(**named)
            else:
                dfs = WorkflowDataFrames(*ordered)
        else:
            dfs = WorkflowDataFrames()
        p = dict(params)
        if sub.has_dfs_input:
            result = sub(dfs, **p)
        elif len(dfs) == 0:
            result = sub(self, **p)
        elif len(dfs) == 1 or not dfs.has_key:
            result = sub(*list(dfs.values()), **p)
        else:
            result = sub(**dfs, **p)
        if sub.has_single_output:
            self._last_df = result
            return result
        if sub.has_multiple_output:
            return result
        return None

    # --- post clauses ---------------------------------------------------- #
    def sql_persist(self, df: Any) -> Any:
        return df.persist() if df is not None else None

    def sql_broadcast(self, df: Any) -> Any:
        return df.broadcast() if df is not None else None

    def sql_weak_checkpoint(self, df: Any) -> Any:
        return df.weak_checkpoint() if df is not None else None

    def sql_strong_checkpoint(self, df: Any) -> Any:
        return df.strong_checkpoint() if df is not None else None

    def sql_deterministic_checkpoint(self, df: Any) -> Any:
        return df.deterministic_checkpoint() if df is not None else None

    def yield_as(self, df: Any, kind: str, name: str, local: bool) -> Any:
        if df is None:
            raise FugueSQLError("nothing to yield")
        if kind == "DATAFRAME":
            df.yield_dataframe_as(name, as_local=local)
        elif kind == "FILE":
            df.yield_file_as(name)
        elif kind == "TABLE":
            df.yield_table_as(name)
        else:
            raise FugueSQLError(f"can't yield {kind}")
        return df
