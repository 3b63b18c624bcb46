"""FugueSQL dialect parser: compiles the extended SQL language into
workflow DAG operations.

Reference parity: ``fugue/sql/_visitors.py`` (which walks an ANTLR tree;
here a hand-written recursive-descent parser emits the same DAG calls).

Supported statement forms::

    name = <body> | name ?? <body> | <body>
    body:
      SELECT ...                         (passed to the SQL engine)
      CREATE [[row],[row]] SCHEMA s
      CREATE USING ext(params) [SCHEMA s]
      LOAD [PARQUET|CSV|JSON] 'path' [(params)] [COLUMNS cols]
      TRANSFORM [dfs] [PREPARTITION ...] USING ext [(params)] [SCHEMA s] [CALLBACK f]
      OUTTRANSFORM [dfs] [PREPARTITION ...] USING ext [(params)]
      PROCESS [dfs] [PREPARTITION ...] USING ext [(params)] [SCHEMA s]
      OUTPUT [dfs] [PREPARTITION ...] USING ext [(params)]
      PRINT [dfs] [ROWS n] [ROWCOUNT] [TITLE 'title']
      SAVE [df] [PREPARTITION ...] OVERWRITE|APPEND|TO [SINGLE] [fmt] 'path' [(params)]
      SAVE AND USE ...
      TAKE n ROW[S] [FROM df] [PREPARTITION ...] [PRESORT cols] [NULL[S] FIRST|LAST]
      ZIP dfs [how] [BY cols] [PRESORT ...]
      DROP ROWS IF ALL|ANY NULL[S] [ON cols] [FROM df] / DROP COLUMNS c1,c2 [IF EXISTS]
      RENAME COLUMNS a:b,c:d [FROM df]
      ALTER COLUMNS a:type [FROM df]
      SAMPLE [REPLACE] n ROWS | x PERCENT [SEED s] [FROM df]
      FILL NULLS PARAMS ... / FILL NULLS (params)
    post-clauses: PERSIST | BROADCAST | [WEAK|STRONG|DETERMINISTIC] CHECKPOINT
                  | YIELD [LOCAL] DATAFRAME|FILE|TABLE AS name
"""
import json
from typing import Any, Callable, Dict, List, Optional, Tuple

from fugue_amd.collections.partition import PartitionSpec
from fugue_amd.exceptions import FugueSQLSyntaxError
from fugue_amd.sql._tokenizer import Token, TokenStream, tokenize
from fugue_amd.utils.params import ParamDict

_STMT_START = {
    "SELECT",
    "CREATE",
    "LOAD",
    "TRANSFORM",
    "OUTTRANSFORM",
    "PROCESS",
    "OUTPUT",
    "PRINT",
    "SAVE",
    "TAKE",
    "ZIP",
    "DROP",
    "RENAME",
    "ALTER",
    "SAMPLE",
    "FILL",
    "CONNECT",
    "WITH",
    "SUB",
}

_POST_CLAUSES = {
    "PERSIST",
    "BROADCAST",
    "CHECKPOINT",
    "WEAK",
    "STRONG",
    "DETERMINISTIC",
    "YIELD",
}


class FugueSQLParser:
    """Parse a FugueSQL script and emit DAG calls on a hook object (the
    FugueSQLWorkflow)."""

    def __init__(self, code: str, hooks: Any):
        self.code = code
        self.ts = TokenStream(tokenize(code))
        self.hooks = hooks  # must provide the visit_* callbacks

    # ------------------------------------------------------------------ #
    def parse(self) -> None:
        while not self.ts.eof:
            while self.ts.take_punct(";"):
                pass
            if self.ts.eof:
                break
            self._parse_statement()

    def _parse_statement(self) -> None:
        ts = self.ts
        assign_name: Optional[str] = None
        t = ts.peek()
        # assignment: name = body  or  name ?? body
        if (
            t is not None
            and t.kind == "NAME"
            and t.upper not in _STMT_START
            and ts.peek(1) is not None
        ):
            nxt = ts.peek(1)
            if nxt.kind == "OP" and nxt.value in ("=", "=="):
                assign_name = ts.next().value
                ts.next()
            elif nxt.kind == "OP" and nxt.value == "??":  # pragma: no cover
                assign_name = ts.next().value
                ts.next()
        t = ts.peek()
        if t is None:
            raise FugueSQLSyntaxError("empty statement")
        kw = t.upper
        if kw == "SELECT" or kw == "WITH":
            df = self._parse_select_statement()
        elif kw == "CONNECT":
            # CONNECT <engine>[(params)] SELECT ... : per-query SQL engine
            # (reference ``_visitors.py:728`` visitFugueSqlEngine)
            ts.next()
            eng = self._parse_extension_name()
            eng_params = self._parse_params()
            nt = ts.peek()
            if nt is None or nt.upper not in ("SELECT", "WITH"):
                raise FugueSQLSyntaxError("CONNECT must be followed by SELECT")
            df = self._parse_select_statement(
                sql_engine=eng, sql_engine_params=eng_params
            )
        elif kw == "CREATE":
            df = self._parse_create()
        elif kw == "LOAD":
            df = self._parse_load()
        elif kw == "TRANSFORM":
            df = self._parse_transform(output=False)
        elif kw == "OUTTRANSFORM":
            df = self._parse_transform(output=True)
        elif kw == "PROCESS":
            df = self._parse_process()
        elif kw == "OUTPUT":
            self._parse_output()
            df = None
        elif kw == "PRINT":
            self._parse_print()
            df = None
        elif kw == "SAVE":
            df = self._parse_save()
        elif kw == "TAKE":
            df = self._parse_take()
        elif kw == "ZIP":
            df = self._parse_zip()
        elif kw == "DROP":
            df = self._parse_drop()
        elif kw == "RENAME":
            df = self._parse_rename()
        elif kw == "ALTER":
            df = self._parse_alter()
        elif kw == "SAMPLE":
            df = self._parse_sample()
        elif kw == "FILL":
            df = self._parse_fill()
        elif kw == "SUB":
            df = self._parse_sub()
        else:
            raise FugueSQLSyntaxError(
                f"unexpected token {t.value!r} at {t.pos} in FugueSQL"
            )
        # post clauses
        df = self._parse_post_clauses(df)
        if assign_name is not None and df is not None:
            self.hooks.set_var(assign_name, df)

    # ------------------------------------------------------------------ #
    def _parse_post_clauses(self, df: Any) -> Any:
        ts = self.ts
        while True:
            if ts.take_kw("PERSIST"):
                df = self.hooks.sql_persist(df)
            elif ts.take_kw("BROADCAST"):
                df = self.hooks.sql_broadcast(df)
            elif ts.match_kw("WEAK") or ts.match_kw("LAZY"):
                ts.next()
                ts.expect_kw("CHECKPOINT")
                df = self.hooks.sql_weak_checkpoint(df)
            elif ts.take_kw("STRONG"):
                ts.expect_kw("CHECKPOINT")
                df = self.hooks.sql_strong_checkpoint(df)
            elif ts.take_kw("DETERMINISTIC"):
                ts.expect_kw("CHECKPOINT")
                df = self.hooks.sql_deterministic_checkpoint(df)
            elif ts.take_kw("CHECKPOINT"):
                df = self.hooks.sql_strong_checkpoint(df)
            elif ts.take_kw("YIELD"):
                local = ts.take_kw("LOCAL")
                kind = ts.next().upper  # DATAFRAME | FILE | TABLE
                ts.expect_kw("AS")
                name = ts.next().value
                df = self.hooks.yield_as(df, kind, name, local)
            else:
                return df

    # ------------------------------------------------------------------ #
    def _parse_select_statement(
        self, sql_engine: Any = None, sql_engine_params: Any = None
    ) -> Any:
        """Collect the SELECT body tokens until a statement/post-clause
        boundary; table references resolved through the hook."""
        ts = self.ts
        start = ts.peek().pos
        depth = 0
        parts: List[Tuple[bool, str]] = []
        seg_start = start
        end = start
        from_context = False
        while not ts.eof:
            t = ts.peek()
            if t.kind == "PUNCT" and t.value == "(":
                depth += 1
            elif t.kind == "PUNCT" and t.value == ")":
                depth -= 1
            elif depth == 0 and t.kind == "PUNCT" and t.value == ";":
                ts.next()
                break
            elif t.kind == "NAME":
                up = t.upper
                if depth == 0:
                    if up in _POST_CLAUSES:
                        break
                    if up in (
                        "TRANSFORM", "PROCESS", "OUTPUT", "PRINT", "SAVE",
                        "TAKE", "ZIP", "OUTTRANSFORM", "LOAD", "CREATE",
                    ):
                        break
                    nxt = ts.peek(1)
                    if (
                        up not in _STMT_START
                        and nxt is not None
                        and nxt.kind == "OP"
                        and nxt.value == "="
                        and self._starts_line(t)
                    ):
                        break  # next statement: NAME = ...
                    if up == "SELECT" and t.pos != start:
                        prev = ts.tokens[ts.pos - 1] if ts.pos > 0 else None
                        if prev is None or prev.upper not in (
                            "UNION", "ALL", "INTERSECT", "EXCEPT", "(",
                        ):
                            break  # a new top-level SELECT statement
                        from_context = False
                # table-ref replacement tracks FROM/JOIN context at ANY
                # depth so subqueries resolve captured frames too
                if up == "SELECT":
                    from_context = False
                elif up in ("FROM", "JOIN"):
                    from_context = True
                elif up in (
                    "WHERE", "GROUP", "HAVING", "ORDER", "LIMIT",
                    "ON", "UNION", "INTERSECT", "EXCEPT",
                ):
                    from_context = False
                elif from_context and self.hooks.has_var(t.value):
                    parts.append((False, self.code[seg_start : t.pos]))
                    ts.next()
                    ref_end = t.src_end
                    if ts.match_punct("["):
                        # dfs[key] / dfs[0]: resolve to the object here
                        obj = self._maybe_index(self.hooks.get_var(t.value))
                        last = ts.tokens[ts.pos - 1]  # the closing ']'
                        ref_end = last.src_end
                        parts.append((True, obj))
                    else:
                        parts.append((True, t.value))
                    # keep the original name visible as a table alias unless
                    # the query supplies its own alias
                    nxt2 = ts.peek()
                    has_alias = nxt2 is not None and (
                        (nxt2.kind == "NAME" and nxt2.upper == "AS")
                        or (
                            nxt2.kind == "NAME"
                            and nxt2.upper not in _STMT_START
                            and nxt2.upper not in _POST_CLAUSES
                            and nxt2.upper
                            not in (
                                "WHERE", "GROUP", "HAVING", "ORDER", "LIMIT",
                                "ON", "UNION", "INTERSECT", "EXCEPT", "JOIN",
                                "INNER", "LEFT", "RIGHT", "FULL", "CROSS",
                                "USING", "AND", "OR",
                            )
                        )
                    )
                    if not has_alias:
                        parts.append((False, f" AS {t.value} "))
                    seg_start = ref_end
                    end = seg_start
                    continue
            ts.next()
            end = t.src_end
        parts.append((False, self.code[seg_start:end]))
        return self.hooks.select_statement(
            [(r, v) for r, v in parts if r or v != ""],
            sql_engine=sql_engine,
            sql_engine_params=sql_engine_params,
        )

    def _token_end(self, idx: int) -> int:
        t = self.ts.tokens[idx]
        return t.src_end

    def _starts_line(self, t: Token) -> bool:
        """Whether this token is the first non-whitespace on its line."""
        i = t.pos - 1
        while i >= 0 and self.code[i] in " \t":
            i -= 1
        return i < 0 or self.code[i] == "\n"

    # ------------------------------------------------------------------ #
    def _parse_schema(self) -> str:
        """Consume schema tokens (after SCHEMA keyword) as raw text until a
        boundary keyword or statement start."""
        ts = self.ts
        start = ts.peek().pos
        end = start
        depth = 0
        while not ts.eof:
            t = ts.peek()
            if t.kind == "PUNCT" and t.value in "([{<":
                depth += 1
            elif t.kind == "PUNCT" and t.value in ")]}>":
                if depth == 0:
                    break
                depth -= 1
            elif depth == 0 and t.kind == "NAME" and t.upper in (
                _STMT_START | _POST_CLAUSES | {"USING", "FROM", "CALLBACK", "PREPARTITION"}
            ):
                break
            elif depth == 0 and t.kind == "PUNCT" and t.value == ";":
                break
            elif (
                depth == 0
                and t.kind == "NAME"
                and self._starts_line(t)
                and ts.peek(1) is not None
                and ts.peek(1).kind == "OP"
                and ts.peek(1).value == "="
            ):
                break  # next statement: NAME = ...
            ts.next()
            end = self._token_end(ts.pos - 1)
        return self.code[start:end].strip()

    def _parse_params(self) -> ParamDict:
        """Parse ``(key=value, ...)`` or ``PARAMS key=value,...`` or a JSON
        object ``{...}``."""
        ts = self.ts
        res = ParamDict()
        if ts.take_kw("PARAMS"):
            pass
        elif not ts.match_punct("(") and not ts.match_punct("{"):
            return res
        if ts.take_punct("("):
            closing = ")"
        elif ts.take_punct("{"):
            closing = "}"
        else:
            closing = None
        while True:
            if closing is not None and ts.take_punct(closing):
                break
            t = ts.peek()
            if t is None:
                break
            key = ts.next().value.strip('"')
            if not (ts.take_punct(":") or self._take_op("=")):
                raise FugueSQLSyntaxError(f"expected = or : after {key}")
            res[key] = self._parse_value()
            if not ts.take_punct(","):
                if closing is not None:
                    ts.expect_punct(closing)
                break
        return res

    def _take_op(self, op: str) -> bool:
        t = self.ts.peek()
        if t is not None and t.kind == "OP" and t.value == op:
            self.ts.next()
            return True
        return False

    def _parse_value(self) -> Any:
        ts = self.ts
        t = ts.peek()
        if t is None:
            raise FugueSQLSyntaxError("expected value")
        if t.kind == "STRING":
            ts.next()
            return t.value[1:-1]
        if t.kind == "NUMBER":
            ts.next()
            return float(t.value) if "." in t.value else int(t.value)
        if t.kind == "OP" and t.value == "-":
            ts.next()
            v = self._parse_value()
            return -v
        if t.kind == "NAME":
            up = t.upper
            if up == "TRUE":
                ts.next()
                return True
            if up == "FALSE":
                ts.next()
                return False
            if up == "NULL" or up == "NONE":
                ts.next()
                return None
            ts.next()
            return t.value
        if t.value == "[":
            ts.next()
            arr = []
            while not ts.take_punct("]"):
                arr.append(self._parse_value())
                ts.take_punct(",")
            return arr
        if t.value == "{":
            ts.next()
            obj = {}
            while not ts.take_punct("}"):
                k = ts.next().value.strip('"')
                if not (ts.take_punct(":") or self._take_op("=")):
                    raise FugueSQLSyntaxError("expected : in dict")
                obj[k] = self._parse_value()
                ts.take_punct(",")
            return obj
        raise FugueSQLSyntaxError(f"unexpected value token {t.value!r}")

    def _parse_df_list(self) -> List[Any]:
        """Parse dataframe references (names or nothing → last df)."""
        ts = self.ts
        dfs: List[Any] = []
        while True:
            t = ts.peek()
            if (
                t is not None
                and t.kind == "NAME"
                and t.upper not in _STMT_START
                and t.upper
                not in ("USING", "PREPARTITION", "ROWS", "ROWCOUNT", "TITLE", "BY", "PRESORT", "FROM", "SCHEMA", "CALLBACK", "OVERWRITE", "APPEND", "TO", "ROW", "AND")
                and self.hooks.has_var(t.value)
            ):
                dfs.append(self._maybe_index(self.hooks.get_var(ts.next().value)))
                if not ts.take_punct(","):
                    break
            else:
                break
        return dfs

    def _maybe_index(self, obj: Any) -> Any:
        """Apply ``[key]`` / ``[0]`` indexing after a dataframes variable
        reference (reference test ``print dfs[a1]``)."""
        ts = self.ts
        while ts.take_punct("["):
            t = ts.next()
            if t.kind == "NUMBER":
                key: Any = int(t.value)
            elif t.kind == "STRING":
                key = t.value[1:-1]
            else:
                key = t.value
            ts.expect_punct("]")
            obj = obj[key]
        return obj

    def _parse_sub(self) -> Any:
        """``SUB [df | key:df, ...] USING module [(params)]`` — invoke a
        :func:`fugue_amd.workflow.module.module` function (reference
        ``_visitors.py:697`` ``visitFugueModuleTask``)."""
        ts = self.ts
        ts.expect_kw("SUB")
        ordered: List[Any] = []
        named: Dict[str, Any] = {}
        while True:
            t = ts.peek()
            if t is None or t.kind != "NAME" or t.upper == "USING":
                break
            nxt = ts.peek(1)
            if nxt is not None and nxt.kind == "PUNCT" and nxt.value == ":":
                key = ts.next().value
                ts.next()  # ':'
                vt = ts.next()
                if vt is None or vt.kind != "NAME" or not self.hooks.has_var(
                    vt.value
                ):
                    where = f" at {vt.pos}" if vt is not None else ""
                    raise FugueSQLSyntaxError(
                        "unknown dataframe variable "
                        f"{vt.value if vt is not None else '<eof>'!r} "
                        f"for SUB input {key!r}{where}"
                    )
                named[key] = self._maybe_index(self.hooks.get_var(vt.value))
            elif self.hooks.has_var(t.value):
                ordered.append(self._maybe_index(self.hooks.get_var(ts.next().value)))
            else:
                break
            if not ts.take_punct(","):
                break
        ts.expect_kw("USING")
        ext = self._parse_extension_name()
        params = self._parse_params()
        return self.hooks.sql_module(ordered, named, ext, params)

    def _parse_prepartition(self) -> Optional[PartitionSpec]:
        ts = self.ts
        if not ts.take_kw("PREPARTITION"):
            return None
        kwargs: Dict[str, Any] = {}
        if ts.take_kw("RAND"):
            kwargs["algo"] = "rand"
        elif ts.take_kw("HASH"):
            kwargs["algo"] = "hash"
        elif ts.take_kw("EVEN"):
            kwargs["algo"] = "even"
        elif ts.take_kw("COARSE"):
            kwargs["algo"] = "coarse"
        t = ts.peek()
        if t is not None and t.kind == "NUMBER":
            kwargs["num"] = int(ts.next().value)
        if ts.take_kw("BY"):
            cols = []
            while True:
                cols.append(ts.next().value)
                if not ts.take_punct(","):
                    break
            kwargs["by"] = cols
        if ts.take_kw("PRESORT"):
            kwargs["presort"] = self._parse_presort_text()
        return PartitionSpec(**kwargs)

    def _parse_presort_text(self) -> str:
        ts = self.ts
        parts = []
        while True:
            name = ts.next().value
            direction = ""
            if ts.take_kw("ASC"):
                direction = " asc"
            elif ts.take_kw("DESC"):
                direction = " desc"
            parts.append(name + direction)
            if not ts.take_punct(","):
                break
        return ",".join(parts)

    # ------------------------------------------------------------------ #
    def _parse_create(self) -> Any:
        ts = self.ts
        ts.expect_kw("CREATE")
        if ts.take_kw("USING"):
            ext = self._parse_extension_name()
            params = self._parse_params()
            schema = None
            if ts.take_kw("SCHEMA"):
                schema = self._parse_schema()
            return self.hooks.sql_create(ext, schema, params)
        # literal data: [[...],[...]] SCHEMA s
        data = self._parse_value()
        ts.expect_kw("SCHEMA")
        schema = self._parse_schema()
        return self.hooks.sql_create_data(data, schema)

    def _parse_extension_name(self) -> str:
        ts = self.ts
        name = ts.next().value
        while ts.match_punct(".") or ts.match_punct(":"):
            sep = ts.next().value
            name += sep + ts.next().value
        return name

    def _parse_load(self) -> Any:
        ts = self.ts
        ts.expect_kw("LOAD")
        fmt = ""
        if ts.match_kw("PARQUET", "CSV", "JSON"):
            fmt = ts.next().value.lower()
        path = ts.next().value[1:-1]
        params = self._parse_params()
        columns = None
        if ts.take_kw("COLUMNS"):
            columns = self._parse_schema()
            if ":" not in columns:
                # a bare column list; names may be backtick-quoted
                columns = [
                    c.strip().strip("`") for c in columns.split(",")
                ]
        return self.hooks.sql_load(path, fmt, columns, params)

    def _parse_transform(self, output: bool) -> Any:
        ts = self.ts
        ts.next()  # TRANSFORM / OUTTRANSFORM
        dfs = self._parse_df_list()
        spec = self._parse_prepartition()
        ts.expect_kw("USING")
        ext = self._parse_extension_name()
        params = self._parse_params()
        schema = None
        if ts.take_kw("SCHEMA"):
            schema = self._parse_schema()
        callback = None
        if ts.take_kw("CALLBACK"):
            callback = self._parse_extension_name()
        if output:
            self.hooks.sql_out_transform(dfs, ext, params, spec, callback)
            return None
        return self.hooks.sql_transform(dfs, ext, schema, params, spec, callback)

    def _parse_process(self) -> Any:
        ts = self.ts
        ts.expect_kw("PROCESS")
        dfs = self._parse_df_list()
        spec = self._parse_prepartition()
        ts.expect_kw("USING")
        ext = self._parse_extension_name()
        params = self._parse_params()
        schema = None
        if ts.take_kw("SCHEMA"):
            schema = self._parse_schema()
        return self.hooks.sql_process(dfs, ext, schema, params, spec)

    def _parse_output(self) -> None:
        ts = self.ts
        ts.expect_kw("OUTPUT")
        dfs = self._parse_df_list()
        spec = self._parse_prepartition()
        ts.expect_kw("USING")
        ext = self._parse_extension_name()
        params = self._parse_params()
        self.hooks.sql_output(dfs, ext, params, spec)

    def _parse_print(self) -> None:
        ts = self.ts
        ts.expect_kw("PRINT")
        dfs = self._parse_df_list()
        n = 10
        with_count = False
        title = None
        while True:
            t = ts.peek()
            if t is not None and t.kind == "NUMBER":
                n = int(ts.next().value)
                ts.take_kw("ROWS")
            elif ts.take_kw("ROWS"):
                n = int(ts.next().value)
            elif ts.take_kw("ROWCOUNT"):
                with_count = True
            elif ts.take_kw("TITLE"):
                title = ts.next().value[1:-1]
            else:
                break
        self.hooks.print_dfs(dfs, n, with_count, title)

    def _parse_save(self) -> Any:
        ts = self.ts
        ts.expect_kw("SAVE")
        and_use = False
        if ts.take_kw("AND"):
            ts.expect_kw("USE")
            and_use = True
        dfs = self._parse_df_list()
        spec = self._parse_prepartition()
        mode = "overwrite"
        if ts.take_kw("OVERWRITE"):
            mode = "overwrite"
        elif ts.take_kw("APPEND"):
            mode = "append"
        elif ts.take_kw("TO"):
            mode = "error"
        single = ts.take_kw("SINGLE")
        fmt = ""
        if ts.match_kw("PARQUET", "CSV", "JSON"):
            fmt = ts.next().value.lower()
        path = ts.next().value[1:-1]
        params = self._parse_params()
        return self.hooks.sql_save(dfs, path, fmt, mode, single, spec, params, and_use)

    def _parse_take(self) -> Any:
        ts = self.ts
        ts.expect_kw("TAKE")
        n = int(ts.next().value)
        ts.take_kw("ROWS") or ts.take_kw("ROW")
        dfs: List[Any] = []
        if ts.take_kw("FROM"):
            dfs = self._parse_df_list()
        spec = self._parse_prepartition()
        presort = None
        if ts.take_kw("PRESORT"):
            presort = self._parse_presort_text()
        na_position = "last"
        if ts.take_kw("NULL") or ts.take_kw("NULLS"):
            if ts.take_kw("FIRST"):
                na_position = "first"
            else:
                ts.take_kw("LAST")
        return self.hooks.sql_take(dfs, n, presort, na_position, spec)

    def _parse_zip(self) -> Any:
        ts = self.ts
        ts.expect_kw("ZIP")
        dfs = self._parse_df_list()
        how = "inner"
        for h in ("INNER", "LEFT", "RIGHT", "FULL", "CROSS"):
            if ts.take_kw(h):
                ts.take_kw("OUTER")
                how = {
                    "INNER": "inner",
                    "LEFT": "left_outer",
                    "RIGHT": "right_outer",
                    "FULL": "full_outer",
                    "CROSS": "cross",
                }[h]
                break
        by: Optional[List[str]] = None
        if ts.take_kw("BY"):
            by = []
            while True:
                by.append(ts.next().value)
                if not ts.take_punct(","):
                    break
        presort = None
        if ts.take_kw("PRESORT"):
            presort = self._parse_presort_text()
        return self.hooks.sql_zip(dfs, how, by, presort)

    def _parse_drop(self) -> Any:
        ts = self.ts
        ts.expect_kw("DROP")
        if ts.take_kw("COLUMNS"):
            cols = []
            while True:
                cols.append(ts.next().value)
                if not ts.take_punct(","):
                    break
            if_exists = False
            if ts.take_kw("IF"):
                ts.expect_kw("EXISTS")
                if_exists = True
            dfs: List[Any] = []
            if ts.take_kw("FROM"):
                dfs = self._parse_df_list()
            return self.hooks.sql_drop_columns(dfs, cols, if_exists)
        ts.expect_kw("ROWS")
        ts.expect_kw("IF")
        how = "any"
        if ts.take_kw("ALL"):
            how = "all"
        else:
            ts.take_kw("ANY")
        ts.take_kw("NULL") or ts.take_kw("NULLS")
        subset = None
        if ts.take_kw("ON"):
            subset = []
            while True:
                subset.append(ts.next().value)
                if not ts.take_punct(","):
                    break
        dfs = []
        if ts.take_kw("FROM"):
            dfs = self._parse_df_list()
        return self.hooks.sql_dropna(dfs, how, subset)

    def _parse_rename(self) -> Any:
        ts = self.ts
        ts.expect_kw("RENAME")
        ts.expect_kw("COLUMNS")
        m: Dict[str, str] = {}
        while True:
            old = ts.next().value
            ts.expect_punct(":")
            new = ts.next().value
            m[old] = new
            if not ts.take_punct(","):
                break
        dfs: List[Any] = []
        if ts.take_kw("FROM"):
            dfs = self._parse_df_list()
        return self.hooks.sql_rename(dfs, m)

    def _parse_alter(self) -> Any:
        ts = self.ts
        ts.expect_kw("ALTER")
        ts.expect_kw("COLUMNS")
        schema = self._parse_schema()
        dfs: List[Any] = []
        if ts.take_kw("FROM"):
            dfs = self._parse_df_list()
        return self.hooks.sql_alter_columns(dfs, schema)

    def _parse_sample(self) -> Any:
        ts = self.ts
        ts.expect_kw("SAMPLE")
        replace = ts.take_kw("REPLACE")
        n = None
        frac = None
        t = ts.next()
        value = float(t.value)
        if ts.take_kw("ROWS"):
            n = int(value)
        elif ts.take_kw("PERCENT"):
            frac = value / 100.0
        seed = None
        if ts.take_kw("SEED"):
            seed = int(ts.next().value)
        dfs: List[Any] = []
        if ts.take_kw("FROM"):
            dfs = self._parse_df_list()
        return self.hooks.sql_sample(dfs, n, frac, replace, seed)

    def _parse_fill(self) -> Any:
        ts = self.ts
        ts.expect_kw("FILL")
        ts.take_kw("NULL") or ts.take_kw("NULLS")
        params = self._parse_params()
        dfs: List[Any] = []
        if ts.take_kw("FROM"):
            dfs = self._parse_df_list()
        return self.hooks.sql_fillna(dfs, dict(params))
