"""Raw SQL statements as (is_table_ref, text) token lists.

Reference parity: ``fugue/collections/sql.py`` — ``StructuredRawSQL``
carries a SQL statement whose table references are symbolic, so engines can
substitute their own temp-table names at execution time.  Dialect
transpilation is a no-op here (no sqlglot dependency); the MI355X SQL
engine consumes a single dialect directly.
"""
import re
import uuid
from typing import Any, Callable, Iterable, List, Optional, Tuple

from fugue_amd.utils.hash import to_uuid


class TempTableName:
    """A unique temp table reference usable inside raw SQL."""

    def __init__(self):
        self.key = "_" + str(uuid.uuid4())[:5]

    @property
    def body(self) -> str:
        return "<tmpdf:" + self.key + ">"

    def __repr__(self) -> str:
        return self.body


_TMP_RE = re.compile(r"<tmpdf:(?P<key>[^>]+)>")


# dialect families that share the grammar this engine's SELECT parser
# accepts (spark-flavored: backtick identifiers, standard functions)
_SPARKLIKE = {"spark", "hive", "databricks", "trino", "presto"}

# function spellings normalized to the internal dialect
_FN_RENAMES = {
    "ifnull": "COALESCE",
    "nvl": "COALESCE",
}


def _builtin_transpile(raw: str, from_dialect: str, to_dialect: str) -> str:
    """Minimal cross-dialect normalization for the supported grammar:
    double-quoted identifiers (duckdb/postgres style) become backticked,
    and a small set of function aliases are rewritten.  Constructs
    outside the engine's grammar surface still fail in the parser with
    a position-annotated error."""
    out = []
    i, n = 0, len(raw)
    while i < n:
        ch = raw[i]
        if ch == "'":  # string literal: copy verbatim ('' escapes)
            j = i + 1
            while j < n:
                if raw[j] == "'" and (j + 1 >= n or raw[j + 1] != "'"):
                    break
                j += 2 if raw[j] == "'" else 1
            out.append(raw[i : j + 1])
            i = j + 1
            continue
        if ch == '"' and from_dialect not in _SPARKLIKE:
            j = raw.find('"', i + 1)
            if j < 0:
                out.append(raw[i:])
                break
            out.append("`" + raw[i + 1 : j] + "`")
            i = j + 1
            continue
        out.append(ch)
        i += 1
    res = "".join(out)
    for name, repl in _FN_RENAMES.items():
        res = re.sub(rf"\b{name}\s*\(", repl + "(", res, flags=re.I)
    return res


def transpile_sql(raw: str, from_dialect: Optional[str], to_dialect: Optional[str]) -> str:
    """Dialect transpile: plugin-overridable; the built-in fallback
    normalizes quote styles and function aliases between the supported
    dialect families (reference used sqlglot, ``fugue/collections/
    sql.py:24``)."""
    from fugue_amd.utils.registry import try_run_plugin

    if (
        from_dialect is None
        or to_dialect is None
        or from_dialect == to_dialect
    ):
        return raw
    ok, res = try_run_plugin("transpile_sql", raw, from_dialect, to_dialect)
    if ok:
        return res
    if from_dialect in _SPARKLIKE and to_dialect in _SPARKLIKE:
        return raw
    return _builtin_transpile(raw, from_dialect, to_dialect)


class StructuredRawSQL:
    """A list of (is_table_reference, text) parts making up one statement."""

    def __init__(
        self,
        statements: Iterable[Tuple[bool, str]],
        dialect: Optional[str] = None,
    ):
        self._statements = list(statements)
        self._dialect = dialect

    @property
    def dialect(self) -> Optional[str]:
        return self._dialect

    def __uuid__(self) -> str:
        return to_uuid(self._dialect, self._statements)

    def construct(
        self,
        name_map: Any = None,
        dialect: Optional[str] = None,
        log: Any = None,
    ) -> str:
        """Construct the final SQL string, mapping symbolic table names
        through ``name_map`` (dict or callable)."""

        def _map(name: str) -> str:
            if name_map is None:
                return name
            if callable(name_map):
                return name_map(name)
            return name_map.get(name, name)

        sql = "".join(
            _map(text) if is_ref else text for is_ref, text in self._statements
        )
        res = transpile_sql(sql, self._dialect, dialect)
        if log is not None:
            log.debug("constructed sql: %s", res)
        return res

    @staticmethod
    def from_expr(
        sql: str,
        prefix: str = "<tmpdf:",
        dialect: Optional[str] = None,
    ) -> "StructuredRawSQL":
        """Parse a SQL string containing ``<tmpdf:key>`` markers into the
        structured form."""
        statements: List[Tuple[bool, str]] = []
        pos = 0
        for m in _TMP_RE.finditer(sql):
            if m.start() > pos:
                statements.append((False, sql[pos : m.start()]))
            statements.append((True, m.group("key")))
            pos = m.end()
        if pos < len(sql):
            statements.append((False, sql[pos:]))
        return StructuredRawSQL(statements, dialect=dialect)
