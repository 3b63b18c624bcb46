"""Jinja-style templating for FugueSQL (reference:
``fugue/sql/_utils.py`` ``fill_sql_template``)."""
import re
from typing import Any, Dict

_VAR_RE = re.compile(r"\{\{\s*([A-Za-z_][A-Za-z_0-9]*)\s*\}\}")


def fill_sql_template(sql: str, params: Dict[str, Any]) -> str:
    if "{{" not in sql:
        return sql
    try:
        from jinja2 import Template

        return Template(sql).render(params)
    except ImportError:
        def _sub(m: "re.Match") -> str:
            name = m.group(1)
            if name in params:
                return str(params[name])
            return m.group(0)

        return _VAR_RE.sub(_sub, sql)
