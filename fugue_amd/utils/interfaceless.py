"""``# schema: a:int,b:str`` comment-annotation parsing.

Mirrors the behavior of the reference's ``fugue/_utils/interfaceless.py:9-67``
(parse a special comment right above a function definition) with a new
implementation based on ``inspect.getsource``.
"""
import inspect
import re
from typing import Any, Callable, Optional

_COMMENT_RE = re.compile(r"^\s*#\s*(\w+)\s*:\s*(.*?)\s*$")


def parse_comment_annotation(func: Callable, annotation: str) -> Optional[str]:
    """Find ``# <annotation>: value`` in the comment block directly above
    ``func``'s definition."""
    try:
        source_file = inspect.getsourcefile(func)
        if source_file is None:
            return None
        lines, start = inspect.getsourcelines(func)
    except (OSError, TypeError):
        return None
    try:
        with open(source_file, "r") as f:
            all_lines = f.readlines()
    except OSError:
        return None
    idx = start - 2  # line above the def (0-based)
    found: Optional[str] = None
    while idx >= 0:
        line = all_lines[idx]
        m = _COMMENT_RE.match(line)
        if m is None:
            if line.strip() == "" or line.strip().startswith("@"):
                idx -= 1
                continue
            break
        if m.group(1).lower() == annotation.lower():
            found = m.group(2)
            break
        idx -= 1
    return found


def parse_output_schema_from_comment(func: Callable) -> Optional[str]:
    value = parse_comment_annotation(func, "schema")
    if value is None or value.strip() == "":
        return None
    return value.strip()
