"""ASCII table pretty printer (reference behavior: ``fugue/_utils/display.py``)."""
from typing import Any, List, Optional


class PrettyTable:
    def __init__(
        self,
        schema_names: List[str],
        rows: List[List[Any]],
        best_width: int = 100,
        truncate_width: int = 500,
    ):
        self.names = schema_names
        self.rows = rows
        self.best_width = best_width
        self.truncate_width = truncate_width

    def _cell(self, v: Any) -> str:
        s = "NULL" if v is None else str(v)
        if len(s) > self.truncate_width:
            s = s[: self.truncate_width - 3] + "..."
        return s

    def to_string(self) -> str:
        table = [list(self.names)] + [
            [self._cell(v) for v in row] for row in self.rows
        ]
        widths = [
            max(len(table[r][c]) for r in range(len(table)))
            for c in range(len(self.names))
        ] if len(self.names) > 0 else []
        lines = []
        for r, row in enumerate(table):
            lines.append(
                "|" + "|".join(cell.ljust(w) for cell, w in zip(row, widths)) + "|"
            )
            if r == 0:
                lines.append("+" + "+".join("-" * w for w in widths) + "+")
        return "\n".join(lines)
