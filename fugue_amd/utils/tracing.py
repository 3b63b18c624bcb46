"""Per-op tracing ranges, visible in rocprofv3 runtime traces.

SURVEY.md §5 (tracing): the reference has only error-site tracing; the
MI355X build adds roctx-compatible ranges around engine operations so a
``rocprofv3 --runtime-trace`` run attributes kernels and collectives to
the relational op that issued them.  ``torch.cuda.nvtx`` maps to roctx on
ROCm builds.
"""
from contextlib import contextmanager
from typing import Iterator

_ENABLED = [True]


def set_tracing(enabled: bool) -> None:
    _ENABLED[0] = enabled


@contextmanager
def op_range(name: str) -> Iterator[None]:
    """Mark an engine operation; no-op when tracing disabled or no GPU."""
    pushed = False
    if _ENABLED[0]:
        try:
            import torch

            if torch.cuda.is_available():
                torch.cuda.nvtx.range_push(name)
                pushed = True
        except Exception:
            pass
    try:
        yield
    finally:
        if pushed:
            import torch

            torch.cuda.nvtx.range_pop()
