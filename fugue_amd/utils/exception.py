"""Traceback rewriting: make DAG-runtime errors point at the user's
compile-time call site (reference behavior: ``fugue/_utils/exception.py``)."""
import sys
import types
from typing import Iterable, List, Optional, Tuple


def frames_to_traceback(
    frame: Optional[types.FrameType],
    limit: int,
    should_prune: Optional[callable] = None,
) -> Optional[types.TracebackType]:
    """Build a TracebackType chain from a live frame stack (innermost first),
    pruning frames whose module matches ``should_prune``."""
    tb: Optional[types.TracebackType] = None
    count = 0
    while frame is not None and count < limit:
        module = frame.f_globals.get("__name__", "")
        if should_prune is None or not should_prune(module):
            tb = types.TracebackType(tb, frame, frame.f_lasti, frame.f_lineno)
            count += 1
        frame = frame.f_back
    return tb


def modify_traceback(
    exc: BaseException,
    extra_tb: Optional[types.TracebackType],
    should_prune: Optional[callable] = None,
) -> BaseException:
    """Append ``extra_tb`` (the compile-time call site) to the exception's
    traceback, pruning framework frames."""
    tb = exc.__traceback__
    frames: List[types.TracebackType] = []
    while tb is not None:
        module = tb.tb_frame.f_globals.get("__name__", "")
        if should_prune is None or not should_prune(module):
            frames.append(tb)
        tb = tb.tb_next
    new_tb = extra_tb
    for t in reversed(frames):
        new_tb = types.TracebackType(new_tb, t.tb_frame, t.tb_lasti, t.tb_lineno)
    return exc.with_traceback(new_tb)


def make_prune_predicate(prefixes: Iterable[str]):
    prefixes = tuple(prefixes)

    def should_prune(module: str) -> bool:
        return any(module == p or module.startswith(p + ".") for p in prefixes)

    return should_prune
