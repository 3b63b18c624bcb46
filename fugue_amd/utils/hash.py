"""Deterministic uuid hashing of nested python values.

Replaces ``triad.utils.hash.to_uuid`` (reference usage: spec-UUID
determinism for workflow tasks, ``fugue/workflow/_tasks.py:85``).  The hash
is stable across processes and runs: it never uses object ids.  Objects may
customize by defining ``__uuid__``.
"""
import uuid
import weakref
from typing import Any, Iterable

_NAMESPACE = uuid.UUID("6b1f84e2-13a1-4a73-9a33-0d67b6f0a1c5")


def _feed(obj: Any, parts: list) -> None:
    if obj is None:
        parts.append("\0N")
    elif hasattr(obj, "__uuid__"):
        parts.append("\0U" + str(obj.__uuid__()))
    elif isinstance(obj, bool):
        parts.append("\0b" + str(obj))
    elif isinstance(obj, (int, float, complex)):
        parts.append("\0n" + repr(obj))
    elif isinstance(obj, str):
        parts.append("\0s" + obj)
    elif isinstance(obj, bytes):
        parts.append("\0y" + obj.hex())
    elif isinstance(obj, dict):
        parts.append("\0d{")
        for k in obj:  # preserve insertion order (it is part of identity)
            _feed(k, parts)
            _feed(obj[k], parts)
        parts.append("}")
    elif isinstance(obj, (list, tuple)):
        parts.append("\0l[")
        for x in obj:
            _feed(x, parts)
        parts.append("]")
    elif _is_raw_frame(obj):
        parts.append("\0D" + _frame_token(obj))
    elif isinstance(obj, set):
        parts.append("\0S[")
        for x in sorted(str(i) for i in obj):
            parts.append(x + ",")
        parts.append("]")
    elif isinstance(obj, Iterable):
        parts.append("\0l[")
        for x in obj:
            _feed(x, parts)
        parts.append("]")
    elif callable(obj):
        name = getattr(obj, "__qualname__", getattr(obj, "__name__", None))
        mod = getattr(obj, "__module__", "")
        if name is None:
            name = type(obj).__qualname__
        parts.append("\0f" + mod + "." + name)
        # distinct closures over the same code differ by their captured
        # values — include them so such functions don't hash-collide
        # (still process-stable: values, never object ids)
        clo = getattr(obj, "__closure__", None)
        if clo:
            for cell in clo:
                try:
                    _feed(cell.cell_contents, parts)
                except Exception:  # pragma: no cover - self cycles etc.
                    parts.append("\0?")
    else:
        parts.append("\0o" + repr(obj))


_SMALL_FRAME_ROWS = 10_000


def _is_raw_frame(obj: Any) -> bool:
    cls = type(obj).__module__ + "." + type(obj).__name__
    if cls in ("pandas.core.frame.DataFrame", "pyarrow.lib.Table"):
        return True
    # fugue local bounded frames: identity must include the data, not
    # just the repr/schema (two distinct literals must hash apart)
    return (
        hasattr(obj, "as_array")
        and hasattr(obj, "schema")
        and getattr(obj, "is_local", False)
        and getattr(obj, "is_bounded", False)
    )


_FRAME_TOKEN_MEMO: dict = {}
_FRAME_TOKEN_MEMO_CAP = 256


def _frame_token(obj: Any) -> str:
    """Identity token for a raw pandas/arrow frame, memoized per object
    identity (frames are treated as immutable, so the same object always
    tokenizes the same — repeated workflow builds over one frame skip
    re-serialization)."""
    key = id(obj)
    hit = _FRAME_TOKEN_MEMO.get(key)
    if hit is not None and hit[0]() is obj:
        return hit[1]
    tok = _frame_token_impl(obj)
    try:
        ref = weakref.ref(obj)
    except TypeError:
        return tok
    if len(_FRAME_TOKEN_MEMO) >= _FRAME_TOKEN_MEMO_CAP:
        _FRAME_TOKEN_MEMO.pop(next(iter(_FRAME_TOKEN_MEMO)))
    _FRAME_TOKEN_MEMO[key] = (ref, tok)
    return tok


def _frame_token_impl(obj: Any) -> str:
    """Content-based for small frames (so identical literals dedup
    deterministically), id-based for large ones (content hashing a
    1e8-row frame is not acceptable — such tasks are simply not
    deterministic across runs)."""
    try:
        n = len(obj)
    except TypeError:
        # pyarrow Table has num_rows; fugue frames have count()
        n = obj.num_rows if hasattr(obj, "num_rows") else obj.count()
    if n <= _SMALL_FRAME_ROWS:
        try:
            import pandas as pd

            if isinstance(obj, pd.DataFrame):
                return (
                    "c" + str(list(obj.columns)) + repr(obj.values.tolist())
                )
            if hasattr(obj, "to_pylist"):  # pyarrow Table
                return "c" + str(obj.schema) + repr(obj.to_pylist())
            # fugue local bounded frame
            return "c" + str(obj.schema) + repr(obj.as_array())
        except Exception:
            pass
    return "i" + str(id(obj))


def to_uuid(*args: Any) -> str:
    parts: list = []
    for a in args:
        _feed(a, parts)
    return str(uuid.uuid5(_NAMESPACE, "".join(parts)))
