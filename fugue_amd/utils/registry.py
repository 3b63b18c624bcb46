"""Conditional-dispatch plugin registry.

Replaces the reference's triad ``conditional_dispatcher`` usage
(``fugue/_utils/registry.py:9``): a named registry of candidate functions,
each guarded by a matcher predicate; dispatch calls the highest-priority
matching candidate.
"""
import threading
from typing import Any, Callable, Dict, List, Optional, Tuple

_LOCK = threading.RLock()
_REGISTRY: Dict[str, List[Tuple[float, Callable[..., bool], Callable]]] = {}


class NoMatchError(ValueError):
    pass


def register_plugin(
    name: str,
    matcher: Callable[..., bool],
    func: Callable,
    priority: float = 1.0,
) -> None:
    with _LOCK:
        _REGISTRY.setdefault(name, []).append((priority, matcher, func))
        # higher priority first; later registrations win ties (stable by -index)
        _REGISTRY[name].sort(key=lambda t: -t[0])


def plugin_candidates(name: str) -> List[Tuple[float, Callable, Callable]]:
    with _LOCK:
        return list(_REGISTRY.get(name, []))


def run_plugin(name: str, *args: Any, **kwargs: Any) -> Any:
    last_err: Optional[Exception] = None
    for _, matcher, func in plugin_candidates(name):
        try:
            ok = matcher(*args, **kwargs)
        except Exception:
            continue
        if ok:
            return func(*args, **kwargs)
    raise NoMatchError(f"no plugin in {name!r} matched {args} {kwargs}")


def try_run_plugin(name: str, *args: Any, **kwargs: Any) -> Tuple[bool, Any]:
    try:
        return True, run_plugin(name, *args, **kwargs)
    except NoMatchError:
        return False, None


class ConditionalDispatcher:
    """A named plugin point with the reference's dispatcher ergonomics
    (``@parse_transformer.candidate(matcher)`` — triad
    ``conditional_dispatcher`` in ``fugue/_utils/registry.py:9``),
    backed by this module's registry."""

    def __init__(self, name: str):
        self.name = name

    def candidate(
        self, matcher: Callable[..., bool], priority: float = 1.0
    ) -> Callable[[Callable], Callable]:
        def deco(func: Callable) -> Callable:
            register_plugin(self.name, matcher, func, priority)
            return func

        return deco

    def run(self, *args: Any, **kwargs: Any) -> Tuple[bool, Any]:
        return try_run_plugin(self.name, *args, **kwargs)

    def __call__(self, *args: Any, **kwargs: Any) -> Any:
        return run_plugin(self.name, *args, **kwargs)


def fugue_plugin(name: str, priority: float = 1.0):
    """Decorator form: ``@fugue_plugin("parse_execution_engine")`` with a
    ``matcher`` attribute on the function, or pass matcher via this factory."""

    def deco(func: Callable) -> Callable:
        matcher = getattr(func, "__plugin_matcher__", lambda *a, **k: True)
        register_plugin(name, matcher, func, priority)
        return func

    return deco
