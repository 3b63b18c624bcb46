"""pytest helpers: backend fixtures + ``@with_backend`` parametrization.

Reference parity: ``fugue/test/plugins.py`` (``with_backend`` :39,
``fugue_test_suite`` :194, ``FugueTestBackend`` :226).  Registered
backends: "native"/"pandas" (CPU) and "hip" (MI355X engine; CPU tensors
when no GPU is present).
"""
import threading
from contextlib import contextmanager
from dataclasses import dataclass
from typing import Any, Callable, Dict, Iterator, List, Optional, Type

import pytest

from fugue_amd.dataframe.utils import _df_eq
from fugue_amd.execution.execution_engine import ExecutionEngine
from fugue_amd.execution.factory import make_execution_engine

_BACKENDS: Dict[str, Type["FugueTestBackend"]] = {}
_LOCK = threading.RLock()
_GLOBAL_TEST_CONF: Dict[str, Any] = {}


def set_global_test_conf(conf: Dict[str, Any]) -> None:
    """Install the parsed ``fugue_test_conf`` pytest-ini mapping (called
    by ``fugue_amd.test.pytest_plugin.pytest_configure``)."""
    with _LOCK:
        _GLOBAL_TEST_CONF.clear()
        _GLOBAL_TEST_CONF.update(conf)


def _backend_conf(name: str, base: Dict[str, Any]) -> Dict[str, Any]:
    """Merge backend class conf with ini entries: ``<name>.key`` entries
    override (prefix stripped); unprefixed entries apply to all backends
    (reference parity: ``fugue/test/plugins.py`` ``fugue_test_conf``)."""
    out = dict(base)
    with _LOCK:
        for k, v in _GLOBAL_TEST_CONF.items():
            if "." in k and k.split(".", 1)[0] in _BACKENDS:
                if k.startswith(name + "."):
                    out[k.split(".", 1)[1]] = v
            else:
                out[k] = v
    return out


@dataclass
class FugueTestContext:
    engine: ExecutionEngine
    session: Any
    name: str


class FugueTestBackend:
    name = ""
    conf: Dict[str, Any] = {}

    @classmethod
    @contextmanager
    def session_context(cls) -> Iterator[Any]:
        yield None

    @classmethod
    @contextmanager
    def context(cls) -> Iterator[FugueTestContext]:
        with cls.session_context() as session:
            engine = make_execution_engine(
                cls.name if session is None else session,
                _backend_conf(cls.name, cls.conf),
            )
            yield FugueTestContext(engine=engine, session=session, name=cls.name)


def fugue_test_backend(cls: Type[FugueTestBackend]) -> Type[FugueTestBackend]:
    """Register a test backend class."""
    with _LOCK:
        _BACKENDS[cls.name] = cls
    return cls


def _get_backend(name: str) -> Type[FugueTestBackend]:
    with _LOCK:
        if name not in _BACKENDS:
            raise KeyError(
                f"test backend {name!r} not registered; "
                f"available: {list(_BACKENDS)}"
            )
        return _BACKENDS[name]


def with_backend(*names: str) -> Callable:
    """Parametrize a test across engines; the test receives a
    ``FugueTestContext`` as its ``backend_context`` argument."""

    def deco(func: Callable) -> Callable:
        @pytest.mark.parametrize("fugue_backend_name", list(names))
        def wrapper(fugue_backend_name: str, *args: Any, **kwargs: Any) -> Any:
            backend = _get_backend(fugue_backend_name)
            with backend.context() as ctx:
                with ctx.engine.as_context():
                    return func(*args, backend_context=ctx, **kwargs)

        wrapper.__name__ = func.__name__
        return wrapper

    return deco


class FugueTestSuite:
    """Base class for user test suites bound to a backend with
    :func:`fugue_test_suite` (reference ``fugue/test/plugins.py:139``):
    exposes ``engine`` (lazily built from the bound backend) and
    ``df_eq``."""

    backend: str = ""
    __test__ = False
    _engine: Any = None

    @property
    def engine(self) -> Any:
        if type(self)._engine is None:
            type(self)._engine = type(self).make_engine()  # type: ignore
        return type(self)._engine

    def df_eq(self, *args: Any, **kwargs: Any) -> bool:
        return _df_eq(*args, **kwargs)


def extract_conf(
    conf: Dict[str, Any], prefix: str, remove_prefix: bool
) -> Dict[str, Any]:
    """Extract config entries under ``prefix`` (reference
    ``fugue/test/plugins.py:315``)."""
    res: Dict[str, Any] = {}
    for k, v in conf.items():
        if k.startswith(prefix):
            res[k[len(prefix):] if remove_prefix else k] = v
    return res


def fugue_test_suite(backend: str, mark_test: Optional[Any] = None) -> Callable:
    """Class decorator binding a conformance suite to a backend."""

    def deco(cls: type) -> type:
        backend_cls = _get_backend(backend)

        @classmethod
        def make_engine(kls) -> ExecutionEngine:  # type: ignore
            with backend_cls.session_context() as session:
                return make_execution_engine(
                    backend if session is None else session, backend_cls.conf
                )

        cls.make_engine = make_engine  # type: ignore
        if not hasattr(cls, "df_eq"):
            cls.df_eq = staticmethod(_df_eq)  # type: ignore
        cls.backend = backend  # type: ignore
        if isinstance(cls, type) and issubclass(cls, FugueTestSuite):
            cls.__test__ = True  # re-enable collection for bound suites
        if mark_test is not None:
            cls = mark_test(cls)
        return cls

    return deco


@fugue_test_backend
class _NativeTestBackend(FugueTestBackend):
    name = "native"


@fugue_test_backend
class _PandasTestBackend(FugueTestBackend):
    name = "pandas"


@fugue_test_backend
class _HipTestBackend(FugueTestBackend):
    name = "hip"
