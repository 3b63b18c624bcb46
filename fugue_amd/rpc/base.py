"""Worker→driver callback channel.

Reference parity: ``fugue/rpc/base.py`` — ``RPCHandler`` lifecycle,
``RPCServer.invoke/register``, in-process ``NativeRPCServer``; the
distributed HTTP variant lives in ``fugue_amd/rpc/http.py`` (stdlib
``http.server`` instead of flask).  For the single-node 8-GPU MI355X
topology the native (in-process) server is the default.
"""
import pickle
import threading
import uuid
from typing import Any, Callable, Dict, Optional

from fugue_amd.utils.convert import to_instance, to_type
from fugue_amd.utils.hash import to_uuid
from fugue_amd.utils.params import ParamDict


class RPCClient:
    """Client interface handed to workers; calling it invokes the
    registered driver-side handler."""

    def __call__(self, *args: Any, **kwargs: Any) -> Any:  # pragma: no cover
        raise NotImplementedError


class RPCHandler(RPCClient):
    """Driver-side handler with start/stop lifecycle."""

    def __init__(self):
        self._lock = threading.RLock()
        self._running = 0

    @property
    def running(self) -> bool:
        return self._running > 0

    def __uuid__(self) -> str:
        raise NotImplementedError  # pragma: no cover

    def start_handler(self) -> None:
        ...

    def stop_handler(self) -> None:
        ...

    def start(self) -> "RPCHandler":
        with self._lock:
            if self._running == 0:
                self.start_handler()
            self._running += 1
        return self

    def stop(self) -> None:
        with self._lock:
            if self._running == 1:
                self.stop_handler()
            self._running -= 1
            if self._running < 0:
                self._running = 0

    def __enter__(self) -> "RPCHandler":
        with self._lock:
            if self._running == 0:
                raise RuntimeError("use handler.start() before entering")
        return self

    def __exit__(self, *args: Any) -> None:
        self.stop()

    def __copy__(self) -> "RPCHandler":
        return self

    def __deepcopy__(self, memo: Any) -> "RPCHandler":
        return self


class EmptyRPCHandler(RPCHandler):
    """Placeholder for no callback."""

    def __uuid__(self) -> str:
        return to_uuid("EmptyRPCHandler")


class RPCFunc(RPCHandler):
    """Wrap a plain callable as a handler."""

    def __init__(self, func: Callable):
        super().__init__()
        if not callable(func):
            raise ValueError(f"{func} is not callable")
        self._func = func

    def __call__(self, *args: Any, **kwargs: Any) -> Any:
        return self._func(*args, **kwargs)

    def __uuid__(self) -> str:
        return to_uuid("RPCFunc", self._func)


def to_rpc_handler(obj: Any) -> RPCHandler:
    if obj is None:
        return EmptyRPCHandler()
    if isinstance(obj, RPCHandler):
        return obj
    if callable(obj):
        return RPCFunc(obj)
    raise ValueError(f"{obj} can't be converted to RPCHandler")


class RPCServer(RPCHandler):
    """Registry of handlers keyed by generated names; transformers get a
    client bound to their handler's key."""

    def __init__(self, conf: Any):
        super().__init__()
        self._conf = ParamDict(conf)
        self._handlers: Dict[str, RPCHandler] = {}

    @property
    def conf(self) -> ParamDict:
        return self._conf

    def __uuid__(self) -> str:
        return to_uuid(str(type(self)), dict(self._conf))

    def make_client(self, handler: Any) -> RPCClient:  # pragma: no cover
        raise NotImplementedError

    def start_server(self) -> None:
        ...

    def stop_server(self) -> None:
        ...

    def start_handler(self) -> None:
        self.start_server()

    def stop_handler(self) -> None:
        self.stop_server()
        with self._lock:
            for v in self._handlers.values():
                if v.running:
                    v.stop()
            self._handlers.clear()

    def invoke(self, key: str, *args: Any, **kwargs: Any) -> Any:
        with self._lock:
            handler = self._handlers[key]
        return handler(*args, **kwargs)

    def register(self, handler: Any) -> str:
        with self._lock:
            key = "_" + str(uuid.uuid4()).split("-")[-1]
            if key in self._handlers:
                raise ValueError(f"handler key {key} already exists")
            self._handlers[key] = to_rpc_handler(handler).start()
            return key


class NativeRPCClient(RPCClient):
    """In-process client; unpicklable by design (single-process only)."""

    def __init__(self, server: "NativeRPCServer", key: str):
        self._key = key
        self._server = server

    def __call__(self, *args: Any, **kwargs: Any) -> Any:
        return self._server.invoke(self._key, *args, **kwargs)

    def __getstate__(self):
        raise pickle.PicklingError(f"{self} is not serializable")


class NativeRPCServer(RPCServer):
    """Reference parity: ``fugue/rpc/base.py:197``."""

    def make_client(self, handler: Any) -> RPCClient:
        key = self.register(handler)
        return NativeRPCClient(self, key)


def make_rpc_server(conf: Any = None) -> RPCServer:
    """Build the configured server (conf key ``fugue.rpc.server``)."""
    conf = ParamDict(conf)
    tp = conf.get_or_none("fugue.rpc.server", str)
    t_server = NativeRPCServer if tp is None else to_type(tp, RPCServer)
    return t_server(conf)  # type: ignore
